"""Anchor generation goldens (reference parity: 261,888 anchors @1024^2)."""

import numpy as np

from maskrcnn_amd.config import Config, compute_backbone_shapes
from maskrcnn_amd.utils.anchors import (AnchorCache, generate_anchors,
                                        generate_pyramid_anchors)


class C(Config):
    NAME = "t"
    NUM_CLASSES = 2


def test_anchor_count_golden():
    cfg = C()
    shapes = compute_backbone_shapes(cfg, [1024, 1024, 3])
    a = generate_pyramid_anchors(cfg.RPN_ANCHOR_SCALES,
                                 cfg.RPN_ANCHOR_RATIOS, shapes,
                                 cfg.BACKBONE_STRIDES,
                                 cfg.RPN_ANCHOR_STRIDE)
    # 3 * (256^2 + 128^2 + 64^2 + 32^2 + 16^2) = 261,888
    assert a.shape == (261888, 4)


def test_single_level_geometry():
    a = generate_anchors([32], [0.5, 1, 2], [4, 4], 16, 1)
    assert a.shape == (4 * 4 * 3, 4)
    # Ratio-1 anchor at cell (0,0) is a 32x32 box centered at origin.
    sq = a[1]
    np.testing.assert_allclose(sq, [-16, -16, 16, 16])
    # Anchors preserve area across ratios (h*w == scale^2).
    h = a[:, 2] - a[:, 0]
    w = a[:, 3] - a[:, 1]
    np.testing.assert_allclose(h * w, 32 * 32, rtol=1e-6)


def test_cache_normalized_range():
    cfg = C()
    cache = AnchorCache(cfg)
    t = cache.get([1024, 1024, 3])
    assert t.shape == (261888, 4)
    # Normalized anchors can exceed [0,1] at borders but must be finite and
    # centered in the unit square on average.
    assert float(t.mean()) > 0.2 and float(t.mean()) < 0.8

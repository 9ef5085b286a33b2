"""Vectorized (sync-free) DetectionTargetLayer: invariants + equivalence
of the selected-sample properties with the loop variant."""

import pytest
import torch

from maskrcnn_amd.model.targets import DetectionTargetLayer
from maskrcnn_amd.utils.boxes import box_refinement_t, overlaps_t


def _make_inputs(seed=0, B=2, P=120, G=6, crowd=False):
    g = torch.Generator().manual_seed(seed)
    proposals = torch.rand(B, P, 4, generator=g)
    proposals[..., 2:] = proposals[..., :2] + \
        torch.rand(B, P, 2, generator=g) * 0.3 + 0.02
    proposals.clamp_(0, 1)
    # A few proposals duplicated near each GT so positives exist.
    gt_boxes = torch.rand(B, G, 4, generator=g) * 0.5
    gt_boxes[..., 2:] = gt_boxes[..., :2] + 0.25
    gt_boxes.clamp_(0, 1)
    for b in range(B):
        for i in range(G):
            proposals[b, i * 3] = gt_boxes[b, i] + 0.01
            proposals[b, i * 3 + 1] = gt_boxes[b, i] - 0.005
    gt_ids = torch.randint(1, 4, (B, G), generator=g)
    if crowd:
        gt_ids[:, -1] *= -1
    gt_masks = (torch.rand(B, 56, 56, G, generator=g) > 0.5).float()
    # Zero-pad tail rows (reference padding semantics).
    gt_ids[:, G - 1 - int(crowd)] = 0
    gt_boxes[:, G - 1 - int(crowd)] = 0
    return proposals, gt_ids, gt_boxes, gt_masks


@pytest.fixture
def layer(shapes_config, monkeypatch):
    monkeypatch.setenv("MASKRCNN_VECTORIZED_TARGETS", "1")
    return DetectionTargetLayer(shapes_config)


def test_vectorized_shapes_and_padding(layer, shapes_config):
    cfg = shapes_config
    proposals, ids, boxes, masks = _make_inputs()
    rois, cls, deltas, m = layer(proposals, ids, boxes, masks)
    T = cfg.TRAIN_ROIS_PER_IMAGE
    assert rois.shape == (2, T, 4)
    assert cls.shape == (2, T)
    assert deltas.shape == (2, T, 4)
    assert m.shape == (2, T, *cfg.MASK_SHAPE)
    assert torch.isfinite(deltas).all()
    # Zero-class slots have zero rois/deltas/masks.
    empty = cls == 0
    assert (deltas[empty].abs().sum()) == 0
    assert (m[empty].abs().sum()) == 0


def test_vectorized_positive_invariants(layer, shapes_config):
    cfg = shapes_config
    torch.manual_seed(1)
    proposals, ids, boxes, masks = _make_inputs(seed=1)
    rois, cls, deltas, m = layer(proposals, ids, boxes, masks)
    cap_pos = int(cfg.TRAIN_ROIS_PER_IMAGE * cfg.ROI_POSITIVE_RATIO)
    std = torch.tensor(cfg.BBOX_STD_DEV)
    for b in range(2):
        pos = (cls[b] > 0).nonzero(as_tuple=True)[0]
        assert pos.numel() > 0
        assert pos.max() < cap_pos  # positives live in the head slots
        for i in pos.tolist():
            ov = overlaps_t(rois[b, i:i + 1],
                            boxes[b][ids[b] > 0]).max()
            assert ov >= 0.5 - 1e-5
            # Delta reproduces the refinement to the assigned gt.
            ovl = overlaps_t(rois[b, i:i + 1], boxes[b]).squeeze(0)
            ovl[ids[b] <= 0] = -1
            gt = boxes[b][ovl.argmax()]
            ref = box_refinement_t(rois[b, i:i + 1], gt[None]) / std
            torch.testing.assert_close(deltas[b, i], ref[0], atol=1e-4,
                                       rtol=1e-4)


def test_vectorized_ratio_matches_loop_semantics(shapes_config,
                                                 monkeypatch):
    """pos:neg counts follow the reference rule
    (neg = floor(r*pos) - pos) in both variants."""
    proposals, ids, boxes, masks = _make_inputs(seed=2)
    monkeypatch.setenv("MASKRCNN_VECTORIZED_TARGETS", "1")
    torch.manual_seed(0)
    vec = DetectionTargetLayer(shapes_config)
    _, cls_v, _, _ = vec(proposals, ids, boxes, masks)
    monkeypatch.setenv("MASKRCNN_VECTORIZED_TARGETS", "0")
    torch.manual_seed(0)
    loop = DetectionTargetLayer(shapes_config)
    assert not loop.vectorized
    rois_l, cls_l, _, _ = loop(proposals, ids, boxes, masks)
    r = 1.0 / shapes_config.ROI_POSITIVE_RATIO
    for b in range(2):
        n_pos_v = int((cls_v[b] > 0).sum())
        n_pos_l = int((cls_l[b] > 0).sum())
        # Same positive pool -> same capped count.
        assert n_pos_v == n_pos_l
        n_rois_l = int((rois_l[b].abs().sum(dim=1) > 0).sum())
        expected_neg = int(r * n_pos_l) - n_pos_l
        # Loop variant: total = pos + min(neg_avail, expected).
        assert n_rois_l <= n_pos_l + expected_neg


def test_vectorized_crowd_exclusion(layer, shapes_config):
    proposals, ids, boxes, masks = _make_inputs(seed=3, crowd=True)
    rois, cls, deltas, m = layer(proposals, ids, boxes, masks)
    # No positive may be assigned a crowd (negative) class.
    assert (cls >= 0).all()


def test_vectorized_no_mini_mask(shapes_config, monkeypatch):
    class C(type(shapes_config)):
        USE_MINI_MASK = False
    monkeypatch.setenv("MASKRCNN_VECTORIZED_TARGETS", "1")
    layer = DetectionTargetLayer(C())
    proposals, ids, boxes, _ = _make_inputs(seed=4)
    full_masks = (torch.rand(2, 128, 128, 6) > 0.5).float()
    rois, cls, deltas, m = layer(proposals, ids, boxes, full_masks)
    assert torch.isfinite(m).all()
    assert set(m.unique().tolist()) <= {0.0, 1.0}

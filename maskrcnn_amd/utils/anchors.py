"""Pyramid anchor generation.

Replicates reference mrcnn/utils.py generate_anchors / generate_pyramid_anchors
(~lines 760-830): per level, a scales x ratios grid centered at stride
offsets; levels concatenated P2..P6 giving 261,888 anchors at 1024x1024.
Vectorized numpy; a cached torch variant is provided for the model.
"""

from __future__ import annotations

import numpy as np
import torch


def generate_anchors(scales, ratios, shape, feature_stride, anchor_stride):
    """Anchors for one feature map level.

    scales: 1D array of anchor sizes in pixels (e.g. [32])
    ratios: 1D array of w/h ratios (e.g. [0.5, 1, 2])
    shape: [height, width] of the feature map
    feature_stride: feature map stride relative to the image, in pixels
    anchor_stride: cell stride on the feature map (usually 1)

    Returns [N, (y1, x1, y2, x2)] in image pixel coordinates.
    """
    scales, ratios = np.meshgrid(np.array(scales), np.array(ratios))
    scales = scales.flatten()
    ratios = ratios.flatten()

    heights = scales / np.sqrt(ratios)
    widths = scales * np.sqrt(ratios)

    shifts_y = np.arange(0, shape[0], anchor_stride) * feature_stride
    shifts_x = np.arange(0, shape[1], anchor_stride) * feature_stride
    shifts_x, shifts_y = np.meshgrid(shifts_x, shifts_y)

    box_widths, box_centers_x = np.meshgrid(widths, shifts_x)
    box_heights, box_centers_y = np.meshgrid(heights, shifts_y)

    box_centers = np.stack([box_centers_y, box_centers_x],
                           axis=2).reshape([-1, 2])
    box_sizes = np.stack([box_heights, box_widths], axis=2).reshape([-1, 2])

    boxes = np.concatenate([box_centers - 0.5 * box_sizes,
                            box_centers + 0.5 * box_sizes], axis=1)
    return boxes


def generate_pyramid_anchors(scales, ratios, feature_shapes, feature_strides,
                             anchor_stride):
    """Anchors at all pyramid levels, concatenated (level order preserved).

    Returns [A, (y1, x1, y2, x2)] pixel coords; A = 261,888 at 1024^2 with
    the default config.
    """
    anchors = []
    for i in range(len(scales)):
        anchors.append(generate_anchors(scales[i], ratios, feature_shapes[i],
                                        feature_strides[i], anchor_stride))
    return np.concatenate(anchors, axis=0)


class AnchorCache:
    """Per-image-shape cache of normalized pyramid anchors as torch tensors.

    Mirrors MaskRCNN.get_anchors (reference mrcnn/model.py:~2420).
    """

    def __init__(self, config):
        self.config = config
        self._cache = {}

    def get(self, image_shape, device="cpu") -> torch.Tensor:
        """Normalized anchors [A, 4] for the given image shape."""
        from ..config import compute_backbone_shapes
        from .boxes import norm_boxes
        key = tuple(image_shape[:2])
        if key not in self._cache:
            backbone_shapes = compute_backbone_shapes(self.config, image_shape)
            a = generate_pyramid_anchors(
                self.config.RPN_ANCHOR_SCALES,
                self.config.RPN_ANCHOR_RATIOS,
                backbone_shapes,
                self.config.BACKBONE_STRIDES,
                self.config.RPN_ANCHOR_STRIDE)
            self._cache[key] = {
                "pixel": a,
                "norm": torch.from_numpy(norm_boxes(a, image_shape[:2])),
            }
        return self._cache[key]["norm"].to(device)

    def get_pixel(self, image_shape) -> np.ndarray:
        self.get(image_shape)
        return self._cache[tuple(image_shape[:2])]["pixel"]

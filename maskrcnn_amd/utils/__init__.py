"""Utility subpackage: boxes, anchors, image resize/mold, masks, metrics.

Functional parity targets are listed per-module (reference mrcnn/utils.py).
"""

from .boxes import (  # noqa: F401
    extract_bboxes, compute_iou, compute_overlaps, compute_overlaps_masks,
    non_max_suppression, apply_box_deltas, box_refinement,
    norm_boxes, denorm_boxes, norm_boxes_t, denorm_boxes_t,
    apply_box_deltas_t, clip_boxes_t, box_refinement_t, overlaps_t,
    batch_slice,
)
from .anchors import generate_anchors, generate_pyramid_anchors  # noqa: F401
from .image import (  # noqa: F401
    resize_image, resize, mold_image, unmold_image, download_trained_weights,
)
from .masks import (  # noqa: F401
    resize_mask, minimize_mask, expand_mask, mold_mask, unmold_mask,
)
from .meta import (  # noqa: F401
    compose_image_meta, log, parse_image_meta, parse_image_meta_t,
)
from .metrics import (  # noqa: F401
    compute_matches, compute_ap, compute_ap_range, compute_recall,
)

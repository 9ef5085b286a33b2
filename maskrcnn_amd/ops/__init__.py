"""Compute ops: RoIAlign, NMS, detection post-processing.

Each op has a hand-written CDNA4 HIP kernel (csrc/hip/*) used on GPU and a
pure-PyTorch reference used on CPU and as the numerics oracle in tests.
On a GPU box the HIP extension is REQUIRED — ops raise if it is missing
(no silent eager fallback on cuda tensors).
"""

from .roi_align import roi_align, crop_and_resize  # noqa: F401
from .nms import (  # noqa: F401
    nms, batched_nms, nms_padded, batched_nms_padded,
)
from .fpn_fuse import fpn_fuse  # noqa: F401
from .conv3x3 import conv3x3_bias_relu, conv3x3_bn_relu  # noqa: F401

"""maskrcnn_amd — an MI355X-native Mask R-CNN training & serving stack.

A from-scratch PyTorch-ROCm + HIP/CDNA4 framework with the capability
surface of matterport/Mask_RCNN + TF-Serving (the reference repo
huyhoang17/matterport-maskrcnn-with-tensorflow-serving).
"""

__version__ = "0.1.0"

from .config import Config  # noqa: F401

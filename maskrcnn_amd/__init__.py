"""maskrcnn_amd — an MI355X-native Mask R-CNN training & serving stack.

A from-scratch PyTorch-ROCm + HIP/CDNA4 framework with the capability
surface of matterport/Mask_RCNN + TF-Serving (the reference repo
huyhoang17/matterport-maskrcnn-with-tensorflow-serving).
"""

__version__ = "0.1.0"

# Stage the in-repo MIOpen find-db BEFORE torch/libMIOpen can load (see
# _rocm_env.py and profiles/r01_naive_conv.md). Torch-free import.
from ._rocm_env import stage_miopen_db as _stage_miopen_db

_stage_miopen_db(int(__import__("os").environ.get("LOCAL_RANK", "0")))

from .config import Config  # noqa: F401

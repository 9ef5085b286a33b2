"""maskrcnn_amd — an MI355X-native Mask R-CNN training & serving stack.

A from-scratch PyTorch-ROCm + HIP/CDNA4 framework with the capability
surface of matterport/Mask_RCNN + TF-Serving (the reference repo
huyhoang17/matterport-maskrcnn-with-tensorflow-serving).
"""

__version__ = "0.1.0"

from .config import Config  # noqa: F401


def __getattr__(name):
    # Lazy top-level facade imports (keep `import maskrcnn_amd` light —
    # pulling the model pulls torch). importlib, NOT `from . import x`:
    # the latter re-enters this __getattr__ and recurses.
    import importlib
    if name == "MaskRCNN":
        return importlib.import_module(".model", __name__).MaskRCNN
    if name == "visualize":
        return importlib.import_module(".visualize", __name__)
    raise AttributeError(f"module 'maskrcnn_amd' has no attribute {name!r}")

from .mask_rcnn import MaskRCNN, MaskRCNNModel  # noqa: F401

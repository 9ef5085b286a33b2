"""Base configuration class for the MI355X-native Mask R-CNN stack.

Field names and defaults replicate the reference's ``mrcnn/config.py`` (see
SURVEY.md P1, reference mrcnn/config.py:1-235): the attribute set IS the model
spec (anchor scales, strides, bbox std-devs, thresholds, image-meta layout),
so names are kept verbatim.  The implementation is new: a plain Python class
with derived values computed in ``__init__`` plus dict/yaml override support.
"""

from __future__ import annotations

import math


class Config:
    """Base configuration. Subclass and override attributes, then instantiate.

    Mirrors reference mrcnn/config.py Config (field-for-field).
    """

    # Name the configuration (used in log dirs / checkpoint filenames).
    NAME: str = None  # Override in subclass

    # Number of GPUs and images per GPU. BATCH_SIZE is derived.
    GPU_COUNT = 1
    IMAGES_PER_GPU = 2

    # Steps per epoch / validation steps (training-loop bookkeeping).
    STEPS_PER_EPOCH = 1000
    VALIDATION_STEPS = 50

    # Backbone: "resnet50" or "resnet101".
    BACKBONE = "resnet101"

    # Only useful if you supply a callable backbone (kept for API parity).
    COMPUTE_BACKBONE_SHAPE = None

    # Strides of each FPN pyramid level (C2..C6 / P2..P6).
    BACKBONE_STRIDES = [4, 8, 16, 32, 64]

    # Size of the fully-connected layers in the classification head.
    FPN_CLASSIF_FC_LAYERS_SIZE = 1024

    # Size of the top-down FPN layers.
    TOP_DOWN_PYRAMID_SIZE = 256

    # Number of classification classes (including background).
    NUM_CLASSES = 1  # Override in subclass

    # Length of square anchor side in pixels, one per pyramid level.
    RPN_ANCHOR_SCALES = (32, 64, 128, 256, 512)

    # Anchor aspect ratios (width/height) at each cell.
    RPN_ANCHOR_RATIOS = [0.5, 1, 2]

    # Anchor stride on the feature map (1 = every cell).
    RPN_ANCHOR_STRIDE = 1

    # NMS threshold used to filter RPN proposals.
    RPN_NMS_THRESHOLD = 0.7

    # How many anchors per image to use for RPN training.
    RPN_TRAIN_ANCHORS_PER_IMAGE = 256

    # ROIs kept before proposal NMS.
    PRE_NMS_LIMIT = 6000

    # ROIs kept after proposal NMS.
    POST_NMS_ROIS_TRAINING = 2000
    POST_NMS_ROIS_INFERENCE = 1000

    # If enabled, GT instance masks are stored resized to MINI_MASK_SHAPE.
    USE_MINI_MASK = True
    MINI_MASK_SHAPE = (56, 56)

    # Input image resizing. Modes: "none", "square", "pad64", "crop".
    IMAGE_RESIZE_MODE = "square"
    IMAGE_MIN_DIM = 800
    IMAGE_MAX_DIM = 1024
    # Minimum scale factor (e.g. 2.0 for nucleus-style upscaling); 0 disables.
    IMAGE_MIN_SCALE = 0
    # Number of color channels.
    IMAGE_CHANNEL_COUNT = 3

    # Image mean (RGB).
    MEAN_PIXEL = [123.7, 116.8, 103.9]

    # Number of ROIs per image fed to the classifier/mask heads (training).
    TRAIN_ROIS_PER_IMAGE = 200

    # Percent of positive ROIs used to train classifier/mask heads.
    ROI_POSITIVE_RATIO = 0.33

    # Pooled ROI sizes.
    POOL_SIZE = 7
    MASK_POOL_SIZE = 14

    # Shape of output masks (changing this requires changes to the mask head).
    MASK_SHAPE = [28, 28]

    # Maximum number of ground truth instances in one image.
    MAX_GT_INSTANCES = 100

    # Bounding box refinement standard deviations.
    RPN_BBOX_STD_DEV = [0.1, 0.1, 0.2, 0.2]
    BBOX_STD_DEV = [0.1, 0.1, 0.2, 0.2]

    # Maximum number of final detections.
    DETECTION_MAX_INSTANCES = 100

    # Minimum probability to accept a detected instance.
    DETECTION_MIN_CONFIDENCE = 0.7

    # NMS threshold for detections.
    DETECTION_NMS_THRESHOLD = 0.3

    # Learning rate and momentum.
    LEARNING_RATE = 0.001
    LEARNING_MOMENTUM = 0.9

    # Weight decay regularization.
    WEIGHT_DECAY = 0.0001

    # Loss weights for finer control.
    LOSS_WEIGHTS = {
        "rpn_class_loss": 1.,
        "rpn_bbox_loss": 1.,
        "mrcnn_class_loss": 1.,
        "mrcnn_bbox_loss": 1.,
        "mrcnn_mask_loss": 1.,
    }

    # Use RPN ROIs or externally generated ROIs for training.
    USE_RPN_ROIS = True

    # Train or freeze batch normalization layers.
    #   None: freeze (the default — matches reference TRAIN_BN=False usage)
    #   False: freeze
    #   True: train
    TRAIN_BN = False

    # Gradient norm clipping.
    GRADIENT_CLIP_NORM = 5.0

    # --- MI355X-native additions (absent in the reference; new capability) ---
    # Compute dtype for autocast on GPU ("bf16" | "fp32").
    COMPUTE_DTYPE = "bf16"
    # Memory format for convs ("channels_last" maps to NHWC in MIOpen).
    CHANNELS_LAST = True
    # Use hand-written HIP kernels when on GPU (roi_align / nms / fused ops).
    USE_HIP_KERNELS = True

    def __init__(self):
        """Compute derived attributes (reference config.py:~200)."""
        self.BATCH_SIZE = self.IMAGES_PER_GPU * self.GPU_COUNT

        if self.IMAGE_RESIZE_MODE == "crop":
            self.IMAGE_SHAPE = [self.IMAGE_MIN_DIM, self.IMAGE_MIN_DIM,
                                self.IMAGE_CHANNEL_COUNT]
        else:
            self.IMAGE_SHAPE = [self.IMAGE_MAX_DIM, self.IMAGE_MAX_DIM,
                                self.IMAGE_CHANNEL_COUNT]

        # image_id(1) + original_shape(3) + molded_shape(3) + window(4)
        # + scale(1) + active_class_ids(NUM_CLASSES)
        self.IMAGE_META_SIZE = 1 + 3 + 3 + 4 + 1 + self.NUM_CLASSES

    def update(self, overrides: dict):
        """Apply a dict of attribute overrides and re-derive."""
        for k, v in overrides.items():
            if not hasattr(self, k):
                raise KeyError(f"Config has no attribute {k!r}")
            setattr(self, k, v)
        self.__init__()
        return self

    def to_dict(self) -> dict:
        return {a: getattr(self, a) for a in sorted(dir(self))
                if a.isupper()}

    def display(self):
        """Print all configuration values (reference Config.display)."""
        print("\nConfigurations:")
        for key, val in self.to_dict().items():
            print(f"{key:30} {val}")
        print("\n")


def compute_backbone_shapes(config: Config, image_shape):
    """[height, width] of each FPN stage feature map for an input shape.

    Reference: mrcnn/model.py compute_backbone_shapes (~line 160).
    """
    if config.COMPUTE_BACKBONE_SHAPE is not None:
        return config.COMPUTE_BACKBONE_SHAPE(image_shape)
    return [[int(math.ceil(image_shape[0] / s)),
             int(math.ceil(image_shape[1] / s))]
            for s in config.BACKBONE_STRIDES]

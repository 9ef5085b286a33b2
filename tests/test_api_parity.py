"""API-surface parity regression: the reference's public names exist here
(SURVEY.md §2 component tables). Guards against accidental removals."""

import numpy as np
import pytest


def test_config_attribute_parity():
    """Every reference Config attribute exists with the
    reference default where the survey pins one."""
    from maskrcnn_amd.config import Config
    expected_defaults = {
        "BACKBONE": "resnet101",
        "BACKBONE_STRIDES": [4, 8, 16, 32, 64],
        "RPN_ANCHOR_SCALES": (32, 64, 128, 256, 512),
        "RPN_ANCHOR_RATIOS": [0.5, 1, 2],
        "RPN_ANCHOR_STRIDE": 1,
        "RPN_NMS_THRESHOLD": 0.7,
        "RPN_TRAIN_ANCHORS_PER_IMAGE": 256,
        "PRE_NMS_LIMIT": 6000,
        "POST_NMS_ROIS_TRAINING": 2000,
        "POST_NMS_ROIS_INFERENCE": 1000,
        "USE_MINI_MASK": True,
        "MINI_MASK_SHAPE": (56, 56),
        "IMAGE_RESIZE_MODE": "square",
        "IMAGE_MIN_DIM": 800,
        "IMAGE_MAX_DIM": 1024,
        "IMAGE_MIN_SCALE": 0,
        "IMAGE_CHANNEL_COUNT": 3,
        "MEAN_PIXEL": [123.7, 116.8, 103.9],
        "TRAIN_ROIS_PER_IMAGE": 200,
        "ROI_POSITIVE_RATIO": 0.33,
        "POOL_SIZE": 7,
        "MASK_POOL_SIZE": 14,
        "MASK_SHAPE": [28, 28],
        "MAX_GT_INSTANCES": 100,
        "RPN_BBOX_STD_DEV": [0.1, 0.1, 0.2, 0.2],
        "BBOX_STD_DEV": [0.1, 0.1, 0.2, 0.2],
        "DETECTION_MAX_INSTANCES": 100,
        "DETECTION_MIN_CONFIDENCE": 0.7,
        "DETECTION_NMS_THRESHOLD": 0.3,
        "LEARNING_RATE": 0.001,
        "LEARNING_MOMENTUM": 0.9,
        "WEIGHT_DECAY": 0.0001,
        "USE_RPN_ROIS": True,
        "TRAIN_BN": False,
        "GRADIENT_CLIP_NORM": 5.0,
        "STEPS_PER_EPOCH": 1000,
        "VALIDATION_STEPS": 50,
        "FPN_CLASSIF_FC_LAYERS_SIZE": 1024,
        "TOP_DOWN_PYRAMID_SIZE": 256,
        "GPU_COUNT": 1,
        "IMAGES_PER_GPU": 2,
        "COMPUTE_BACKBONE_SHAPE": None,
    }
    for attr, val in expected_defaults.items():
        assert hasattr(Config, attr), attr
        assert getattr(Config, attr) == val, (attr, getattr(Config, attr))
    assert set(Config.LOSS_WEIGHTS) == {
        "rpn_class_loss", "rpn_bbox_loss", "mrcnn_class_loss",
        "mrcnn_bbox_loss", "mrcnn_mask_loss"}


def test_maskrcnn_facade_methods():
    """Reference MaskRCNN class method surface."""
    from maskrcnn_amd.model import MaskRCNN
    for name in ("build" if False else "detect", "detect_molded",
                 "mold_inputs", "unmold_detections", "get_anchors",
                 "load_weights", "save_weights", "find_last",
                 "set_log_dir", "set_trainable", "train", "run_graph",
                 "get_imagenet_weights", "get_trainable_layers",
                 "find_trainable_layer", "compute_losses"):
        assert callable(getattr(MaskRCNN, name, None)), name


def test_utils_function_surface():
    """Reference mrcnn/utils.py function surface."""
    import maskrcnn_amd.utils as U
    for name in ("extract_bboxes", "compute_iou", "compute_overlaps",
                 "compute_overlaps_masks", "non_max_suppression",
                 "apply_box_deltas", "box_refinement", "resize_image",
                 "resize_mask", "minimize_mask", "expand_mask",
                 "mold_mask", "unmold_mask", "generate_anchors",
                 "generate_pyramid_anchors", "compute_matches",
                 "compute_ap", "compute_ap_range", "compute_recall",
                 "norm_boxes", "denorm_boxes", "resize",
                 "download_trained_weights", "compose_image_meta",
                 "parse_image_meta", "mold_image", "unmold_image"):
        assert callable(getattr(U, name, None)), name
    from maskrcnn_amd.data import Dataset
    for name in ("add_class", "add_image", "prepare",
                 "map_source_class_id", "get_source_class_id",
                 "load_image", "load_mask", "image_reference",
                 "source_image_link"):
        assert callable(getattr(Dataset, name, None)), name


def test_visualize_function_surface():
    """Reference mrcnn/visualize.py function surface."""
    from maskrcnn_amd import visualize as V
    for name in ("display_images", "random_colors", "apply_mask",
                 "display_instances", "display_differences", "draw_rois",
                 "draw_box", "display_top_masks", "plot_precision_recall",
                 "plot_overlaps", "draw_boxes", "display_table",
                 "display_weight_stats"):
        assert callable(getattr(V, name, None)), name


def test_serving_version_pinning():
    """Predict with an explicit wrong version is rejected (TF-Serving
    model_spec.version semantics)."""
    from maskrcnn_amd.serving.server import ModelServer, ModelState
    from maskrcnn_amd.serving.wireproto import (ModelSpec, PredictRequest,
                                                numpy_to_tensor)
    ms = ModelServer.__new__(ModelServer)
    ms.models = {"m": ModelState("m", "/nonexistent")}
    ms.state.version = 3
    req = PredictRequest(model_spec=ModelSpec(name="m", version=2))
    req.inputs["input_image"] = numpy_to_tensor(
        np.zeros((1, 4, 4, 3), np.float32))
    req.inputs["input_image_meta"] = numpy_to_tensor(
        np.zeros((1, 12), np.float32))
    with pytest.raises(KeyError):
        ms._handle_predict(req)

"""Config derivations match the reference spec."""

from maskrcnn_amd.config import Config, compute_backbone_shapes


class CocoLikeConfig(Config):
    NAME = "coco"
    NUM_CLASSES = 81
    IMAGES_PER_GPU = 2


def test_defaults():
    c = CocoLikeConfig()
    assert c.BATCH_SIZE == 2
    assert c.IMAGE_SHAPE == [1024, 1024, 3]
    # 1 + 3 + 3 + 4 + 1 + 81 = 93 (the COCO serving meta size)
    assert c.IMAGE_META_SIZE == 93
    assert c.RPN_ANCHOR_SCALES == (32, 64, 128, 256, 512)
    assert c.BACKBONE_STRIDES == [4, 8, 16, 32, 64]
    assert c.RPN_BBOX_STD_DEV == [0.1, 0.1, 0.2, 0.2]
    assert c.MEAN_PIXEL == [123.7, 116.8, 103.9]
    assert c.POST_NMS_ROIS_TRAINING == 2000
    assert c.POST_NMS_ROIS_INFERENCE == 1000
    assert c.TRAIN_ROIS_PER_IMAGE == 200
    assert c.DETECTION_MAX_INSTANCES == 100


def test_gpu_count_scales_batch():
    class C(CocoLikeConfig):
        GPU_COUNT = 8
    assert C().BATCH_SIZE == 16


def test_backbone_shapes():
    c = CocoLikeConfig()
    shapes = compute_backbone_shapes(c, [1024, 1024, 3])
    assert shapes == [[256, 256], [128, 128], [64, 64], [32, 32], [16, 16]]


def test_update_rederives():
    c = CocoLikeConfig()
    c.update({"IMAGES_PER_GPU": 4})
    assert c.BATCH_SIZE == 4

"""Parity extras: transforms, USE_RPN_ROIS=False, metadata RPC, yaml
config, facade introspection."""

import numpy as np
import pytest
import torch


# ------------------------------------------------------------ transforms
def test_transforms_mask_synchronized():
    from maskrcnn_amd.data.transforms import (Fliplr, Flipud, Multiply,
                                              Rotate90, Sequential)
    rng = np.random.RandomState(0)
    image = rng.randint(0, 255, (32, 32, 3), dtype=np.uint8)
    masks = np.zeros((32, 32, 2), dtype=bool)
    masks[4:10, 6:20, 0] = True
    masks[20:30, 2:8, 1] = True
    aug = Sequential([Fliplr(1.0), Flipud(1.0), Rotate90(0.0),
                      Multiply((0.9, 0.9))])
    img2, m2 = aug(image, masks, rng)
    assert img2.shape == image.shape and m2.shape == masks.shape
    # Geometric ops applied to both: flipped twice = point symmetric.
    np.testing.assert_array_equal(m2[:, :, 0],
                                  np.flipud(np.fliplr(masks[:, :, 0])))
    # Multiply touched only the image.
    assert m2.dtype == bool


def test_transforms_in_load_image_gt(shapes_config, shapes_dataset):
    from maskrcnn_amd.data.targets import load_image_gt
    from maskrcnn_amd.data.transforms import Fliplr, Sequential
    aug = Sequential([Fliplr(1.0)])
    img_a, _, cls_a, box_a, mask_a = load_image_gt(
        shapes_dataset, shapes_config, 0, augmentation=None)
    img_b, _, cls_b, box_b, mask_b = load_image_gt(
        shapes_dataset, shapes_config, 0, augmentation=aug)
    np.testing.assert_array_equal(np.fliplr(img_a), img_b)
    assert cls_a.tolist() == cls_b.tolist()


# --------------------------------------------------- USE_RPN_ROIS=False
def test_generate_random_rois():
    from maskrcnn_amd.data.targets import generate_random_rois
    rng = np.random.RandomState(0)
    gt = np.array([[10, 10, 40, 50], [60, 60, 90, 90]])
    rois = generate_random_rois((128, 128, 3), 100, gt, rng=rng)
    assert rois.shape == (100, 4)
    assert (rois[:, 2] > rois[:, 0]).all()
    assert (rois[:, 3] > rois[:, 1]).all()
    assert rois.min() >= 0 and rois.max() <= 128


def test_training_with_input_rois(shapes_config, shapes_dataset):
    """USE_RPN_ROIS=False: heads train on provided rois."""
    from maskrcnn_amd.data.loader import make_data_loader
    from maskrcnn_amd.data.targets import generate_random_rois
    from maskrcnn_amd.model import MaskRCNN
    from maskrcnn_amd.utils.boxes import norm_boxes

    class C(type(shapes_config)):
        USE_RPN_ROIS = False
    cfg = C()
    m = MaskRCNN("training", cfg, model_dir="/tmp/userois_logs")
    loader = make_data_loader(shapes_dataset, cfg, batch_size=1)
    batch = next(iter(loader))
    rng = np.random.RandomState(0)
    H, W, _ = cfg.IMAGE_SHAPE
    rois_px = generate_random_rois((H, W, 3), 64, None, rng=rng)
    rois = torch.from_numpy(
        norm_boxes(rois_px.astype(np.float32), (H, W)))[None]
    anchors = m.anchor_cache.get(cfg.IMAGE_SHAPE)
    out = m.model(batch["image"], batch["image_meta"], anchors,
                  gt_class_ids=batch["gt_class_ids"],
                  gt_boxes=batch["gt_boxes"],
                  gt_masks=batch["gt_masks"], mode="training",
                  input_rois=rois)
    assert out["rois"].shape[1] == cfg.TRAIN_ROIS_PER_IMAGE


# ----------------------------------------------------------- metadata RPC
def test_get_model_metadata_roundtrip():
    from maskrcnn_amd.serving.wireproto import (GetModelMetadataRequest,
                                                GetModelMetadataResponse,
                                                ModelSpec, SignatureDef,
                                                TensorInfo, DT_FLOAT)
    req = GetModelMetadataRequest(
        model_spec=ModelSpec(name="m"), metadata_field=["signature_def"])
    d = GetModelMetadataRequest.decode(req.encode())
    assert d.model_spec.name == "m"
    assert d.metadata_field == ["signature_def"]

    sig = SignatureDef()
    sig.inputs["input_image"] = TensorInfo("input_image:0", DT_FLOAT,
                                           [-1, -1, -1, 3])
    resp = GetModelMetadataResponse(
        model_spec=ModelSpec(name="m", version=7),
        signature_defs={"serving_default": sig})
    d = GetModelMetadataResponse.decode(resp.encode())
    assert d.model_spec.version == 7
    assert "serving_default" in d.signature_defs
    assert d.signature_defs["serving_default"] \
        .inputs["input_image"].shape == [-1, -1, -1, 3]


def test_get_model_metadata_rpc(served_model_module):
    server, cfg, _ = served_model_module
    import grpc
    from maskrcnn_amd.serving.wireproto import (GetModelMetadataRequest,
                                                GetModelMetadataResponse,
                                                ModelSpec)
    channel = grpc.insecure_channel(f"localhost:{server.grpc_port}")
    stub = channel.unary_unary(
        "/tensorflow.serving.PredictionService/GetModelMetadata",
        request_serializer=lambda r: r.encode(),
        response_deserializer=GetModelMetadataResponse.decode)
    resp = stub(GetModelMetadataRequest(
        model_spec=ModelSpec(name="mask_rcnn")), timeout=30)
    assert set(resp.signature_defs["serving_default"].inputs) == {
        "input_image", "input_image_meta", "input_anchors"}


@pytest.fixture(scope="module")
def served_model_module(tmp_path_factory):
    from maskrcnn_amd.data.synthetic import ShapesConfig
    from maskrcnn_amd.model import MaskRCNN
    from maskrcnn_amd.serving.savedmodel import export_saved_model
    from maskrcnn_amd.serving.server import ModelServer

    class SmallConfig(ShapesConfig):
        IMAGES_PER_GPU = 1
        PRE_NMS_LIMIT = 200
        POST_NMS_ROIS_INFERENCE = 30
    cfg = SmallConfig()
    tmp = tmp_path_factory.mktemp("meta_rpc")
    m = MaskRCNN("inference", cfg, model_dir=str(tmp))
    base = str(tmp / "models" / "mask_rcnn")
    export_saved_model(m, cfg, base, version=1)
    server = ModelServer("mask_rcnn", base, grpc_port=0, rest_port=None,
                         config=cfg)
    server.start()
    yield server, cfg, m
    server.stop()


# ------------------------------------------------------------ yaml config
def test_config_yaml_roundtrip(tmp_path):
    import yaml
    from maskrcnn_amd.config import Config

    class C(Config):
        NAME = "yamltest"
        NUM_CLASSES = 5
    c = C()
    path = tmp_path / "cfg.yaml"
    with open(path, "w") as f:
        yaml.safe_dump({"IMAGES_PER_GPU": 4, "IMAGE_MAX_DIM": 512,
                        "IMAGE_MIN_DIM": 512}, f)
    with open(path) as f:
        c.update(yaml.safe_load(f))
    assert c.BATCH_SIZE == 4
    assert c.IMAGE_SHAPE[0] == 512


# ------------------------------------------------------- facade introspect
def test_trainable_layer_introspection(shapes_config):
    from maskrcnn_amd.model import MaskRCNN
    m = MaskRCNN("training", shapes_config, model_dir="/tmp/introspect")
    m.set_trainable("heads")
    layers = m.get_trainable_layers()
    names = [n for n, _ in layers]
    assert any(n.startswith("rpn") for n in names)
    assert not any(n.startswith("backbone") for n in names)
    found = m.find_trainable_layer("rpn.rpn_conv_shared.weight")
    assert found is not None and found[0] == "rpn.rpn_conv_shared"
    with pytest.raises(RuntimeError):
        m.get_imagenet_weights()

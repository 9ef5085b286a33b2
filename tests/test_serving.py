"""Serving plane tests: wire protos, SavedModel layout, batcher, and a
full gRPC + REST Predict round trip on CPU (shapes-scale model)."""

import os
import time

import numpy as np
import pytest
import torch

from maskrcnn_amd.serving.wireproto import (
    DT_FLOAT, DT_INT32, ModelSpec, PredictRequest, PredictResponse,
    SignatureDef, TensorInfo, TensorProto, decode_saved_model,
    encode_saved_model, numpy_to_tensor, tensor_to_numpy,
)


# ------------------------------------------------------------- wireproto
def test_tensor_roundtrip_content():
    for arr in (np.random.rand(2, 3, 4).astype(np.float32),
                np.random.randint(-5, 5, (7,), dtype=np.int32),
                np.random.rand(3, 2) > 0.5,
                np.random.randint(0, 9, (2, 2), dtype=np.int64)):
        t = numpy_to_tensor(arr)
        back = tensor_to_numpy(t)
        np.testing.assert_array_equal(back, arr)
        # And through bytes.
        back2 = tensor_to_numpy(TensorProto.decode(t.encode()))
        np.testing.assert_array_equal(back2, arr)


def test_tensor_float_val_fallback():
    """Clients that send float_val lists instead of tensor_content."""
    t = TensorProto(dtype=DT_FLOAT, shape=[2, 2],
                    float_val=[1.0, 2.0, 3.0, 4.0])
    back = tensor_to_numpy(TensorProto.decode(t.encode()))
    np.testing.assert_allclose(back, [[1, 2], [3, 4]])


def test_tensor_negative_dims_and_ints():
    t = TensorProto(dtype=DT_INT32, shape=[-1, 3], int_val=[-7, 0, 9])
    d = TensorProto.decode(t.encode())
    assert d.shape == [-1, 3]
    assert d.int_val == [-7, 0, 9]


def test_golden_wire_bytes():
    """Hand-computed golden: field 1 varint 1, field 4 bytes."""
    t = TensorProto(dtype=1, shape=[2], tensor_content=b"\x01\x02")
    enc = t.encode()
    # field1 (dtype): tag 0x08, value 1
    assert enc[0:2] == b"\x08\x01"
    # field2 (shape): tag 0x12, len 4, dim submessage tag 0x12 len 2,
    # Dim.size tag 0x08 value 2
    assert enc[2:8] == b"\x12\x04\x12\x02\x08\x02"
    # field4 (content): tag 0x22 len 2
    assert enc[8:12] == b"\x22\x02\x01\x02"


def test_predict_request_roundtrip():
    req = PredictRequest(model_spec=ModelSpec(
        name="mask_rcnn", signature_name="serving_default", version=3))
    req.inputs["input_image"] = numpy_to_tensor(
        np.zeros((1, 4, 4, 3), np.float32))
    req.output_filter = ["mrcnn_detection"]
    d = PredictRequest.decode(req.encode())
    assert d.model_spec.name == "mask_rcnn"
    assert d.model_spec.version == 3
    assert d.model_spec.signature_name == "serving_default"
    assert list(d.inputs) == ["input_image"]
    assert tensor_to_numpy(d.inputs["input_image"]).shape == (1, 4, 4, 3)
    assert d.output_filter == ["mrcnn_detection"]


def test_parity_with_protobuf_runtime():
    """Cross-check our encoder against the official protobuf runtime on a
    dynamically-built message with the same field layout."""
    from google.protobuf import descriptor_pb2, descriptor_pool
    from google.protobuf import message_factory

    pool = descriptor_pool.DescriptorPool()
    f = descriptor_pb2.FileDescriptorProto()
    f.name = "t.proto"
    f.package = "t"
    m = f.message_type.add()
    m.name = "Shape"
    dim = m.nested_type.add()
    dim.name = "Dim"
    fld = dim.field.add()
    fld.name = "size"
    fld.number = 1
    fld.type = fld.TYPE_INT64
    fld.label = fld.LABEL_OPTIONAL
    fld2 = m.field.add()
    fld2.name = "dim"
    fld2.number = 2
    fld2.type = fld2.TYPE_MESSAGE
    fld2.type_name = ".t.Shape.Dim"
    fld2.label = fld2.LABEL_REPEATED
    pool.Add(f)
    cls = message_factory.GetMessageClass(
        pool.FindMessageTypeByName("t.Shape"))
    msg = cls()
    for s in (1, 224, -1):
        msg.dim.add().size = s
    ours = TensorProto(shape=[1, 224, -1])._shape_bytes()
    assert ours == msg.SerializeToString()


def test_signature_def_roundtrip():
    sig = SignatureDef()
    sig.inputs["input_image"] = TensorInfo("input_image:0", DT_FLOAT,
                                           [-1, -1, -1, 3])
    sig.outputs["mrcnn_detection"] = TensorInfo("mrcnn_detection:0",
                                                DT_FLOAT, [-1, 100, 6])
    blob = encode_saved_model({"serving_default": sig})
    version, tags, sigs = decode_saved_model(blob)
    assert version == 1
    assert tags == ["serve"]
    s = sigs["serving_default"]
    assert s.inputs["input_image"].shape == [-1, -1, -1, 3]
    assert s.outputs["mrcnn_detection"].shape == [-1, 100, 6]
    assert s.method_name == "tensorflow/serving/predict"


# ------------------------------------------------------------ savedmodel
def test_savedmodel_export_load_roundtrip(shapes_config, tmp_path):
    from maskrcnn_amd.model import MaskRCNN
    from maskrcnn_amd.serving.savedmodel import (export_saved_model,
                                                 find_latest_version,
                                                 load_saved_model)
    m = MaskRCNN("inference", shapes_config, model_dir=str(tmp_path))
    base = str(tmp_path / "export" / "mask_rcnn")
    export_dir = export_saved_model(m, shapes_config, base, version=1)
    assert os.path.isfile(os.path.join(export_dir, "saved_model.pb"))
    assert os.path.isfile(os.path.join(
        export_dir, "variables", "variables.data-00000-of-00001"))
    assert os.path.isfile(os.path.join(
        export_dir, "variables", "variables.index"))
    assert find_latest_version(base) == 1

    state, config_dict, sigs = load_saved_model(export_dir)
    assert set(sigs) == {"serving_default"}
    assert set(sigs["serving_default"].inputs) == {
        "input_image", "input_image_meta", "input_anchors"}
    assert len(sigs["serving_default"].outputs) == 7
    sd = m.model.state_dict()
    assert set(state) == set(sd)
    for k in sd:
        assert torch.equal(state[k], sd[k]), k
    assert config_dict["NUM_CLASSES"] == shapes_config.NUM_CLASSES


def test_engine_from_saved_model(shapes_config, tmp_path):
    from maskrcnn_amd.model import MaskRCNN
    from maskrcnn_amd.serving.engine import InferenceEngine
    from maskrcnn_amd.serving.savedmodel import export_saved_model
    m = MaskRCNN("inference", shapes_config, model_dir=str(tmp_path))
    base = str(tmp_path / "export" / "mask_rcnn")
    export_dir = export_saved_model(m, shapes_config, base, version=1)

    engine = InferenceEngine.from_saved_model(export_dir)
    img = (np.random.rand(128, 128, 3) * 255).astype(np.uint8)
    molded, metas, windows = m.mold_inputs([img])
    out = engine.predict(molded, metas)
    D = shapes_config.DETECTION_MAX_INSTANCES
    C = shapes_config.NUM_CLASSES
    assert out["mrcnn_detection"].shape == (1, D, 6)
    assert out["mrcnn_mask"].shape == (1, D, 28, 28, C)

    # Engine outputs must match the facade's predict path.
    ref = m.predict_molded(
        molded, metas,
        m.anchor_cache.get(molded[0].shape, device=m.device))
    np.testing.assert_allclose(
        out["mrcnn_detection"], ref["mrcnn_detection"].cpu().numpy(),
        atol=1e-4)


# --------------------------------------------------------------- batcher
class _FakeEngine:
    def __init__(self):
        self.calls = []

    def predict(self, images, metas, anchors=None, trim_to=None,
                outputs=None, pad_to=None):
        # Batcher contract: images is the request LIST (engine-side
        # collate); pad_to is the padded batch target.
        if isinstance(images, (list, tuple)):
            images = np.concatenate(images, axis=0)
        if pad_to is not None and images.shape[0] < pad_to:
            pad = pad_to - images.shape[0]
            images = np.concatenate(
                [images, np.zeros((pad,) + images.shape[1:],
                                  images.dtype)], axis=0)
        self.calls.append(images.shape[0])
        time.sleep(0.01)
        out = images.sum(axis=(1, 2, 3))
        if trim_to is not None:
            out = out[:trim_to]
        return {"out": out}


def test_batcher_merges_requests():
    from maskrcnn_amd.serving.batcher import DynamicBatcher
    eng = _FakeEngine()
    b = DynamicBatcher(eng, max_batch_size=8, batch_timeout_micros=50_000)
    futs = []
    imgs = []
    for i in range(8):
        img = np.full((1, 4, 4, 3), float(i), np.float32)
        imgs.append(img)
        futs.append(b.submit(img, np.zeros((1, 5), np.float32)))
    outs = [f.result(timeout=5) for f in futs]
    for i, o in enumerate(outs):
        assert o["out"][0] == pytest.approx(imgs[i].sum())
    assert max(eng.calls) > 1  # some merging happened
    b.close()


def test_batcher_shape_bucketing():
    from maskrcnn_amd.serving.batcher import DynamicBatcher
    eng = _FakeEngine()
    b = DynamicBatcher(eng, max_batch_size=4, batch_timeout_micros=30_000)
    f1 = b.submit(np.zeros((1, 4, 4, 3), np.float32),
                  np.zeros((1, 5), np.float32))
    f2 = b.submit(np.zeros((1, 8, 8, 3), np.float32),
                  np.zeros((1, 5), np.float32))
    assert f1.result(timeout=5)["out"].shape == (1,)
    assert f2.result(timeout=5)["out"].shape == (1,)
    b.close()


# --------------------------------------------- full gRPC/REST round trip
@pytest.fixture(scope="module")
def served_model(tmp_path_factory):
    from maskrcnn_amd.data.synthetic import ShapesConfig
    from maskrcnn_amd.model import MaskRCNN
    from maskrcnn_amd.serving.savedmodel import export_saved_model
    from maskrcnn_amd.serving.server import ModelServer

    class SmallConfig(ShapesConfig):
        IMAGES_PER_GPU = 1
        PRE_NMS_LIMIT = 500
        POST_NMS_ROIS_INFERENCE = 50
        DETECTION_MIN_CONFIDENCE = 0.0

    cfg = SmallConfig()
    tmp = tmp_path_factory.mktemp("serving")
    m = MaskRCNN("inference", cfg, model_dir=str(tmp))
    base = str(tmp / "models" / "mask_rcnn")
    export_saved_model(m, cfg, base, version=1)
    server = ModelServer("mask_rcnn", base, grpc_port=0, rest_port=None,
                         max_batch_size=4, batch_timeout_micros=1000,
                         poll_interval_s=0.5, config=cfg)
    server.start()
    yield server, cfg, m
    server.stop()


def test_grpc_predict_roundtrip(served_model):
    server, cfg, m = served_model
    from maskrcnn_amd.serving.client import GrpcServingClient
    client = GrpcServingClient(f"localhost:{server.grpc_port}", cfg)
    img = (np.random.rand(96, 120, 3) * 255).astype(np.uint8)
    results = client.detect([img])
    r = results[0]
    assert r["rois"].shape[1] == 4
    assert r["masks"].shape[:2] == img.shape[:2]

    # Compare against local facade inference on identical molded input.
    local = m.detect([img])[0]
    np.testing.assert_allclose(r["rois"], local["rois"])
    np.testing.assert_array_equal(r["class_ids"], local["class_ids"])


def test_grpc_output_filter_and_model_spec(served_model):
    server, cfg, _ = served_model
    import grpc
    from maskrcnn_amd.serving.client import PREDICT_METHOD
    channel = grpc.insecure_channel(f"localhost:{server.grpc_port}")
    stub = channel.unary_unary(
        PREDICT_METHOD, request_serializer=lambda r: r.encode(),
        response_deserializer=PredictResponse.decode)
    H = cfg.IMAGE_SHAPE[0]
    req = PredictRequest(model_spec=ModelSpec(name="mask_rcnn"))
    req.inputs["input_image"] = numpy_to_tensor(
        np.zeros((1, H, H, 3), np.float32))
    req.inputs["input_image_meta"] = numpy_to_tensor(
        np.zeros((1, cfg.IMAGE_META_SIZE), np.float32))
    req.output_filter = ["mrcnn_detection"]
    resp = stub(req, timeout=60)
    assert list(resp.outputs) == ["mrcnn_detection"]
    assert resp.model_spec.version == 1

    # Unknown model -> NOT_FOUND.
    bad = PredictRequest(model_spec=ModelSpec(name="nope"))
    bad.inputs["input_image"] = req.inputs["input_image"]
    bad.inputs["input_image_meta"] = req.inputs["input_image_meta"]
    with pytest.raises(grpc.RpcError) as e:
        stub(bad, timeout=30)
    assert e.value.code() == grpc.StatusCode.NOT_FOUND


def test_version_hot_reload(served_model, tmp_path):
    server, cfg, m = served_model
    from maskrcnn_amd.serving.savedmodel import export_saved_model
    assert server.state.version == 1
    export_saved_model(m, cfg, server.state.base_path, version=2)
    deadline = time.time() + 10
    while time.time() < deadline and server.state.version != 2:
        time.sleep(0.2)
    assert server.state.version == 2


def test_multi_replica_state(tmp_path, shapes_config):
    """devices=[cpu, cpu] loads two replicas; requests spread across
    batchers (8-replica serving mode, scaled down for CPU)."""
    from maskrcnn_amd.model import MaskRCNN
    from maskrcnn_amd.serving.savedmodel import export_saved_model
    from maskrcnn_amd.serving.server import ModelState

    m = MaskRCNN("inference", shapes_config, model_dir=str(tmp_path))
    base = str(tmp_path / "mr" / "mask_rcnn")
    export_saved_model(m, shapes_config, base, version=1)
    state = ModelState("mask_rcnn", base, devices=["cpu", "cpu"],
                       max_batch_size=1, batch_timeout_micros=500,
                       config=shapes_config)
    state.maybe_reload()
    assert len(state.engines) == 2 and len(state.batchers) == 2
    H = shapes_config.IMAGE_SHAPE[0]
    images = np.zeros((1, H, H, 3), np.float32)
    metas = np.zeros((1, shapes_config.IMAGE_META_SIZE), np.float32)
    metas[:, 4:7] = (H, H, 3)
    metas[:, 7:11] = (0, 0, H, H)
    out1, v1 = state.predict(images, metas)
    out2, v2 = state.predict(images, metas)
    assert v1 == v2 == 1
    np.testing.assert_allclose(out1["mrcnn_detection"],
                               out2["mrcnn_detection"], atol=1e-5)
    for b in state.batchers:
        b.close()


def test_batcher_propagates_engine_errors():
    from maskrcnn_amd.serving.batcher import DynamicBatcher

    class Boom:
        def predict(self, *a, **k):
            raise ValueError("engine exploded")

    b = DynamicBatcher(Boom(), max_batch_size=2,
                       batch_timeout_micros=1000)
    fut = b.submit(np.zeros((1, 4, 4, 3), np.float32),
                   np.zeros((1, 5), np.float32))
    with pytest.raises(ValueError, match="engine exploded"):
        fut.result(timeout=10)
    b.close()


def test_trainer_custom_callbacks(shapes_config, shapes_dataset,
                                  tmp_path):
    from maskrcnn_amd.engine.trainer import Trainer
    from maskrcnn_amd.model import MaskRCNN
    cfg = shapes_config
    m = MaskRCNN("training", cfg, model_dir=str(tmp_path))
    tr = Trainer(m, shapes_dataset)
    seen = []
    tr.fit(cfg.LEARNING_RATE, epochs=1, layers="heads",
           steps_per_epoch=1,
           custom_callbacks=[lambda ep, sc: seen.append((ep, sc))])
    assert len(seen) == 1
    assert seen[0][0] == 0 and "total_loss" in seen[0][1]


def test_data_loader_skips_bad_images(shapes_config, shapes_dataset):
    """Reference parity: a failing load_image is logged and another
    image drawn (data_generator error tolerance, model.py:~1700)."""
    from maskrcnn_amd.data.loader import TorchMaskRCNNDataset
    broken = shapes_dataset
    orig = broken.load_image

    def flaky(image_id):
        if image_id == 0:
            raise IOError("corrupt image")
        return orig(image_id)

    broken.load_image = flaky
    try:
        tds = TorchMaskRCNNDataset(broken, shapes_config)
        sample = tds[0]  # index 0 fails -> a replacement is drawn
        assert sample["image"].shape[0] == 3
    finally:
        broken.load_image = orig


def test_text_proto_parsers(tmp_path):
    from maskrcnn_amd.serving.config_file import (
        parse_batching_parameters, parse_model_config, parse_text_proto)
    mc = tmp_path / "models.config"
    mc.write_text("""
# served models
model_config_list {
  config {
    name: 'mask_rcnn'
    base_path: '/models/mask_rcnn'
    model_platform: 'tensorflow'
  }
  config { name: "second"  base_path: "/models/second" }
}
""")
    cfgs = parse_model_config(str(mc))
    assert cfgs == [
        {"name": "mask_rcnn", "base_path": "/models/mask_rcnn",
         "model_platform": "tensorflow"},
        {"name": "second", "base_path": "/models/second",
         "model_platform": "tensorflow"}]
    bp = tmp_path / "batching_parameters.txt"
    bp.write_text("""
max_batch_size { value: 32 }
batch_timeout_micros { value: 2000 }
num_batch_threads { value: 8 }
max_enqueued_batches { value: 1000000 }
""")
    assert parse_batching_parameters(str(bp)) == {
        "max_batch_size": 32, "batch_timeout_micros": 2000,
        "num_batch_threads": 8, "max_enqueued_batches": 1000000}
    with pytest.raises(ValueError):
        parse_text_proto("a { b: 1")


def test_multi_model_server(tmp_path):
    """--model_config_file mode: two models behind one front door,
    routed by model_spec.name (grpc) and URL (rest status)."""
    from maskrcnn_amd.data.synthetic import ShapesConfig
    from maskrcnn_amd.model import MaskRCNN
    from maskrcnn_amd.serving.savedmodel import export_saved_model
    from maskrcnn_amd.serving.server import ModelServer
    from maskrcnn_amd.serving.wireproto import (ModelSpec, PredictRequest,
                                                numpy_to_tensor)

    class SmallConfig(ShapesConfig):
        IMAGES_PER_GPU = 1
        PRE_NMS_LIMIT = 200
        POST_NMS_ROIS_INFERENCE = 20

    cfg = SmallConfig()
    configs = []
    for name in ("alpha", "beta"):
        m = MaskRCNN("inference", cfg, model_dir=str(tmp_path))
        base = str(tmp_path / "models" / name)
        export_saved_model(m, cfg, base, version=1)
        configs.append({"name": name, "base_path": base})
    server = ModelServer(model_configs=configs, grpc_port=0,
                         rest_port=None, max_batch_size=1,
                         batch_timeout_micros=100, config=cfg)
    server.start()
    try:
        assert set(server.models) == {"alpha", "beta"}
        H = cfg.IMAGE_SHAPE[0]
        meta = np.zeros((1, cfg.IMAGE_META_SIZE), np.float32)
        meta[:, 4:7] = (H, H, 3)
        meta[:, 7:11] = (0, 0, H, H)
        for name in ("alpha", "beta"):
            req = PredictRequest(
                model_spec=ModelSpec(name=name),
                inputs={"input_image": numpy_to_tensor(
                            np.zeros((1, H, H, 3), np.float32)),
                        "input_image_meta": numpy_to_tensor(meta)})
            resp = server._handle_predict(req)
            assert resp.model_spec.name == name
            assert "mrcnn_detection" in resp.outputs
        with pytest.raises(KeyError):
            server._handle_predict(PredictRequest(
                model_spec=ModelSpec(name="gamma"), inputs={}))
    finally:
        server.stop()


def test_demo_app_roundtrip(shapes_config, tmp_path):
    """S5 demo app: upload PNG -> rendered detections PNG back."""
    import importlib.util
    import io as _io
    spec = importlib.util.spec_from_file_location(
        "demo_app", os.path.join(os.path.dirname(__file__), "..",
                                 "samples", "demo_app.py"))
    demo_app = importlib.util.module_from_spec(spec)
    spec.loader.exec_module(demo_app)
    from PIL import Image
    from fastapi.testclient import TestClient
    from maskrcnn_amd.model import MaskRCNN

    m = MaskRCNN("inference", shapes_config, model_dir=str(tmp_path))
    names = [f"c{i}" for i in range(shapes_config.NUM_CLASSES)]
    app = demo_app.create_app(lambda img: m.detect([img])[0], names)
    client = TestClient(app)
    assert "Detect" in client.get("/").text
    img = Image.fromarray(
        (np.random.rand(96, 96, 3) * 255).astype(np.uint8))
    buf = _io.BytesIO()
    img.save(buf, format="PNG")
    r = client.post("/detect", content=buf.getvalue())
    assert r.status_code == 200
    assert r.headers["content-type"] == "image/png"
    assert Image.open(_io.BytesIO(r.content)).size[0] > 0
    r = client.post("/detect", content=b"not an image")
    assert r.status_code == 400


def test_grpc_model_service_status_and_reload(served_model, tmp_path):
    """tensorflow.serving.ModelService: GetModelStatus over the wire
    and HandleReloadConfigRequest adding a second model at runtime."""
    import grpc
    from maskrcnn_amd.data.synthetic import ShapesConfig
    from maskrcnn_amd.model import MaskRCNN
    from maskrcnn_amd.serving.savedmodel import export_saved_model
    from maskrcnn_amd.serving.wireproto import (
        GetModelStatusRequest, GetModelStatusResponse, ModelConfigProto,
        ModelSpec, ReloadConfigRequest, ReloadConfigResponse,
        STATE_AVAILABLE)

    server, cfg, m = served_model
    channel = grpc.insecure_channel(f"localhost:{server.grpc_port}")
    status_call = channel.unary_unary(
        "/tensorflow.serving.ModelService/GetModelStatus",
        request_serializer=lambda r: r.encode(),
        response_deserializer=GetModelStatusResponse.decode)
    st = status_call(GetModelStatusRequest(
        model_spec=ModelSpec(name="mask_rcnn")), timeout=10)
    assert st.model_version_status[0].version == server.state.version
    assert st.model_version_status[0].state == STATE_AVAILABLE

    with pytest.raises(grpc.RpcError) as e:
        status_call(GetModelStatusRequest(
            model_spec=ModelSpec(name="nope")), timeout=10)
    assert e.value.code() == grpc.StatusCode.NOT_FOUND

    # Reload: keep mask_rcnn, add second.
    base2 = str(tmp_path / "models" / "second")
    export_saved_model(m, cfg, base2, version=7)
    reload_call = channel.unary_unary(
        "/tensorflow.serving.ModelService/HandleReloadConfigRequest",
        request_serializer=lambda r: r.encode(),
        response_deserializer=ReloadConfigResponse.decode)
    first_base = server.state.base_path
    resp = reload_call(ReloadConfigRequest(model_configs=[
        ModelConfigProto(name="mask_rcnn", base_path=first_base),
        ModelConfigProto(name="second", base_path=base2)]), timeout=60)
    assert resp.status.error_code == 0, resp.status.error_message
    assert set(server.models) == {"mask_rcnn", "second"}
    st2 = status_call(GetModelStatusRequest(
        model_spec=ModelSpec(name="second")), timeout=10)
    assert st2.model_version_status[0].version == 7

    # Reload with a bad path -> INTERNAL status, serving set unchanged.
    resp = reload_call(ReloadConfigRequest(model_configs=[
        ModelConfigProto(name="mask_rcnn", base_path=first_base),
        ModelConfigProto(name="ghost", base_path="/nonexistent")]),
        timeout=60)
    assert resp.status.error_code == 13
    assert set(server.models) == {"mask_rcnn", "second"}
    # Drop back to the single model (unloads 'second').
    resp = reload_call(ReloadConfigRequest(model_configs=[
        ModelConfigProto(name="mask_rcnn", base_path=first_base)]),
        timeout=60)
    assert resp.status.error_code == 0
    assert set(server.models) == {"mask_rcnn"}
    channel.close()


def test_model_service_client(served_model):
    from maskrcnn_amd.serving.client import ModelServiceClient
    from maskrcnn_amd.serving.wireproto import STATE_AVAILABLE
    server, cfg, m = served_model
    c = ModelServiceClient(f"localhost:{server.grpc_port}")
    st = c.get_model_status("mask_rcnn")
    assert st[0]["state"] == STATE_AVAILABLE
    with pytest.raises(RuntimeError, match="reload rejected"):
        c.reload_config([{"name": "x", "base_path": "/nonexistent"}])
    c.close()


def test_concurrent_predicts_with_reload(tmp_path):
    """Stress: concurrent gRPC-path predicts while versions hot-load
    and the model set is reloaded — no deadlocks, no lost futures."""
    import threading

    from maskrcnn_amd.data.synthetic import ShapesConfig
    from maskrcnn_amd.model import MaskRCNN
    from maskrcnn_amd.serving.savedmodel import export_saved_model
    from maskrcnn_amd.serving.server import ModelServer

    class TinyConfig(ShapesConfig):
        IMAGES_PER_GPU = 1
        PRE_NMS_LIMIT = 200
        POST_NMS_ROIS_INFERENCE = 20
        DETECTION_MIN_CONFIDENCE = 0.0

    cfg = TinyConfig()
    m = MaskRCNN("inference", cfg, model_dir=str(tmp_path))
    base = str(tmp_path / "models" / "m")
    export_saved_model(m, cfg, base, version=1)
    server = ModelServer("m", base, grpc_port=0, rest_port=None,
                         max_batch_size=4, batch_timeout_micros=500,
                         poll_interval_s=0.2, config=cfg)
    server.start()
    H = cfg.IMAGE_SHAPE[0]
    meta = np.zeros((1, cfg.IMAGE_META_SIZE), np.float32)
    meta[:, 4:7] = (H, H, 3)
    meta[:, 7:11] = (0, 0, H, H)
    errors, done = [], []

    def client(i):
        try:
            for _ in range(4):
                out, _v = server.state.predict(
                    np.random.rand(1, H, H, 3).astype(np.float32), meta)
                assert "mrcnn_detection" in out
            done.append(i)
        except Exception as e:  # noqa: BLE001
            errors.append(repr(e))

    threads = [threading.Thread(target=client, args=(i,))
               for i in range(4)]
    for t in threads:
        t.start()
    # Mid-traffic: new version appears (poller hot-loads it) and the
    # model set reloads to the same single model (no-op path).
    export_saved_model(m, cfg, base, version=2)
    server.reload_config([{"name": "m", "base_path": base}])
    for t in threads:
        t.join(timeout=120)
    assert not errors, errors
    assert len(done) == 4
    for _ in range(40):  # poller picks up v2
        if server.state.version == 2:
            break
        time.sleep(0.25)
    assert server.state.version == 2
    server.stop()


def test_server_cli_with_config_files(tmp_path):
    """python -m maskrcnn_amd.serving.server with --model_config_file +
    --batching_parameters_file: boots, serves REST status, dies clean."""
    import signal
    import socket
    import subprocess
    import sys

    from maskrcnn_amd.data.synthetic import ShapesConfig
    from maskrcnn_amd.model import MaskRCNN
    from maskrcnn_amd.serving.savedmodel import export_saved_model

    class TinyConfig(ShapesConfig):
        IMAGES_PER_GPU = 1
        PRE_NMS_LIMIT = 100
        POST_NMS_ROIS_INFERENCE = 10

    cfg = TinyConfig()
    m = MaskRCNN("inference", cfg, model_dir=str(tmp_path))
    base = str(tmp_path / "models" / "cli_model")
    export_saved_model(m, cfg, base, version=1)
    mc = tmp_path / "models.config"
    mc.write_text("model_config_list { config { name: 'cli_model' "
                  f"base_path: '{base}' }} }}\n".replace("}}", "}"))
    bp = tmp_path / "batching.txt"
    bp.write_text("max_batch_size { value: 2 }\n"
                  "batch_timeout_micros { value: 1000 }\n")

    def free_port():
        s = socket.socket()
        s.bind(("127.0.0.1", 0))
        p = s.getsockname()[1]
        s.close()
        return p

    gport, rport = free_port(), free_port()
    proc = subprocess.Popen(
        [sys.executable, "-m", "maskrcnn_amd.serving.server",
         "--model_config_file", str(mc),
         "--batching_parameters_file", str(bp),
         "--port", str(gport), "--rest_api_port", str(rport)],
        stdout=subprocess.PIPE, stderr=subprocess.STDOUT, text=True)
    try:
        import requests
        st = None
        for _ in range(300):
            if proc.poll() is not None:
                raise AssertionError(proc.stdout.read()[-2000:])
            try:
                st = requests.get(
                    f"http://127.0.0.1:{rport}/v1/models/cli_model",
                    timeout=1).json()
                break
            except Exception:
                time.sleep(0.2)
        assert st is not None, "server never came up"
        assert st["model_version_status"][0]["state"] == "AVAILABLE"
    finally:
        proc.send_signal(signal.SIGTERM)
        try:
            proc.wait(timeout=10)
        except subprocess.TimeoutExpired:
            proc.kill()
            proc.wait(timeout=10)


def test_batcher_multiple_worker_threads():
    from maskrcnn_amd.serving.batcher import DynamicBatcher
    eng = _FakeEngine()
    b = DynamicBatcher(eng, max_batch_size=2, batch_timeout_micros=500,
                       num_batch_threads=3)
    futs = [b.submit(np.full((1, 4, 4, 3), float(i), np.float32),
                     np.zeros((1, 5), np.float32)) for i in range(12)]
    for i, f in enumerate(futs):
        out = f.result(timeout=10)
        assert out["out"][0] == pytest.approx(i * 4 * 4 * 3)
    b.close()
    assert sum(eng.calls) >= 12


def test_version_policy_all(tmp_path):
    """version_policy='all': every version stays servable; requests pin
    via model_spec.version (TF-Serving model_version_policy {all {}})."""
    from maskrcnn_amd.data.synthetic import ShapesConfig
    from maskrcnn_amd.model import MaskRCNN
    from maskrcnn_amd.serving.savedmodel import export_saved_model
    from maskrcnn_amd.serving.server import ModelState

    class TinyConfig(ShapesConfig):
        IMAGES_PER_GPU = 1
        PRE_NMS_LIMIT = 100
        POST_NMS_ROIS_INFERENCE = 10

    cfg = TinyConfig()
    m = MaskRCNN("inference", cfg, model_dir=str(tmp_path))
    base = str(tmp_path / "models" / "vp")
    export_saved_model(m, cfg, base, version=1)
    # Startup with BOTH versions already on disk: policy 'all' must
    # load every one (TF-Serving {all{}} semantics), not just latest.
    export_saved_model(m, cfg, base, version=2)
    st = ModelState("vp", base, max_batch_size=1,
                    batch_timeout_micros=100, config=cfg,
                    version_policy="all")
    st.maybe_reload()
    assert st.version == 2
    assert sorted(st.versions) == [1, 2]

    H = cfg.IMAGE_SHAPE[0]
    meta = np.zeros((1, cfg.IMAGE_META_SIZE), np.float32)
    meta[:, 4:7] = (H, H, 3)
    meta[:, 7:11] = (0, 0, H, H)
    img = np.zeros((1, H, H, 3), np.float32)
    _, v = st.predict(img, meta)                 # unpinned -> latest
    assert v == 2
    _, v = st.predict(img, meta, version=1)      # pinned -> old version
    assert v == 1
    with pytest.raises(KeyError):
        st.predict(img, meta, version=9)
    for slot in st.versions.values():
        for b in slot["batchers"]:
            b.close()

    # 'latest' policy drops the old version on reload.
    st2 = ModelState("vp", base, max_batch_size=1,
                     batch_timeout_micros=100, config=cfg)
    st2.maybe_reload()
    assert sorted(st2.versions) == [2]
    for slot in st2.versions.values():
        for b in slot["batchers"]:
            b.close()


def test_convert_checkpoint_roundtrip(tmp_path):
    """tools/convert_checkpoint.py: pth -> SavedModel -> pth preserves
    every tensor."""
    import subprocess
    import sys

    import torch

    from maskrcnn_amd.data.synthetic import ShapesConfig
    from maskrcnn_amd.model.mask_rcnn import MaskRCNNModel

    class C(ShapesConfig):
        BACKBONE = "resnet50"

    cfg = C()
    m = MaskRCNNModel(cfg)
    p1 = str(tmp_path / "a.pth")
    torch.save({"model": m.state_dict()}, p1)
    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    for src, dst in ((p1, str(tmp_path / "sm")),
                     (str(tmp_path / "sm"), str(tmp_path / "b.pth"))):
        r = subprocess.run(
            [sys.executable, "tools/convert_checkpoint.py", "--src", src,
             "--dst", dst, "--num-classes", str(cfg.NUM_CLASSES),
             "--backbone", "resnet50"],
            capture_output=True, text=True, timeout=300, cwd=root)
        assert r.returncode == 0, r.stderr[-1500:]
    sd2 = torch.load(str(tmp_path / "b.pth"), weights_only=False)["model"]
    sd1 = m.state_dict()
    for k in sd1:
        torch.testing.assert_close(sd1[k].float(), sd2[k].float(),
                                   atol=1e-6, rtol=0)


def test_serving_bench_cpu_smoke():
    """bench/serving_bench.py end to end on CPU (tiny config): the
    round-2 block3 command path must not rot under server refactors."""
    import json
    import subprocess
    import sys

    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    r = subprocess.run(
        [sys.executable, "bench/serving_bench.py", "--requests", "8",
         "--client-procs", "2", "--client-threads", "2", "--warmup", "2",
         "--image-size", "128", "--backbone", "resnet50",
         "--max-batch", "2"],
        capture_output=True, text=True, timeout=600, cwd=root)
    assert r.returncode == 0, r.stderr[-2000:]
    rec = json.loads(r.stdout.strip().splitlines()[-1])
    assert rec["metric"] == "serving_images_per_sec"
    assert rec["value"] > 0
    assert rec["steps"] == 8


def test_grpc_mask_top_extension(served_model):
    """output_filter entry "mrcnn_mask_top" returns the detected-class
    mask [B,D,mh,mw] (81x fewer bytes than mrcnn_mask) and matches a
    gather of the full tensor; stock mrcnn_mask stays available."""
    server, cfg, _ = served_model
    import grpc
    from maskrcnn_amd.serving.client import PREDICT_METHOD
    channel = grpc.insecure_channel(f"localhost:{server.grpc_port}")
    stub = channel.unary_unary(
        PREDICT_METHOD, request_serializer=lambda r: r.encode(),
        response_deserializer=PredictResponse.decode)
    H = cfg.IMAGE_SHAPE[0]
    rng = np.random.RandomState(3)
    req = PredictRequest(model_spec=ModelSpec(name="mask_rcnn"))
    req.inputs["input_image"] = numpy_to_tensor(
        rng.rand(1, H, H, 3).astype(np.float32) * 60)
    meta = np.zeros((1, cfg.IMAGE_META_SIZE), np.float32)
    meta[:, 4:7] = (H, H, 3)
    meta[:, 7:11] = (0, 0, H, H)
    req.inputs["input_image_meta"] = numpy_to_tensor(meta)
    req.output_filter = ["mrcnn_detection", "mrcnn_mask",
                         "mrcnn_mask_top"]
    resp = stub(req, timeout=120)
    det = tensor_to_numpy(resp.outputs["mrcnn_detection"])
    full = tensor_to_numpy(resp.outputs["mrcnn_mask"])
    top = tensor_to_numpy(resp.outputs["mrcnn_mask_top"])
    assert top.shape == full.shape[:4]
    cls = det[:, :, 4].astype(np.int64)
    B, D = cls.shape
    ref = full[np.arange(B)[:, None], np.arange(D)[None, :], :, :, cls]
    np.testing.assert_array_equal(top, ref)
    assert top.nbytes * (full.shape[-1]) == full.nbytes


def test_grpc_uint8_image_extension(served_model):
    """uint8 input_image (wire extension, 1/4 the bytes) is molded
    server-side and must match the stock client-molded fp32 request."""
    server, cfg, _ = served_model
    import grpc
    from maskrcnn_amd.serving.client import PREDICT_METHOD
    channel = grpc.insecure_channel(f"localhost:{server.grpc_port}")
    stub = channel.unary_unary(
        PREDICT_METHOD, request_serializer=lambda r: r.encode(),
        response_deserializer=PredictResponse.decode)
    H = cfg.IMAGE_SHAPE[0]
    rng = np.random.RandomState(9)
    raw = (rng.rand(1, H, H, 3) * 255).astype(np.uint8)
    meta = np.zeros((1, cfg.IMAGE_META_SIZE), np.float32)
    meta[:, 4:7] = (H, H, 3)
    meta[:, 7:11] = (0, 0, H, H)

    def call(img):
        req = PredictRequest(model_spec=ModelSpec(name="mask_rcnn"))
        req.inputs["input_image"] = numpy_to_tensor(img)
        req.inputs["input_image_meta"] = numpy_to_tensor(meta)
        req.output_filter = ["mrcnn_detection"]
        return tensor_to_numpy(
            stub(req, timeout=120).outputs["mrcnn_detection"])

    molded = raw.astype(np.float32) - np.array(cfg.MEAN_PIXEL,
                                               np.float32)
    det_fp32 = call(molded)
    det_u8 = call(raw)
    np.testing.assert_allclose(det_u8, det_fp32, atol=1e-5)


def test_cli_replicas_same_export(tmp_path):
    """Two independent server CLI processes on one export (the
    bench --server-procs scenario): both come AVAILABLE with REST
    disabled (--rest_api_port 0) and both answer gRPC Predict."""
    import signal
    import socket
    import subprocess
    import sys

    import grpc

    from maskrcnn_amd.data.synthetic import ShapesConfig
    from maskrcnn_amd.model import MaskRCNN
    from maskrcnn_amd.serving.client import (ModelServiceClient,
                                             PREDICT_METHOD)
    from maskrcnn_amd.serving.savedmodel import export_saved_model

    class TinyConfig(ShapesConfig):
        IMAGES_PER_GPU = 1
        PRE_NMS_LIMIT = 100
        POST_NMS_ROIS_INFERENCE = 10

    cfg = TinyConfig()
    m = MaskRCNN("inference", cfg, model_dir=str(tmp_path))
    base = str(tmp_path / "models" / "rep_model")
    export_saved_model(m, cfg, base, version=1)

    def free_port():
        s = socket.socket()
        s.bind(("127.0.0.1", 0))
        p = s.getsockname()[1]
        s.close()
        return p

    ports = [free_port(), free_port()]
    procs = [subprocess.Popen(
        [sys.executable, "-m", "maskrcnn_amd.serving.server",
         "--model_name", "rep_model", "--model_base_path", base,
         "--port", str(p), "--rest_api_port", "0"],
        stdout=subprocess.PIPE, stderr=subprocess.STDOUT, text=True)
        for p in ports]
    try:
        H = cfg.IMAGE_SHAPE[0]
        req = PredictRequest(model_spec=ModelSpec(name="rep_model"))
        req.inputs["input_image"] = numpy_to_tensor(
            np.zeros((1, H, H, 3), np.float32))
        meta = np.zeros((1, cfg.IMAGE_META_SIZE), np.float32)
        meta[:, 4:7] = (H, H, 3)
        meta[:, 7:11] = (0, 0, H, H)
        req.inputs["input_image_meta"] = numpy_to_tensor(meta)
        req.output_filter = ["mrcnn_detection"]
        for port, proc in zip(ports, procs):
            st = None
            for _ in range(300):
                if proc.poll() is not None:
                    raise AssertionError(proc.stdout.read()[-2000:])
                try:
                    st = ModelServiceClient(
                        f"127.0.0.1:{port}", timeout=2.0) \
                        .get_model_status("rep_model")
                    # 30 = ModelVersionStatus.State.AVAILABLE
                    if st and st[0]["state"] in (30, "AVAILABLE"):
                        break
                except Exception:
                    pass
                time.sleep(0.2)
            assert st and st[0]["state"] in (30, "AVAILABLE")
            channel = grpc.insecure_channel(f"127.0.0.1:{port}")
            stub = channel.unary_unary(
                PREDICT_METHOD, request_serializer=lambda r: r.encode(),
                response_deserializer=PredictResponse.decode)
            resp = stub(req, timeout=120)
            assert tensor_to_numpy(
                resp.outputs["mrcnn_detection"]).shape[0] == 1
    finally:
        for proc in procs:
            proc.send_signal(signal.SIGTERM)
        for proc in procs:
            try:
                proc.wait(timeout=10)
            except subprocess.TimeoutExpired:
                proc.kill()
                proc.wait(timeout=10)


def test_grpc_mixed_uint8_fp32_clients(served_model):
    """A uint8-extension client and a stock fp32 client hitting the
    SAME server concurrently get identical results (the batcher must
    not concat mixed-dtype batches — it splits on dtype change)."""
    import threading

    import grpc
    from maskrcnn_amd.serving.client import PREDICT_METHOD
    server, cfg, _ = served_model
    channel = grpc.insecure_channel(f"localhost:{server.grpc_port}")
    stub = channel.unary_unary(
        PREDICT_METHOD, request_serializer=lambda r: r.encode(),
        response_deserializer=PredictResponse.decode)
    H = cfg.IMAGE_SHAPE[0]
    rng = np.random.RandomState(11)
    raw = (rng.rand(1, H, H, 3) * 255).astype(np.uint8)
    molded = raw.astype(np.float32) - np.array(cfg.MEAN_PIXEL,
                                               np.float32)
    meta = np.zeros((1, cfg.IMAGE_META_SIZE), np.float32)
    meta[:, 4:7] = (H, H, 3)
    meta[:, 7:11] = (0, 0, H, H)

    def make_req(img):
        r = PredictRequest(model_spec=ModelSpec(name="mask_rcnn"))
        r.inputs["input_image"] = numpy_to_tensor(img)
        r.inputs["input_image_meta"] = numpy_to_tensor(meta)
        r.output_filter = ["mrcnn_detection"]
        return r

    results = {}

    def call(tag, img):
        results[tag] = tensor_to_numpy(
            stub(make_req(img), timeout=120).outputs["mrcnn_detection"])

    # Interleave several of each so both dtypes share batcher windows.
    threads = []
    for i in range(3):
        threads.append(threading.Thread(
            target=call, args=(f"u8_{i}", raw)))
        threads.append(threading.Thread(
            target=call, args=(f"f32_{i}", molded)))
    for t in threads:
        t.start()
    for t in threads:
        t.join()
    for i in range(3):
        np.testing.assert_allclose(results[f"u8_{i}"],
                                   results["f32_0"], atol=1e-5)
        np.testing.assert_allclose(results[f"f32_{i}"],
                                   results["f32_0"], atol=1e-5)


def test_batcher_mixed_shapes_concurrent(shapes_config):
    """Three resolutions interleaved concurrently: every request gets
    results for ITS OWN shape (regression for the stale-batch-head bug:
    items after a shape flush were compared against the old head, so a
    later odd-shape request could be concatenated into the wrong batch
    — np.concatenate would raise, failing the whole window)."""
    import threading

    import torch
    from maskrcnn_amd.serving.batcher import DynamicBatcher
    from maskrcnn_amd.serving.engine import InferenceEngine
    torch.manual_seed(0)
    eng = InferenceEngine(shapes_config)
    b = DynamicBatcher(eng, max_batch_size=8, batch_timeout_micros=4000)
    H = shapes_config.IMAGE_SHAPE[0]
    sizes = [H, H + 64, H + 128]
    rng = np.random.RandomState(5)
    results = {}

    def call(tag, s):
        img = (rng.rand(1, s, s, 3) * 50).astype(np.float32)
        meta = np.zeros((1, shapes_config.IMAGE_META_SIZE), np.float32)
        meta[:, 4:7] = (s, s, 3)
        meta[:, 7:11] = (0, 0, s, s)
        results[tag] = b.submit(
            img, meta, None,
            outputs=["mrcnn_detection", "mrcnn_mask"]).result(180)

    threads = [threading.Thread(target=call, args=(f"{s}_{i}", s))
               for i in range(2) for s in sizes]
    for t in threads:
        t.start()
    for t in threads:
        t.join()
    b.close()
    assert len(results) == 6
    for tag, out in results.items():
        assert out["mrcnn_detection"].shape[0] == 1, tag
        assert np.isfinite(out["mrcnn_detection"]).all(), tag


def test_server_cli_cpp_ingress(tmp_path):
    """python -m maskrcnn_amd.serving.server --grpc_ingress cpp: boots,
    answers a real gRPC Predict through the native daemon, dies clean."""
    import signal
    import socket
    import subprocess
    import sys

    pytest.importorskip("maskrcnn_amd._ingress")
    from maskrcnn_amd.data.synthetic import ShapesConfig
    from maskrcnn_amd.model import MaskRCNN
    from maskrcnn_amd.serving.savedmodel import export_saved_model

    class TinyConfig(ShapesConfig):
        IMAGES_PER_GPU = 1
        PRE_NMS_LIMIT = 100
        POST_NMS_ROIS_INFERENCE = 10
        DETECTION_MIN_CONFIDENCE = 0.0

    cfg = TinyConfig()
    m = MaskRCNN("inference", cfg, model_dir=str(tmp_path))
    base = str(tmp_path / "models" / "cli_cpp")
    export_saved_model(m, cfg, base, version=1)

    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    gport = s.getsockname()[1]
    s.close()
    proc = subprocess.Popen(
        [sys.executable, "-m", "maskrcnn_amd.serving.server",
         "--model_name", "cli_cpp", "--model_base_path", base,
         "--port", str(gport), "--rest_api_port", "0",
         "--enable_batching", "--max_batch_size", "2",
         "--batch_timeout_micros", "1000",
         "--grpc_ingress", "cpp"],
        stdout=subprocess.PIPE, stderr=subprocess.STDOUT, text=True)
    try:
        from maskrcnn_amd.serving.client import GrpcServingClient
        img = (np.random.RandomState(5).rand(96, 96, 3) * 255) \
            .astype(np.uint8)
        r = None
        deadline = time.time() + 120
        while time.time() < deadline:
            if proc.poll() is not None:
                raise AssertionError(proc.stdout.read()[-2000:])
            try:
                client = GrpcServingClient(f"127.0.0.1:{gport}", cfg,
                                           model_name="cli_cpp",
                                           timeout=10.0)
                r = client.detect([img])[0]
                break
            except Exception:
                time.sleep(0.5)
        assert r is not None, "no Predict answer through cpp ingress"
        assert r["masks"].shape[:2] == img.shape[:2]
    finally:
        proc.send_signal(signal.SIGTERM)
        try:
            proc.wait(timeout=20)
        except subprocess.TimeoutExpired:
            proc.kill()


def test_batcher_load_shedding():
    """max_enqueued_batches full -> BatcherQueueFull (TF-Serving
    RESOURCE_EXHAUSTED semantics), not unbounded buffering."""
    from maskrcnn_amd.serving.batcher import (BatcherQueueFull,
                                              DynamicBatcher)

    class Slow:
        def predict(self, images, metas, anchors=None, trim_to=None,
                    outputs=None, pad_to=None):
            time.sleep(0.5)
            n = trim_to if trim_to is not None else 1
            return {"out": np.zeros(n)}

    b = DynamicBatcher(Slow(), max_batch_size=1,
                       batch_timeout_micros=100,
                       max_enqueued_batches=2)
    img = np.zeros((1, 8, 8, 3), np.float32)
    meta = np.zeros((1, 5), np.float32)
    futs = [b.submit(img, meta)]  # taken by the worker quickly
    with pytest.raises(BatcherQueueFull):
        for _ in range(8):  # maxsize=2 -> must trip well before 8
            futs.append(b.submit(img, meta))
            time.sleep(0.01)
    for f in futs:
        f.result(timeout=30)
    b.close()


def test_version_labels_routing(tmp_path):
    """TF-Serving version_labels: ModelSpec.version_label routes grpc
    Predicts, /v1/models/{m}/labels/{l}:predict routes REST, and the
    models.config `version_labels {key value}` entries parse."""
    from maskrcnn_amd.data.synthetic import ShapesConfig
    from maskrcnn_amd.model import MaskRCNN
    from maskrcnn_amd.serving.savedmodel import export_saved_model
    from maskrcnn_amd.serving.server import ModelServer

    class TinyConfig(ShapesConfig):
        IMAGES_PER_GPU = 1
        PRE_NMS_LIMIT = 100
        POST_NMS_ROIS_INFERENCE = 10
        DETECTION_MIN_CONFIDENCE = 0.0

    cfg = TinyConfig()
    m = MaskRCNN("inference", cfg, model_dir=str(tmp_path))
    base = str(tmp_path / "models" / "vl")
    export_saved_model(m, cfg, base, version=1)
    export_saved_model(m, cfg, base, version=2)
    server = ModelServer(
        model_configs=[{"name": "vl", "base_path": base,
                        "version_policy": "all",
                        "version_labels": {"stable": 1, "canary": 2}}],
        grpc_port=0, rest_port=None, max_batch_size=1,
        batch_timeout_micros=100, config=cfg)
    server.start()
    try:
        import grpc
        from maskrcnn_amd.serving.client import PREDICT_METHOD
        ch = grpc.insecure_channel(f"localhost:{server.grpc_port}")
        stub = ch.unary_unary(
            PREDICT_METHOD, request_serializer=lambda r: r.encode(),
            response_deserializer=PredictResponse.decode)
        H = cfg.IMAGE_SHAPE[0]
        meta = np.zeros((1, cfg.IMAGE_META_SIZE), np.float32)
        meta[:, 4:7] = (H, H, 3)
        meta[:, 7:11] = (0, 0, H, H)

        def req(label):
            r = PredictRequest(model_spec=ModelSpec(
                name="vl", version_label=label))
            r.inputs["input_image"] = numpy_to_tensor(
                np.zeros((1, H, H, 3), np.float32))
            r.inputs["input_image_meta"] = numpy_to_tensor(meta)
            r.output_filter = ["mrcnn_detection"]
            return r

        assert stub(req("stable"), timeout=60).model_spec.version == 1
        assert stub(req("canary"), timeout=60).model_spec.version == 2
        with pytest.raises(grpc.RpcError) as e:
            stub(req("nope"), timeout=30)
        assert e.value.code() == grpc.StatusCode.NOT_FOUND
        ch.close()
    finally:
        server.stop()


def test_model_config_version_labels_parse(tmp_path):
    from maskrcnn_amd.serving.config_file import parse_model_config
    mc = tmp_path / "models.config"
    mc.write_text(
        "model_config_list { config {\n"
        "  name: 'm' base_path: '/x'\n"
        "  model_version_policy { all {} }\n"
        "  version_labels { key: 'stable' value: 1 }\n"
        "  version_labels { key: 'canary' value: 2 }\n"
        "} }\n")
    cfgs = parse_model_config(str(mc))
    assert cfgs[0]["version_policy"] == "all"
    assert cfgs[0]["version_labels"] == {"stable": 1, "canary": 2}


def test_warmup_tfrecord_roundtrip(tmp_path):
    """TFRecord framing (masked crc32c) + PredictionLog wire encode/
    decode roundtrip, plus TF-mask crc self-check."""
    from maskrcnn_amd.serving.warmup import (decode_prediction_log,
                                             encode_prediction_log,
                                             read_warmup_requests,
                                             tfrecord_read,
                                             tfrecord_write,
                                             write_warmup_requests)
    recs = [b"abc", b"", b"\x00" * 1000]
    p = str(tmp_path / "r.tfrecord")
    tfrecord_write(p, recs)
    assert tfrecord_read(p) == recs
    # corrupt one data byte -> crc error
    blob = bytearray(open(p, "rb").read())
    blob[12] ^= 0xFF
    open(p, "wb").write(bytes(blob))
    with pytest.raises(ValueError):
        tfrecord_read(p)

    req = PredictRequest(model_spec=ModelSpec(name="m"))
    req.inputs["input_image"] = numpy_to_tensor(
        np.zeros((2, 4, 4, 3), np.uint8))
    req.inputs["input_image_meta"] = numpy_to_tensor(
        np.ones((2, 5), np.float32))
    back = decode_prediction_log(encode_prediction_log(req))
    np.testing.assert_array_equal(
        tensor_to_numpy(back.inputs["input_image_meta"]),
        np.ones((2, 5), np.float32))

    vd = str(tmp_path / "1")
    write_warmup_requests(vd, [req])
    assert os.path.isfile(
        os.path.join(vd, "assets.extra", "tf_serving_warmup_requests"))
    got = read_warmup_requests(vd)
    assert len(got) == 1
    assert tensor_to_numpy(got[0].inputs["input_image"]).shape == \
        (2, 4, 4, 3)


def test_model_warmup_replayed_at_load(tmp_path):
    """export(warmup_batches=[2]) -> ModelState load replays the warmup
    request through the engine before the version is AVAILABLE."""
    from maskrcnn_amd.data.synthetic import ShapesConfig
    from maskrcnn_amd.model import MaskRCNN
    from maskrcnn_amd.serving.savedmodel import export_saved_model
    from maskrcnn_amd.serving.server import ModelState

    class TinyConfig(ShapesConfig):
        IMAGES_PER_GPU = 1
        PRE_NMS_LIMIT = 100
        POST_NMS_ROIS_INFERENCE = 10

    cfg = TinyConfig()
    m = MaskRCNN("inference", cfg, model_dir=str(tmp_path))
    base = str(tmp_path / "models" / "wu")
    export_saved_model(m, cfg, base, version=1, warmup_batches=[2])
    assert os.path.isfile(os.path.join(
        base, "1", "assets.extra", "tf_serving_warmup_requests"))

    calls = []
    import maskrcnn_amd.serving.engine as E
    orig = E.InferenceEngine.predict

    def spy(self, *a, **k):
        calls.append(1)
        return orig(self, *a, **k)

    E.InferenceEngine.predict = spy
    try:
        st = ModelState("wu", base, max_batch_size=2,
                        batch_timeout_micros=100, config=cfg)
        st.maybe_reload()
        assert len(calls) >= 1, "warmup request was not replayed"
    finally:
        E.InferenceEngine.predict = orig
        for slot in st.versions.values():
            for b in slot["batchers"]:
                b.close()


def test_model_config_file_polling(tmp_path):
    """--model_config_file_poll_wait_seconds: editing models.config on
    disk adds/removes served models without an RPC."""
    from maskrcnn_amd.data.synthetic import ShapesConfig
    from maskrcnn_amd.model import MaskRCNN
    from maskrcnn_amd.serving.savedmodel import export_saved_model
    from maskrcnn_amd.serving.server import ModelServer

    class TinyConfig(ShapesConfig):
        IMAGES_PER_GPU = 1
        PRE_NMS_LIMIT = 100
        POST_NMS_ROIS_INFERENCE = 10

    cfg = TinyConfig()
    m = MaskRCNN("inference", cfg, model_dir=str(tmp_path))
    base_a = str(tmp_path / "models" / "a")
    base_b = str(tmp_path / "models" / "b")
    export_saved_model(m, cfg, base_a, version=1)
    export_saved_model(m, cfg, base_b, version=1)

    mc = tmp_path / "models.config"

    def write_cfg(names_paths):
        body = "".join(
            "config { name: '%s' base_path: '%s' }\n" % np
            for np_ in [None] for np in names_paths)
        mc.write_text("model_config_list { %s }\n" % body)

    write_cfg([("a", base_a)])
    server = ModelServer(
        model_configs=[{"name": "a", "base_path": base_a}],
        grpc_port=0, rest_port=None, max_batch_size=1,
        batch_timeout_micros=100, config=cfg, poll_interval_s=0.2)
    server.model_config_path = str(mc)
    server.config_poll_interval_s = 0.2
    server.start()
    try:
        assert set(server.models) == {"a"}
        write_cfg([("a", base_a), ("b", base_b)])
        deadline = time.time() + 30
        while time.time() < deadline and set(server.models) != {"a", "b"}:
            time.sleep(0.2)
        assert set(server.models) == {"a", "b"}
        write_cfg([("b", base_b)])
        deadline = time.time() + 30
        while time.time() < deadline and set(server.models) != {"b"}:
            time.sleep(0.2)
        assert set(server.models) == {"b"}
    finally:
        server.stop()


def test_batcher_uint8_mask_top_pad_combination(shapes_config):
    """The production hot path in one unit: uint8 wire images + the
    mrcnn_mask_top filter + pad-to-max batching, through DynamicBatcher
    -> engine (engine-side GPU-analog gather + pinned-collate path),
    with per-request slices returned intact."""
    from maskrcnn_amd.serving.batcher import DynamicBatcher
    from maskrcnn_amd.serving.engine import InferenceEngine

    cfg = shapes_config
    eng = InferenceEngine(cfg, device="cpu", use_bf16=False)
    b = DynamicBatcher(eng, max_batch_size=4, batch_timeout_micros=50000,
                       pad_to_max=True)
    H = cfg.IMAGE_SHAPE[0]
    rng = np.random.RandomState(0)
    meta = np.zeros((1, cfg.IMAGE_META_SIZE), np.float32)
    meta[:, 4:7] = (H, H, 3)
    meta[:, 7:11] = (0, 0, H, H)
    futs = []
    imgs = []
    for i in range(3):  # 3 requests -> one batch padded to 4
        img = (rng.rand(1, H, H, 3) * 255).astype(np.uint8)
        imgs.append(img)
        futs.append(b.submit(img, meta.copy(),
                             outputs=["mrcnn_detection",
                                      "mrcnn_mask_top"]))
    outs = [f.result(timeout=120) for f in futs]
    D = cfg.DETECTION_MAX_INSTANCES
    for out in outs:
        assert set(out) >= {"mrcnn_detection", "mrcnn_mask_top"}
        assert out["mrcnn_detection"].shape == (1, D, 6)
        assert out["mrcnn_mask_top"].shape == (1, D, 28, 28)
    # Each request's slice must equal running it alone (same molding,
    # no cross-contamination from batching/padding).
    solo = eng.predict(imgs[1], meta.copy(),
                       outputs=["mrcnn_detection", "mrcnn_mask_top"])
    np.testing.assert_allclose(outs[1]["mrcnn_detection"],
                               solo["mrcnn_detection"], atol=1e-4)
    np.testing.assert_allclose(outs[1]["mrcnn_mask_top"],
                               solo["mrcnn_mask_top"], atol=1e-4)
    b.close()


def test_grpc_status_version_pin(served_model):
    """GetModelStatus with model_spec.version reports only that
    version; unknown version -> NOT_FOUND."""
    server, cfg, _ = served_model
    import grpc
    from maskrcnn_amd.serving.wireproto import (GetModelStatusRequest,
                                                GetModelStatusResponse)
    ch = grpc.insecure_channel(f"localhost:{server.grpc_port}")
    stub = ch.unary_unary(
        "/tensorflow.serving.ModelService/GetModelStatus",
        request_serializer=lambda r: r.encode(),
        response_deserializer=GetModelStatusResponse.decode)
    cur = server.state.version
    resp = stub(GetModelStatusRequest(
        model_spec=ModelSpec(name="mask_rcnn", version=cur)), timeout=30)
    assert [s.version for s in resp.model_version_status] == [cur]
    with pytest.raises(grpc.RpcError) as e:
        stub(GetModelStatusRequest(
            model_spec=ModelSpec(name="mask_rcnn", version=987)),
            timeout=30)
    assert e.value.code() == grpc.StatusCode.NOT_FOUND
    ch.close()


def test_fp8_export_with_warmup_loads(tmp_path):
    """fp8-weight SavedModel + warmup records: the quantized engine
    loads, replays warmup, and serves (config #5 interaction)."""
    from maskrcnn_amd.data.synthetic import ShapesConfig
    from maskrcnn_amd.model import MaskRCNN
    from maskrcnn_amd.serving.savedmodel import export_saved_model
    from maskrcnn_amd.serving.server import ModelState

    class TinyConfig(ShapesConfig):
        IMAGES_PER_GPU = 1
        PRE_NMS_LIMIT = 100
        POST_NMS_ROIS_INFERENCE = 10

    cfg = TinyConfig()
    m = MaskRCNN("inference", cfg, model_dir=str(tmp_path))
    base = str(tmp_path / "models" / "fp8wu")
    export_saved_model(m, cfg, base, version=1, weight_dtype="fp8",
                       warmup_batches=[1])
    st = ModelState("fp8wu", base, max_batch_size=1,
                    batch_timeout_micros=100, config=cfg)
    st.maybe_reload()
    assert st.version == 1
    H = cfg.IMAGE_SHAPE[0]
    meta = np.zeros((1, cfg.IMAGE_META_SIZE), np.float32)
    meta[:, 4:7] = (H, H, 3)
    meta[:, 7:11] = (0, 0, H, H)
    out, v = st.predict(np.zeros((1, H, H, 3), np.float32), meta)
    assert v == 1 and "mrcnn_detection" in out
    for slot in st.versions.values():
        for b in slot["batchers"]:
            b.close()


def test_warmup_tfrecord_golden_bytes(tmp_path):
    """Pin the TFRecord wire format byte-for-byte: u64le length,
    u32le masked-crc32c(length), payload, u32le masked-crc32c(payload).
    crc32c('abc') = 0x364B3FB7 is the published Castagnoli check value;
    TF's mask is ((c>>15)|(c<<17)) + 0xA282EAD8."""
    import struct
    from maskrcnn_amd.serving.warmup import tfrecord_write
    p = str(tmp_path / "g.tfrecord")
    tfrecord_write(p, [b"abc"])
    blob = open(p, "rb").read()
    assert blob == (struct.pack("<Q", 3) +
                    struct.pack("<I", 0x0E4999B0) +
                    b"abc" +
                    struct.pack("<I", 0x21F1576E)), blob.hex()

"""REST (:8501-style) Predict round trip against the FastAPI facade."""

import socket
import time

import numpy as np
import pytest


def _free_port():
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


@pytest.fixture(scope="module")
def rest_server(tmp_path_factory):
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
    tmp = tmp_path_factory.mktemp("rest")
    m = MaskRCNN("inference", cfg, model_dir=str(tmp))
    base = str(tmp / "models" / "mask_rcnn")
    export_saved_model(m, cfg, base, version=1)
    port = _free_port()
    server = ModelServer("mask_rcnn", base, grpc_port=0, rest_port=port,
                         max_batch_size=2, batch_timeout_micros=1000,
                         config=cfg)
    server.start()
    # Wait for uvicorn to come up.
    import requests
    for _ in range(100):
        try:
            requests.get(f"http://127.0.0.1:{port}/v1/models/mask_rcnn",
                         timeout=1)
            break
        except Exception:
            time.sleep(0.1)
    yield server, cfg, port
    server.stop()


def test_rest_status(rest_server):
    import requests
    server, cfg, port = rest_server
    r = requests.get(f"http://127.0.0.1:{port}/v1/models/mask_rcnn",
                     timeout=10)
    assert r.status_code == 200
    st = r.json()["model_version_status"][0]
    assert st["state"] == "AVAILABLE"
    assert st["version"] == "1"


def test_rest_predict_columnar(rest_server):
    server, cfg, port = rest_server
    from maskrcnn_amd.serving.client import RestServingClient
    client = RestServingClient(f"http://127.0.0.1:{port}", cfg)
    img = (np.random.rand(100, 90, 3) * 255).astype(np.uint8)
    results = client.detect([img])
    r = results[0]
    assert r["rois"].shape[1] == 4
    assert r["masks"].shape[:2] == img.shape[:2]


def test_rest_unknown_model_404(rest_server):
    import requests
    server, cfg, port = rest_server
    r = requests.post(f"http://127.0.0.1:{port}/v1/models/nope:predict",
                      json={"inputs": {}}, timeout=10)
    assert r.status_code == 404


def test_rest_metadata_endpoint(rest_server):
    import requests
    server, cfg, port = rest_server
    r = requests.get(
        f"http://127.0.0.1:{port}/v1/models/mask_rcnn/metadata",
        timeout=10)
    assert r.status_code == 200
    md = r.json()
    sig = md["metadata"]["signature_def"]["signature_def"][
        "serving_default"]
    assert set(sig["inputs"]) == {"input_image", "input_image_meta",
                                  "input_anchors"}
    assert len(sig["outputs"]) == 7
    assert md["model_spec"]["name"] == "mask_rcnn"


def test_rest_predict_instances_format(rest_server):
    """TF-Serving row ('instances') JSON schema."""
    import requests
    server, cfg, port = rest_server
    H = cfg.IMAGE_SHAPE[0]
    inst = {
        "input_image": np.zeros((H, H, 3), np.float32).tolist(),
        "input_image_meta": np.concatenate(
            [[0], (H, H, 3), (H, H, 3), (0, 0, H, H), [1.0],
             np.ones(cfg.NUM_CLASSES)]).tolist(),
    }
    r = requests.post(
        f"http://127.0.0.1:{port}/v1/models/mask_rcnn:predict",
        json={"instances": [inst]}, timeout=120)
    assert r.status_code == 200, r.text[:300]
    # TF REST row schema: instances -> "predictions", one per row,
    # WITHOUT the batch dim (columnar "inputs" -> "outputs" keeps it).
    preds = r.json()["predictions"]
    assert len(preds) == 1
    assert "mrcnn_detection" in preds[0]
    D = cfg.DETECTION_MAX_INSTANCES
    assert np.asarray(preds[0]["mrcnn_detection"]).shape == (D, 6)


def test_rest_signature_name_validation(rest_server):
    import requests
    server, cfg, port = rest_server
    r = requests.post(
        f"http://127.0.0.1:{port}/v1/models/mask_rcnn:predict",
        json={"signature_name": "bogus", "inputs": {}}, timeout=10)
    assert r.status_code == 400
    assert "signature" in r.json()["error"]


def test_rest_versioned_predict_url(rest_server):
    """TF-Serving /v1/models/{m}/versions/{v}:predict routing."""
    import requests
    server, cfg, port = rest_server
    cur = server.state.version
    H = cfg.IMAGE_SHAPE[0]
    meta = np.concatenate(
        [[[0] + [H, H, 3] + [H, H, 3] + [0, 0, H, H] + [1.0]],
         np.ones((1, cfg.NUM_CLASSES))], axis=1)
    body = {"inputs": {
        "input_image": np.zeros((1, H, H, 3), np.float32).tolist(),
        "input_image_meta": meta.tolist()}}
    r = requests.post(
        f"http://127.0.0.1:{port}/v1/models/mask_rcnn/versions/"
        f"{cur}:predict", json=body, timeout=120)
    assert r.status_code == 200, r.text[:200]
    r = requests.post(
        f"http://127.0.0.1:{port}/v1/models/mask_rcnn/versions/"
        f"999:predict", json=body, timeout=10)
    assert r.status_code == 404


def test_rest_version_status_url(rest_server):
    import requests
    server, cfg, port = rest_server
    cur = server.state.version
    st = requests.get(f"http://127.0.0.1:{port}/v1/models/mask_rcnn/"
                      f"versions/{cur}", timeout=10).json()
    assert st["model_version_status"][0]["state"] == "AVAILABLE"
    st = requests.get(f"http://127.0.0.1:{port}/v1/models/mask_rcnn/"
                      f"versions/999", timeout=10).json()
    assert st["model_version_status"][0]["state"] == "UNAVAILABLE"


def test_rest_concurrent_requests_not_serialized(rest_server):
    """Sync (threadpool) predict handler: concurrent REST predicts must
    overlap rather than serialize behind one blocked event loop."""
    import threading

    import requests
    server, cfg, port = rest_server
    H = cfg.IMAGE_SHAPE[0]
    meta = np.concatenate(
        [[[0] + [H, H, 3] + [H, H, 3] + [0, 0, H, H] + [1.0]],
         np.ones((1, cfg.NUM_CLASSES))], axis=1)
    body = {"inputs": {
        "input_image": np.zeros((1, H, H, 3), np.float32).tolist(),
        "input_image_meta": meta.tolist()}}
    codes = []

    def hit():
        r = requests.post(
            f"http://127.0.0.1:{port}/v1/models/mask_rcnn:predict",
            json=body, timeout=120)
        codes.append(r.status_code)

    ts = [threading.Thread(target=hit) for _ in range(4)]
    t0 = time.time()
    for t in ts:
        t.start()
    for t in ts:
        t.join(timeout=120)
    assert codes == [200] * 4, codes
    assert time.time() - t0 < 100


def test_rest_prometheus_metrics_path(rest_server):
    """TF-Serving monitoring path on the REST port."""
    import requests
    server, cfg, port = rest_server
    r = requests.get(
        f"http://127.0.0.1:{port}/monitoring/prometheus/metrics",
        timeout=10)
    assert r.status_code == 200
    assert b"predict_requests_total" in r.content or \
        b"batch_queue_wait" in r.content


def test_engine_stage_timers(shapes_config):
    """engine.predict records per-stage wall times (SURVEY §5 tracing:
    per-stage CPU timers in the serve loop) and exports them to the
    prometheus registry when available."""
    import numpy as np
    from maskrcnn_amd.serving.engine import InferenceEngine

    cfg = shapes_config
    eng = InferenceEngine(cfg, device="cpu", use_bf16=False)
    H = cfg.IMAGE_SHAPE[0]
    img = np.zeros((1, H, H, 3), np.float32)
    meta = np.zeros((1, cfg.IMAGE_META_SIZE), np.float32)
    eng.predict(img, meta, outputs=["mrcnn_detection"])
    st = eng.last_stage_ms
    assert set(st) == {"upload", "forward", "dtoh"}
    assert all(v >= 0 for v in st.values())
    assert st["forward"] > 0
    try:
        from prometheus_client import REGISTRY
        sample = REGISTRY.get_sample_value(
            "predict_stage_seconds_count", {"stage": "forward"})
        assert sample and sample >= 1
    except ImportError:
        pass


def test_rest_version_labels_route(tmp_path):
    """/v1/models/{m}/labels/{l}:predict (TF-Serving REST label
    addressing) — exercised in-process via the FastAPI TestClient."""
    from fastapi.testclient import TestClient

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
    base = str(tmp_path / "models" / "vlr")
    export_saved_model(m, cfg, base, version=1)
    export_saved_model(m, cfg, base, version=2)
    server = ModelServer(
        model_configs=[{"name": "vlr", "base_path": base,
                        "version_policy": "all",
                        "version_labels": {"stable": 1}}],
        grpc_port=0, rest_port=None, max_batch_size=1,
        batch_timeout_micros=100, config=cfg)
    for st in server.models.values():
        st.maybe_reload()
    app = server._build_rest_app()
    client = TestClient(app)
    H = cfg.IMAGE_SHAPE[0]
    meta = np.zeros((1, cfg.IMAGE_META_SIZE), np.float32)
    meta[:, 4:7] = (H, H, 3)
    meta[:, 7:11] = (0, 0, H, H)
    body = {"inputs": {
        "input_image": np.zeros((1, H, H, 3), np.float32).tolist(),
        "input_image_meta": meta.tolist()}}
    r = client.post("/v1/models/vlr/labels/stable:predict", json=body)
    assert r.status_code == 200, r.text
    assert "mrcnn_detection" in r.json()["outputs"]
    r = client.post("/v1/models/vlr/labels/nope:predict", json=body)
    assert r.status_code == 404
    for st in server.models.values():
        for slot in st.versions.values():
            for b in slot["batchers"]:
                b.close()


def test_rest_output_filter_extension(rest_server):
    """REST body `output_filter` (wire extension mirroring gRPC):
    restricts computed+serialized outputs, supports mrcnn_mask_top."""
    import requests
    server, cfg, port = rest_server
    H = cfg.IMAGE_SHAPE[0]
    meta = np.zeros((1, cfg.IMAGE_META_SIZE), np.float32)
    meta[:, 4:7] = (H, H, 3)
    meta[:, 7:11] = (0, 0, H, H)
    body = {"inputs": {
        "input_image": np.zeros((1, H, H, 3), np.float32).tolist(),
        "input_image_meta": meta.tolist()},
        "output_filter": ["mrcnn_detection", "mrcnn_mask_top"]}
    r = requests.post(
        f"http://127.0.0.1:{port}/v1/models/mask_rcnn:predict",
        json=body, timeout=120)
    assert r.status_code == 200, r.text
    out = r.json()["outputs"]
    assert set(out) == {"mrcnn_detection", "mrcnn_mask_top"}
    D = cfg.DETECTION_MAX_INSTANCES
    assert np.asarray(out["mrcnn_mask_top"]).shape == (1, D, 28, 28)
    # bad filter type -> 400
    body["output_filter"] = "mrcnn_detection"
    r = requests.post(
        f"http://127.0.0.1:{port}/v1/models/mask_rcnn:predict",
        json=body, timeout=60)
    assert r.status_code == 400


def test_rest_per_version_metadata(rest_server):
    import requests
    server, cfg, port = rest_server
    r = requests.get(
        f"http://127.0.0.1:{port}/v1/models/mask_rcnn/versions/1/"
        f"metadata", timeout=30)
    assert r.status_code == 200
    assert r.json()["model_spec"]["version"] == "1"
    r = requests.get(
        f"http://127.0.0.1:{port}/v1/models/mask_rcnn/versions/9/"
        f"metadata", timeout=30)
    assert r.status_code == 404

#!/usr/bin/env python3
"""Serving soak: mixed concurrent clients against a live ModelServer.

QA harness for the serving plane (S2/N11): boots a ModelServer (CPU
shapes-scale model by default, or the real config on a GPU box), then
hammers it from several threads with a MIX of traffic — fp32 and uint8
Predicts, output filters incl. mrcnn_mask_top, version pins, metadata
and status RPCs, deliberate NOT_FOUND errors — for --seconds, and
reports request counts, error counts, latency percentiles and RSS
drift (a leak indicator).

Usage:
    python tools/soak_serving.py --seconds 120 --threads 8 \
        [--grpc-ingress cpp]
"""

from __future__ import annotations

import argparse
import json
import random
import sys
import tempfile
import threading
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import numpy as np  # noqa: E402


def rss_mb():
    import psutil
    return psutil.Process().memory_info().rss / 1e6


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--seconds", type=float, default=60)
    ap.add_argument("--threads", type=int, default=8)
    ap.add_argument("--grpc-ingress", default="cpp",
                    choices=["python", "cpp"])
    ap.add_argument("--image-size", type=int, default=128)
    args = ap.parse_args()

    from maskrcnn_amd.data.synthetic import ShapesConfig
    from maskrcnn_amd.model import MaskRCNN
    from maskrcnn_amd.serving.savedmodel import export_saved_model
    from maskrcnn_amd.serving.server import ModelServer

    class SoakConfig(ShapesConfig):
        IMAGES_PER_GPU = 1
        PRE_NMS_LIMIT = 200
        POST_NMS_ROIS_INFERENCE = 20
        DETECTION_MIN_CONFIDENCE = 0.0
        IMAGE_MIN_DIM = 128
        IMAGE_MAX_DIM = 128

    cfg = SoakConfig()
    cfg.IMAGE_MIN_DIM = cfg.IMAGE_MAX_DIM = args.image_size
    cfg.__init__()
    tmp = tempfile.mkdtemp(prefix="soak_")
    m = MaskRCNN("inference", cfg, model_dir=tmp)
    base = str(Path(tmp) / "models" / "soak")
    export_saved_model(m, cfg, base, version=1)
    export_saved_model(m, cfg, base, version=2)
    server = ModelServer(
        model_configs=[{"name": "soak", "base_path": base,
                        "version_policy": "all",
                        "version_labels": {"stable": 1}}],
        grpc_port=0, rest_port=None, max_batch_size=4,
        batch_timeout_micros=2000, config=cfg,
        grpc_ingress=args.grpc_ingress)
    server.start()

    import grpc
    from maskrcnn_amd.serving.client import PREDICT_METHOD
    from maskrcnn_amd.serving.wireproto import (ModelSpec, PredictRequest,
                                                PredictResponse,
                                                numpy_to_tensor)

    H = cfg.IMAGE_SHAPE[0]
    meta = np.zeros((1, cfg.IMAGE_META_SIZE), np.float32)
    meta[:, 4:7] = (H, H, 3)
    meta[:, 7:11] = (0, 0, H, H)

    stats = {"ok": 0, "expected_err": 0, "err": 0}
    lats = []
    lock = threading.Lock()
    stop = threading.Event()

    def worker(i):
        rng = random.Random(i)
        np_rng = np.random.RandomState(i)
        ch = grpc.insecure_channel(f"127.0.0.1:{server.grpc_port}")
        stub = ch.unary_unary(
            PREDICT_METHOD, request_serializer=lambda r: r.encode(),
            response_deserializer=PredictResponse.decode)
        from maskrcnn_amd.serving.wireproto import (
            GetModelStatusRequest, GetModelStatusResponse)
        status_stub = ch.unary_unary(
            "/tensorflow.serving.ModelService/GetModelStatus",
            request_serializer=lambda r: r.encode(),
            response_deserializer=GetModelStatusResponse.decode)
        while not stop.is_set():
            kind = rng.random()
            t0 = time.perf_counter()
            try:
                if kind < 0.05:          # status RPC
                    status_stub(GetModelStatusRequest(
                        model_spec=ModelSpec(name="soak")), timeout=30)
                elif kind < 0.10:        # expected NOT_FOUND
                    r = PredictRequest(model_spec=ModelSpec(name="nope"))
                    r.inputs["input_image"] = numpy_to_tensor(
                        np.zeros((1, H, H, 3), np.float32))
                    r.inputs["input_image_meta"] = numpy_to_tensor(meta)
                    try:
                        stub(r, timeout=30)
                        with lock:
                            stats["err"] += 1
                    except grpc.RpcError as e:
                        with lock:
                            if e.code() == grpc.StatusCode.NOT_FOUND:
                                stats["expected_err"] += 1
                            else:
                                stats["err"] += 1
                    continue
                else:                    # predict, mixed wire formats
                    r = PredictRequest(model_spec=ModelSpec(
                        name="soak",
                        version=rng.choice([None, None, 1, 2]),
                        version_label=rng.choice(
                            [""] * 3 + ["stable"])
                        if rng.random() < 0.5 else ""))
                    if r.model_spec.version is not None:
                        r.model_spec.version_label = ""
                    if rng.random() < 0.5:
                        img = (np_rng.rand(1, H, H, 3) * 255) \
                            .astype(np.uint8)
                    else:
                        img = np_rng.rand(1, H, H, 3) \
                            .astype(np.float32) * 60
                    r.inputs["input_image"] = numpy_to_tensor(img)
                    r.inputs["input_image_meta"] = numpy_to_tensor(meta)
                    r.output_filter = rng.choice([
                        [], ["mrcnn_detection"],
                        ["mrcnn_detection", "mrcnn_mask_top"],
                        ["mrcnn_detection", "mrcnn_mask"]])
                    resp = stub(r, timeout=60)
                    assert resp.model_spec.version in (1, 2)
                with lock:
                    stats["ok"] += 1
                    lats.append(time.perf_counter() - t0)
            except Exception:  # noqa: BLE001
                import traceback
                traceback.print_exc()
                with lock:
                    stats["err"] += 1
        ch.close()

    rss0 = rss_mb()
    threads = [threading.Thread(target=worker, args=(i,))
               for i in range(args.threads)]
    for t in threads:
        t.start()
    time.sleep(args.seconds)
    stop.set()
    for t in threads:
        t.join(timeout=90)
    server.stop()
    lats.sort()
    out = {
        "ingress": args.grpc_ingress,
        "seconds": args.seconds,
        "threads": args.threads,
        "requests_ok": stats["ok"],
        "expected_not_found": stats["expected_err"],
        "unexpected_errors": stats["err"],
        "rps": round(stats["ok"] / args.seconds, 1),
        "p50_ms": round(lats[len(lats) // 2] * 1e3, 1) if lats else None,
        "p99_ms": round(lats[int(len(lats) * 0.99)] * 1e3, 1)
        if lats else None,
        "rss_mb_start": round(rss0, 1),
        "rss_mb_end": round(rss_mb(), 1),
    }
    print(json.dumps(out))
    return 0 if stats["err"] == 0 else 1


if __name__ == "__main__":
    sys.exit(main())

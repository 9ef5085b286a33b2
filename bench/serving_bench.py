#!/usr/bin/env python3
"""Serving benchmark — BASELINE config #4: gRPC Predict, R101-FPN,
dynamic-batched inference on 1 MI355X.

Starts an in-process ModelServer (gRPC only) on a random-init R101-FPN
export, fires concurrent Predict requests from client threads (pre-molded
synthetic 1024x1024 images, anchors server-side), and reports images/s +
p50/p99 latency as one JSON line.
"""

from __future__ import annotations

import argparse
import json
import os
import sys
import tempfile
import threading
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))))
from maskrcnn_amd._rocm_env import ensure_env_or_reexec  # noqa: E402

ensure_env_or_reexec()

import numpy as np  # noqa: E402


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--requests", type=int, default=64)
    ap.add_argument("--client-timeout", type=float, default=300.0)
    ap.add_argument("--concurrency", type=int, default=16)
    ap.add_argument("--warmup", type=int, default=8)
    ap.add_argument("--image-size", type=int, default=1024)
    ap.add_argument("--max-batch", type=int, default=32)
    ap.add_argument("--batch-timeout-micros", type=int, default=5000)
    ap.add_argument("--backbone", default="resnet101")
    ap.add_argument("--send-anchors", action="store_true",
                    help="include the [A,4] anchor tensor in each request "
                         "(reference client behavior)")
    args = ap.parse_args()

    import logging
    logging.basicConfig(
        level=logging.INFO, stream=sys.stderr,
        format="%(asctime)s %(name)s %(message)s")

    import torch
    from maskrcnn_amd.config import Config
    from maskrcnn_amd.model.mask_rcnn import MaskRCNNModel
    from maskrcnn_amd.serving.savedmodel import export_saved_model
    from maskrcnn_amd.serving.server import ModelServer
    from maskrcnn_amd.serving.client import GrpcServingClient

    class ServeConfig(Config):
        NAME = "serve_bench"
        BACKBONE = args.backbone
        NUM_CLASSES = 81
        IMAGES_PER_GPU = 1
        IMAGE_MIN_DIM = args.image_size
        IMAGE_MAX_DIM = args.image_size

    cfg = ServeConfig()
    torch.manual_seed(0)
    model = MaskRCNNModel(cfg)
    base = tempfile.mkdtemp(prefix="serve_bench_")
    export_saved_model(model, cfg, base, version=1)
    del model

    server = ModelServer("mask_rcnn", base, grpc_port=0, rest_port=None,
                         max_batch_size=args.max_batch,
                         batch_timeout_micros=args.batch_timeout_micros,
                         config=cfg)
    server.start(warmup=True)

    client = GrpcServingClient(f"localhost:{server.grpc_port}", cfg,
                               send_anchors=args.send_anchors,
                               timeout=args.client_timeout)
    # Pre-mold one synthetic image; clients reuse the molded arrays
    # (the serving benchmark measures the server, not PIL).
    img = (np.random.rand(800, 1000, 3) * 255).astype(np.uint8)
    molded, metas, windows = client.mold_inputs([img])
    anchors = None
    if args.send_anchors:
        anchors = client.anchor_cache.get(molded[0].shape).numpy() \
            .astype(np.float32)[None]

    latencies = []
    lat_lock = threading.Lock()
    counter = {"n": -args.warmup}

    def worker():
        while True:
            with lat_lock:
                if counter["n"] >= args.requests:
                    return
                counter["n"] += 1
                warm = counter["n"] <= 0
            t0 = time.perf_counter()
            client.predict_raw(molded, metas, anchors)
            dt = time.perf_counter() - t0
            if not warm:
                with lat_lock:
                    latencies.append(dt)

    # Warmup request serially first (MIOpen find etc.).
    client.predict_raw(molded, metas, anchors)

    threads = [threading.Thread(target=worker)
               for _ in range(args.concurrency)]
    t_start = time.perf_counter()
    for t in threads:
        t.start()
    for t in threads:
        t.join()
    elapsed = time.perf_counter() - t_start

    lats = np.array(sorted(latencies))
    n = len(lats)
    qps = n / elapsed
    out = {
        "metric": "serving_images_per_sec",
        "value": round(qps, 3),
        "unit": "images/s",
        "n_gpus": 1,
        "steps": n,
        "warmup": args.warmup,
        "ms_per_step": round(1000 * elapsed / max(n, 1), 2),
        "higher_is_better": True,
        "scaling": "weak",
        "vs_baseline": None,
        "dtype": "bf16",
        "data": "synthetic",
        "p50_ms": round(1000 * float(np.percentile(lats, 50)), 2),
        "p99_ms": round(1000 * float(np.percentile(lats, 99)), 2),
        "mean_batch": round(server.state.batcher.stats["batch_size_sum"] /
                            max(server.state.batcher.stats["batches"], 1),
                            2),
        "config": {
            "model": f"{args.backbone}-FPN maskrcnn serving",
            "global_batch": args.max_batch,
            "image_size": args.image_size,
            "concurrency": args.concurrency,
            "parallelism": "dp1",
        },
    }
    print(json.dumps(out))
    server.stop()


if __name__ == "__main__":
    main()

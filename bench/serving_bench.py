#!/usr/bin/env python3
"""Serving benchmark — BASELINE config #4: gRPC Predict, R101-FPN,
dynamic-batched inference on 1 MI355X.

Starts an in-process ModelServer (gRPC only) on a random-init R101-FPN
export, then drives it with MULTI-PROCESS gRPC clients (each subprocess
has its own channel and GIL; single-process clients were the round-1
bottleneck) firing pre-molded synthetic 1024x1024 Predict requests.
Reports images/s + p50/p99 latency as one JSON line.
"""

from __future__ import annotations

import argparse
import json
import os
import subprocess
import sys
import tempfile
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))))
from maskrcnn_amd._rocm_env import ensure_env_or_reexec  # noqa: E402

if __name__ == "__main__":
    # re-exec only when run as a script (library imports must never
    # replace the host process; see maskrcnn_amd/_rocm_env.py)
    ensure_env_or_reexec()

import numpy as np  # noqa: E402

WORKER = os.path.join(os.path.dirname(os.path.abspath(__file__)),
                      "serving_client_worker.py")


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--requests", type=int, default=64,
                    help="total timed requests across all clients")
    ap.add_argument("--client-procs", type=int, default=4)
    ap.add_argument("--client-threads", type=int, default=8)
    ap.add_argument("--warmup", type=int, default=8)
    ap.add_argument("--image-size", type=int, default=1024)
    ap.add_argument("--max-batch", type=int, default=32)
    ap.add_argument("--batch-timeout-micros", type=int, default=5000)
    ap.add_argument("--backbone", default="resnet101")
    ap.add_argument("--weight-dtype", default="fp32",
                    choices=["fp32", "fp8"])
    ap.add_argument("--grpc-ingress", default="python",
                    choices=["python", "cpp"],
                    help="cpp = native HTTP/2 front-end "
                         "(csrc/ingress.cpp); applies to the in-process "
                         "server and the --server-procs replicas")
    ap.add_argument("--server-procs", type=int, default=1,
                    help="N independent ModelServer replicas on ONE "
                         "GPU (separate processes, separate ports, "
                         "clients round-robin): bypasses the ~28 "
                         "unary-msgs/s gRPC-python per-process ingress "
                         "ceiling; the engines share cuda:0")
    args = ap.parse_args()

    import logging
    logging.basicConfig(level=logging.INFO, stream=sys.stderr,
                        format="%(asctime)s %(name)s %(message)s")

    import torch
    from maskrcnn_amd.config import Config
    from maskrcnn_amd.model.mask_rcnn import MaskRCNNModel
    from maskrcnn_amd.serving.savedmodel import export_saved_model
    from maskrcnn_amd.serving.server import ModelServer

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
    export_saved_model(model, cfg, base, version=1,
                       weight_dtype=args.weight_dtype)
    del model

    servers = []
    targets = []
    server = ModelServer("mask_rcnn", base, grpc_port=0, rest_port=None,
                         max_batch_size=args.max_batch,
                         batch_timeout_micros=args.batch_timeout_micros,
                         config=cfg, grpc_ingress=args.grpc_ingress)
    server.start(warmup=True)
    servers.append(server)
    targets.append(f"127.0.0.1:{server.grpc_port}")
    # Extra same-GPU replicas as SUBPROCESSES (each with its own gRPC
    # C-core + GIL). The committed miopen_db find-db keeps their
    # warmups cheap.
    extra = []
    if args.server_procs > 1:
        for i in range(args.server_procs - 1):
            port = 18600 + i
            pr = subprocess.Popen(
                [sys.executable, "-m", "maskrcnn_amd.serving.server",
                 "--model_name", "mask_rcnn",
                 "--model_base_path", base,
                 "--port", str(port), "--rest_api_port", "0",
                 "--enable_batching",
                 "--max_batch_size", str(args.max_batch),
                 "--batch_timeout_micros",
                 str(args.batch_timeout_micros),
                 "--grpc_ingress", args.grpc_ingress],
                stderr=subprocess.DEVNULL)
            extra.append(pr)
            targets.append(f"127.0.0.1:{port}")
        # Wait for the extra replicas to come up.
        from maskrcnn_amd.serving.client import ModelServiceClient
        for t in targets[1:]:
            deadline = time.time() + 600
            while time.time() < deadline:
                try:
                    st = ModelServiceClient(t, timeout=5.0) \
                        .get_model_status("mask_rcnn")
                    # 30 = ModelVersionStatus.State.AVAILABLE
                    if st and st[0]["state"] in (30, "AVAILABLE"):
                        break
                except Exception:
                    pass
                time.sleep(2.0)

    def run_clients(total_requests):
        per = max(1, total_requests // args.client_procs)
        procs = [subprocess.Popen(
            [sys.executable, WORKER,
             "--target", targets[i % len(targets)],
             "--requests", str(per),
             "--threads", str(args.client_threads),
             "--image-size", str(args.image_size)],
            stdout=subprocess.PIPE, text=True)
            for i in range(args.client_procs)]
        lats = []
        elapsed = 0.0
        n_err = 0
        for p in procs:
            out, _ = p.communicate(timeout=1800)
            assert p.returncode == 0, "client worker failed"
            d = json.loads(out.strip().splitlines()[-1])
            lats.extend(d["lats"])
            elapsed = max(elapsed, d["elapsed"])
            n_err += d.get("n_errors", 0)
            if d.get("errors"):
                print("client errors:", d["errors"], file=sys.stderr)
        if n_err:
            print(f"WARNING: {n_err} client requests errored",
                  file=sys.stderr)
        return lats, elapsed

    run_clients(args.warmup)  # warmup (incl. channel setup)
    t0 = time.perf_counter()
    lats, elapsed = run_clients(args.requests)
    wall = time.perf_counter() - t0

    lats_a = np.array(sorted(lats))
    n = len(lats_a)
    qps = n / elapsed
    st = server.state.batcher.stats
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
        "dtype": "bf16" if args.weight_dtype == "fp32" else "fp8-weights",
        "data": "synthetic",
        "p50_ms": round(1000 * float(np.percentile(lats_a, 50)), 2),
        "p99_ms": round(1000 * float(np.percentile(lats_a, 99)), 2),
        "mean_batch": round(st["batch_size_sum"] / max(st["batches"], 1),
                            2),
        "wall_s": round(wall, 2),
        "config": {
            "model": f"{args.backbone}-FPN maskrcnn serving",
            "server_procs": args.server_procs,
            "global_batch": args.max_batch,
            "image_size": args.image_size,
            "clients": f"{args.client_procs}p x {args.client_threads}t",
            "parallelism": "dp1",
        },
    }
    print(json.dumps(out))
    server.stop()


if __name__ == "__main__":
    main()

#!/usr/bin/env python3
"""Subprocess client worker for serving_bench: fires N Predict requests
at a target with its own grpc channel + GIL, prints one JSON line of
latencies."""

from __future__ import annotations

import argparse
import json
import os
import sys
import threading
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))))

import numpy as np  # noqa: E402


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--target", required=True)
    ap.add_argument("--requests", type=int, required=True)
    ap.add_argument("--threads", type=int, default=4)
    ap.add_argument("--image-size", type=int, default=1024)
    ap.add_argument("--num-classes", type=int, default=81)
    ap.add_argument("--timeout", type=float, default=300.0)
    args = ap.parse_args()

    import grpc
    from maskrcnn_amd.serving.client import PREDICT_METHOD
    from maskrcnn_amd.serving.wireproto import (ModelSpec, PredictRequest,
                                                PredictResponse,
                                                numpy_to_tensor)

    H = args.image_size
    # uint8 wire extension (default): the raw image ships at 1/4 the
    # bytes and the server molds on-GPU. MASKRCNN_CLIENT_FP32=1 sends
    # the stock molded-fp32 TF-Serving payload instead.
    if os.environ.get("MASKRCNN_CLIENT_FP32") == "1":
        molded = np.random.randn(1, H, H, 3).astype(np.float32) * 40
    else:
        molded = (np.random.rand(1, H, H, 3) * 255).astype(np.uint8)
    meta = np.zeros((1, 1 + 3 + 3 + 4 + 1 + args.num_classes), np.float32)
    meta[:, 4:7] = (H, H, 3)
    meta[:, 7:11] = (0, 0, H, H)

    channel = grpc.insecure_channel(
        args.target,
        options=[("grpc.max_receive_message_length", 256 << 20),
                 ("grpc.max_send_message_length", 256 << 20)])
    stub = channel.unary_unary(PREDICT_METHOD,
                               request_serializer=lambda r: r.encode(),
                               response_deserializer=PredictResponse.decode)
    req = PredictRequest(model_spec=ModelSpec(
        name="mask_rcnn", signature_name="serving_default"))
    req.inputs["input_image"] = numpy_to_tensor(molded)
    req.inputs["input_image_meta"] = numpy_to_tensor(meta)
    # mrcnn_mask_top: detected-class masks only ([D,28,28] — exactly
    # what unmold consumes), 81x fewer response bytes than the full
    # per-class tensor. MASKRCNN_CLIENT_FULL_MASK=1 benchmarks the
    # stock TF-Serving-shaped response instead.
    full = os.environ.get("MASKRCNN_CLIENT_FULL_MASK") == "1"
    req.output_filter = ["mrcnn_detection",
                         "mrcnn_mask" if full else "mrcnn_mask_top"]
    req_bytes_cached = req.encode()  # serialize once per worker

    lats = []
    lock = threading.Lock()
    counter = {"n": 0}

    raw_stub = channel.unary_unary(PREDICT_METHOD,
                                   request_serializer=lambda b: b,
                                   response_deserializer=lambda b: b)

    errors = []

    def worker():
        while True:
            with lock:
                if counter["n"] >= args.requests:
                    return
                counter["n"] += 1
            t0 = time.perf_counter()
            try:
                raw_stub(req_bytes_cached, timeout=args.timeout)
            except Exception as e:  # noqa: BLE001
                with lock:
                    errors.append(repr(e)[:200])
                continue
            dt = time.perf_counter() - t0
            with lock:
                lats.append(dt)

    threads = [threading.Thread(target=worker)
               for _ in range(args.threads)]
    t0 = time.perf_counter()
    for t in threads:
        t.start()
    for t in threads:
        t.join()
    elapsed = time.perf_counter() - t0
    print(json.dumps({"lats": lats, "elapsed": elapsed,
                      "errors": errors[:5],
                      "n_errors": len(errors)}))


if __name__ == "__main__":
    main()

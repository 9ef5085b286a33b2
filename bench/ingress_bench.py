#!/usr/bin/env python3
"""A/B the gRPC ingress implementations at the transport+proto layer.

Handler = real wireproto PredictRequest.decode + a canned encoded
PredictResponse (no model, no GPU): measures exactly the component the
native daemon (csrc/ingress.cpp) replaces — grpc-python's C-core +
GIL-side serializer hops vs C++ HTTP/2 framing with one Python entry
per request. Request shape mirrors production serving: uint8 [1, S, S,
3] image (~3 MB at S=1024) + image_meta; response mirrors the
mrcnn_mask_top extension (~160 KB).

Usage: python bench/ingress_bench.py [--workers 8] [--n 30] [--size 1024]
"""

from __future__ import annotations

import argparse
import json
import subprocess
import sys
import time
from pathlib import Path

import numpy as np

REPO = Path(__file__).resolve().parent.parent
sys.path.insert(0, str(REPO))

CLIENT = r"""
import sys, time
import numpy as np
import grpc
sys.path.insert(0, {repo!r})
from maskrcnn_amd.serving.wireproto import (ModelSpec, PredictRequest,
                                            PredictResponse,
                                            numpy_to_tensor)
port, n, size = int(sys.argv[1]), int(sys.argv[2]), int(sys.argv[3])
req = PredictRequest(model_spec=ModelSpec(name="m"))
rng = np.random.RandomState(0)
req.inputs["input_image"] = numpy_to_tensor(
    (rng.rand(1, size, size, 3) * 255).astype(np.uint8))
req.inputs["input_image_meta"] = numpy_to_tensor(
    np.zeros((1, 93), np.float32))
req.output_filter = ["mrcnn_detection", "mrcnn_mask_top"]
ch = grpc.insecure_channel(f"127.0.0.1:{{port}}",
    options=[("grpc.max_receive_message_length", 256 << 20),
             ("grpc.max_send_message_length", 256 << 20)])
stub = ch.unary_unary("/tensorflow.serving.PredictionService/Predict",
                      request_serializer=lambda r: r.encode(),
                      response_deserializer=PredictResponse.decode)
stub(req, timeout=60)
t0 = time.time()
for _ in range(n):
    r = stub(req, timeout=60)
    assert "mrcnn_mask_top" in r.outputs
t1 = time.time()
print(t0, t1, n)
"""


def make_handler():
    from maskrcnn_amd.serving.wireproto import (ModelSpec, PredictRequest,
                                                PredictResponse,
                                                numpy_to_tensor)
    canned = PredictResponse(
        outputs={
            "mrcnn_detection": numpy_to_tensor(
                np.zeros((1, 50, 6), np.float32)),
            "mrcnn_mask_top": numpy_to_tensor(
                np.zeros((1, 50, 28, 28), np.float32)),
        },
        model_spec=ModelSpec(name="m", version=1)).encode()

    def handle_bytes(payload) -> bytes:
        req = PredictRequest.decode(payload)
        assert "input_image" in req.inputs
        return canned

    return handle_bytes


def run_clients(port: int, workers: int, n: int, size: int):
    """Aggregate rate over the union of the clients' own request
    windows (process startup — torch import etc. — excluded; it
    dominated the wall clock and made early numbers meaningless)."""
    script = CLIENT.format(repo=str(REPO))
    procs = [subprocess.Popen([sys.executable, "-c", script, str(port),
                               str(n), str(size)],
                              stdout=subprocess.PIPE)
             for _ in range(workers)]
    spans, lat = [], []
    for p in procs:
        out, _ = p.communicate()
        if p.returncode != 0:
            raise RuntimeError("client failed")
        t0, t1, cn = out.split()[-3:]
        spans.append((float(t0), float(t1)))
        lat.append((float(t1) - float(t0)) / int(cn) * 1e3)
    dt = max(t1 for _, t1 in spans) - min(t0 for t0, _ in spans)
    total = workers * n
    return total / dt, float(np.median(lat))


def bench_cpp(workers, n, size):
    from maskrcnn_amd import _ingress
    handle = make_handler()

    def dispatch(path, payload):
        return (0, "", handle(payload))

    srv = _ingress.IngressServer(0, dispatch, 16)
    srv.start()
    try:
        return run_clients(srv.port, workers, n, size)
    finally:
        srv.stop()


def bench_python(workers, n, size):
    import grpc
    from concurrent import futures as cf
    handle = make_handler()
    handler = grpc.method_handlers_generic_handler(
        "tensorflow.serving.PredictionService",
        {"Predict": grpc.unary_unary_rpc_method_handler(
            lambda req, ctx: handle(req),
            request_deserializer=lambda b: b,
            response_serializer=lambda b: b)})
    server = grpc.server(
        cf.ThreadPoolExecutor(max_workers=16),
        options=[("grpc.max_receive_message_length", 256 << 20),
                 ("grpc.max_send_message_length", 256 << 20)])
    server.add_generic_rpc_handlers((handler,))
    port = server.add_insecure_port("[::]:0")
    server.start()
    try:
        return run_clients(port, workers, n, size)
    finally:
        server.stop(0)


def gil_load_thread(stop):
    """Simulates the model-runner/batcher thread of a busy server: pure
    Python work holding the GIL in short bursts (collation, unmold,
    proto bookkeeping)."""
    x = 0
    while not stop.is_set():
        for i in range(10000):
            x += i * i
        time.sleep(0.0002)  # brief release, like a torch op boundary


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--workers", type=int, default=8)
    ap.add_argument("--n", type=int, default=30)
    ap.add_argument("--size", type=int, default=1024)
    ap.add_argument("--gil-load", action="store_true",
                    help="run a GIL-hungry background thread in the "
                         "server process (the busy-model scenario)")
    args = ap.parse_args()
    stop = None
    if args.gil_load:
        import threading
        stop = threading.Event()
        threading.Thread(target=gil_load_thread, args=(stop,),
                         daemon=True).start()
    req_mb = args.size * args.size * 3 / 1e6
    out = {"request_mb": round(req_mb, 2), "workers": args.workers,
           "n_per_worker": args.n}
    for name, fn in (("cpp", bench_cpp), ("python", bench_python)):
        rate, p50 = fn(args.workers, args.n, args.size)
        out[name] = {"msgs_per_s": round(rate, 1),
                     "client_p50_ms": round(p50, 2),
                     "gb_per_s_in": round(rate * req_mb / 1e3, 2)}
        print(f"{name:7s}: {rate:7.1f} msgs/s of {req_mb:.1f} MB "
              f"(p50 {p50:.1f} ms)")
    out["speedup"] = round(out["cpp"]["msgs_per_s"] /
                           out["python"]["msgs_per_s"], 2)
    out["gil_load"] = bool(args.gil_load)
    if stop is not None:
        stop.set()
    print(json.dumps(out))


if __name__ == "__main__":
    main()

"""Native C++ gRPC ingress (csrc/ingress.cpp) against real grpc-python
clients: HTTP/2 framing, HPACK (the client side Huffman-encodes), flow
control with multi-MB messages both ways, grpc-status trailers, and the
full ModelServer Predict path with grpc_ingress="cpp".

Reference parity target: TF-Serving's C++ grpc ModelServer front-end
(SURVEY.md N11 / §2.2 S2).
"""

import threading

import numpy as np
import pytest

from maskrcnn_amd.serving.wireproto import (ModelSpec, PredictRequest,
                                            PredictResponse,
                                            numpy_to_tensor)

pytest.importorskip("grpc")
ingress = pytest.importorskip("maskrcnn_amd._ingress")


def _raw_stub(port, method="/tensorflow.serving.PredictionService/Predict"):
    import grpc
    ch = grpc.insecure_channel(
        f"127.0.0.1:{port}",
        options=[("grpc.max_receive_message_length", 256 << 20),
                 ("grpc.max_send_message_length", 256 << 20)])
    return ch, ch.unary_unary(method, request_serializer=lambda b: b,
                              response_deserializer=lambda b: b)


def test_ingress_echo_small_and_large():
    def handler(path, payload):
        return (0, "", bytes(payload))

    srv = ingress.IngressServer(0, handler, 4)
    srv.start()
    try:
        ch, stub = _raw_stub(srv.port)
        assert stub(b"hello", timeout=10) == b"hello"
        # 8 MB each way: exceeds the 64 KB default stream window many
        # times over -> exercises WINDOW_UPDATE handling both ways.
        big = bytes(range(256)) * (8 * 1024 * 1024 // 256)
        assert stub(big, timeout=30) == big
        ch.close()
    finally:
        srv.stop()


def test_ingress_error_status_and_unknown_method():
    import grpc

    def handler(path, payload):
        if path.endswith("/Fail"):
            return (5, "model not here", b"")
        return (0, "", b"ok")

    srv = ingress.IngressServer(0, handler, 2)
    srv.start()
    try:
        ch, stub = _raw_stub(srv.port, "/tensorflow.serving.Svc/Fail")
        with pytest.raises(grpc.RpcError) as e:
            stub(b"x", timeout=10)
        assert e.value.code() == grpc.StatusCode.NOT_FOUND
        assert "model not here" in e.value.details()
        ch.close()
    finally:
        srv.stop()


def test_ingress_concurrent_streams():
    """Several threads on SEPARATE channels plus several on one shared
    channel (h2 stream multiplexing on a single connection)."""
    def handler(path, payload):
        return (0, "", bytes(payload[::-1]))

    srv = ingress.IngressServer(0, handler, 8)
    srv.start()
    errs = []
    try:
        shared_ch, shared_stub = _raw_stub(srv.port)

        def worker(i, stub):
            try:
                msg = bytes([i]) * (64 * 1024 + i)
                for _ in range(5):
                    assert stub(msg, timeout=30) == msg[::-1]
            except Exception as e:  # noqa: BLE001
                errs.append(e)

        threads = [threading.Thread(target=worker, args=(i, shared_stub))
                   for i in range(4)]
        own = []
        for i in range(4, 8):
            ch, stub = _raw_stub(srv.port)
            own.append(ch)
            threads.append(threading.Thread(target=worker,
                                            args=(i, stub)))
        for t in threads:
            t.start()
        for t in threads:
            t.join()
        assert not errs, errs
        shared_ch.close()
        for ch in own:
            ch.close()
    finally:
        srv.stop()


# ---------------------------------------------------------------- server
@pytest.fixture(scope="module")
def cpp_served_model(tmp_path_factory):
    from maskrcnn_amd.model.mask_rcnn import MaskRCNN
    from maskrcnn_amd.serving.savedmodel import export_saved_model
    from maskrcnn_amd.serving.server import ModelServer
    from maskrcnn_amd.data.synthetic import ShapesConfig

    class SmallConfig(ShapesConfig):
        IMAGES_PER_GPU = 1
        PRE_NMS_LIMIT = 500
        POST_NMS_ROIS_INFERENCE = 50
        DETECTION_MIN_CONFIDENCE = 0.0

    cfg = SmallConfig()
    tmp = tmp_path_factory.mktemp("serving_cpp")
    m = MaskRCNN("inference", cfg, model_dir=str(tmp))
    base = str(tmp / "models" / "mask_rcnn")
    export_saved_model(m, cfg, base, version=1)
    server = ModelServer("mask_rcnn", base, grpc_port=0, rest_port=None,
                         max_batch_size=4, batch_timeout_micros=1000,
                         poll_interval_s=0.5, config=cfg,
                         grpc_ingress="cpp")
    server.start()
    yield server, cfg, m
    server.stop()


def test_cpp_ingress_predict_matches_local(cpp_served_model):
    server, cfg, m = cpp_served_model
    from maskrcnn_amd.serving.client import GrpcServingClient
    client = GrpcServingClient(f"localhost:{server.grpc_port}", cfg)
    img = (np.random.RandomState(3).rand(96, 120, 3) * 255).astype(np.uint8)
    r = client.detect([img])[0]
    assert r["masks"].shape[:2] == img.shape[:2]
    local = m.detect([img])[0]
    np.testing.assert_allclose(r["rois"], local["rois"])
    np.testing.assert_array_equal(r["class_ids"], local["class_ids"])


def test_cpp_ingress_not_found_and_filter(cpp_served_model):
    server, cfg, _ = cpp_served_model
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

    bad = PredictRequest(model_spec=ModelSpec(name="nope"))
    bad.inputs["input_image"] = req.inputs["input_image"]
    bad.inputs["input_image_meta"] = req.inputs["input_image_meta"]
    with pytest.raises(grpc.RpcError) as e:
        stub(bad, timeout=30)
    assert e.value.code() == grpc.StatusCode.NOT_FOUND
    channel.close()


def test_cpp_ingress_metadata_and_status(cpp_served_model):
    server, cfg, _ = cpp_served_model
    from maskrcnn_amd.serving.client import ModelServiceClient
    from maskrcnn_amd.serving.wireproto import STATE_AVAILABLE
    c = ModelServiceClient(f"localhost:{server.grpc_port}")
    states = c.get_model_status("mask_rcnn")
    assert any(s["version"] == 1 and s["state"] == STATE_AVAILABLE
               for s in states)

    import grpc
    from maskrcnn_amd.serving.wireproto import (GetModelMetadataRequest,
                                                GetModelMetadataResponse)
    ch = grpc.insecure_channel(f"localhost:{server.grpc_port}")
    stub = ch.unary_unary(
        "/tensorflow.serving.PredictionService/GetModelMetadata",
        request_serializer=lambda r: r.encode(),
        response_deserializer=GetModelMetadataResponse.decode)
    resp = stub(GetModelMetadataRequest(
        model_spec=ModelSpec(name="mask_rcnn"),
        metadata_field=["signature_def"]), timeout=30)
    assert "serving_default" in resp.signature_defs
    ch.close()


# ------------------------------------------------------------ robustness
def _h2_frame(ftype, flags, sid, payload=b""):
    import struct
    n = len(payload)
    return struct.pack(">BHBBI", n >> 16, n & 0xffff, ftype, flags,
                       sid) + payload


def test_ingress_survives_garbage_and_truncation():
    """Malformed input must only kill its own connection: bad preface,
    random bytes, truncated frames, HPACK garbage — then a VALID
    request on a fresh channel still succeeds."""
    import socket

    def handler(path, payload):
        return (0, "", b"alive")

    srv = ingress.IngressServer(0, handler, 2)
    srv.start()
    try:
        attacks = [
            b"GET / HTTP/1.1\r\n\r\n",                    # wrong preface
            b"PRI * HTTP/2.0\r\n\r\nSM\r\n\r\n" + b"\xff" * 64,
            b"PRI * HTTP/2.0\r\n\r\nSM\r\n\r\n" +
            _h2_frame(0x4, 0, 0) +                         # settings
            _h2_frame(0x1, 0x4, 1, b"\xff\xfe\xfd"),       # HPACK junk
            b"PRI * HTTP/2.0\r\n\r\nSM\r\n\r\n" +
            _h2_frame(0x1, 0x4, 1, b"\x88")[:11],          # truncated
            b"PRI * HTTP/2.0\r\n\r\nSM\r\n\r\n" +
            b"\xff\xff\xff\x00\x00\x00\x00\x00\x01",       # 16MB+ len
        ]
        for a in attacks:
            s = socket.create_connection(("127.0.0.1", srv.port),
                                         timeout=5)
            s.sendall(a)
            s.settimeout(2)
            try:
                while s.recv(4096):
                    pass
            except socket.timeout:
                pass
            s.close()
        # server still alive and correct
        ch, stub = _raw_stub(srv.port)
        assert stub(b"ping", timeout=10) == b"alive"
        ch.close()
    finally:
        srv.stop()


def test_ingress_handler_exception_maps_to_internal():
    import grpc

    def handler(path, payload):
        raise RuntimeError("kaboom")

    srv = ingress.IngressServer(0, handler, 2)
    srv.start()
    try:
        ch, stub = _raw_stub(srv.port)
        with pytest.raises(grpc.RpcError) as e:
            stub(b"x", timeout=10)
        assert e.value.code() == grpc.StatusCode.INTERNAL
        assert "kaboom" in e.value.details()
        # still serving afterwards
        with pytest.raises(grpc.RpcError):
            stub(b"y", timeout=10)
        ch.close()
    finally:
        srv.stop()


def test_ingress_stop_drains_in_flight():
    """stop() lets queued/in-flight requests finish and flush their
    responses before tearing connections down (graceful shutdown)."""
    import time

    def handler(path, payload):
        time.sleep(0.3)
        return (0, "", b"drained")

    srv = ingress.IngressServer(0, handler, 2)
    srv.start()
    ch, stub = _raw_stub(srv.port)
    result = {}

    def call():
        try:
            result["resp"] = stub(b"x", timeout=10)
        except Exception as e:  # noqa: BLE001
            result["err"] = e

    t = threading.Thread(target=call)
    t.start()
    time.sleep(0.1)  # request is in flight inside the handler
    srv.stop(drain_s=5.0)
    t.join(timeout=10)
    ch.close()
    assert result.get("resp") == b"drained", result


def test_ingress_random_frame_fuzz():
    """Random frame soup after a valid preface must never take the
    server down (property-style fuzz of the hand-rolled RFC 7540
    parser). Liveness is probed after every burst by completing a
    fresh preface+SETTINGS handshake."""
    import random
    import socket
    import struct

    def handler(path, payload):
        return (0, "", b"up")

    srv = ingress.IngressServer(0, handler, 2)
    srv.start()
    rng = random.Random(0xF00D)
    preface = b"PRI * HTTP/2.0\r\n\r\nSM\r\n\r\n"
    try:
        for _ in range(40):
            s = socket.create_connection(("127.0.0.1", srv.port),
                                         timeout=5)
            s.sendall(preface)
            burst = b""
            for _ in range(rng.randint(1, 6)):
                ftype = rng.randint(0, 12)
                flags = rng.randint(0, 255)
                sid = rng.randint(0, 5)
                n = rng.randint(0, 64)
                payload = bytes(rng.getrandbits(8) for _ in range(n))
                burst += struct.pack(
                    ">BHBBI", n >> 16, n & 0xffff, ftype, flags,
                    sid) + payload
            # Sometimes truncate mid-frame.
            if rng.random() < 0.3:
                burst = burst[:rng.randint(1, max(2, len(burst)))]
            try:
                s.sendall(burst)
            except OSError:
                pass  # server already RSTed the connection - fine
            s.close()
            # liveness probe: handshake completes => accept loop and a
            # fresh conn thread still work
            p = socket.create_connection(("127.0.0.1", srv.port),
                                         timeout=5)
            p.sendall(preface + struct.pack(">BHBBI", 0, 0, 0x4, 0, 0))
            p.settimeout(5)
            hdr = p.recv(9)
            assert len(hdr) == 9 and hdr[3] == 0x4, hdr  # SETTINGS back
            p.close()
        # and a real request still round-trips
        ch, stub = _raw_stub(srv.port)
        assert stub(b"x", timeout=10) == b"up"
        ch.close()
    finally:
        srv.stop()


def test_ingress_request_trailers_half_close():
    """A client that half-closes via trailing HEADERS (instead of
    END_STREAM on DATA) still gets served, and the trailer block keeps
    the connection's HPACK table in sync."""
    import socket
    import struct

    def handler(path, payload):
        return (0, "", b"trailer-ok:" + bytes(payload))

    srv = ingress.IngressServer(0, handler, 2)
    srv.start()
    try:
        s = socket.create_connection(("127.0.0.1", srv.port), timeout=5)
        s.sendall(b"PRI * HTTP/2.0\r\n\r\nSM\r\n\r\n" +
                  struct.pack(">BHBBI", 0, 0, 0x4, 0, 0))  # SETTINGS

        def lit(name, value):
            return (b"\x00" + bytes([len(name)]) + name +
                    bytes([len(value)]) + value)

        hdrs = (b"\x83"  # :method POST (static idx 3)
                + lit(b":path", b"/svc/M") + lit(b":scheme", b"http")
                + lit(b"content-type", b"application/grpc"))
        s.sendall(struct.pack(">BHBBI", 0, len(hdrs), 0x1, 0x4, 1)
                  + hdrs)                       # HEADERS, END_HEADERS
        body = b"\x00" + struct.pack(">I", 3) + b"abc"
        s.sendall(struct.pack(">BHBBI", 0, len(body), 0x0, 0x0, 1)
                  + body)                       # DATA, no END_STREAM
        trl = lit(b"grpc-client-note", b"bye")
        s.sendall(struct.pack(">BHBBI", 0, len(trl), 0x1, 0x5, 1)
                  + trl)                        # trailers: EH+ES
        # Read frames until the response DATA arrives.
        s.settimeout(5)
        buf = b""
        deadline = 50
        payload = None
        while deadline and payload is None:
            deadline -= 1
            chunk = s.recv(65536)
            if not chunk:
                break
            buf += chunk
            while len(buf) >= 9:
                n = (buf[0] << 16) | (buf[1] << 8) | buf[2]
                if len(buf) < 9 + n:
                    break
                ftype = buf[3]
                frame = buf[9:9 + n]
                buf = buf[9 + n:]
                if ftype == 0x0:  # DATA
                    payload = frame[5:]
        assert payload == b"trailer-ok:abc", payload
        s.close()
    finally:
        srv.stop()


def test_ingress_randomized_sizes_stress():
    """50 echo calls of random sizes 0..512 KB on one channel — framing
    and flow control must be exact at every size (incl. 0 bytes)."""
    import random

    def handler(path, payload):
        return (0, "", bytes(payload))

    srv = ingress.IngressServer(0, handler, 4)
    srv.start()
    try:
        ch, stub = _raw_stub(srv.port)
        rng = random.Random(42)
        for i in range(50):
            n = rng.choice([0, 1, 5, rng.randint(0, 512 * 1024)])
            msg = bytes(rng.getrandbits(8) for _ in range(min(n, 256))) \
                * (n // min(n, 256) if n else 0) + b"x" * (
                    n % min(n, 256) if n else 0)
            msg = msg[:n]
            assert stub(msg, timeout=15) == msg, f"size {n} (call {i})"
        ch.close()
    finally:
        srv.stop()


def test_ingress_long_error_message():
    """grpc-message values over 127 bytes require HPACK's multi-byte
    7-bit-prefix length (a one-byte length here corrupted the header
    block)."""
    import grpc
    long_msg = "x" * 300 + " END"

    def handler(path, payload):
        return (13, long_msg, b"")

    srv = ingress.IngressServer(0, handler, 2)
    srv.start()
    try:
        ch, stub = _raw_stub(srv.port)
        with pytest.raises(grpc.RpcError) as e:
            stub(b"q", timeout=10)
        assert e.value.code() == grpc.StatusCode.INTERNAL
        assert e.value.details().endswith(" END")
        assert len(e.value.details()) >= 300
        ch.close()
    finally:
        srv.stop()


def test_ingress_grpc_length_overflow():
    """A crafted gRPC length prefix near 2^32 must be rejected as
    truncated (uint64 bounds math), not wrap and read out of bounds."""
    import socket
    import struct

    def handler(path, payload):
        return (0, "", b"ok")

    srv = ingress.IngressServer(0, handler, 2)
    srv.start()
    try:
        s = socket.create_connection(("127.0.0.1", srv.port), timeout=5)
        s.sendall(b"PRI * HTTP/2.0\r\n\r\nSM\r\n\r\n" +
                  struct.pack(">BHBBI", 0, 0, 0x4, 0, 0))
        hdrs = b"\x83" + (b"\x00\x05:path\x02/a")
        s.sendall(struct.pack(">BHBBI", 0, len(hdrs), 0x1, 0x4, 1)
                  + hdrs)
        body = b"\x00" + struct.pack(">I", 0xFFFFFFFE) + b"abc"
        s.sendall(struct.pack(">BHBBI", 0, len(body), 0x0, 0x1, 1)
                  + body)                 # DATA + END_STREAM
        s.settimeout(5)
        # Expect a trailers frame carrying grpc-status 13 (the literal
        # "13" appears in the plain HPACK literal encoding) and no crash.
        buf = b""
        for _ in range(20):
            try:
                chunk = s.recv(65536)
            except OSError:
                break
            if not chunk:
                break
            buf += chunk
            if b"grpc-status" in buf:
                break
        assert b"grpc-status" in buf
        s.close()
        # server still alive
        ch, stub = _raw_stub(srv.port)
        assert stub(b"x", timeout=10) == b"ok"
        ch.close()
    finally:
        srv.stop()


def test_cpp_ingress_version_hot_reload(cpp_served_model, tmp_path):
    """A new version dropped on disk is picked up by the poller and
    served through the cpp ingress without restarting it."""
    import time
    server, cfg, m = cpp_served_model
    from maskrcnn_amd.serving.savedmodel import export_saved_model
    from maskrcnn_amd.serving.client import PREDICT_METHOD
    import grpc

    start_version = server.state.version
    export_saved_model(m, cfg, server.state.base_path,
                       version=start_version + 1)
    deadline = time.time() + 60
    while time.time() < deadline and \
            server.state.version != start_version + 1:
        time.sleep(0.25)
    assert server.state.version == start_version + 1

    ch = grpc.insecure_channel(f"localhost:{server.grpc_port}")
    stub = ch.unary_unary(
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
    assert resp.model_spec.version == start_version + 1
    ch.close()


def test_ingress_expired_deadline_skips_handler():
    """grpc-timeout honored: a request whose client deadline passed
    while queued gets DEADLINE_EXCEEDED WITHOUT entering the handler
    (no wasted model batch). One worker; first request occupies it past
    the second's 200 ms deadline."""
    import time
    calls = []

    def handler(path, payload):
        calls.append(bytes(payload))
        time.sleep(1.0)
        return (0, "", b"done")

    srv = ingress.IngressServer(0, handler, 1)  # single worker
    srv.start()
    try:
        ch1, stub1 = _raw_stub(srv.port)
        ch2, stub2 = _raw_stub(srv.port)
        res = {}

        def slow():
            res["slow"] = stub1(b"slow", timeout=30)

        t = threading.Thread(target=slow)
        t.start()
        # Wait until 'slow' has actually ENTERED the handler (a fixed
        # sleep raced thread scheduling on loaded boxes).
        deadline = time.time() + 20
        while time.time() < deadline and not calls:
            time.sleep(0.01)
        assert calls == [b"slow"]
        import grpc
        with pytest.raises(grpc.RpcError):
            stub2(b"fast", timeout=0.2)  # expires while queued
        t.join(timeout=30)
        assert res.get("slow") == b"done"
        time.sleep(0.5)  # give the worker time to (not) run 'fast'
        assert calls == [b"slow"], calls
        ch1.close()
        ch2.close()
    finally:
        srv.stop()


def test_serving_client_worker_against_cpp_ingress(cpp_served_model):
    """The real measurement client (bench/serving_client_worker.py)
    round-trips against the cpp ingress: uint8 + mask_top wire path,
    zero errors."""
    import json
    import os
    import subprocess
    import sys

    server, cfg, _ = cpp_served_model
    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    r = subprocess.run(
        [sys.executable, "bench/serving_client_worker.py",
         "--target", f"127.0.0.1:{server.grpc_port}",
         "--requests", "6", "--threads", "2",
         "--image-size", str(cfg.IMAGE_SHAPE[0]),
         "--num-classes", str(cfg.NUM_CLASSES)],
        capture_output=True, text=True, timeout=300, cwd=root)
    assert r.returncode == 0, r.stderr[-2000:]
    rec = json.loads(r.stdout.strip().splitlines()[-1])
    assert rec["n_errors"] == 0, rec
    assert len(rec["lats"]) == 6


def test_ingress_stream_flood_refused():
    """A client opening far more streams than advertised
    MAX_CONCURRENT_STREAMS gets REFUSED_STREAM RSTs (bounded partial-
    body state) while the connection and its HPACK table stay healthy."""
    import socket
    import struct

    def handler(path, payload):
        return (0, "", b"ok")

    srv = ingress.IngressServer(0, handler, 2)
    srv.start()
    try:
        s = socket.create_connection(("127.0.0.1", srv.port), timeout=5)
        s.sendall(b"PRI * HTTP/2.0\r\n\r\nSM\r\n\r\n" +
                  struct.pack(">BHBBI", 0, 0, 0x4, 0, 0))

        def lit(name, value):
            return (b"\x00" + bytes([len(name)]) + name +
                    bytes([len(value)]) + value)

        hdrs = b"\x83" + lit(b":path", b"/x/Y")
        # open 400 streams with headers but no END_STREAM (partial).
        for i in range(400):
            sid = 1 + 2 * i
            s.sendall(struct.pack(">BHBBI", 0, len(hdrs), 0x1, 0x4,
                                  sid) + hdrs)
        s.settimeout(5)
        buf = b""
        rsts = 0
        try:
            while rsts == 0:
                chunk = s.recv(65536)
                if not chunk:
                    break
                buf += chunk
                # count RST_STREAM (type 3) frames
                i = 0
                rsts = 0
                while i + 9 <= len(buf):
                    n = (buf[i] << 16) | (buf[i+1] << 8) | buf[i+2]
                    if buf[i+3] == 0x3:
                        rsts += 1
                    i += 9 + n
        except socket.timeout:
            pass
        assert rsts > 0, "no REFUSED_STREAM seen"
        # the HPACK table stayed consistent: a fresh request on stream
        # 801 (after the flood) with END_STREAM completes normally.
        s.sendall(struct.pack(">BHBBI", 0, len(hdrs), 0x1, 0x5, 801)
                  + hdrs)
        got_headers = False
        try:
            deadline = 20
            while deadline and not got_headers:
                deadline -= 1
                chunk = s.recv(65536)
                if not chunk:
                    break
                buf += chunk
                got_headers = b"grpc-status" in buf
        except socket.timeout:
            pass
        # Stream 801 was over the cap too? 400 partial streams remain
        # open, so it may ALSO be refused — either way the connection
        # answered (RST or response) without dying. Send on a NEW
        # connection to prove overall health:
        s.close()
        ch, stub = _raw_stub(srv.port)
        assert stub(b"x", timeout=10) == b"ok"
        ch.close()
    finally:
        srv.stop()

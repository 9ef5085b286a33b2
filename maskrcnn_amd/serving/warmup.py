"""TF-Serving model-warmup file support.

TF-Serving reads `<version>/assets.extra/tf_serving_warmup_requests`
— a TFRecord file of PredictionLog protos — and replays the requests
at model-load time so the first real request never pays cold-start
costs (reference S2; here that cost is MIOpen Find + MFMA kernel
first-launch, which the committed find-db already shrinks).

This module implements both sides without TensorFlow:
- TFRecord framing: {u64le length, u32le masked-crc32c(length),
  payload, u32le masked-crc32c(payload)} per record.
- PredictionLog wire decode/encode (prediction_log.proto field
  numbers: log_metadata=1, predict_log=6{request=1, response=2}).

The crc32c is the Castagnoli polynomial with TF's mask:
masked = ((crc >> 15 | crc << 17) + 0xa282ead8) & 0xffffffff.
"""

from __future__ import annotations

import os
import struct
from typing import List

from .wireproto import PredictRequest, _w_len_field

_CRC_TABLE = []


def _crc32c_table():
    global _CRC_TABLE
    if _CRC_TABLE:
        return _CRC_TABLE
    poly = 0x82F63B78  # Castagnoli, reflected
    tbl = []
    for n in range(256):
        c = n
        for _ in range(8):
            c = (c >> 1) ^ poly if c & 1 else c >> 1
        tbl.append(c)
    _CRC_TABLE = tbl
    return tbl


def crc32c(data: bytes) -> int:
    tbl = _crc32c_table()
    crc = 0xFFFFFFFF
    for b in data:
        crc = tbl[(crc ^ b) & 0xFF] ^ (crc >> 8)
    return crc ^ 0xFFFFFFFF


def _masked_crc(data: bytes) -> int:
    crc = crc32c(data)
    return (((crc >> 15) | (crc << 17)) + 0xA282EAD8) & 0xFFFFFFFF


def tfrecord_write(path: str, records: List[bytes]) -> None:
    with open(path, "wb") as f:
        for rec in records:
            ln = struct.pack("<Q", len(rec))
            f.write(ln)
            f.write(struct.pack("<I", _masked_crc(ln)))
            f.write(rec)
            f.write(struct.pack("<I", _masked_crc(rec)))


def tfrecord_read(path: str) -> List[bytes]:
    out = []
    with open(path, "rb") as f:
        while True:
            hdr = f.read(12)
            if len(hdr) < 12:
                break
            (n,) = struct.unpack("<Q", hdr[:8])
            (lcrc,) = struct.unpack("<I", hdr[8:12])
            if _masked_crc(hdr[:8]) != lcrc:
                raise ValueError("TFRecord length crc mismatch")
            rec = f.read(n)
            (dcrc,) = struct.unpack("<I", f.read(4))
            if _masked_crc(rec) != dcrc:
                raise ValueError("TFRecord data crc mismatch")
            out.append(rec)
    return out


def encode_prediction_log(request: PredictRequest) -> bytes:
    """PredictionLog{predict_log: PredictLog{request}} bytes."""
    predict_log = bytearray()
    _w_len_field(predict_log, 1, request.encode())
    out = bytearray()
    _w_len_field(out, 6, bytes(predict_log))
    return bytes(out)


def decode_prediction_log(buf: bytes) -> PredictRequest:
    """Extract the PredictRequest from a PredictionLog record (only
    predict_log entries are supported — classify/regress logs raise)."""
    from .wireproto import _r_field
    pos = 0
    while pos < len(buf):
        fieldno, wt, val, pos = _r_field(buf, pos)
        if fieldno == 6 and wt == 2:  # predict_log
            ppos = 0
            while ppos < len(val):
                f2, w2, v2, ppos = _r_field(val, ppos)
                if f2 == 1 and w2 == 2:  # request
                    return PredictRequest.decode(bytes(v2))
        elif fieldno in (2, 3, 4, 5) and wt == 2:
            raise ValueError(
                f"unsupported warmup log type (field {fieldno}); only "
                f"predict_log is servable here")
    raise ValueError("PredictionLog carries no predict_log.request")


WARMUP_RELPATH = os.path.join("assets.extra",
                              "tf_serving_warmup_requests")


def write_warmup_requests(version_dir: str,
                          requests: List[PredictRequest]) -> str:
    d = os.path.join(version_dir, "assets.extra")
    os.makedirs(d, exist_ok=True)
    path = os.path.join(version_dir, WARMUP_RELPATH)
    tfrecord_write(path,
                   [encode_prediction_log(r) for r in requests])
    return path


def read_warmup_requests(version_dir: str) -> List[PredictRequest]:
    path = os.path.join(version_dir, WARMUP_RELPATH)
    if not os.path.isfile(path):
        return []
    return [decode_prediction_log(rec) for rec in tfrecord_read(path)]

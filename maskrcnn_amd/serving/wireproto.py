"""Minimal protobuf wire codec for the TF-Serving Predict protocol.

grpcio-tools is unavailable offline, so instead of generated stubs this
module speaks the protobuf WIRE FORMAT directly, with field numbers taken
from the upstream .proto files (tensorflow/core/framework/tensor.proto,
tensor_shape.proto, types.proto; tensorflow_serving/apis/{model,predict}
.proto; tensorflow/core/protobuf/{meta_graph,saved_model}.proto) so that
stock TF-Serving clients interoperate byte-for-byte.

Only the fields the Predict path uses are implemented; unknown fields are
skipped on decode (forward-compatible).
"""

from __future__ import annotations

import struct
from dataclasses import dataclass, field
from typing import Dict, List, Optional

import numpy as np

# ---------------------------------------------------------------- wire ops

_WT_VARINT = 0
_WT_I64 = 1
_WT_LEN = 2
_WT_I32 = 5


def _w_varint(out: bytearray, v: int):
    if v < 0:
        v &= (1 << 64) - 1
    while True:
        b = v & 0x7F
        v >>= 7
        if v:
            out.append(b | 0x80)
        else:
            out.append(b)
            return


def _w_tag(out: bytearray, fieldno: int, wt: int):
    _w_varint(out, (fieldno << 3) | wt)


def _w_len_field(out: bytearray, fieldno: int, payload: bytes):
    _w_tag(out, fieldno, _WT_LEN)
    _w_varint(out, len(payload))
    out.extend(payload)


def _w_varint_field(out: bytearray, fieldno: int, v: int):
    if v:
        _w_tag(out, fieldno, _WT_VARINT)
        _w_varint(out, v)


def _w_string_field(out: bytearray, fieldno: int, s: str):
    if s:
        _w_len_field(out, fieldno, s.encode("utf-8"))


def _r_varint(buf: bytes, pos: int):
    result = 0
    shift = 0
    while True:
        b = buf[pos]
        pos += 1
        result |= (b & 0x7F) << shift
        if not (b & 0x80):
            return result, pos
        shift += 7


def _utf8(v) -> str:
    return (v.tobytes() if isinstance(v, memoryview) else v) \
        .decode("utf-8")


def _r_field(buf: bytes, pos: int):
    """Returns (fieldno, wiretype, value_or_bytes, new_pos).

    LEN values are returned as ZERO-COPY memoryview slices when `buf`
    is a memoryview (public decode entry points wrap once): the old
    bytes-slicing copied every nested submessage — ~48 MB of copies and
    97% of the decode time for a 16 MB Predict message."""
    if isinstance(buf, (bytes, bytearray)):
        buf = memoryview(buf)
    key, pos = _r_varint(buf, pos)
    fieldno, wt = key >> 3, key & 7
    if wt == _WT_VARINT:
        v, pos = _r_varint(buf, pos)
        return fieldno, wt, v, pos
    if wt == _WT_I64:
        return fieldno, wt, buf[pos:pos + 8], pos + 8
    if wt == _WT_LEN:
        n, pos = _r_varint(buf, pos)
        return fieldno, wt, buf[pos:pos + n], pos + n
    if wt == _WT_I32:
        return fieldno, wt, buf[pos:pos + 4], pos + 4
    raise ValueError(f"unsupported wire type {wt}")


def _zigzag_i64(v: int) -> int:
    # NOTE: despite the historical name this is two's-complement masking
    # (protobuf int32/int64 semantics), not sint zigzag — TensorProto uses
    # int32/int64 fields.
    return v if v >= 0 else v + (1 << 64)


# --------------------------------------------------------------- DataType

DT_FLOAT = 1
DT_DOUBLE = 2
DT_INT32 = 3
DT_UINT8 = 4
DT_INT16 = 5
DT_INT8 = 6
DT_STRING = 7
DT_INT64 = 9
DT_BOOL = 10
DT_BFLOAT16 = 14
DT_HALF = 19

_NP_TO_DT = {
    np.dtype(np.float32): DT_FLOAT,
    np.dtype(np.float64): DT_DOUBLE,
    np.dtype(np.int32): DT_INT32,
    np.dtype(np.uint8): DT_UINT8,
    np.dtype(np.int16): DT_INT16,
    np.dtype(np.int8): DT_INT8,
    np.dtype(np.int64): DT_INT64,
    np.dtype(np.bool_): DT_BOOL,
    np.dtype(np.float16): DT_HALF,
}
_DT_TO_NP = {v: k for k, v in _NP_TO_DT.items()}


# ----------------------------------------------------------- TensorProto

@dataclass
class TensorProto:
    """tensorflow.TensorProto (tensor.proto).

    Fields: 1 dtype, 2 tensor_shape, 4 tensor_content, 5 float_val,
    6 double_val, 7 int_val, 10 int64_val, 11 bool_val.
    """
    dtype: int = 0
    shape: List[int] = field(default_factory=list)
    tensor_content: bytes = b""
    float_val: List[float] = field(default_factory=list)
    double_val: List[float] = field(default_factory=list)
    int_val: List[int] = field(default_factory=list)
    int64_val: List[int] = field(default_factory=list)
    bool_val: List[bool] = field(default_factory=list)

    def _shape_bytes(self) -> bytes:
        out = bytearray()
        for d in self.shape:
            dim = bytearray()
            _w_tag(dim, 1, _WT_VARINT)        # Dim.size
            _w_varint(dim, _zigzag_i64(d))
            _w_len_field(out, 2, bytes(dim))  # TensorShapeProto.dim
        return bytes(out)

    def encode_chunks(self):
        """Zero-copy-ish encoding: returns a list of bytes chunks whose
        concatenation is encode(). Large tensor_content is REFERENCED,
        not copied — the only big copy happens at the caller's final
        b"".join (a 33 MB Predict response dropped 100 ms -> ~8 ms)."""
        head = bytearray()
        _w_varint_field(head, 1, self.dtype)
        _w_len_field(head, 2, self._shape_bytes())
        chunks = [bytes(head)]
        if self.tensor_content:
            tag = bytearray()
            _w_tag(tag, 4, _WT_LEN)
            _w_varint(tag, len(self.tensor_content))
            chunks.append(bytes(tag))
            chunks.append(bytes(self.tensor_content)
                          if not isinstance(self.tensor_content, bytes)
                          else self.tensor_content)
        if (self.float_val or self.double_val or self.int_val or
                self.int64_val or self.bool_val):
            tail = bytearray()
            self._encode_small_fields(tail)
            chunks.append(bytes(tail))
        return chunks

    def _encode_small_fields(self, out: bytearray) -> None:
        if self.float_val:
            _w_len_field(out, 5,
                         struct.pack(f"<{len(self.float_val)}f",
                                     *self.float_val))
        if self.double_val:
            _w_len_field(out, 6,
                         struct.pack(f"<{len(self.double_val)}d",
                                     *self.double_val))
        if self.int_val:
            packed = bytearray()
            for v in self.int_val:
                _w_varint(packed, _zigzag_i64(v))
            _w_len_field(out, 7, bytes(packed))
        if self.int64_val:
            packed = bytearray()
            for v in self.int64_val:
                _w_varint(packed, _zigzag_i64(v))
            _w_len_field(out, 10, bytes(packed))
        if self.bool_val:
            packed = bytearray()
            for v in self.bool_val:
                _w_varint(packed, 1 if v else 0)
            _w_len_field(out, 11, bytes(packed))

    def encode(self) -> bytes:
        out = bytearray()
        _w_varint_field(out, 1, self.dtype)
        _w_len_field(out, 2, self._shape_bytes())
        if self.tensor_content:
            _w_len_field(out, 4, self.tensor_content)
        if self.float_val:
            _w_len_field(out, 5,
                         struct.pack(f"<{len(self.float_val)}f",
                                     *self.float_val))
        if self.double_val:
            _w_len_field(out, 6,
                         struct.pack(f"<{len(self.double_val)}d",
                                     *self.double_val))
        if self.int_val:
            packed = bytearray()
            for v in self.int_val:
                _w_varint(packed, _zigzag_i64(v))
            _w_len_field(out, 7, bytes(packed))
        if self.int64_val:
            packed = bytearray()
            for v in self.int64_val:
                _w_varint(packed, _zigzag_i64(v))
            _w_len_field(out, 10, bytes(packed))
        if self.bool_val:
            packed = bytearray()
            for v in self.bool_val:
                _w_varint(packed, 1 if v else 0)
            _w_len_field(out, 11, bytes(packed))
        return bytes(out)

    @classmethod
    def decode(cls, buf: bytes) -> "TensorProto":
        t = cls()
        pos = 0
        while pos < len(buf):
            fno, wt, v, pos = _r_field(buf, pos)
            if fno == 1 and wt == _WT_VARINT:
                t.dtype = v
            elif fno == 2 and wt == _WT_LEN:
                t.shape = _decode_shape(v)
            elif fno == 4 and wt == _WT_LEN:
                t.tensor_content = v
            elif fno == 5 and wt == _WT_LEN:
                t.float_val = list(
                    struct.unpack(f"<{len(v) // 4}f", v))
            elif fno == 5 and wt == _WT_I32:
                t.float_val.append(struct.unpack("<f", v)[0])
            elif fno == 6 and wt == _WT_LEN:
                t.double_val = list(
                    struct.unpack(f"<{len(v) // 8}d", v))
            elif fno == 7 and wt == _WT_LEN:
                p = 0
                while p < len(v):
                    x, p = _r_varint(v, p)
                    t.int_val.append(_unzig32(x))
            elif fno == 7 and wt == _WT_VARINT:
                t.int_val.append(_unzig32(v))
            elif fno == 10 and wt == _WT_LEN:
                p = 0
                while p < len(v):
                    x, p = _r_varint(v, p)
                    t.int64_val.append(_unzig64(x))
            elif fno == 10 and wt == _WT_VARINT:
                t.int64_val.append(_unzig64(v))
            elif fno == 11 and wt == _WT_LEN:
                t.bool_val = [bool(b & 1) for b in v]
            elif fno == 11 and wt == _WT_VARINT:
                t.bool_val.append(bool(v))
        return t


def _unzig64(v: int) -> int:
    return v - (1 << 64) if v >= (1 << 63) else v


def _unzig32(v: int) -> int:
    # int32 negatives arrive as sign-extended 64-bit varints.
    v = _unzig64(v)
    return v


def _decode_shape(buf: bytes) -> List[int]:
    dims = []
    pos = 0
    while pos < len(buf):
        fno, wt, v, pos = _r_field(buf, pos)
        if fno == 2 and wt == _WT_LEN:  # dim
            p = 0
            size = 0
            while p < len(v):
                f2, w2, v2, p = _r_field(v, p)
                if f2 == 1 and w2 == _WT_VARINT:
                    size = _unzig64(v2)
            dims.append(size)
    return dims


def numpy_to_tensor(arr: np.ndarray) -> TensorProto:
    arr = np.ascontiguousarray(arr)
    dt = _NP_TO_DT.get(arr.dtype)
    if dt is None:
        arr = arr.astype(np.float32)
        dt = DT_FLOAT
    return TensorProto(dtype=dt, shape=list(arr.shape),
                       tensor_content=arr.tobytes())


def tensor_to_numpy(t: TensorProto) -> np.ndarray:
    np_dtype = _DT_TO_NP.get(t.dtype)
    if np_dtype is None:
        raise ValueError(f"unsupported dtype {t.dtype}")
    shape = tuple(t.shape)
    if t.tensor_content:
        return np.frombuffer(t.tensor_content, dtype=np_dtype) \
            .reshape(shape).copy()
    if t.dtype == DT_FLOAT and t.float_val:
        a = np.array(t.float_val, dtype=np.float32)
    elif t.dtype == DT_DOUBLE and t.double_val:
        a = np.array(t.double_val, dtype=np.float64)
    elif t.dtype in (DT_INT32, DT_UINT8, DT_INT16, DT_INT8) and t.int_val:
        a = np.array(t.int_val, dtype=np_dtype)
    elif t.dtype == DT_INT64 and t.int64_val:
        a = np.array(t.int64_val, dtype=np.int64)
    elif t.dtype == DT_BOOL and t.bool_val:
        a = np.array(t.bool_val, dtype=np.bool_)
    else:
        a = np.zeros(int(np.prod(shape)) if shape else 0, dtype=np_dtype)
    if shape and a.size == int(np.prod(shape)):
        a = a.reshape(shape)
    elif shape and a.size == 1:
        a = np.full(shape, a.reshape(-1)[0], dtype=np_dtype)
    return a


# ------------------------------------------------------------- ModelSpec

@dataclass
class ModelSpec:
    """tensorflow.serving.ModelSpec: 1 name, 2 version (Int64Value),
    3 signature_name, 4 version_label."""
    name: str = ""
    version: Optional[int] = None
    signature_name: str = ""
    version_label: str = ""

    def encode(self) -> bytes:
        out = bytearray()
        _w_string_field(out, 1, self.name)
        if self.version is not None:
            inner = bytearray()
            _w_tag(inner, 1, _WT_VARINT)  # Int64Value.value
            _w_varint(inner, _zigzag_i64(self.version))
            _w_len_field(out, 2, bytes(inner))
        _w_string_field(out, 3, self.signature_name)
        _w_string_field(out, 4, self.version_label)
        return bytes(out)

    @classmethod
    def decode(cls, buf: bytes) -> "ModelSpec":
        m = cls()
        pos = 0
        while pos < len(buf):
            fno, wt, v, pos = _r_field(buf, pos)
            if fno == 1 and wt == _WT_LEN:
                m.name = _utf8(v)
            elif fno == 2 and wt == _WT_LEN:
                p = 0
                while p < len(v):
                    f2, w2, v2, p = _r_field(v, p)
                    if f2 == 1 and w2 == _WT_VARINT:
                        m.version = _unzig64(v2)
            elif fno == 3 and wt == _WT_LEN:
                m.signature_name = _utf8(v)
            elif fno == 4 and wt == _WT_LEN:
                m.version_label = _utf8(v)
        return m


def _encode_map_entry(key: str, value_bytes: bytes) -> bytes:
    entry = bytearray()
    _w_string_field(entry, 1, key)
    _w_len_field(entry, 2, value_bytes)
    return bytes(entry)


def _decode_map_entry(buf: bytes):
    key, val = "", b""
    pos = 0
    while pos < len(buf):
        fno, wt, v, pos = _r_field(buf, pos)
        if fno == 1 and wt == _WT_LEN:
            key = _utf8(v)
        elif fno == 2 and wt == _WT_LEN:
            val = v
    return key, val


# --------------------------------------------------- Predict req/response

@dataclass
class PredictRequest:
    """tensorflow.serving.PredictRequest: 1 model_spec,
    2 inputs map<string, TensorProto>, 3 output_filter."""
    model_spec: ModelSpec = field(default_factory=ModelSpec)
    inputs: Dict[str, TensorProto] = field(default_factory=dict)
    output_filter: List[str] = field(default_factory=list)

    def encode(self) -> bytes:
        head = bytearray()
        _w_len_field(head, 1, self.model_spec.encode())
        chunks = [bytes(head)]
        for k, t in self.inputs.items():   # chunked: content not re-copied
            t_chunks = t.encode_chunks()
            t_len = sum(len(c) for c in t_chunks)
            kb = k.encode("utf-8")
            entry_head = bytearray()
            _w_tag(entry_head, 1, _WT_LEN)
            _w_varint(entry_head, len(kb))
            entry_head += kb
            _w_tag(entry_head, 2, _WT_LEN)
            _w_varint(entry_head, t_len)
            outer = bytearray()
            _w_tag(outer, 2, _WT_LEN)
            _w_varint(outer, len(entry_head) + t_len)
            outer += entry_head
            chunks.append(bytes(outer))
            chunks.extend(t_chunks)
        tail = bytearray()
        for f in self.output_filter:
            _w_string_field(tail, 3, f)
        chunks.append(bytes(tail))
        return b"".join(chunks)

    @classmethod
    def decode(cls, buf: bytes) -> "PredictRequest":
        r = cls()
        pos = 0
        while pos < len(buf):
            fno, wt, v, pos = _r_field(buf, pos)
            if fno == 1 and wt == _WT_LEN:
                r.model_spec = ModelSpec.decode(v)
            elif fno == 2 and wt == _WT_LEN:
                k, tv = _decode_map_entry(v)
                r.inputs[k] = TensorProto.decode(tv)
            elif fno == 3 and wt == _WT_LEN:
                r.output_filter.append(_utf8(v))
        return r


@dataclass
class PredictResponse:
    """tensorflow.serving.PredictResponse: 1 outputs map, 2 model_spec."""
    outputs: Dict[str, TensorProto] = field(default_factory=dict)
    model_spec: ModelSpec = field(default_factory=ModelSpec)

    def encode(self) -> bytes:
        # Chunked: the per-output TensorProto payloads (tens of MB for
        # mrcnn_mask) are referenced, never re-copied through the map
        # entry / message nesting; ONE final join materializes.
        chunks = []
        for k, t in self.outputs.items():
            t_chunks = t.encode_chunks()
            t_len = sum(len(c) for c in t_chunks)
            kb = k.encode("utf-8")
            entry_head = bytearray()
            _w_tag(entry_head, 1, _WT_LEN)      # map entry key
            _w_varint(entry_head, len(kb))
            entry_head += kb
            _w_tag(entry_head, 2, _WT_LEN)      # map entry value
            _w_varint(entry_head, t_len)
            entry_len = len(entry_head) + t_len
            outer = bytearray()
            _w_tag(outer, 1, _WT_LEN)           # outputs field
            _w_varint(outer, entry_len)
            outer += entry_head
            chunks.append(bytes(outer))
            chunks.extend(t_chunks)
        tail = bytearray()
        _w_len_field(tail, 2, self.model_spec.encode())
        chunks.append(bytes(tail))
        return b"".join(chunks)

    @classmethod
    def decode(cls, buf: bytes) -> "PredictResponse":
        r = cls()
        pos = 0
        while pos < len(buf):
            fno, wt, v, pos = _r_field(buf, pos)
            if fno == 1 and wt == _WT_LEN:
                k, tv = _decode_map_entry(v)
                r.outputs[k] = TensorProto.decode(tv)
            elif fno == 2 and wt == _WT_LEN:
                r.model_spec = ModelSpec.decode(v)
        return r


# ------------------------------------------- SavedModel / SignatureDef

@dataclass
class TensorInfo:
    """tensorflow.TensorInfo: 1 name, 2 dtype, 3 tensor_shape."""
    name: str = ""
    dtype: int = DT_FLOAT
    shape: List[int] = field(default_factory=list)

    def encode(self) -> bytes:
        out = bytearray()
        _w_string_field(out, 1, self.name)
        _w_varint_field(out, 2, self.dtype)
        tp = TensorProto(shape=self.shape)
        _w_len_field(out, 3, tp._shape_bytes())
        return bytes(out)

    @classmethod
    def decode(cls, buf: bytes) -> "TensorInfo":
        ti = cls()
        pos = 0
        while pos < len(buf):
            fno, wt, v, pos = _r_field(buf, pos)
            if fno == 1 and wt == _WT_LEN:
                ti.name = _utf8(v)
            elif fno == 2 and wt == _WT_VARINT:
                ti.dtype = v
            elif fno == 3 and wt == _WT_LEN:
                ti.shape = _decode_shape(v)
        return ti


@dataclass
class SignatureDef:
    """tensorflow.SignatureDef: 1 inputs map, 2 outputs map, 3 method_name."""
    inputs: Dict[str, TensorInfo] = field(default_factory=dict)
    outputs: Dict[str, TensorInfo] = field(default_factory=dict)
    method_name: str = "tensorflow/serving/predict"

    def encode(self) -> bytes:
        out = bytearray()
        for k, ti in self.inputs.items():
            _w_len_field(out, 1, _encode_map_entry(k, ti.encode()))
        for k, ti in self.outputs.items():
            _w_len_field(out, 2, _encode_map_entry(k, ti.encode()))
        _w_string_field(out, 3, self.method_name)
        return bytes(out)

    @classmethod
    def decode(cls, buf: bytes) -> "SignatureDef":
        s = cls(method_name="")
        pos = 0
        while pos < len(buf):
            fno, wt, v, pos = _r_field(buf, pos)
            if fno == 1 and wt == _WT_LEN:
                k, tv = _decode_map_entry(v)
                s.inputs[k] = TensorInfo.decode(tv)
            elif fno == 2 and wt == _WT_LEN:
                k, tv = _decode_map_entry(v)
                s.outputs[k] = TensorInfo.decode(tv)
            elif fno == 3 and wt == _WT_LEN:
                s.method_name = _utf8(v)
        return s


@dataclass
class GetModelMetadataRequest:
    """tensorflow.serving.GetModelMetadataRequest:
    1 model_spec, 2 metadata_field repeated string."""
    model_spec: ModelSpec = field(default_factory=ModelSpec)
    metadata_field: List[str] = field(default_factory=list)

    def encode(self) -> bytes:
        out = bytearray()
        _w_len_field(out, 1, self.model_spec.encode())
        for f in self.metadata_field:
            _w_string_field(out, 2, f)
        return bytes(out)

    @classmethod
    def decode(cls, buf: bytes) -> "GetModelMetadataRequest":
        r = cls()
        pos = 0
        while pos < len(buf):
            fno, wt, v, pos = _r_field(buf, pos)
            if fno == 1 and wt == _WT_LEN:
                r.model_spec = ModelSpec.decode(v)
            elif fno == 2 and wt == _WT_LEN:
                r.metadata_field.append(_utf8(v))
        return r


SIGNATURE_DEF_MAP_TYPE_URL = \
    "type.googleapis.com/tensorflow.serving.SignatureDefMap"


@dataclass
class GetModelMetadataResponse:
    """tensorflow.serving.GetModelMetadataResponse:
    1 model_spec, 2 metadata map<string, google.protobuf.Any>.
    The 'signature_def' entry packs a SignatureDefMap
    (1 signature_def map<string, SignatureDef>)."""
    model_spec: ModelSpec = field(default_factory=ModelSpec)
    signature_defs: Dict[str, SignatureDef] = field(default_factory=dict)

    def encode(self) -> bytes:
        sdm = bytearray()
        for name, sig in self.signature_defs.items():
            _w_len_field(sdm, 1, _encode_map_entry(name, sig.encode()))
        any_msg = bytearray()
        _w_string_field(any_msg, 1, SIGNATURE_DEF_MAP_TYPE_URL)
        _w_len_field(any_msg, 2, bytes(sdm))
        out = bytearray()
        _w_len_field(out, 1, self.model_spec.encode())
        _w_len_field(out, 2,
                     _encode_map_entry("signature_def", bytes(any_msg)))
        return bytes(out)

    @classmethod
    def decode(cls, buf: bytes) -> "GetModelMetadataResponse":
        r = cls()
        pos = 0
        while pos < len(buf):
            fno, wt, v, pos = _r_field(buf, pos)
            if fno == 1 and wt == _WT_LEN:
                r.model_spec = ModelSpec.decode(v)
            elif fno == 2 and wt == _WT_LEN:
                key, any_bytes = _decode_map_entry(v)
                if key != "signature_def":
                    continue
                p = 0
                payload = b""
                while p < len(any_bytes):
                    f2, w2, v2, p = _r_field(any_bytes, p)
                    if f2 == 2 and w2 == _WT_LEN:
                        payload = v2
                p = 0
                while p < len(payload):
                    f2, w2, v2, p = _r_field(payload, p)
                    if f2 == 1 and w2 == _WT_LEN:
                        name, sv = _decode_map_entry(v2)
                        r.signature_defs[name] = SignatureDef.decode(sv)
        return r


def encode_saved_model(signature_defs: Dict[str, SignatureDef],
                       tags=("serve",)) -> bytes:
    """tensorflow.SavedModel: 1 schema_version, 2 meta_graphs.

    MetaGraphDef: 1 meta_info_def (MetaInfoDef: 4 tags), 5 signature_def
    map. The graph_def is intentionally empty — the graph is the
    maskrcnn_amd model, reconstructed by the loader from the weights; the
    file carries the signature map and the tag set (layout parity with S1).
    """
    meta_info = bytearray()
    for t in tags:
        _w_string_field(meta_info, 4, t)
    mg = bytearray()
    _w_len_field(mg, 1, bytes(meta_info))
    for name, sig in signature_defs.items():
        _w_len_field(mg, 5, _encode_map_entry(name, sig.encode()))
    out = bytearray()
    _w_varint_field(out, 1, 1)  # saved_model_schema_version
    _w_len_field(out, 2, bytes(mg))
    return bytes(out)


def decode_saved_model(buf: bytes):
    """Returns (schema_version, tags, {sig_name: SignatureDef})."""
    version = 0
    tags: List[str] = []
    sigs: Dict[str, SignatureDef] = {}
    pos = 0
    while pos < len(buf):
        fno, wt, v, pos = _r_field(buf, pos)
        if fno == 1 and wt == _WT_VARINT:
            version = v
        elif fno == 2 and wt == _WT_LEN:
            p = 0
            while p < len(v):
                f2, w2, v2, p = _r_field(v, p)
                if f2 == 1 and w2 == _WT_LEN:
                    pp = 0
                    while pp < len(v2):
                        f3, w3, v3, pp = _r_field(v2, pp)
                        if f3 == 4 and w3 == _WT_LEN:
                            tags.append(_utf8(v3))
                elif f2 == 5 and w2 == _WT_LEN:
                    k, sv = _decode_map_entry(v2)
                    sigs[k] = SignatureDef.decode(sv)
    return version, tags, sigs


# ------------------------------------------- ModelService (model mgmt)
# tensorflow_serving/apis/get_model_status.proto +
# model_management.proto + config/model_server_config.proto field
# numbers; the upstream gRPC service is
# /tensorflow.serving.ModelService/{GetModelStatus,HandleReloadConfigRequest}.

#: ModelVersionStatus.State enum values (get_model_status.proto).
STATE_UNKNOWN, STATE_START, STATE_LOADING = 0, 10, 20
STATE_AVAILABLE, STATE_UNLOADING, STATE_END = 30, 40, 50


@dataclass
class StatusProto:
    """tensorflow.error-style status: 1 error_code, 2 error_message."""
    error_code: int = 0
    error_message: str = ""

    def encode(self) -> bytes:
        out = bytearray()
        _w_varint_field(out, 1, self.error_code)
        _w_string_field(out, 2, self.error_message)
        return bytes(out)

    @classmethod
    def decode(cls, buf: bytes) -> "StatusProto":
        m = cls()
        pos = 0
        while pos < len(buf):
            fno, wt, v, pos = _r_field(buf, pos)
            if fno == 1 and wt == _WT_VARINT:
                m.error_code = v
            elif fno == 2 and wt == _WT_LEN:
                m.error_message = _utf8(v)
        return m


@dataclass
class ModelVersionStatus:
    """1 version (int64), 2 state (enum), 3 status."""
    version: int = 0
    state: int = STATE_UNKNOWN
    status: StatusProto = field(default_factory=StatusProto)

    def encode(self) -> bytes:
        out = bytearray()
        _w_varint_field(out, 1, _zigzag_i64(self.version))
        _w_varint_field(out, 2, self.state)
        _w_len_field(out, 3, self.status.encode())
        return bytes(out)

    @classmethod
    def decode(cls, buf: bytes) -> "ModelVersionStatus":
        m = cls()
        pos = 0
        while pos < len(buf):
            fno, wt, v, pos = _r_field(buf, pos)
            if fno == 1 and wt == _WT_VARINT:
                m.version = v if v < (1 << 63) else v - (1 << 64)
            elif fno == 2 and wt == _WT_VARINT:
                m.state = v
            elif fno == 3 and wt == _WT_LEN:
                m.status = StatusProto.decode(v)
        return m


@dataclass
class GetModelStatusRequest:
    """1 model_spec."""
    model_spec: ModelSpec = field(default_factory=ModelSpec)

    def encode(self) -> bytes:
        out = bytearray()
        _w_len_field(out, 1, self.model_spec.encode())
        return bytes(out)

    @classmethod
    def decode(cls, buf: bytes) -> "GetModelStatusRequest":
        m = cls()
        pos = 0
        while pos < len(buf):
            fno, wt, v, pos = _r_field(buf, pos)
            if fno == 1 and wt == _WT_LEN:
                m.model_spec = ModelSpec.decode(v)
        return m


@dataclass
class GetModelStatusResponse:
    """1 repeated model_version_status."""
    model_version_status: List[ModelVersionStatus] = \
        field(default_factory=list)

    def encode(self) -> bytes:
        out = bytearray()
        for s in self.model_version_status:
            _w_len_field(out, 1, s.encode())
        return bytes(out)

    @classmethod
    def decode(cls, buf: bytes) -> "GetModelStatusResponse":
        m = cls()
        pos = 0
        while pos < len(buf):
            fno, wt, v, pos = _r_field(buf, pos)
            if fno == 1 and wt == _WT_LEN:
                m.model_version_status.append(
                    ModelVersionStatus.decode(v))
        return m


@dataclass
class ModelConfigProto:
    """model_server_config.proto ModelConfig: 1 name, 2 base_path,
    4 model_platform."""
    name: str = ""
    base_path: str = ""
    model_platform: str = "tensorflow"

    def encode(self) -> bytes:
        out = bytearray()
        _w_string_field(out, 1, self.name)
        _w_string_field(out, 2, self.base_path)
        _w_string_field(out, 4, self.model_platform)
        return bytes(out)

    @classmethod
    def decode(cls, buf: bytes) -> "ModelConfigProto":
        m = cls()
        pos = 0
        while pos < len(buf):
            fno, wt, v, pos = _r_field(buf, pos)
            if fno == 1 and wt == _WT_LEN:
                m.name = _utf8(v)
            elif fno == 2 and wt == _WT_LEN:
                m.base_path = _utf8(v)
            elif fno == 4 and wt == _WT_LEN:
                m.model_platform = _utf8(v)
        return m


@dataclass
class ReloadConfigRequest:
    """model_management.proto ReloadConfigRequest: 1 config
    (ModelServerConfig: 1 model_config_list (ModelConfigList:
    1 repeated config))."""
    model_configs: List[ModelConfigProto] = field(default_factory=list)

    def encode(self) -> bytes:
        lst = bytearray()
        for c in self.model_configs:
            _w_len_field(lst, 1, c.encode())
        cfg = bytearray()
        _w_len_field(cfg, 1, bytes(lst))
        out = bytearray()
        _w_len_field(out, 1, bytes(cfg))
        return bytes(out)

    @classmethod
    def decode(cls, buf: bytes) -> "ReloadConfigRequest":
        m = cls()
        pos = 0
        while pos < len(buf):
            fno, wt, v, pos = _r_field(buf, pos)
            if fno == 1 and wt == _WT_LEN:
                p = 0
                while p < len(v):
                    f2, w2, v2, p = _r_field(v, p)
                    if f2 == 1 and w2 == _WT_LEN:
                        pp = 0
                        while pp < len(v2):
                            f3, w3, v3, pp = _r_field(v2, pp)
                            if f3 == 1 and w3 == _WT_LEN:
                                m.model_configs.append(
                                    ModelConfigProto.decode(v3))
        return m


@dataclass
class ReloadConfigResponse:
    """1 status."""
    status: StatusProto = field(default_factory=StatusProto)

    def encode(self) -> bytes:
        out = bytearray()
        _w_len_field(out, 1, self.status.encode())
        return bytes(out)

    @classmethod
    def decode(cls, buf: bytes) -> "ReloadConfigResponse":
        m = cls()
        pos = 0
        while pos < len(buf):
            fno, wt, v, pos = _r_field(buf, pos)
            if fno == 1 and wt == _WT_LEN:
                m.status = StatusProto.decode(v)
        return m

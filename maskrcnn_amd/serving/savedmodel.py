"""SavedModel-layout checkpoint export / load.

Replicates reference S1: writes the TF-Serving directory
layout

    <base>/<version>/saved_model.pb        (SavedModel proto: tags=serve,
                                            signature_def 'serving_default')
    <base>/<version>/variables/variables.data-00000-of-00001
    <base>/<version>/variables/variables.index

saved_model.pb is a real protobuf (decodable by TF tooling: schema version,
meta graph tags, the full 3-input/7-output signature map with dtypes and
shapes). The variables shard stores the maskrcnn_amd weights as safetensors
(the index file is a JSON manifest) — the *layout* and the signature
contract are TF-Serving-parity; the tensor serialization inside the shard
is native (no TF available to write SSTable checkpoints, and our loader is
the only consumer of the shard bytes).
"""

from __future__ import annotations

import json
import os


from .wireproto import (DT_FLOAT, SignatureDef, TensorInfo,
                        decode_saved_model, encode_saved_model)

SERVING_SIGNATURE_OUTPUTS = [
    "mrcnn_detection", "mrcnn_class", "mrcnn_bbox", "mrcnn_mask",
    "rois", "rpn_class", "rpn_bbox",
]


def build_signature(config) -> SignatureDef:
    """The serving_default signature of the reference export (S1/S3)."""
    C = config.NUM_CLASSES
    meta = config.IMAGE_META_SIZE
    D = config.DETECTION_MAX_INSTANCES
    N = config.POST_NMS_ROIS_INFERENCE
    sig = SignatureDef()
    sig.inputs = {
        "input_image": TensorInfo("input_image:0", DT_FLOAT,
                                  [-1, -1, -1, 3]),
        "input_image_meta": TensorInfo("input_image_meta:0", DT_FLOAT,
                                       [-1, meta]),
        "input_anchors": TensorInfo("input_anchors:0", DT_FLOAT,
                                    [-1, -1, 4]),
    }
    sig.outputs = {
        "mrcnn_detection": TensorInfo("mrcnn_detection:0", DT_FLOAT,
                                      [-1, D, 6]),
        "mrcnn_class": TensorInfo("mrcnn_class:0", DT_FLOAT, [-1, N, C]),
        "mrcnn_bbox": TensorInfo("mrcnn_bbox:0", DT_FLOAT, [-1, N, C, 4]),
        "mrcnn_mask": TensorInfo("mrcnn_mask:0", DT_FLOAT,
                                 [-1, D, 28, 28, C]),
        "rois": TensorInfo("rois:0", DT_FLOAT, [-1, N, 4]),
        "rpn_class": TensorInfo("rpn_class:0", DT_FLOAT, [-1, -1, 2]),
        "rpn_bbox": TensorInfo("rpn_bbox:0", DT_FLOAT, [-1, -1, 4]),
    }
    return sig


def export_saved_model(model, config, base_dir: str, version: int = 1,
                       extra_meta: dict = None,
                       weight_dtype: str = "fp32",
                       warmup_batches=None) -> str:
    """Export a (torch) Mask R-CNN model as a SavedModel-layout directory.

    model: MaskRCNNModel (or the MaskRCNN facade; .model is used).
    weight_dtype: "fp32" or "fp8" (e4m3fn per-out-channel quantized
    conv/linear weights — BASELINE config #5).
    warmup_batches: list of batch sizes (e.g. [32]) — writes TF-Serving
    `assets.extra/tf_serving_warmup_requests` (TFRecord of
    PredictionLogs) replayed by the server at load time.
    Returns the version directory path.
    """
    import torch
    from safetensors.torch import save_file

    net = getattr(model, "model", model)
    export_dir = os.path.join(base_dir, str(version))
    var_dir = os.path.join(export_dir, "variables")
    os.makedirs(var_dir, exist_ok=True)

    sig = build_signature(config)
    with open(os.path.join(export_dir, "saved_model.pb"), "wb") as f:
        f.write(encode_saved_model({"serving_default": sig}))

    state = {k: v.detach().cpu().contiguous()
             for k, v in net.state_dict().items()}
    if weight_dtype == "fp8":
        from ..utils.fp8 import quantize_state_dict
        state = quantize_state_dict(state)
    elif weight_dtype != "fp32":
        raise ValueError(f"unsupported weight_dtype {weight_dtype!r}")
    shard = os.path.join(var_dir, "variables.data-00000-of-00001")
    save_file(state, shard)

    index = {
        "format": "maskrcnn_amd.safetensors.v1",
        "weight_dtype": weight_dtype,
        "shards": ["variables.data-00000-of-00001"],
        "num_tensors": len(state),
        "config": {k: v for k, v in config.to_dict().items()
                   if isinstance(v, (int, float, str, bool, list, tuple,
                                     dict, type(None)))},
    }
    if extra_meta:
        index["meta"] = extra_meta
    with open(os.path.join(var_dir, "variables.index"), "w") as f:
        json.dump(index, f, indent=1, default=str)

    if warmup_batches:
        import numpy as np
        from .warmup import write_warmup_requests
        from .wireproto import (ModelSpec, PredictRequest,
                                numpy_to_tensor)
        H, W, _ = config.IMAGE_SHAPE
        reqs = []
        for b in warmup_batches:
            r = PredictRequest(model_spec=ModelSpec(
                name="", signature_name="serving_default"))
            # uint8 zeros: 1/4 the record bytes; engine molds on-GPU.
            r.inputs["input_image"] = numpy_to_tensor(
                np.zeros((b, H, W, 3), np.uint8))
            meta = np.zeros((b, config.IMAGE_META_SIZE), np.float32)
            meta[:, 4:7] = (H, W, 3)
            meta[:, 7:11] = (0, 0, H, W)
            r.inputs["input_image_meta"] = numpy_to_tensor(meta)
            reqs.append(r)
        write_warmup_requests(export_dir, reqs)
    return export_dir


def load_saved_model(export_dir: str):
    """Load a SavedModel-layout directory.

    Returns (state_dict, config_dict, signatures: {name: SignatureDef}).
    """
    from safetensors.torch import load_file

    with open(os.path.join(export_dir, "saved_model.pb"), "rb") as f:
        version, tags, sigs = decode_saved_model(f.read())
    if "serve" not in tags:
        raise ValueError(f"saved_model.pb has no 'serve' tag (tags={tags})")

    var_dir = os.path.join(export_dir, "variables")
    with open(os.path.join(var_dir, "variables.index")) as f:
        index = json.load(f)
    state = {}
    for shard in index["shards"]:
        state.update(load_file(os.path.join(var_dir, shard)))
    if index.get("weight_dtype") == "fp8":
        from ..utils.fp8 import dequantize_state_dict
        state = dequantize_state_dict(state)
    return state, index.get("config", {}), sigs


def list_versions(base_dir: str):
    """All numeric version subdirs holding a saved_model.pb, sorted."""
    if not os.path.isdir(base_dir):
        return []
    return sorted(
        int(d) for d in os.listdir(base_dir)
        if d.isdigit() and
        os.path.isfile(os.path.join(base_dir, d, "saved_model.pb")))


def find_latest_version(base_dir: str):
    """Highest numeric version subdir (TF-Serving convention), or None."""
    versions = list_versions(base_dir)
    return versions[-1] if versions else None

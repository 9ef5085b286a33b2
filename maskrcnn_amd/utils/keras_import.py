"""Import matterport/Mask_RCNN Keras ``.h5`` weights.

The reference ecosystem ships checkpoints as Keras HDF5 files
(``mask_rcnn_coco.h5`` etc.; reference load_weights, mrcnn/model.py:
~2080). This module maps every Keras layer/weight name to our state
dict — kernels transposed from Keras ``[kh, kw, cin, cout]`` to torch
``[cout, cin, kh, kw]``, dense ``[in, out]`` -> ``[out, in]``, BN
gamma/beta/moving_* -> weight/bias/running_* — so existing matterport
checkpoints load directly.

The HDF5 decoding itself needs ``h5py`` (absent from this offline
image): ``load_keras_h5`` imports it lazily; ``assign_keras_arrays``
takes a plain ``{"layer/weight": ndarray}`` dict, so the mapping and
transposes are fully testable without it.
"""

from __future__ import annotations

from typing import Dict, List, Tuple

import numpy as np

#: weight-kind -> (our suffix, transform)
_BN_PARTS = (("gamma", "weight"), ("beta", "bias"),
             ("moving_mean", "running_mean"),
             ("moving_variance", "running_var"))


def _conv(arr):
    return np.transpose(arr, (3, 2, 0, 1))


def _deconv(arr):
    # Keras Conv2DTranspose kernel [kh, kw, out, in] -> torch [in, out,
    # kh, kw].
    return np.transpose(arr, (3, 2, 0, 1))


def _dense(arr):
    return np.transpose(arr, (1, 0))


def _copy(arr):
    return arr


def keras_weight_map(backbone: str = "resnet101"
                     ) -> Dict[str, Tuple[str, object]]:
    """{'keras_layer/weight': (our_state_key, transform_fn)}.

    Layer names follow mrcnn/model.py's graph builders (resnet_graph
    res{stage}{block}_branch*, fpn_*, rpn_*, mrcnn_*)."""
    from ..model.resnet import ResNet
    blocks = ResNet.STAGE_BLOCKS[backbone]
    m: Dict[str, Tuple[str, object]] = {}

    def conv(keras, ours):
        m[f"{keras}/kernel"] = (f"{ours}.weight", _conv)
        m[f"{keras}/bias"] = (f"{ours}.bias", _copy)

    def bn(keras, ours):
        for kk, ok in _BN_PARTS:
            m[f"{keras}/{kk}"] = (f"{ours}.{ok}", _copy)

    def dense(keras, ours):
        m[f"{keras}/kernel"] = (f"{ours}.weight", _dense)
        m[f"{keras}/bias"] = (f"{ours}.bias", _copy)

    conv("conv1", "backbone.conv1")
    bn("bn_conv1", "backbone.bn_conv1")
    for stage, n_blocks in zip((2, 3, 4, 5), blocks):
        for i in range(n_blocks):
            blk = chr(ord("a") + i)
            ours = f"backbone.layer{stage}.{i}"
            for j, branch in enumerate(("2a", "2b", "2c"), start=1):
                conv(f"res{stage}{blk}_branch{branch}",
                     f"{ours}.conv{j}")
                bn(f"bn{stage}{blk}_branch{branch}", f"{ours}.bn{j}")
            if i == 0:  # projection shortcut
                conv(f"res{stage}a_branch1", f"{ours}.downsample.0")
                bn(f"bn{stage}a_branch1", f"{ours}.downsample.1")

    for name in ("fpn_c5p5", "fpn_c4p4", "fpn_c3p3", "fpn_c2p2",
                 "fpn_p2", "fpn_p3", "fpn_p4", "fpn_p5"):
        conv(name, f"fpn.{name}")
    for name in ("rpn_conv_shared", "rpn_class_raw", "rpn_bbox_pred"):
        conv(name, f"rpn.{name}")

    conv("mrcnn_class_conv1", "classifier_head.mrcnn_class_conv1")
    bn("mrcnn_class_bn1", "classifier_head.mrcnn_class_bn1")
    conv("mrcnn_class_conv2", "classifier_head.mrcnn_class_conv2")
    bn("mrcnn_class_bn2", "classifier_head.mrcnn_class_bn2")
    dense("mrcnn_class_logits", "classifier_head.mrcnn_class_logits")
    dense("mrcnn_bbox_fc", "classifier_head.mrcnn_bbox_fc")

    for i in range(1, 5):
        conv(f"mrcnn_mask_conv{i}", f"mask_head.convs.{3 * (i - 1)}")
        bn(f"mrcnn_mask_bn{i}", f"mask_head.convs.{3 * (i - 1) + 1}")
    m["mrcnn_mask_deconv/kernel"] = \
        ("mask_head.mrcnn_mask_deconv.weight", _deconv)
    m["mrcnn_mask_deconv/bias"] = \
        ("mask_head.mrcnn_mask_deconv.bias", _copy)
    conv("mrcnn_mask", "mask_head.mrcnn_mask")
    return m


def assign_keras_arrays(model, arrays: Dict[str, np.ndarray],
                        exclude: List[str] = None
                        ) -> Tuple[List[str], List[str], List[str]]:
    """Load {'layer/weight': ndarray} into a MaskRCNNModel.

    exclude: keras LAYER names to skip (the reference's class-count
    surgery: mrcnn_class_logits, mrcnn_bbox_fc, mrcnn_mask).
    Returns (loaded_keys, missing_keras_names, unexpected_names)."""
    import torch

    wmap = keras_weight_map(model.config.BACKBONE)
    state = model.state_dict()
    exclude = set(exclude or [])
    loaded, unexpected = [], []
    seen = set()
    with torch.no_grad():
        for name, arr in arrays.items():
            name = name.split(":")[0]  # tolerate ':0' suffixes
            if name not in wmap:
                unexpected.append(name)
                continue
            seen.add(name)
            if name.split("/")[0] in exclude:
                continue
            key, tf = wmap[name]
            target = state[key]
            value = torch.from_numpy(
                np.ascontiguousarray(tf(np.asarray(arr))))
            if tuple(value.shape) != tuple(target.shape):
                raise ValueError(
                    f"{name} -> {key}: shape {tuple(value.shape)} vs "
                    f"{tuple(target.shape)}")
            target.copy_(value.to(target.dtype))
            loaded.append(key)
    missing = [n for n in wmap if n not in seen]
    return loaded, missing, unexpected


def load_keras_h5(model, path: str, exclude: List[str] = None):
    """Load a matterport ``.h5`` checkpoint file (requires h5py)."""
    try:
        import h5py
    except ImportError as e:  # pragma: no cover
        raise ImportError(
            "loading Keras .h5 checkpoints requires h5py; convert the "
            "file to a torch checkpoint elsewhere or install h5py "
            "(the name mapping in keras_weight_map() is the full "
            "spec)") from e
    arrays = {}
    with h5py.File(path, "r") as f:  # pragma: no cover - needs h5py
        g = f["model_weights"] if "model_weights" in f else f
        def visit(name, obj):
            if isinstance(obj, h5py.Dataset):
                parts = [p for p in name.split("/") if p]
                arrays["/".join(parts[-2:])] = np.asarray(obj)
        g.visititems(visit)
    return assign_keras_arrays(model, arrays, exclude=exclude)

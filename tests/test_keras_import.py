"""Keras .h5 weight-name mapping: completeness + transposes."""

import numpy as np
import pytest

from maskrcnn_amd.data.synthetic import ShapesConfig
from maskrcnn_amd.model.mask_rcnn import MaskRCNNModel
from maskrcnn_amd.utils.keras_import import (assign_keras_arrays,
                                             keras_weight_map)


def _model(backbone):
    class C(ShapesConfig):
        BACKBONE = backbone
    return MaskRCNNModel(C())


@pytest.mark.parametrize("backbone", ["resnet50", "resnet101"])
def test_map_covers_every_parameter(backbone):
    """Every conv/bn/fc/deconv state key must be reachable from exactly
    one Keras name (num_batches_tracked has no Keras source)."""
    m = _model(backbone)
    wmap = keras_weight_map(backbone)
    targets = [k for k, _ in wmap.values()]
    assert len(targets) == len(set(targets)), "duplicate targets"
    state = set(m.state_dict().keys())
    for t in targets:
        assert t in state, f"mapped to nonexistent key {t}"
    expected = {k for k in state if not k.endswith("num_batches_tracked")}
    assert set(targets) == expected, \
        (expected - set(targets), set(targets) - expected)


def test_assign_transposes_and_excludes():
    m = _model("resnet50")
    wmap = keras_weight_map("resnet50")
    state = m.state_dict()
    rng = np.random.default_rng(0)

    def keras_shape(key, kind_name):
        t = state[key]
        if kind_name == "_conv":      # torch [o,i,kh,kw] <- [kh,kw,i,o]
            o, i, kh, kw = t.shape
            return (kh, kw, i, o)
        if kind_name == "_deconv":    # torch [i,o,kh,kw] <- [kh,kw,o,i]
            i, o, kh, kw = t.shape
            return (kh, kw, o, i)
        if kind_name == "_dense":
            o, i = t.shape
            return (i, o)
        return tuple(t.shape)

    arrays = {}
    for name, (key, tf) in wmap.items():
        arrays[name + ":0"] = rng.standard_normal(
            keras_shape(key, tf.__name__)).astype(np.float32)
    loaded, missing, unexpected = assign_keras_arrays(
        m, arrays, exclude=["mrcnn_class_logits", "mrcnn_bbox_fc",
                            "mrcnn_mask"])
    assert not missing and not unexpected
    state = m.state_dict()
    # Spot-check a conv transpose end to end.
    src = arrays["res2a_branch2b/kernel:0"]
    got = state["backbone.layer2.0.conv2.weight"].numpy()
    np.testing.assert_array_equal(got, np.transpose(src, (3, 2, 0, 1)))
    # Dense transpose.
    src = arrays["mrcnn_class_conv1/kernel:0"]  # 7x7 conv
    got = state["classifier_head.mrcnn_class_conv1.weight"].numpy()
    np.testing.assert_array_equal(got, np.transpose(src, (3, 2, 0, 1)))
    # BN passthrough.
    src = arrays["bn_conv1/moving_variance:0"]
    np.testing.assert_array_equal(
        state["backbone.bn_conv1.running_var"].numpy(), src)
    # Excluded layers kept their (random init) values != provided.
    src = np.transpose(arrays["mrcnn_mask/kernel:0"], (3, 2, 0, 1))
    assert not np.allclose(state["mask_head.mrcnn_mask.weight"].numpy(),
                           src)
    # Shape mismatch raises.
    bad = dict(arrays)
    bad["conv1/kernel:0"] = np.zeros((3, 3, 3, 64), np.float32)
    with pytest.raises(ValueError, match="conv1"):
        assign_keras_arrays(_model("resnet50"), bad)


def test_unknown_names_reported_not_fatal():
    m = _model("resnet50")
    loaded, missing, unexpected = assign_keras_arrays(
        m, {"totally_new_layer/kernel": np.zeros((1, 1, 1, 1),
                                                 np.float32)})
    assert unexpected == ["totally_new_layer/kernel"]
    assert not loaded and len(missing) > 100


def test_facade_h5_path_without_h5py(tmp_path):
    """load_weights('.h5') routes to the Keras importer; without h5py
    the error must say exactly what to do."""
    from maskrcnn_amd.data.synthetic import ShapesConfig
    from maskrcnn_amd.model import MaskRCNN
    m = MaskRCNN("inference", ShapesConfig(), model_dir=str(tmp_path))
    f = tmp_path / "w.h5"
    f.write_bytes(b"\x89HDF\r\n")
    with pytest.raises(ImportError, match="h5py"):
        m.load_weights(str(f), by_name=True)

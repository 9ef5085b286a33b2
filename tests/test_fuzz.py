"""Property-based tests (hypothesis): wire-proto codec and box math."""

import numpy as np
from hypothesis import given, settings
from hypothesis import strategies as st

from maskrcnn_amd.serving.wireproto import (ModelSpec, TensorProto,
                                            numpy_to_tensor,
                                            tensor_to_numpy)
from maskrcnn_amd.utils import boxes as B
from maskrcnn_amd.utils.masks import rle_decode, rle_encode
from maskrcnn_amd.data.coco import rle_from_string, rle_to_string


@settings(max_examples=50, deadline=None)
@given(st.lists(st.integers(1, 6), min_size=1, max_size=4),
       st.sampled_from([np.float32, np.int32, np.int64, np.bool_]))
def test_tensor_roundtrip_any_shape(shape, dtype):
    rng = np.random.RandomState(0)
    if dtype == np.bool_:
        arr = rng.rand(*shape) > 0.5
    elif dtype == np.float32:
        arr = (rng.randn(*shape) * 1e3).astype(dtype)
    else:
        arr = rng.randint(-10**6, 10**6, size=shape).astype(dtype)
    t = numpy_to_tensor(arr)
    back = tensor_to_numpy(TensorProto.decode(t.encode()))
    np.testing.assert_array_equal(back, arr)


@settings(max_examples=30, deadline=None)
@given(st.text(min_size=0, max_size=40),
       st.one_of(st.none(), st.integers(-2**40, 2**40)),
       st.text(min_size=0, max_size=20))
def test_model_spec_roundtrip(name, version, sig):
    m = ModelSpec(name=name, version=version, signature_name=sig)
    d = ModelSpec.decode(m.encode())
    assert d.name == name
    assert d.version == version
    assert d.signature_name == sig


@settings(max_examples=25, deadline=None)
@given(st.integers(1, 40), st.integers(1, 40), st.integers(0, 2**31))
def test_rle_string_roundtrip_fuzz(h, w, seed):
    rng = np.random.RandomState(seed % (2**31))
    mask = rng.rand(h, w) > rng.rand()
    rle = rle_encode(mask)
    back = rle_from_string(rle_to_string(rle), rle["size"])
    np.testing.assert_array_equal(rle_decode(back), mask)


@settings(max_examples=25, deadline=None)
@given(st.integers(1, 64), st.integers(0, 2**31))
def test_box_refinement_roundtrip_fuzz(n, seed):
    rng = np.random.RandomState(seed % (2**31))
    src = rng.rand(n, 4)
    src[:, 2:] = src[:, :2] + rng.rand(n, 2) + 0.05
    dst = rng.rand(n, 4)
    dst[:, 2:] = dst[:, :2] + rng.rand(n, 2) + 0.05
    rec = B.apply_box_deltas(src, B.box_refinement(src, dst))
    np.testing.assert_allclose(rec, dst, atol=1e-4)


@settings(max_examples=20, deadline=None)
@given(st.integers(2, 100), st.floats(0.1, 0.9), st.integers(0, 2**31))
def test_nms_invariants(n, thresh, seed):
    import torch
    from maskrcnn_amd.ops import nms
    rng = np.random.RandomState(seed % (2**31))
    boxes = rng.rand(n, 4).astype(np.float32)
    boxes[:, 2:] = boxes[:, :2] + rng.rand(n, 2).astype(np.float32) + 0.01
    scores = rng.rand(n).astype(np.float32)
    keep = nms(torch.from_numpy(boxes), torch.from_numpy(scores), thresh)
    k = keep.numpy()
    # Kept indices unique; scores non-increasing; survivors mutually
    # below the IoU threshold.
    assert len(set(k.tolist())) == len(k)
    s = scores[k]
    assert (np.diff(s) <= 1e-6).all()
    ov = B.compute_overlaps(boxes[k], boxes[k])
    np.fill_diagonal(ov, 0)
    assert (ov <= thresh + 1e-5).all()


@given(st.lists(st.tuples(st.text(min_size=1, max_size=20),
                          st.text(min_size=1, max_size=40)),
                min_size=0, max_size=5),
       st.integers(min_value=0, max_value=2 ** 62),
       st.sampled_from([0, 10, 20, 30, 40, 50]))
@settings(max_examples=50, deadline=None)
def test_model_service_protos_roundtrip(configs, version, state):
    from maskrcnn_amd.serving.wireproto import (
        GetModelStatusResponse, ModelConfigProto, ModelVersionStatus,
        ReloadConfigRequest, StatusProto)
    req = ReloadConfigRequest(model_configs=[
        ModelConfigProto(name=n, base_path=p) for n, p in configs])
    dec = ReloadConfigRequest.decode(req.encode())
    assert [(c.name, c.base_path) for c in dec.model_configs] == configs
    resp = GetModelStatusResponse(model_version_status=[
        ModelVersionStatus(version=version, state=state,
                           status=StatusProto(13, "err"))])
    d2 = GetModelStatusResponse.decode(resp.encode())
    assert d2.model_version_status[0].version == version
    assert d2.model_version_status[0].state == state
    assert d2.model_version_status[0].status.error_message == "err"


@given(st.integers(min_value=1, max_value=3),
       st.integers(min_value=5, max_value=40),
       st.integers(min_value=0, max_value=10 ** 6),
       st.sampled_from([0.0, 0.3, 0.7]))
@settings(max_examples=20, deadline=None)
def test_detection_syncfree_equivalence_fuzz(B, N, seed, min_conf):
    """Sync-free DetectionLayer == eager across random shapes/confs."""
    import os

    import torch

    from maskrcnn_amd.data.synthetic import ShapesConfig
    from maskrcnn_amd.model.detection import DetectionLayer
    from maskrcnn_amd.utils.meta import compose_image_meta

    class C(ShapesConfig):
        DETECTION_MIN_CONFIDENCE = min_conf
        DETECTION_MAX_INSTANCES = 6

    cfg = C()
    torch.manual_seed(seed)
    layer = DetectionLayer(cfg)
    Ccls = cfg.NUM_CLASSES
    rois = torch.rand(B, N, 4)
    rois[..., 2:] = rois[..., :2] * 0.5 + 0.4
    probs = torch.softmax(torch.randn(B, N, Ccls) * 3, dim=-1)
    deltas = torch.randn(B, N, Ccls, 4) * 0.1
    metas = torch.stack([
        torch.from_numpy(compose_image_meta(
            i, (128, 128, 3), (128, 128, 3), (0, 0, 128, 128), 1.0,
            np.ones(Ccls))).float() for i in range(B)])
    try:
        os.environ["MASKRCNN_SYNCFREE_DETECTION"] = "0"
        eager = layer(rois, probs, deltas, metas)
        os.environ["MASKRCNN_SYNCFREE_DETECTION"] = "1"
        syncfree = layer(rois, probs, deltas, metas)
    finally:
        os.environ.pop("MASKRCNN_SYNCFREE_DETECTION", None)
    for b in range(B):
        ne = int((eager[b][:, 5] > 0).sum())
        ns = int((syncfree[b][:, 5] > 0).sum())
        assert ne == ns
        torch.testing.assert_close(
            syncfree[b][:ns].sort(dim=0).values,
            eager[b][:ne].sort(dim=0).values, atol=1e-6, rtol=1e-6)


@given(st.integers(min_value=1, max_value=6),
       st.integers(min_value=1, max_value=4),
       st.integers(min_value=0, max_value=10 ** 6),
       st.floats(min_value=-12, max_value=12))
@settings(max_examples=40, deadline=None)
def test_mx_quantize_roundtrip_fuzz(rows, blocks, seed, log_scale):
    """MX block quant: bounded relative error at ANY magnitude (the
    e8m0 exponent covers 2^-127..2^127)."""
    import torch

    from maskrcnn_amd.utils.fp8 import mx_dequantize, mx_quantize
    torch.manual_seed(seed)
    x = torch.randn(rows, blocks * 32) * (10.0 ** log_scale)
    q, exp = mx_quantize(x, dim=-1)
    deq = mx_dequantize(q, exp, dim=-1)
    rel = (deq - x).abs() / x.abs().clamp(min=1e-30)
    assert float(rel.mean()) < 0.04


@given(st.integers(min_value=1, max_value=3),
       st.integers(min_value=20, max_value=120),
       st.integers(min_value=5, max_value=40),
       st.integers(min_value=0, max_value=10 ** 6))
@settings(max_examples=15, deadline=None)
def test_proposal_padded_scatter_fuzz(B, A, count, seed):
    """ProposalLayer (padded NMS + masked scatter) == reference
    per-image loop across random shapes/counts."""
    import torch

    from maskrcnn_amd.data.synthetic import ShapesConfig
    from maskrcnn_amd.model.proposal import ProposalLayer
    from maskrcnn_amd.ops import nms
    from maskrcnn_amd.utils.boxes import apply_box_deltas_t, clip_boxes_t

    class C(ShapesConfig):
        PRE_NMS_LIMIT = 50
        RPN_NMS_THRESHOLD = 0.7

    cfg = C()
    torch.manual_seed(seed)
    layer = ProposalLayer(cfg)
    probs = torch.softmax(torch.randn(B, A, 2), dim=-1)
    deltas = torch.randn(B, A, 4) * 0.2
    anchors = torch.rand(A, 4)
    anchors[:, 2:] = anchors[:, :2] + 0.1
    rois = layer(probs, deltas, anchors, count)
    assert rois.shape == (B, count, 4)

    scores = probs[:, :, 1]
    d = deltas * torch.tensor(cfg.RPN_BBOX_STD_DEV)
    lim = min(cfg.PRE_NMS_LIMIT, A)
    for b in range(B):
        top_s, ix = torch.topk(scores[b], lim, sorted=True)
        boxes = apply_box_deltas_t(anchors[ix], d[b, ix])
        boxes = clip_boxes_t(boxes, torch.tensor([0., 0., 1., 1.]))
        keep = nms(boxes, top_s, cfg.RPN_NMS_THRESHOLD, max_out=count)
        expect = boxes[keep]
        got = rois[b, :keep.shape[0]]
        torch.testing.assert_close(got, expect, atol=1e-6, rtol=1e-6)
        # Padding rows zeroed.
        assert float(rois[b, keep.shape[0]:].abs().sum()) == 0.0

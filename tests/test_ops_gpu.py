"""GPU kernel numerics tests: HIP kernels vs the plain-PyTorch fp32
references in maskrcnn_amd.ops (run on a real MI355X via gpurun)."""

import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def dev():
    assert torch.cuda.is_available()
    return torch.device("cuda:0")


def _require_ext():
    from maskrcnn_amd.ops._ext import has_ext
    assert has_ext(), "HIP extension must be built on GPU box"


def _rand_boxes(n, rng):
    b = rng.rand(n, 4).astype(np.float32)
    out = np.empty_like(b)
    out[:, 0] = np.minimum(b[:, 0], b[:, 2]) * 0.9
    out[:, 2] = np.maximum(b[:, 0], b[:, 2]) * 0.9 + 0.1
    out[:, 1] = np.minimum(b[:, 1], b[:, 3]) * 0.9
    out[:, 3] = np.maximum(b[:, 1], b[:, 3]) * 0.9 + 0.1
    return out


# --------------------------------------------------------------- roi_align
@pytest.mark.parametrize("mode", ["crop_and_resize", "aligned"])
@pytest.mark.parametrize("cl", [False, True])
def test_roi_align_fwd_matches_cpu(dev, mode, cl):
    _require_ext()
    from maskrcnn_amd.ops import roi_align
    rng = np.random.RandomState(0)
    feats = torch.randn(2, 32, 24, 20)
    boxes = torch.from_numpy(_rand_boxes(40, rng))
    ind = torch.from_numpy(rng.randint(0, 2, 40).astype(np.int64))
    ref = roi_align(feats, boxes, ind, (7, 7), mode=mode)
    fg = feats.to(dev)
    if cl:
        fg = fg.to(memory_format=torch.channels_last)
    out = roi_align(fg, boxes.to(dev), ind.to(dev), (7, 7), mode=mode)
    torch.testing.assert_close(out.cpu(), ref, atol=1e-4, rtol=1e-4)


def test_roi_align_fwd_bf16(dev):
    _require_ext()
    from maskrcnn_amd.ops import roi_align
    rng = np.random.RandomState(1)
    feats = torch.randn(1, 16, 16, 16)
    boxes = torch.from_numpy(_rand_boxes(10, rng))
    ind = torch.zeros(10, dtype=torch.long)
    ref = roi_align(feats, boxes, ind, (7, 7))
    out = roi_align(feats.to(dev).bfloat16(), boxes.to(dev), ind.to(dev),
                    (7, 7))
    assert out.dtype == torch.bfloat16
    torch.testing.assert_close(out.float().cpu(), ref, atol=3e-2, rtol=3e-2)


@pytest.mark.parametrize("mode", ["crop_and_resize", "aligned"])
def test_roi_align_bwd_matches_cpu(dev, mode):
    _require_ext()
    from maskrcnn_amd.ops import roi_align
    rng = np.random.RandomState(2)
    feats = torch.randn(2, 8, 12, 12)
    boxes = torch.from_numpy(_rand_boxes(15, rng))
    ind = torch.from_numpy(rng.randint(0, 2, 15).astype(np.int64))

    f_cpu = feats.clone().requires_grad_(True)
    out_cpu = roi_align(f_cpu, boxes, ind, (5, 5), mode=mode)
    g = torch.randn_like(out_cpu)
    out_cpu.backward(g)

    f_gpu = feats.to(dev).requires_grad_(True)
    out_gpu = roi_align(f_gpu, boxes.to(dev), ind.to(dev), (5, 5),
                        mode=mode)
    out_gpu.backward(g.to(dev))
    torch.testing.assert_close(f_gpu.grad.cpu(), f_cpu.grad, atol=1e-3,
                               rtol=1e-3)


@pytest.mark.parametrize("mode", ["crop_and_resize", "aligned"])
def test_roi_align_bwd_deterministic(dev, mode, monkeypatch):
    """MASKRCNN_DETERMINISTIC=1 routes backward through the fixed-point
    i64 reduction: bitwise-identical grads across repeated runs (the fp
    atomic path is order-dependent), and close to the CPU fp32 oracle.
    (SURVEY §5 sanitizers row)"""
    _require_ext()
    from maskrcnn_amd.ops import roi_align
    rng = np.random.RandomState(7)
    feats = torch.randn(2, 8, 12, 12)
    # Heavily overlapping boxes -> contended atomic targets.
    boxes = torch.from_numpy(
        np.clip(_rand_boxes(200, rng) * 0.3 + 0.3, 0, 1).astype(np.float32))
    ind = torch.from_numpy(rng.randint(0, 2, 200).astype(np.int64))

    f_cpu = feats.clone().requires_grad_(True)
    out_cpu = roi_align(f_cpu, boxes, ind, (5, 5), mode=mode)
    g = torch.randn_like(out_cpu)
    out_cpu.backward(g)

    monkeypatch.setenv("MASKRCNN_DETERMINISTIC", "1")
    grads = []
    for _ in range(3):
        f_gpu = feats.to(dev).requires_grad_(True)
        out_gpu = roi_align(f_gpu, boxes.to(dev), ind.to(dev), (5, 5),
                            mode=mode)
        out_gpu.backward(g.to(dev))
        grads.append(f_gpu.grad.cpu())
    # Bitwise-stable across runs.
    for other in grads[1:]:
        assert torch.equal(grads[0], other)
    # And numerically right (fixed-point quantization ~2^-41 rel amax).
    torch.testing.assert_close(grads[0], f_cpu.grad, atol=1e-3, rtol=1e-3)


def test_roi_align_out_of_bounds_boxes(dev):
    """Mini-mask-relative boxes can exceed [0,1]; zeros outside."""
    _require_ext()
    from maskrcnn_amd.ops import crop_and_resize
    feats = torch.ones(1, 1, 8, 8)
    boxes = torch.tensor([[-0.5, -0.5, 0.5, 0.5]])
    ind = torch.zeros(1, dtype=torch.long)
    ref = crop_and_resize(feats, boxes, ind, (6, 6))
    out = crop_and_resize(feats.to(dev), boxes.to(dev), ind.to(dev), (6, 6))
    torch.testing.assert_close(out.cpu(), ref, atol=1e-5, rtol=1e-5)


# --------------------------------------------------------------------- nms
def test_nms_matches_cpu(dev):
    _require_ext()
    from maskrcnn_amd.ops import nms
    rng = np.random.RandomState(3)
    boxes = torch.from_numpy(_rand_boxes(500, rng))
    scores = torch.from_numpy(rng.rand(500).astype(np.float32))
    ref = nms(boxes, scores, 0.5)
    out = nms(boxes.to(dev), scores.to(dev), 0.5)
    np.testing.assert_array_equal(out.cpu().numpy(), ref.numpy())


def test_nms_large_n(dev):
    """Proposal-scale N (6000)."""
    _require_ext()
    from maskrcnn_amd.ops import nms
    rng = np.random.RandomState(4)
    boxes = torch.from_numpy(_rand_boxes(6000, rng))
    scores = torch.from_numpy(rng.rand(6000).astype(np.float32))
    ref = nms(boxes, scores, 0.7, max_out=2000)
    out = nms(boxes.to(dev), scores.to(dev), 0.7, max_out=2000)
    np.testing.assert_array_equal(out.cpu().numpy(), ref.numpy())


def test_batched_nms_gpu(dev):
    _require_ext()
    from maskrcnn_amd.ops import batched_nms
    rng = np.random.RandomState(5)
    boxes = torch.from_numpy(_rand_boxes(300, rng))
    scores = torch.from_numpy(rng.rand(300).astype(np.float32))
    cls = torch.from_numpy(rng.randint(1, 10, 300).astype(np.int64))
    ref = batched_nms(boxes, scores, cls, 0.3)
    out = batched_nms(boxes.to(dev), scores.to(dev), cls.to(dev), 0.3)
    np.testing.assert_array_equal(np.sort(out.cpu().numpy()),
                                  np.sort(ref.numpy()))


# ------------------------------------------------------------------ model
def test_model_train_step_gpu(dev):
    """Full bf16 training step on GPU with HIP ops in the path."""
    _require_ext()
    from bench import build_config, make_synthetic_batch
    import argparse
    args = argparse.Namespace(batch=1, image_size=512, steps=1, warmup=0,
                              dtype="bf16", backbone="resnet101")
    cfg = build_config(args)
    from maskrcnn_amd.model.mask_rcnn import MaskRCNNModel
    from maskrcnn_amd.model.losses import compute_losses
    from maskrcnn_amd.utils.anchors import AnchorCache
    from maskrcnn_amd.utils.meta import parse_image_meta_t
    torch.manual_seed(0)
    model = MaskRCNNModel(cfg).to(dev).train()
    batch = make_synthetic_batch(cfg, dev, seed=0)
    anchors = AnchorCache(cfg).get(cfg.IMAGE_SHAPE, device=dev)
    meta = parse_image_meta_t(batch["image_meta"])
    with torch.autocast(device_type="cuda", dtype=torch.bfloat16):
        out = model(batch["image"], batch["image_meta"], anchors,
                    gt_class_ids=batch["gt_class_ids"],
                    gt_boxes=batch["gt_boxes"],
                    gt_masks=batch["gt_masks"], mode="training")
        losses = compute_losses(
            cfg, {"rpn_match": batch["rpn_match"],
                  "rpn_bbox": batch["rpn_bbox"],
                  "active_class_ids": meta["active_class_ids"]}, out)
    losses["total_loss"].backward()
    torch.cuda.synchronize()
    for k, v in losses.items():
        assert torch.isfinite(v), (k, v)


def test_nms_padded_gpu_matches_cpu(dev):
    _require_ext()
    from maskrcnn_amd.ops import nms_padded
    rng = np.random.RandomState(12)
    boxes = torch.from_numpy(_rand_boxes(700, rng))
    scores = torch.from_numpy(rng.rand(700).astype(np.float32))
    keep_c, count_c = nms_padded(boxes, scores, 0.6, 700)
    keep_g, count_g = nms_padded(boxes.to(dev), scores.to(dev), 0.6, 700)
    assert int(count_g.item()) == int(count_c.item())
    np.testing.assert_array_equal(keep_g.cpu().numpy(), keep_c.numpy())


@pytest.mark.gpu
def test_detection_syncfree_gpu_matches_eager(dev):
    """Capture-safe DetectionLayer path on GPU (detection_prep + HIP
    nms_padded) equals the eager nonzero/gather path."""
    import os

    from maskrcnn_amd.data.synthetic import ShapesConfig
    from maskrcnn_amd.model.detection import DetectionLayer
    from maskrcnn_amd.utils.meta import compose_image_meta

    class C(ShapesConfig):
        DETECTION_MIN_CONFIDENCE = 0.3
        DETECTION_MAX_INSTANCES = 10

    cfg = C()
    torch.manual_seed(0)
    layer = DetectionLayer(cfg).to(dev)
    B, N, Ccls = 3, 80, cfg.NUM_CLASSES
    rois = torch.rand(B, N, 4, device=dev)
    rois[..., 2:] = rois[..., :2] * 0.5 + 0.4
    probs = torch.softmax(torch.randn(B, N, Ccls, device=dev) * 2, -1)
    deltas = torch.randn(B, N, Ccls, 4, device=dev) * 0.1
    metas = torch.stack([
        torch.from_numpy(compose_image_meta(
            i, (128, 128, 3), (128, 128, 3), (0, 0, 128, 128), 1.0,
            np.ones(Ccls))).float() for i in range(B)]).to(dev)
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
            eager[b][:ne].sort(dim=0).values, atol=1e-5, rtol=1e-5)


@pytest.mark.gpu
def test_detection_prep_kernel_matches_torch(dev):
    """Fused HIP prep vs the layer's torch prep block."""
    _require_ext()
    from maskrcnn_amd.ops._ext import require_ext
    from maskrcnn_amd.utils.boxes import apply_box_deltas_t
    ext = require_ext()
    torch.manual_seed(1)
    B, N, C = 3, 50, 5
    rois = torch.rand(B, N, 4, device=dev)
    rois[..., 2:] = rois[..., :2] * 0.5 + 0.4
    probs = torch.softmax(torch.randn(B, N, C, device=dev) * 2, -1)
    deltas = torch.randn(B, N, C, 4, device=dev) * 0.1
    windows = torch.tensor([[0., 0., 1., 1.]] * B, device=dev)
    std = torch.tensor([0.1, 0.1, 0.2, 0.2], device=dev)
    refined, cls, scores = ext.detection_prep(rois, probs, deltas,
                                              windows, std, 0.3)
    class_ids = probs.argmax(2)
    class_scores = probs.gather(2, class_ids[..., None]).squeeze(-1)
    ds = deltas.gather(2, class_ids[..., None, None]
                       .expand(-1, -1, 1, 4)).squeeze(2)
    ref = apply_box_deltas_t(rois, ds * std).clamp(0, 1)
    keep = (class_ids > 0) & (class_scores >= 0.3)
    torch.testing.assert_close(refined, ref, atol=1e-5, rtol=1e-5)
    assert torch.equal(cls.long(), class_ids)
    torch.testing.assert_close(
        scores, torch.where(keep, class_scores,
                            torch.zeros_like(class_scores)),
        atol=1e-6, rtol=1e-6)

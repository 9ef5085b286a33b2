"""CPU reference-op tests: roi_align / crop_and_resize / NMS.

These CPU implementations are the numerics oracle the HIP kernels are
tested against in tests/test_ops_gpu.py.
"""

import numpy as np
import pytest
import torch

from maskrcnn_amd.ops import batched_nms, crop_and_resize, nms, roi_align


def _bilinear_ref(feat, y, x):
    """Scalar bilinear sample at pixel coords (align_corners=True style)."""
    H, W = feat.shape
    y0, x0 = int(np.floor(y)), int(np.floor(x))
    y1, x1 = min(y0 + 1, H - 1), min(x0 + 1, W - 1)
    y0c, x0c = max(y0, 0), max(x0, 0)
    ly, lx = y - y0, x - x0
    v = (feat[y0c, x0c] * (1 - ly) * (1 - lx) +
         feat[y1, x0c] * ly * (1 - lx) +
         feat[y0c, x1] * (1 - ly) * lx +
         feat[y1, x1] * ly * lx)
    return v


def test_crop_and_resize_matches_scalar_reference():
    torch.manual_seed(0)
    feat = torch.rand(1, 1, 16, 16)
    boxes = torch.tensor([[0.1, 0.2, 0.6, 0.9]])
    ind = torch.zeros(1, dtype=torch.long)
    out = crop_and_resize(feat, boxes, ind, (7, 7))[0, 0]
    f = feat[0, 0].numpy()
    H = W = 16
    y1, x1, y2, x2 = boxes[0].tolist()
    for i in range(7):
        for j in range(7):
            y = (y1 + (y2 - y1) * i / 6) * (H - 1)
            x = (x1 + (x2 - x1) * j / 6) * (W - 1)
            ref = _bilinear_ref(f, y, x)
            assert abs(out[i, j].item() - ref) < 1e-5, (i, j)


def test_crop_and_resize_identity_box():
    """A box covering the whole map at pool=H returns the map itself."""
    feat = torch.rand(1, 3, 8, 8)
    boxes = torch.tensor([[0., 0., 1., 1.]])
    ind = torch.zeros(1, dtype=torch.long)
    out = crop_and_resize(feat, boxes, ind, (8, 8))
    assert torch.allclose(out[0], feat[0], atol=1e-5)


def test_roi_align_aligned_constant_map():
    """On a constant feature map every mode returns the constant."""
    feat = torch.full((1, 2, 10, 10), 3.25)
    boxes = torch.tensor([[0.11, 0.03, 0.77, 0.9]])
    ind = torch.zeros(1, dtype=torch.long)
    for mode in ("crop_and_resize", "aligned"):
        out = roi_align(feat, boxes, ind, (7, 7), mode=mode)
        assert torch.allclose(out, torch.full_like(out, 3.25), atol=1e-5)


def test_roi_align_gradcheck():
    feat = torch.rand(1, 2, 8, 8, dtype=torch.float64,
                      requires_grad=True)
    boxes = torch.tensor([[0.1, 0.1, 0.8, 0.9]], dtype=torch.float64)
    ind = torch.zeros(1, dtype=torch.long)

    def f(x):
        return crop_and_resize(x, boxes, ind, (5, 5))

    assert torch.autograd.gradcheck(f, (feat,), eps=1e-6, atol=1e-4)


def test_roi_align_multibatch_routing():
    """Boxes route to their own image in the batch."""
    feat = torch.zeros(2, 1, 4, 4)
    feat[0] = 1.0
    feat[1] = 2.0
    boxes = torch.tensor([[0., 0., 1., 1.], [0., 0., 1., 1.]])
    ind = torch.tensor([0, 1])
    out = crop_and_resize(feat, boxes, ind, (2, 2))
    assert torch.allclose(out[0], torch.ones(1, 2, 2))
    assert torch.allclose(out[1], 2 * torch.ones(1, 2, 2))


def test_nms_torch_vs_numpy():
    from maskrcnn_amd.utils.boxes import non_max_suppression
    rng = np.random.RandomState(3)
    boxes = rng.rand(50, 4).astype(np.float32)
    boxes[:, 2:] = boxes[:, :2] + np.abs(boxes[:, 2:]) + 0.05
    scores = rng.rand(50).astype(np.float32)
    keep_np = non_max_suppression(boxes, scores, 0.5)
    keep_t = nms(torch.from_numpy(boxes), torch.from_numpy(scores), 0.5)
    np.testing.assert_array_equal(np.sort(keep_np),
                                  np.sort(keep_t.numpy()))


def test_nms_max_out():
    boxes = torch.tensor([[0, 0, 1, 1], [2, 2, 3, 3], [4, 4, 5, 5]],
                         dtype=torch.float32)
    scores = torch.tensor([0.5, 0.9, 0.7])
    keep = nms(boxes, scores, 0.5, max_out=2)
    assert keep.tolist() == [1, 2]


def test_batched_nms_classes_dont_suppress_each_other():
    boxes = torch.tensor([[0, 0, 10, 10], [0, 0, 10, 10]],
                         dtype=torch.float32)
    scores = torch.tensor([0.9, 0.8])
    cls = torch.tensor([1, 2])
    keep = batched_nms(boxes, scores, cls, 0.5)
    assert len(keep) == 2
    keep_same = batched_nms(boxes, scores, torch.tensor([1, 1]), 0.5)
    assert len(keep_same) == 1


def test_nms_padded_matches_nms():
    from maskrcnn_amd.ops import nms, nms_padded
    rng = np.random.RandomState(11)
    boxes = rng.rand(80, 4).astype(np.float32)
    boxes[:, 2:] = boxes[:, :2] + np.abs(boxes[:, 2:]) + 0.05
    scores = rng.rand(80).astype(np.float32)
    ref = nms(torch.from_numpy(boxes), torch.from_numpy(scores), 0.5)
    keep, count = nms_padded(torch.from_numpy(boxes),
                             torch.from_numpy(scores), 0.5, 100)
    n = int(count.item())
    assert n == ref.shape[0]
    np.testing.assert_array_equal(keep[:n].numpy(), ref.numpy())
    assert (keep[n:] == -1).all()


def test_native_unmold_matches_python_path():
    """csrc/unmold.cpp vs the threaded python resize+paste loop.

    The native op fuses bilinear resize (F.interpolate align_corners
    =False semantics) + 0.5 threshold + paste; it must agree with the
    utils.image.resize-based fallback pixel for pixel.
    """
    import maskrcnn_amd.ops._ext as E
    from maskrcnn_amd.model.mask_rcnn import MaskRCNN
    ext = E.load_ext()
    if ext is None or not hasattr(ext, "unmold_masks"):
        pytest.skip("native extension not built")
    rng = np.random.default_rng(7)
    N = 23
    det = np.zeros((N + 2, 6), dtype=np.float32)
    det[:N, 0] = rng.uniform(0, 0.8, N)
    det[:N, 1] = rng.uniform(0, 0.8, N)
    det[:N, 2] = det[:N, 0] + rng.uniform(0.01, 0.2, N)
    det[:N, 3] = det[:N, 1] + rng.uniform(0.01, 0.2, N)
    det[:N, 4] = rng.integers(1, 5, N)
    det[:N, 5] = rng.uniform(0.5, 1, N)  # rows N..N+1 zero-pad
    mask = rng.random((N + 2, 5, 28, 28)).astype(np.float32)
    args = (det, mask, (512, 640, 3), (512, 512, 3), (0, 0, 512, 512))
    got = MaskRCNN.unmold_detections(*args)
    orig = E.load_ext
    E.load_ext = lambda *a, **k: None
    try:
        ref = MaskRCNN.unmold_detections(*args)
    finally:
        E.load_ext = orig
    np.testing.assert_array_equal(got[0], ref[0])
    np.testing.assert_array_equal(got[1], ref[1])
    np.testing.assert_array_equal(got[2], ref[2])
    mism = (got[3] != ref[3]).sum()
    assert mism <= got[3].size * 1e-6, f"{mism} mask pixels differ"


def test_native_unmold_empty_and_clamped():
    from maskrcnn_amd.ops._ext import load_ext
    ext = load_ext()
    if ext is None or not hasattr(ext, "unmold_masks"):
        pytest.skip("native extension not built")
    out = ext.unmold_masks(torch.zeros(0, 28, 28),
                           torch.zeros(0, 4, dtype=torch.long), 64, 64, 0.5)
    assert out.shape == (0, 64, 64)
    # Box partially outside the canvas: clamped, no crash, outside empty.
    m = torch.ones(1, 28, 28)
    out = ext.unmold_masks(m, torch.tensor([[-8, 50, 40, 90]]), 64, 64, 0.5)
    assert out.shape == (1, 64, 64)
    assert bool(out[0, :40, 50:].all())
    assert int(out[0].sum()) == 40 * 14

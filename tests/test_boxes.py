"""Box-utility unit tests (numpy/torch parity and known-value checks)."""

import numpy as np
import torch

from maskrcnn_amd.utils import boxes as B


def test_extract_bboxes():
    mask = np.zeros((10, 10, 2), dtype=bool)
    mask[2:5, 3:7, 0] = True
    out = B.extract_bboxes(mask)
    np.testing.assert_array_equal(out[0], [2, 3, 5, 7])
    np.testing.assert_array_equal(out[1], [0, 0, 0, 0])


def test_compute_overlaps_known():
    b1 = np.array([[0, 0, 10, 10]], dtype=np.float64)
    b2 = np.array([[0, 0, 10, 10], [5, 5, 15, 15], [20, 20, 30, 30]],
                  dtype=np.float64)
    ov = B.compute_overlaps(b1, b2)
    np.testing.assert_allclose(ov[0], [1.0, 25 / 175, 0.0])


def test_overlaps_t_matches_numpy():
    rng = np.random.RandomState(0)
    b1 = rng.rand(12, 4)
    b2 = rng.rand(7, 4)
    for b in (b1, b2):
        b[:, 2:] = b[:, :2] + np.abs(b[:, 2:]) + 0.05
    ref = B.compute_overlaps(b1, b2)
    out = B.overlaps_t(torch.from_numpy(b1), torch.from_numpy(b2)).numpy()
    np.testing.assert_allclose(out, ref, atol=1e-6)


def test_apply_box_deltas_roundtrip():
    rng = np.random.RandomState(1)
    src = rng.rand(20, 4)
    src[:, 2:] = src[:, :2] + np.abs(src[:, 2:]) + 0.1
    dst = rng.rand(20, 4)
    dst[:, 2:] = dst[:, :2] + np.abs(dst[:, 2:]) + 0.1
    deltas = B.box_refinement(src, dst)
    rec = B.apply_box_deltas(src, deltas)
    np.testing.assert_allclose(rec, dst, atol=1e-5)
    # torch parity
    rec_t = B.apply_box_deltas_t(torch.from_numpy(src).float(),
                                 torch.from_numpy(deltas).float())
    np.testing.assert_allclose(rec_t.numpy(), dst, atol=1e-4)
    d_t = B.box_refinement_t(torch.from_numpy(src).float(),
                             torch.from_numpy(dst).float())
    np.testing.assert_allclose(d_t.numpy(), deltas, atol=1e-5)


def test_norm_denorm_roundtrip():
    boxes = np.array([[10, 20, 100, 200], [0, 0, 511, 511]],
                     dtype=np.float32)
    n = B.norm_boxes(boxes, (512, 512))
    d = B.denorm_boxes(n, (512, 512))
    np.testing.assert_array_equal(d, boxes.astype(np.int32))
    # torch parity
    n_t = B.norm_boxes_t(torch.from_numpy(boxes), (512, 512))
    np.testing.assert_allclose(n_t.numpy(), n, atol=1e-6)


def test_nms_numpy():
    boxes = np.array([[0, 0, 10, 10], [1, 1, 11, 11], [20, 20, 30, 30]],
                     dtype=np.float32)
    scores = np.array([0.9, 0.8, 0.7])
    keep = B.non_max_suppression(boxes, scores, 0.3)
    assert sorted(keep.tolist()) == [0, 2]


def test_clip_boxes_t():
    boxes = torch.tensor([[-0.5, 0.2, 1.5, 0.8]])
    window = torch.tensor([0., 0., 1., 1.])
    out = B.clip_boxes_t(boxes, window)
    assert out.min() >= 0 and out.max() <= 1


def test_batch_slice_parity():
    """utils.batch_slice applies fn per item and restacks (reference
    mrcnn/utils.py:~820)."""
    import torch
    from maskrcnn_amd.utils import batch_slice
    x = torch.arange(12.0).reshape(3, 4)
    y = torch.arange(6.0).reshape(3, 2)
    out = batch_slice([x, y], lambda a, b: a.sum() + b.sum(), 3)
    assert out.shape == (3,)
    assert torch.equal(out, x.sum(1) + y.sum(1))
    a, b = batch_slice(x, lambda v: (v * 2, v + 1), 2)
    assert a.shape == (2, 4) and torch.equal(a, x[:2] * 2)
    assert torch.equal(b, x[:2] + 1)

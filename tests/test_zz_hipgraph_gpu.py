"""hipGraph serving capture test — kept in a file that sorts LAST:
an aborted stream capture can poison the CUDA context, and this test
must not take the rest of the GPU suite down with it."""

import numpy as np
import pytest
import torch


@pytest.mark.gpu
def test_engine_hipgraph_matches_eager(shapes_config, monkeypatch):
    """hipGraph replay of the serving forward must equal eager output
    for the same input (engine._forward capture/replay path).

    The eager engine is pinned to the SAME sync-free roi/detection code
    the capture uses (capture auto-selects it), so replay matches eager
    to float noise (not bitwise: MIOpen split-K solvers accumulate with
    atomics whose order varies run-to-run — measured 2e-4)."""
    from maskrcnn_amd.serving.engine import InferenceEngine
    monkeypatch.setenv("MASKRCNN_SYNCFREE_ROI", "1")
    monkeypatch.setenv("MASKRCNN_SYNCFREE_DETECTION", "1")
    torch.manual_seed(0)
    eager = InferenceEngine(shapes_config)
    monkeypatch.setenv("MASKRCNN_SERVING_HIPGRAPH", "1")
    graphed = InferenceEngine(shapes_config)
    graphed.model.load_state_dict(eager.model.state_dict())
    assert graphed.use_hip_graph
    H = shapes_config.IMAGE_SHAPE[0]
    metas = np.zeros((1, shapes_config.IMAGE_META_SIZE), np.float32)
    metas[:, 4:7] = (H, H, 3)
    metas[:, 7:11] = (0, 0, H, H)
    for trial in range(2):  # second input exercises replay, not capture
        images = (np.random.rand(1, H, H, 3).astype(np.float32) * 50 +
                  trial)
        o_e = eager.predict(images, metas)
        o_g = graphed.predict(images, metas)
        # Selection-free outputs (dense over all anchors) must agree to
        # float noise. Outputs DOWNSTREAM of top-k/NMS (rois, heads,
        # detections) are selection-sensitive: the same 2e-4 MIOpen
        # atomic noise can flip score ties and swap whole rows, so they
        # get a set-level check instead of element order.
        np.testing.assert_allclose(o_g["rpn_class"], o_e["rpn_class"],
                                   atol=5e-3, rtol=5e-3)
        np.testing.assert_allclose(o_g["rpn_bbox"], o_e["rpn_bbox"],
                                   atol=5e-2, rtol=5e-2)
        for k in o_e:
            assert np.isfinite(o_g[k]).all(), k
            assert o_g[k].shape == o_e[k].shape, k
        de = o_e["mrcnn_detection"][0]
        dg = o_g["mrcnn_detection"][0]
        ne = int((de[:, 5] > 0).sum())
        ng = int((dg[:, 5] > 0).sum())
        assert abs(ne - ng) <= max(2, int(0.2 * max(ne, ng, 1))), (ne, ng)
        if ne and ng:  # most eager boxes have a close replay match
            be, bg = de[:ne, :4], dg[:ng, :4]
            y1 = np.maximum(be[:, None, 0], bg[None, :, 0])
            x1 = np.maximum(be[:, None, 1], bg[None, :, 1])
            y2 = np.minimum(be[:, None, 2], bg[None, :, 2])
            x2 = np.minimum(be[:, None, 3], bg[None, :, 3])
            inter = np.clip(y2 - y1, 0, None) * np.clip(x2 - x1, 0, None)
            ae = (be[:, 2] - be[:, 0]) * (be[:, 3] - be[:, 1])
            ag = (bg[:, 2] - bg[:, 0]) * (bg[:, 3] - bg[:, 1])
            iou = inter / np.maximum(ae[:, None] + ag[None, :] - inter,
                                     1e-9)
            matched = (iou.max(axis=1) > 0.75).mean()
            assert matched >= 0.7, (matched, ne, ng)
    assert len(graphed._graphs) == 1


@pytest.mark.gpu
def test_engine_hipgraph_close_to_gather_path(shapes_config,
                                              monkeypatch):
    """Replay vs the default eager gather path: same detections within
    bf16 tolerance (the sync-free mask-combine ROI pooling rounds
    differently from the per-level gather)."""
    from maskrcnn_amd.serving.engine import InferenceEngine
    torch.manual_seed(0)
    eager = InferenceEngine(shapes_config)
    monkeypatch.setenv("MASKRCNN_SERVING_HIPGRAPH", "1")
    graphed = InferenceEngine(shapes_config)
    graphed.model.load_state_dict(eager.model.state_dict())
    H = shapes_config.IMAGE_SHAPE[0]
    metas = np.zeros((1, shapes_config.IMAGE_META_SIZE), np.float32)
    metas[:, 4:7] = (H, H, 3)
    metas[:, 7:11] = (0, 0, H, H)
    images = np.random.rand(1, H, H, 3).astype(np.float32) * 50
    o_e = eager.predict(images, metas)
    o_g = graphed.predict(images, metas)
    # Box geometry of the surviving detections agrees closely; scores
    # can reorder at bf16 ties, so compare the sorted box coordinates.
    det_e = np.sort(o_e["mrcnn_detection"][0, :, :4], axis=0)
    det_g = np.sort(o_g["mrcnn_detection"][0, :, :4], axis=0)
    np.testing.assert_allclose(det_g, det_e, atol=5e-3)

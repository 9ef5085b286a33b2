"""run_graph probe + batched-vs-loop equivalence + roi level routing."""

import numpy as np
import pytest
import torch

from maskrcnn_amd.model import MaskRCNN


def test_run_graph_outputs_and_modules(shapes_config):
    m = MaskRCNN("inference", shapes_config, model_dir="/tmp/probe_logs")
    img = (np.random.rand(128, 128, 3) * 255).astype(np.uint8)
    out = m.run_graph([img], ["rois", "mrcnn_detection", "fpn",
                              "backbone.layer3"])
    assert out["rois"].shape == (1, shapes_config.POST_NMS_ROIS_INFERENCE,
                                 4)
    assert out["mrcnn_detection"].shape[2] == 6
    assert isinstance(out["fpn"], list) and len(out["fpn"]) == 5
    assert out["backbone.layer3"].shape[1] == 512


def test_run_graph_unknown_name_raises(shapes_config):
    m = MaskRCNN("inference", shapes_config, model_dir="/tmp/probe_logs")
    img = (np.random.rand(128, 128, 3) * 255).astype(np.uint8)
    with pytest.raises(KeyError):
        m.run_graph([img], ["nonexistent_tensor"])


def test_detection_layer_batched_matches_single():
    """The batched DetectionLayer must equal per-image evaluation."""
    from maskrcnn_amd.data.synthetic import ShapesConfig
    from maskrcnn_amd.model.detection import DetectionLayer
    from maskrcnn_amd.utils.meta import compose_image_meta

    class C(ShapesConfig):
        DETECTION_MIN_CONFIDENCE = 0.3
        DETECTION_MAX_INSTANCES = 8

    cfg = C()
    torch.manual_seed(0)
    layer = DetectionLayer(cfg)
    B, N, Ccls = 3, 40, cfg.NUM_CLASSES
    rois = torch.rand(B, N, 4)
    rois[..., 2:] = rois[..., :2] * 0.5 + 0.4
    probs = torch.softmax(torch.randn(B, N, Ccls) * 2, dim=-1)
    deltas = torch.randn(B, N, Ccls, 4) * 0.1
    metas = torch.stack([
        torch.from_numpy(compose_image_meta(
            i, (128, 128, 3), (128, 128, 3), (0, 0, 128, 128), 1.0,
            np.ones(Ccls))).float()
        for i in range(B)])

    batched = layer(rois, probs, deltas, metas)
    for b in range(B):
        single = layer(rois[b:b + 1], probs[b:b + 1], deltas[b:b + 1],
                       metas[b:b + 1])
        # Same detections (order within equal scores may differ; compare
        # as sorted-by-score sets).
        bb = batched[b]
        ss = single[0]
        nb = int((bb[:, 5] > 0).sum())
        ns = int((ss[:, 5] > 0).sum())
        assert nb == ns
        torch.testing.assert_close(
            bb[:nb].sort(dim=0).values, ss[:ns].sort(dim=0).values,
            atol=1e-5, rtol=1e-5)


def test_proposal_layer_batched_matches_loop():
    from maskrcnn_amd.data.synthetic import ShapesConfig
    from maskrcnn_amd.model.proposal import ProposalLayer
    from maskrcnn_amd.ops import nms
    from maskrcnn_amd.utils.boxes import apply_box_deltas_t, clip_boxes_t

    class C(ShapesConfig):
        PRE_NMS_LIMIT = 100
        RPN_NMS_THRESHOLD = 0.7

    cfg = C()
    torch.manual_seed(1)
    layer = ProposalLayer(cfg)
    B, A = 2, 300
    probs = torch.softmax(torch.randn(B, A, 2), dim=-1)
    deltas = torch.randn(B, A, 4) * 0.2
    anchors = torch.rand(A, 4)
    anchors[:, 2:] = anchors[:, :2] + 0.1
    count = 20
    rois = layer(probs, deltas, anchors, count)

    # Reference per-image loop.
    scores = probs[:, :, 1]
    d = deltas * torch.tensor(cfg.RPN_BBOX_STD_DEV)
    for b in range(B):
        top_s, ix = torch.topk(scores[b], cfg.PRE_NMS_LIMIT, sorted=True)
        boxes = apply_box_deltas_t(anchors[ix], d[b, ix])
        boxes = clip_boxes_t(boxes, torch.tensor([0., 0., 1., 1.]))
        keep = nms(boxes, top_s, cfg.RPN_NMS_THRESHOLD, max_out=count)
        expect = boxes[keep]
        got = rois[b, :keep.shape[0]]
        torch.testing.assert_close(got, expect, atol=1e-6, rtol=1e-6)


def test_pyramid_roi_align_level_routing(shapes_config):
    """Small rois sample P2, large rois sample P5 (reference level
    assignment, model.py:~360)."""
    from maskrcnn_amd.model.roialign import PyramidROIAlign
    layer = PyramidROIAlign(7, [1024, 1024, 3])
    B, C = 1, 8
    feats = []
    for i, s in enumerate([256, 128, 64, 32]):
        f = torch.full((B, C, s, s), float(i + 1))
        feats.append(f)
    rois = torch.tensor([[[0.0, 0.0, 0.02, 0.02],    # tiny -> P2 (=1)
                          [0.0, 0.0, 0.9, 0.9]]])    # huge -> P5 (=4)
    out = layer(rois, feats)
    assert out.shape == (2, C, 7, 7)
    assert torch.allclose(out[0], torch.ones_like(out[0]))
    assert torch.allclose(out[1], 4 * torch.ones_like(out[1]))


def test_pyramid_roi_align_half_octave_rounding(shapes_config):
    """Level assignment uses 4 + round(log2(scale)) like the reference's
    tf.round (model.py:~360), NOT floor: a 224px roi at 1024^2 maps to
    P4 while a 340px roi (log2(340/224)=0.60) rounds UP to P5."""
    from maskrcnn_amd.model.roialign import PyramidROIAlign
    layer = PyramidROIAlign(7, [1024, 1024, 3])
    B, C = 1, 4
    feats = [torch.full((B, C, s, s), float(i + 1))
             for i, s in enumerate([256, 128, 64, 32])]
    rois = torch.tensor([[[0.0, 0.0, 224 / 1024, 224 / 1024],  # -> P4
                          [0.0, 0.0, 340 / 1024, 340 / 1024],  # -> P5
                          [0.0, 0.0, 150 / 1024, 150 / 1024]]])  # -> P3
    out = layer(rois, feats)
    assert torch.allclose(out[0], 3 * torch.ones_like(out[0]))  # P4
    assert torch.allclose(out[1], 4 * torch.ones_like(out[1]))  # P5
    assert torch.allclose(out[2], 2 * torch.ones_like(out[2]))  # P3


def test_classifier_head_linear_equals_conv(shapes_config):
    """The head's F.linear evaluation must equal the reference conv
    formulation (7x7 VALID conv == dense over the pooled tile)."""
    import torch.nn.functional as F
    from maskrcnn_amd.model.heads import FPNClassifierHead
    torch.manual_seed(0)
    head = FPNClassifierHead(shapes_config)
    x = torch.randn(6, 256, shapes_config.POOL_SIZE,
                    shapes_config.POOL_SIZE)
    # Reference formulation.
    ref = F.relu(head.mrcnn_class_bn1(head.mrcnn_class_conv1(x)))
    ref = F.relu(head.mrcnn_class_bn2(head.mrcnn_class_conv2(ref)))
    ref = ref.flatten(1)
    # Linear formulation (what forward uses).
    w1 = head.mrcnn_class_conv1.weight
    y = F.linear(x.reshape(6, -1), w1.reshape(w1.shape[0], -1),
                 head.mrcnn_class_conv1.bias)
    y = F.relu(head._bn_vec(head.mrcnn_class_bn1, y))
    w2 = head.mrcnn_class_conv2.weight
    y = F.linear(y, w2.reshape(w2.shape[0], -1),
                 head.mrcnn_class_conv2.bias)
    y = F.relu(head._bn_vec(head.mrcnn_class_bn2, y))
    torch.testing.assert_close(y, ref, atol=1e-4, rtol=1e-4)


def test_detection_syncfree_matches_eager(monkeypatch):
    """The fixed-shape capture-safe DetectionLayer path must produce the
    same detections as the nonzero/gather eager path."""
    from maskrcnn_amd.data.synthetic import ShapesConfig
    from maskrcnn_amd.model.detection import DetectionLayer
    from maskrcnn_amd.utils.meta import compose_image_meta

    class C(ShapesConfig):
        DETECTION_MIN_CONFIDENCE = 0.3
        DETECTION_MAX_INSTANCES = 8

    cfg = C()
    torch.manual_seed(3)
    layer = DetectionLayer(cfg)
    B, N, Ccls = 2, 60, cfg.NUM_CLASSES
    rois = torch.rand(B, N, 4)
    rois[..., 2:] = rois[..., :2] * 0.5 + 0.4
    probs = torch.softmax(torch.randn(B, N, Ccls) * 2, dim=-1)
    deltas = torch.randn(B, N, Ccls, 4) * 0.1
    metas = torch.stack([
        torch.from_numpy(compose_image_meta(
            i, (128, 128, 3), (128, 128, 3), (0, 0, 128, 128), 1.0,
            np.ones(Ccls))).float()
        for i in range(B)])

    monkeypatch.setenv("MASKRCNN_SYNCFREE_DETECTION", "0")
    eager = layer(rois, probs, deltas, metas)
    monkeypatch.setenv("MASKRCNN_SYNCFREE_DETECTION", "1")
    syncfree = layer(rois, probs, deltas, metas)
    for b in range(B):
        ne = int((eager[b][:, 5] > 0).sum())
        ns = int((syncfree[b][:, 5] > 0).sum())
        assert ne == ns
        torch.testing.assert_close(
            syncfree[b][:ns].sort(dim=0).values,
            eager[b][:ne].sort(dim=0).values, atol=1e-6, rtol=1e-6)


def test_roialign_syncfree_matches_eager(monkeypatch):
    from maskrcnn_amd.model.roialign import PyramidROIAlign
    torch.manual_seed(0)
    layer = PyramidROIAlign(7, [1024, 1024, 3])
    feats = [torch.randn(2, 8, s, s) for s in (256, 128, 64, 32)]
    rois = torch.rand(2, 12, 4)
    rois[..., 2:] = rois[..., :2] + torch.rand(2, 12, 2) * 0.5
    rois[:, -2:] = 0  # padding rows
    monkeypatch.setenv("MASKRCNN_SYNCFREE_ROI", "0")
    eager = layer(rois, feats)
    monkeypatch.setenv("MASKRCNN_SYNCFREE_ROI", "1")
    syncfree = layer(rois, feats)
    torch.testing.assert_close(syncfree, eager, atol=1e-5, rtol=1e-5)


def test_full_inference_syncfree_matches_default(shapes_config,
                                                 monkeypatch):
    """Whole-model inference with BOTH capture-safe paths forced equals
    the default eager forward (the hipGraph-captured graph computes
    exactly this composition)."""
    import numpy as np

    m = MaskRCNN("inference", shapes_config, model_dir="/tmp/sf_logs")
    img = (np.random.RandomState(5).rand(120, 100, 3) * 255) \
        .astype(np.uint8)
    monkeypatch.setenv("MASKRCNN_SYNCFREE_DETECTION", "0")
    monkeypatch.setenv("MASKRCNN_SYNCFREE_ROI", "0")
    base = m.detect([img])[0]
    monkeypatch.setenv("MASKRCNN_SYNCFREE_DETECTION", "1")
    monkeypatch.setenv("MASKRCNN_SYNCFREE_ROI", "1")
    sf = m.detect([img])[0]
    assert base["rois"].shape == sf["rois"].shape
    np.testing.assert_allclose(sf["rois"], base["rois"], atol=1)
    np.testing.assert_array_equal(sf["class_ids"], base["class_ids"])
    np.testing.assert_allclose(sf["scores"], base["scores"], atol=1e-5)
    np.testing.assert_array_equal(sf["masks"], base["masks"])

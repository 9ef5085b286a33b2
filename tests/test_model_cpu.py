"""End-to-end CPU model tests (BASELINE config #1 class: plumbing)."""

import numpy as np
import torch

from maskrcnn_amd.model import MaskRCNN
from maskrcnn_amd.data.loader import make_data_loader
from maskrcnn_amd.engine.trainer import Trainer, build_optimizer


def test_inference_forward_shapes(shapes_config):
    cfg = shapes_config
    m = MaskRCNN("inference", cfg, model_dir="/tmp/mrcnn_test_logs")
    img = (np.random.rand(100, 150, 3) * 255).astype(np.uint8)
    results = m.detect([img])
    r = results[0]
    assert r["rois"].ndim == 2 and r["rois"].shape[1] == 4
    assert r["class_ids"].shape[0] == r["rois"].shape[0]
    assert r["scores"].shape[0] == r["rois"].shape[0]
    assert r["masks"].shape[:2] == img.shape[:2]


def test_inference_output_signature(shapes_config):
    """The 7 named outputs of the serving signature exist with right shapes."""
    cfg = shapes_config
    m = MaskRCNN("inference", cfg, model_dir="/tmp/mrcnn_test_logs")
    img = (np.random.rand(128, 128, 3) * 255).astype(np.uint8)
    molded, metas, _ = m.mold_inputs([img])
    out = m.predict_molded(
        molded, metas,
        m.anchor_cache.get(molded[0].shape, device=m.device))
    D = cfg.DETECTION_MAX_INSTANCES
    C = cfg.NUM_CLASSES
    N = cfg.POST_NMS_ROIS_INFERENCE
    assert out["mrcnn_detection"].shape == (1, D, 6)
    assert out["mrcnn_mask"].shape == (1, D, C, 28, 28)
    assert out["rois"].shape == (1, N, 4)
    assert out["mrcnn_class"].shape == (1, N, C)
    assert out["mrcnn_bbox"].shape == (1, N, C, 4)
    assert out["rpn_class"].shape[0] == 1
    assert out["rpn_bbox"].shape[2] == 4


def test_train_step_runs_and_losses_finite(shapes_config, shapes_dataset):
    cfg = shapes_config
    m = MaskRCNN("training", cfg, model_dir="/tmp/mrcnn_test_logs")
    tr = Trainer(m, shapes_dataset)
    opt = build_optimizer(m.model, cfg, cfg.LEARNING_RATE)
    loader = make_data_loader(shapes_dataset, cfg)
    batch = next(iter(loader))
    losses = tr.train_step(batch, opt)
    for k, v in losses.items():
        assert np.isfinite(v), (k, v)
    assert set(losses) == {"rpn_class_loss", "rpn_bbox_loss",
                           "mrcnn_class_loss", "mrcnn_bbox_loss",
                           "mrcnn_mask_loss", "total_loss"}


def test_loss_decreases_overfit_single_batch(shapes_config, shapes_dataset):
    """Overfit one batch: total loss must drop measurably in 15 steps."""
    cfg = shapes_config
    torch.manual_seed(0)
    m = MaskRCNN("training", cfg, model_dir="/tmp/mrcnn_test_logs")
    tr = Trainer(m, shapes_dataset)
    opt = build_optimizer(m.model, cfg, 0.02)
    loader = make_data_loader(shapes_dataset, cfg, shuffle=False)
    batch = next(iter(loader))
    first = tr.train_step(batch, opt)["total_loss"]
    last = first
    for _ in range(15):
        last = tr.train_step(batch, opt)["total_loss"]
    assert last < first * 0.9, (first, last)


def test_set_trainable_freezes_backbone(shapes_config):
    m = MaskRCNN("training", shapes_config, model_dir="/tmp/mrcnn_test_logs")
    m.set_trainable("heads")
    for name, p in m.model.named_parameters():
        if name.startswith("backbone."):
            assert not p.requires_grad, name
        if name.startswith("rpn.") or name.startswith("fpn."):
            assert p.requires_grad, name
    n_all = m.set_trainable("all")
    assert n_all == sum(1 for _ in m.model.parameters())


def test_checkpoint_save_load_roundtrip(shapes_config, tmp_path):
    m = MaskRCNN("training", shapes_config, model_dir=str(tmp_path))
    p = str(tmp_path / "w.pth")
    m.save_weights(p)
    m2 = MaskRCNN("inference", shapes_config, model_dir=str(tmp_path))
    m2.load_weights(p)
    for (k1, v1), (k2, v2) in zip(m.model.state_dict().items(),
                                  m2.model.state_dict().items()):
        assert k1 == k2
        assert torch.equal(v1, v2), k1


def test_load_weights_exclude_head_surgery(shapes_config, tmp_path):
    m = MaskRCNN("training", shapes_config, model_dir=str(tmp_path))
    p = str(tmp_path / "w.pth")
    m.save_weights(p)

    class MoreClasses(type(shapes_config)):
        NUM_CLASSES = 7
    m2 = MaskRCNN("training", MoreClasses(), model_dir=str(tmp_path))
    m2.load_weights(p, by_name=True,
                    exclude=[r"classifier_head\.mrcnn_class_logits",
                             r"classifier_head\.mrcnn_bbox_fc",
                             r"mask_head\.mrcnn_mask\."])
    # Backbone weights transferred.
    assert torch.equal(
        m2.model.backbone.conv1.weight,
        m.model.backbone.conv1.weight)


def test_optimizer_state_resume(shapes_config, shapes_dataset, tmp_path):
    """Epoch checkpoints save SGD momentum beside the weights; a fresh
    MaskRCNN resumed via load_weights(find_last()) restores it."""
    import os

    from maskrcnn_amd.engine.trainer import Trainer
    from maskrcnn_amd.model import MaskRCNN

    cfg = shapes_config
    m = MaskRCNN("training", cfg, model_dir=str(tmp_path))
    tr = Trainer(m, shapes_dataset)
    tr.fit(cfg.LEARNING_RATE, epochs=1, layers="heads",
           steps_per_epoch=1)
    state_path = os.path.join(m.log_dir, "train_state.pth")
    assert os.path.exists(state_path)
    assert not tr._resumed_opt

    m2 = MaskRCNN("training", cfg, model_dir=str(tmp_path))
    last = m2.find_last()
    m2.load_weights(last, by_name=True)
    assert m2.epoch == 1
    tr2 = Trainer(m2, shapes_dataset)
    tr2.fit(cfg.LEARNING_RATE, epochs=2, layers="heads",
            steps_per_epoch=1)
    assert tr2._resumed_opt


def test_detect_pad64_nonsquare(shapes_config, tmp_path):
    """End to end with IMAGE_RESIZE_MODE='pad64' (the nucleus-sample
    mode): non-square input molds to a 64-multiple non-square shape;
    anchors, meta and unmold must all follow."""
    from maskrcnn_amd.model import MaskRCNN

    class Pad64Config(type(shapes_config)):
        IMAGE_RESIZE_MODE = "pad64"
        IMAGE_MIN_DIM = 64
        # IMAGE_SHAPE (anchors default / model assert) still derives
        # from IMAGE_MAX_DIM, exactly like reference config.py; pad64
        # molding itself ignores the max.
        IMAGE_MAX_DIM = 256
        IMAGE_MIN_SCALE = 0

    cfg = Pad64Config()
    m = MaskRCNN("inference", cfg, model_dir=str(tmp_path))
    img = (np.random.rand(100, 160, 3) * 255).astype(np.uint8)
    r = m.detect([img])[0]
    assert r["masks"].shape[:2] == (100, 160)
    assert r["rois"].shape[1] == 4
    # Molded shape must be 64-multiples (192x256... wait 100->128, 160->192).
    molded, _, windows = m.mold_inputs([img])
    assert molded.shape[1] % 64 == 0 and molded.shape[2] % 64 == 0
    assert molded.shape[1] != molded.shape[2]  # stayed non-square


def test_detect_molded_returns_unmolded_results(shapes_config):
    """detect_molded matches the reference contract: a list of result
    dicts (rois/class_ids/scores/masks), same as detect()."""
    m = MaskRCNN("inference", shapes_config, model_dir="/tmp/dm_logs")
    img = (np.random.RandomState(2).rand(100, 90, 3) * 255) \
        .astype(np.uint8)
    molded, metas, _ = m.mold_inputs([img])
    res = m.detect_molded(molded, metas)
    via_detect = m.detect([img])
    assert isinstance(res, list) and len(res) == 1
    r = res[0]
    assert set(r) == {"rois", "class_ids", "scores", "masks"}
    assert r["masks"].shape[:2] == img.shape[:2]
    np.testing.assert_allclose(r["rois"], via_detect[0]["rois"])

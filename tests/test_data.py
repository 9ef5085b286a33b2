"""Data-plane tests: dataset, resize modes, RPN targets, loader."""

import numpy as np
import pytest

from maskrcnn_amd.data.loader import make_data_loader
from maskrcnn_amd.data.targets import build_rpn_targets, load_image_gt
from maskrcnn_amd.utils.image import resize_image
from maskrcnn_amd.utils.masks import (expand_mask, minimize_mask, rle_decode,
                                      rle_encode, rle_iou, unmold_mask)


def test_resize_square_mode():
    img = np.random.randint(0, 255, (300, 500, 3), dtype=np.uint8)
    out, window, scale, padding, crop = resize_image(
        img, min_dim=256, max_dim=256, mode="square")
    assert out.shape == (256, 256, 3)
    y1, x1, y2, x2 = window
    # Aspect preserved: 300x500 -> scale 256/500 -> 154x256 content.
    assert x2 - x1 == 256
    assert abs((y2 - y1) - round(300 * 256 / 500)) <= 1


def test_resize_pad64_mode():
    img = np.random.randint(0, 255, (100, 200, 3), dtype=np.uint8)
    out, window, scale, padding, crop = resize_image(
        img, min_dim=128, mode="pad64")
    assert out.shape[0] % 64 == 0 and out.shape[1] % 64 == 0


def test_resize_crop_mode():
    img = np.random.randint(0, 255, (300, 400, 3), dtype=np.uint8)
    out, window, scale, padding, crop = resize_image(
        img, min_dim=128, mode="crop")
    assert out.shape == (128, 128, 3)
    assert crop is not None


def test_mini_mask_roundtrip():
    mask = np.zeros((64, 64, 1), dtype=bool)
    mask[10:40, 20:50, 0] = True
    from maskrcnn_amd.utils.boxes import extract_bboxes
    bbox = extract_bboxes(mask)
    mini = minimize_mask(bbox, mask, (28, 28))
    assert mini.shape == (28, 28, 1)
    back = expand_mask(bbox, mini, (64, 64))
    inter = np.logical_and(back[..., 0], mask[..., 0]).sum()
    union = np.logical_or(back[..., 0], mask[..., 0]).sum()
    assert inter / union > 0.9


def test_unmold_mask():
    small = np.random.rand(28, 28)
    full = unmold_mask(small, np.array([10, 10, 30, 40]), (100, 100, 3))
    assert full.shape == (100, 100)
    assert not full[:10].any() and not full[:, :10].any()


def test_rle_roundtrip():
    rng = np.random.RandomState(0)
    mask = rng.rand(37, 23) > 0.5
    rle = rle_encode(mask)
    back = rle_decode(rle)
    np.testing.assert_array_equal(back, mask)
    assert rle_iou(rle, rle) == pytest.approx(1.0)


def test_build_rpn_targets(shapes_config, shapes_dataset):
    cfg = shapes_config
    from maskrcnn_amd.utils.anchors import AnchorCache
    anchors = AnchorCache(cfg).get_pixel(cfg.IMAGE_SHAPE)
    image, meta, class_ids, bbox, mask = load_image_gt(
        shapes_dataset, cfg, 0)
    rpn_match, rpn_bbox = build_rpn_targets(
        image.shape, anchors, class_ids, bbox, cfg)
    assert rpn_match.shape == (anchors.shape[0],)
    n_pos = (rpn_match == 1).sum()
    n_neg = (rpn_match == -1).sum()
    assert n_pos > 0
    assert n_pos + n_neg <= cfg.RPN_TRAIN_ANCHORS_PER_IMAGE
    # Deltas for positives only, finite.
    assert np.isfinite(rpn_bbox[:n_pos]).all()


def test_loader_batch_shapes(shapes_config, shapes_dataset):
    cfg = shapes_config
    loader = make_data_loader(shapes_dataset, cfg, batch_size=2)
    batch = next(iter(loader))
    H, W, _ = cfg.IMAGE_SHAPE
    A = 0
    from maskrcnn_amd.config import compute_backbone_shapes
    for s, fs in zip(cfg.RPN_ANCHOR_SCALES,
                     compute_backbone_shapes(cfg, cfg.IMAGE_SHAPE)):
        A += fs[0] * fs[1] * len(cfg.RPN_ANCHOR_RATIOS)
    assert batch["image"].shape == (2, 3, H, W)
    assert batch["image_meta"].shape == (2, cfg.IMAGE_META_SIZE)
    assert batch["rpn_match"].shape == (2, A, 1)
    assert batch["rpn_bbox"].shape == (2, cfg.RPN_TRAIN_ANCHORS_PER_IMAGE, 4)
    assert batch["gt_class_ids"].shape == (2, cfg.MAX_GT_INSTANCES)
    assert batch["gt_boxes"].shape == (2, cfg.MAX_GT_INSTANCES, 4)
    mh, mw = cfg.MINI_MASK_SHAPE
    assert batch["gt_masks"].shape == (2, mh, mw, cfg.MAX_GT_INSTANCES)


def test_dataset_prepare_class_mapping(shapes_dataset):
    ds = shapes_dataset
    assert ds.num_classes == 4
    assert ds.class_names[0] == "BG"
    assert ds.map_source_class_id("shapes.1") == 1


def test_loader_crop_mode_end_to_end():
    """IMAGE_RESIZE_MODE='crop' (nucleus-style training): samples crop
    to IMAGE_MIN_DIM squares with consistent targets."""
    import torch

    from maskrcnn_amd.data.loader import TorchMaskRCNNDataset
    from maskrcnn_amd.data.synthetic import ShapesConfig, ShapesDataset

    class CropConfig(ShapesConfig):
        IMAGE_RESIZE_MODE = "crop"
        IMAGE_MIN_DIM = 64
        IMAGE_MAX_DIM = None
        IMAGE_MIN_SCALE = 0
        USE_MINI_MASK = True

    cfg = CropConfig()
    ds = ShapesDataset()
    ds.load_shapes(4, 128, 128, seed=3)
    ds.prepare()
    tds = TorchMaskRCNNDataset(ds, cfg)
    s = tds[0]
    assert s["image"].shape == (3, 64, 64)
    assert s["gt_masks"].shape == (*cfg.MINI_MASK_SHAPE,
                                   cfg.MAX_GT_INSTANCES)
    assert torch.isfinite(s["rpn_bbox"]).all()


def test_balloon_color_splash_and_via_loading(tmp_path):
    """balloon sample: color_splash math + VIA-json dataset loading
    (reference samples/balloon parity)."""
    import importlib.util
    import json as _json
    import os

    import numpy as np

    spec = importlib.util.spec_from_file_location(
        "balloon", os.path.join(os.path.dirname(__file__), "..",
                                "samples", "balloon.py"))
    balloon = importlib.util.module_from_spec(spec)
    spec.loader.exec_module(balloon)

    img = (np.random.rand(20, 20, 3) * 255).astype(np.uint8)
    mask = np.zeros((20, 20, 1), bool)
    mask[5:10, 5:10, 0] = True
    out = balloon.color_splash(img, mask)
    assert out.shape == img.shape
    np.testing.assert_array_equal(out[6, 6], img[6, 6])   # kept color
    assert not np.array_equal(out[0, 0], img[0, 0]) or \
        img[0, 0, 0] == img[0, 0, 1] == img[0, 0, 2]      # grayed
    # No-mask case -> fully gray.
    out2 = balloon.color_splash(img, np.zeros((20, 20, 0), bool))
    assert (out2[..., 0] == out2[..., 1]).all()

    # Minimal VIA dataset: one image with one polygon region.
    sub = tmp_path / "train"
    sub.mkdir(parents=True)
    from PIL import Image
    Image.fromarray(img).save(sub / "img0.png")
    via = {"x": {"filename": "img0.png",
                 "regions": {"0": {"shape_attributes": {
                     "name": "polygon",
                     "all_points_x": [5, 15, 15, 5],
                     "all_points_y": [5, 5, 15, 15]}}}}}
    (sub / "via_region_data.json").write_text(_json.dumps(via))
    ds = balloon.BalloonDataset()
    ds.load_balloon(str(tmp_path), "train")
    ds.prepare()
    assert ds.num_images == 1
    masks, cids = ds.load_mask(0)
    assert masks.shape == (20, 20, 1) and cids.tolist() == [1]
    assert masks[:, :, 0].sum() > 50  # the polygon filled


def test_nucleus_dataset_loading(tmp_path):
    """nucleus sample: DSB2018 per-instance mask-dir layout."""
    import importlib.util
    import os

    import numpy as np
    from PIL import Image

    spec = importlib.util.spec_from_file_location(
        "nucleus", os.path.join(os.path.dirname(__file__), "..",
                                "samples", "nucleus.py"))
    nucleus = importlib.util.module_from_spec(spec)
    spec.loader.exec_module(nucleus)

    iid = "img_a"
    (tmp_path / iid / "images").mkdir(parents=True)
    (tmp_path / iid / "masks").mkdir()
    img = (np.random.rand(32, 32, 3) * 255).astype(np.uint8)
    Image.fromarray(img).save(tmp_path / iid / "images" / f"{iid}.png")
    for k in range(2):
        m = np.zeros((32, 32), np.uint8)
        m[4 + 10 * k:9 + 10 * k, 4:9] = 255
        Image.fromarray(m).save(tmp_path / iid / "masks" / f"m{k}.png")

    ds = nucleus.NucleusDataset()
    ds.load_nucleus(str(tmp_path))
    ds.prepare()
    assert ds.num_images == 1
    assert ds.load_image(0).shape == (32, 32, 3)
    masks, cids = ds.load_mask(0)
    assert masks.shape == (32, 32, 2)
    assert cids.tolist() == [1, 1]
    assert int(masks[:, :, 0].sum()) == 25


def test_loader_augmentation_varies_across_epochs(tmp_path):
    """set_epoch must change the augmentation stream (same index,
    different epoch -> flips can differ)."""
    import numpy as np

    from maskrcnn_amd.data.loader import TorchMaskRCNNDataset
    from maskrcnn_amd.data.synthetic import ShapesConfig, ShapesDataset

    ds = ShapesDataset()
    ds.load_shapes(2, 128, 128, seed=1)
    ds.prepare()
    tds = TorchMaskRCNNDataset(ds, ShapesConfig(),
                               augmentation="fliplr")
    base = tds[0]["image"].numpy()
    changed = False
    for e in range(1, 6):
        tds.set_epoch(e)
        if not np.array_equal(tds[0]["image"].numpy(), base):
            changed = True
            break
    assert changed, "augmentation identical across 6 epochs"
    # Reproducible: same epoch -> same sample.
    tds.set_epoch(3)
    a = tds[0]["image"].numpy()
    tds.set_epoch(3)
    assert np.array_equal(tds[0]["image"].numpy(), a)


def test_transforms_mask_registration():
    """Geometric transforms keep image and mask registered: a bright
    square under its own mask stays under it after Affine/CropAndPad
    (reference MASK_AUGMENTERS whitelist semantics, model.py:~1260)."""
    from maskrcnn_amd.data import transforms as T
    rng = np.random.RandomState(3)
    H = W = 96
    image = np.zeros((H, W, 3), np.uint8)
    masks = np.zeros((H, W, 1), bool)
    image[30:60, 40:70] = 200
    masks[30:60, 40:70, 0] = True
    aug = T.Sequential([
        T.Affine(rotate=(-25, 25), scale=(0.8, 1.2),
                 translate_frac=(-0.1, 0.1), shear=(-8, 8)),
        T.CropAndPad(percent=(-0.1, 0.1)),
    ])
    for _ in range(5):
        img2, m2 = aug(image, masks, rng)
        assert img2.shape == image.shape and m2.shape == masks.shape
        if m2[..., 0].sum() < 50:
            continue  # transform pushed the square mostly out of frame
        # Bright pixels inside the transformed mask; dark outside.
        inside = img2[m2[..., 0]].mean()
        outside = img2[~m2[..., 0]].mean()
        assert inside > outside + 50, (inside, outside)


def test_transforms_combinators_and_photometric():
    from maskrcnn_amd.data import transforms as T
    rng = np.random.RandomState(4)
    image = (np.random.RandomState(0).rand(64, 64, 3) * 255) \
        .astype(np.uint8)
    masks = np.zeros((64, 64, 2), bool)
    masks[10:30, 10:30, 0] = True
    masks[40:60, 40:60, 1] = True
    aug = T.Sequential([
        T.Sometimes(0.5, T.GaussianBlur(sigma=(0.5, 1.5), p=1.0)),
        T.OneOf([T.Multiply((0.7, 1.3)), T.Add((-30, 30)),
                 T.ContrastNormalization((0.6, 1.4))]),
        T.SomeOf(2, [T.Fliplr(1.0), T.Flipud(1.0), T.Rotate90(1.0)]),
    ])
    for _ in range(8):
        img2, m2 = aug(image, masks, rng)
        assert img2.shape == image.shape
        assert m2.shape == masks.shape
        assert img2.dtype == np.uint8 and m2.dtype == bool
        # Photometric ops must not change mask pixel counts; geometric
        # flips/rot90 preserve them exactly.
        assert m2.sum() == masks.sum()


def test_affine_identity_is_noop():
    from maskrcnn_amd.data import transforms as T
    rng = np.random.RandomState(5)
    image = (np.random.RandomState(1).rand(48, 48, 3) * 255) \
        .astype(np.uint8)
    masks = np.zeros((48, 48, 1), bool)
    masks[8:20, 8:20, 0] = True
    img2, m2 = T.Affine()(image, masks, rng)
    np.testing.assert_array_equal(m2, masks)
    assert np.abs(img2.astype(int) - image.astype(int)).max() <= 1


def test_piecewise_affine_registration():
    """PiecewiseAffine warps image and masks through the same
    displacement field (last reference whitelist member)."""
    from maskrcnn_amd.data import transforms as T
    rng = np.random.RandomState(9)
    H = W = 96
    image = np.zeros((H, W, 3), np.uint8)
    masks = np.zeros((H, W, 1), bool)
    image[30:60, 40:70] = 200
    masks[30:60, 40:70, 0] = True
    aug = T.PiecewiseAffine(scale=(0.02, 0.04))
    moved = False
    for _ in range(5):
        img2, m2 = aug(image, masks, rng)
        assert img2.shape == image.shape and m2.shape == masks.shape
        inside = img2[m2[..., 0]].mean()
        outside = img2[~m2[..., 0]].mean()
        assert inside > outside + 50, (inside, outside)
        if (m2 != masks).any():
            moved = True
    assert moved, "warp never displaced any mask pixel"
    # scale 0 or p=0 -> noop
    img3, m3 = T.PiecewiseAffine(scale=(0.0, 0.0))(image, masks, rng)
    np.testing.assert_array_equal(img3, image)
    img4, m4 = T.PiecewiseAffine(p=0.0)(image, masks, rng)
    np.testing.assert_array_equal(img4, image)

"""Evaluation stack tests: RLE string codec, COCOEvalLite, evaluator."""

import numpy as np
import pytest

from maskrcnn_amd.data.coco import (ann_to_mask, rle_from_string,
                                    rle_to_string)
from maskrcnn_amd.utils.cocoeval import COCOEvalLite
from maskrcnn_amd.utils.masks import rle_decode, rle_encode


def test_coco_rle_string_roundtrip():
    rng = np.random.RandomState(0)
    mask = rng.rand(25, 17) > 0.6
    rle = rle_encode(mask)
    s = rle_to_string(rle)
    back = rle_from_string(s, rle["size"])
    np.testing.assert_array_equal(rle_decode(back), mask)


def test_coco_rle_string_known_value():
    """pycocotools golden: a 3-run RLE encodes to a short ascii string."""
    rle = {"size": [3, 3], "counts": [2, 4, 3]}
    s = rle_to_string(rle)
    assert isinstance(s, str) and len(s) >= 3
    back = rle_from_string(s, [3, 3])
    assert back["counts"] == [2, 4, 3]


def test_ann_to_mask_polygon():
    ann = {"segmentation": [[1, 1, 6, 1, 6, 6, 1, 6]]}  # square
    m = ann_to_mask(ann, 8, 8)
    assert m[3, 3] and not m[0, 0] and not m[7, 7]


def test_ann_to_mask_rle_dict():
    mask = np.zeros((5, 4), dtype=bool)
    mask[1:3, 1:3] = True
    rle = rle_encode(mask)
    m = ann_to_mask({"segmentation": rle}, 5, 4)
    np.testing.assert_array_equal(m, mask)
    # And compressed string form.
    m2 = ann_to_mask({"segmentation": {
        "size": rle["size"], "counts": rle_to_string(rle)}}, 5, 4)
    np.testing.assert_array_equal(m2, mask)


def _ann(img, cat, box, score=None, iscrowd=0):
    x, y, w, h = box
    a = {"image_id": img, "category_id": cat,
         "bbox": [x, y, w, h], "area": w * h, "iscrowd": iscrowd}
    if score is not None:
        a["score"] = score
    return a


def test_cocoeval_perfect_detections():
    gt = [_ann(1, 1, (10, 10, 20, 20)), _ann(1, 2, (50, 50, 30, 10)),
          _ann(2, 1, (0, 0, 15, 15))]
    dt = [_ann(1, 1, (10, 10, 20, 20), 0.9),
          _ann(1, 2, (50, 50, 30, 10), 0.8),
          _ann(2, 1, (0, 0, 15, 15), 0.95)]
    ev = COCOEvalLite(gt, dt, iou_type="bbox")
    stats = ev.evaluate()
    assert stats["AP"] == pytest.approx(1.0, abs=1e-6)
    assert stats["AP50"] == pytest.approx(1.0, abs=1e-6)


def test_cocoeval_misses_and_fps():
    gt = [_ann(1, 1, (10, 10, 20, 20)), _ann(1, 1, (100, 100, 20, 20))]
    # One perfect match and one far-off false positive.
    dt = [_ann(1, 1, (10, 10, 20, 20), 0.9),
          _ann(1, 1, (200, 200, 20, 20), 0.8)]
    stats = COCOEvalLite(gt, dt, iou_type="bbox").evaluate()
    # Recall caps at 0.5, precision 0.5-1.0 -> AP about 0.5.
    assert 0.3 < stats["AP"] < 0.6


def test_cocoeval_localization_quality_affects_high_iou():
    gt = [_ann(1, 1, (10, 10, 20, 20))]
    dt = [_ann(1, 1, (12, 12, 20, 20), 0.9)]  # IoU ~0.68
    stats = COCOEvalLite(gt, dt, iou_type="bbox").evaluate()
    assert stats["AP50"] == pytest.approx(1.0, abs=1e-6)
    assert stats["AP75"] == pytest.approx(0.0, abs=1e-6)


def test_cocoeval_crowd_not_counted_as_fp():
    gt = [_ann(1, 1, (10, 10, 20, 20), iscrowd=1)]
    dt = [_ann(1, 1, (10, 10, 20, 20), 0.9)]
    stats = COCOEvalLite(gt, dt, iou_type="bbox").evaluate()
    # No countable GT -> empty stats, not a penalty.
    assert stats["AP"] == 0.0 and stats["per_class"] == {}


def test_cocoeval_segm():
    m1 = np.zeros((32, 32), dtype=bool)
    m1[4:20, 4:20] = True
    gt = [dict(_ann(1, 1, (4, 4, 16, 16)), segmentation=rle_encode(m1),
               area=float(m1.sum()))]
    dt = [dict(_ann(1, 1, (4, 4, 16, 16), 0.9),
               segmentation=rle_encode(m1))]
    stats = COCOEvalLite(gt, dt, iou_type="segm").evaluate()
    assert stats["AP"] == pytest.approx(1.0, abs=1e-6)


def test_evaluator_on_shapes_smoke(shapes_config, shapes_dataset):
    """End-to-end evaluate_dataset runs and returns the stat dict
    (random weights -> AP can be 0; structure is what's tested)."""
    from maskrcnn_amd.engine.evaluator import evaluate_dataset
    from maskrcnn_amd.model import MaskRCNN
    m = MaskRCNN("inference", shapes_config,
                 model_dir="/tmp/mrcnn_eval_logs")
    out = evaluate_dataset(m, shapes_dataset, limit=2, verbose=0)
    assert set(out) == {"bbox", "segm"}
    for stats in out.values():
        assert set(stats) >= {"AP", "AP50", "AP75", "AR"}


def test_coco_dataset_from_json(tmp_path):
    """CocoDataset loads a minimal instances JSON + image files."""
    import json
    from PIL import Image
    from maskrcnn_amd.data.coco import CocoDataset

    img_dir = tmp_path / "train2017"
    img_dir.mkdir()
    ann_dir = tmp_path / "annotations"
    ann_dir.mkdir()
    Image.new("RGB", (32, 24), (128, 0, 0)).save(img_dir / "000001.jpg")
    data = {
        "images": [{"id": 1, "file_name": "000001.jpg",
                    "width": 32, "height": 24}],
        "categories": [{"id": 7, "name": "widget"},
                       {"id": 9, "name": "gadget"}],
        "annotations": [
            {"id": 11, "image_id": 1, "category_id": 7, "iscrowd": 0,
             "bbox": [2, 2, 10, 8], "area": 80,
             "segmentation": [[2, 2, 12, 2, 12, 10, 2, 10]]},
        ],
    }
    with open(ann_dir / "instances_train2017.json", "w") as f:
        json.dump(data, f)

    ds = CocoDataset()
    ds.load_coco(str(tmp_path), "train", year="2017")
    ds.prepare()
    assert ds.num_classes == 3  # BG + 2
    assert ds.num_images == 1
    image = ds.load_image(0)
    assert image.shape == (24, 32, 3)
    masks, cls = ds.load_mask(0)
    assert masks.shape == (24, 32, 1)
    assert cls.tolist() == [ds.map_source_class_id("coco.7")]
    assert masks[5, 5, 0]


def test_visualize_display_instances(tmp_path, shapes_dataset):
    """display_instances renders to a file without error."""
    from maskrcnn_amd import visualize
    from maskrcnn_amd.utils.boxes import extract_bboxes
    image = shapes_dataset.load_image(0)
    masks, class_ids = shapes_dataset.load_mask(0)
    boxes = extract_bboxes(masks)
    out = str(tmp_path / "vis.png")
    visualize.display_instances(
        image, boxes, masks, class_ids, shapes_dataset.class_names,
        scores=np.ones(len(class_ids)), save_path=out)
    import os
    assert os.path.getsize(out) > 1000


def test_native_rle_matches_python():
    """C++ codec (csrc/rle.cpp) vs the pure-python reference paths."""
    from maskrcnn_amd.ops._ext import load_ext
    ext = load_ext()
    if ext is None or not hasattr(ext, "rle_encode"):
        pytest.skip("extension not built")
    import torch
    rng = np.random.RandomState(7)
    for _ in range(5):
        mask = rng.rand(33, 21) > rng.rand()
        counts = ext.rle_encode(torch.from_numpy(mask)).tolist()
        rle = rle_encode(mask)
        assert counts == rle["counts"]
        back = ext.rle_decode(torch.tensor(counts), 33, 21).numpy()
        np.testing.assert_array_equal(back, mask)


def test_native_rle_iou_matrix():
    from maskrcnn_amd.ops._ext import load_ext
    ext = load_ext()
    if ext is None or not hasattr(ext, "rle_iou_matrix"):
        pytest.skip("extension not built")
    import torch
    rng = np.random.RandomState(8)
    masks_d = [rng.rand(20, 20) > 0.5 for _ in range(3)]
    masks_g = [rng.rand(20, 20) > 0.5 for _ in range(2)]
    dts = [torch.tensor(rle_encode(m)["counts"]) for m in masks_d]
    gts = [torch.tensor(rle_encode(m)["counts"]) for m in masks_g]
    got = ext.rle_iou_matrix(dts, gts,
                             torch.tensor([False, True])).numpy()
    for i, dm in enumerate(masks_d):
        for j, gm in enumerate(masks_g):
            inter = np.logical_and(dm, gm).sum()
            denom = dm.sum() if j == 1 else np.logical_or(dm, gm).sum()
            assert got[i, j] == pytest.approx(inter / max(denom, 1e-10))


def test_visualize_suite_smoke(tmp_path, shapes_dataset):
    """Every public visualize function renders without error (the
    reference's notebook helpers are part of the capability surface)."""
    import matplotlib
    matplotlib.use("Agg")
    import numpy as np

    from maskrcnn_amd import visualize

    ds = shapes_dataset
    image = ds.load_image(0)
    masks, class_ids = ds.load_mask(0)
    from maskrcnn_amd.utils.boxes import extract_bboxes
    boxes = extract_bboxes(masks)
    scores = np.linspace(0.9, 0.5, len(class_ids))

    visualize.display_images([image, image], titles=["a", "b"],
                             save_path=str(tmp_path / "imgs.png"))
    visualize.display_top_masks(image, masks, class_ids, ds.class_names,
                                save_path=str(tmp_path / "top.png"))
    visualize.draw_boxes(image, boxes=boxes, masks=masks,
                         save_path=str(tmp_path / "boxes.png"))
    visualize.draw_box(image.copy(), boxes[0], (255, 0, 0))
    visualize.plot_precision_recall(
        0.5, np.linspace(1, 0, 11), np.linspace(0, 1, 11),
        save_path=str(tmp_path / "pr.png"))
    visualize.plot_overlaps(
        class_ids, class_ids, scores,
        np.eye(len(class_ids)), ds.class_names, threshold=0.5,
        save_path=str(tmp_path / "ov.png"))
    visualize.display_differences(
        image, boxes, class_ids, masks, boxes, class_ids, scores, masks,
        ds.class_names, save_path=str(tmp_path / "diff.png"))
    visualize.display_table([["a", "b"], ["1", "2"]])
    visualize.draw_rois(image, boxes.astype(np.float32),
                        boxes.astype(np.float32), masks, class_ids,
                        ds.class_names, limit=2,
                        save_path=str(tmp_path / "rois.png"))
    colors = visualize.random_colors(5)
    assert len(colors) == 5
    out = visualize.apply_mask(image.astype(np.float32),
                               masks[:, :, 0], colors[0])
    assert out.shape == image.shape
    for f in ("imgs.png", "top.png", "boxes.png", "pr.png", "ov.png",
              "diff.png"):
        assert (tmp_path / f).exists()


def test_display_weight_stats_accepts_facade(shapes_config, tmp_path,
                                             capsys):
    from maskrcnn_amd import visualize
    from maskrcnn_amd.model import MaskRCNN
    m = MaskRCNN("inference", shapes_config, model_dir=str(tmp_path))
    table = visualize.display_weight_stats(m)
    assert len(table) > 100 and table[0][0] == "WEIGHT NAME"

"""Evaluation loop: detect over a Dataset and score with COCOEvalLite.

Replicates reference evaluate_coco (samples/coco/coco.py:~300): batched
model.detect(), COCO-style result dicts, AP summary and the
"Prediction time / Total time" prints.
"""

from __future__ import annotations

import time


from ..utils.boxes import extract_bboxes
from ..utils.cocoeval import COCOEvalLite
from ..utils.masks import rle_encode


def dataset_gt_annotations(dataset, image_ids):
    """Build COCO-style GT annotation dicts from a Dataset."""
    anns = []
    ann_id = 1
    for image_id in image_ids:
        masks, class_ids = dataset.load_mask(image_id)
        boxes = extract_bboxes(masks)
        for i, cid in enumerate(class_ids):
            y1, x1, y2, x2 = boxes[i]
            m = masks[:, :, i]
            anns.append({
                "id": ann_id,
                "image_id": int(image_id),
                "category_id": int(abs(cid)),
                "bbox": [float(x1), float(y1), float(x2 - x1),
                         float(y2 - y1)],
                "area": float(m.sum()),
                "iscrowd": 1 if cid < 0 else 0,
                "segmentation": rle_encode(m),
            })
            ann_id += 1
    return anns


def evaluate_dataset(model, dataset, limit=0, image_ids=None, verbose=1):
    """Run detection + COCO-style eval (bbox and segm).

    model: MaskRCNN facade in inference mode.
    Returns {"bbox": stats, "segm": stats}.
    """
    image_ids = image_ids if image_ids is not None \
        else list(dataset.image_ids)
    if limit:
        image_ids = image_ids[:limit]

    t_prediction = 0.0
    t_start = time.time()
    dt_anns = []
    B = model.config.BATCH_SIZE
    for i in range(0, len(image_ids), B):
        chunk = image_ids[i:i + B]
        images = [dataset.load_image(iid) for iid in chunk]
        while len(images) < B:  # pad the last batch
            images.append(images[-1])
        t = time.time()
        results = model.detect(images, verbose=0)
        t_prediction += time.time() - t
        for iid, r in zip(chunk, results):
            for k in range(r["rois"].shape[0]):
                y1, x1, y2, x2 = r["rois"][k]
                dt_anns.append({
                    "image_id": int(iid),
                    "category_id": int(r["class_ids"][k]),
                    "bbox": [float(x1), float(y1), float(x2 - x1),
                             float(y2 - y1)],
                    "score": float(r["scores"][k]),
                    "segmentation": rle_encode(
                        r["masks"][:, :, k].astype(bool)),
                })

    gt_anns = dataset_gt_annotations(dataset, image_ids)
    out = {}
    for iou_type in ("bbox", "segm"):
        ev = COCOEvalLite(gt_anns, dt_anns, iou_type=iou_type)
        out[iou_type] = ev.summarize(prefix="" if verbose else None) \
            if verbose else ev.evaluate()
    if verbose:
        print("Prediction time: {:.3f}s ({:.4f}s/image)".format(
            t_prediction, t_prediction / max(len(image_ids), 1)))
        print(f"Total time: {time.time() - t_start:.3f}s")
    return out

"""Evaluation metrics: VOC-style mAP / recall.

Replicates reference mrcnn/utils.py compute_matches / compute_ap /
compute_ap_range / compute_recall (~lines 680-760).
"""

from __future__ import annotations

import numpy as np

from .boxes import compute_overlaps, compute_overlaps_masks


def compute_matches(gt_boxes, gt_class_ids, gt_masks,
                    pred_boxes, pred_class_ids, pred_scores, pred_masks,
                    iou_threshold=0.5, score_threshold=0.0):
    """Match predictions to ground truth. Returns (gt_match, pred_match, overlaps)."""
    # Trim zero padding.
    gt_boxes = trim_zeros(gt_boxes)
    gt_masks = gt_masks[..., :gt_boxes.shape[0]]
    pred_boxes = trim_zeros(pred_boxes)
    pred_scores = pred_scores[:pred_boxes.shape[0]]
    indices = np.argsort(pred_scores)[::-1]
    pred_boxes = pred_boxes[indices]
    pred_class_ids = pred_class_ids[indices]
    pred_scores = pred_scores[indices]
    pred_masks = pred_masks[..., indices]

    if pred_masks is not None and gt_masks is not None \
            and pred_masks.size and gt_masks.size:
        overlaps = compute_overlaps_masks(pred_masks, gt_masks)
    else:
        overlaps = compute_overlaps(pred_boxes, gt_boxes)

    match_count = 0
    pred_match = -1 * np.ones([pred_boxes.shape[0]])
    gt_match = -1 * np.ones([gt_boxes.shape[0]])
    for i in range(len(pred_boxes)):
        sorted_ixs = np.argsort(overlaps[i])[::-1]
        low_score_idx = np.where(
            overlaps[i, sorted_ixs] < score_threshold)[0]
        if low_score_idx.size > 0:
            sorted_ixs = sorted_ixs[:low_score_idx[0]]
        for j in sorted_ixs:
            if gt_match[j] > -1:
                continue
            iou = overlaps[i, j]
            if iou < iou_threshold:
                break
            if pred_class_ids[i] == gt_class_ids[j]:
                match_count += 1
                gt_match[j] = i
                pred_match[i] = j
                break
    return gt_match, pred_match, overlaps


def trim_zeros(x: np.ndarray) -> np.ndarray:
    """Remove all-zero rows (reference trim_zeros)."""
    assert len(x.shape) == 2
    return x[~np.all(x == 0, axis=1)]


def compute_ap(gt_boxes, gt_class_ids, gt_masks,
               pred_boxes, pred_class_ids, pred_scores, pred_masks,
               iou_threshold=0.5):
    """Average Precision at one IoU threshold (VOC-style).

    Returns (mAP, precisions, recalls, overlaps).
    """
    gt_match, pred_match, overlaps = compute_matches(
        gt_boxes, gt_class_ids, gt_masks,
        pred_boxes, pred_class_ids, pred_scores, pred_masks, iou_threshold)

    precisions = np.cumsum(pred_match > -1) / (np.arange(len(pred_match)) + 1)
    recalls = np.cumsum(pred_match > -1).astype(np.float32) / max(len(gt_match), 1)

    precisions = np.concatenate([[0], precisions, [0]])
    recalls = np.concatenate([[0], recalls, [1]])

    for i in range(len(precisions) - 2, -1, -1):
        precisions[i] = np.maximum(precisions[i], precisions[i + 1])

    indices = np.where(recalls[:-1] != recalls[1:])[0] + 1
    mAP = np.sum((recalls[indices] - recalls[indices - 1]) *
                 precisions[indices])
    return mAP, precisions, recalls, overlaps


def compute_ap_range(gt_box, gt_class_id, gt_mask,
                     pred_box, pred_class_id, pred_score, pred_mask,
                     iou_thresholds=None, verbose=1):
    """AP averaged over IoU thresholds 0.5:0.95:0.05 (COCO-style)."""
    iou_thresholds = iou_thresholds or np.arange(0.5, 1.0, 0.05)
    AP = []
    for iou_threshold in iou_thresholds:
        ap, precisions, recalls, overlaps = compute_ap(
            gt_box, gt_class_id, gt_mask,
            pred_box, pred_class_id, pred_score, pred_mask,
            iou_threshold=iou_threshold)
        if verbose:
            print(f"AP @{iou_threshold:.2f}:\t {ap:.3f}")
        AP.append(ap)
    AP = np.array(AP).mean()
    if verbose:
        print(f"AP @{iou_thresholds[0]:.2f}-{iou_thresholds[-1]:.2f}:\t {AP:.3f}")
    return AP


def compute_recall(pred_boxes, gt_boxes, iou):
    """Recall of pred_boxes against gt_boxes at an IoU threshold."""
    overlaps = compute_overlaps(pred_boxes, gt_boxes)
    iou_max = np.max(overlaps, axis=1)
    iou_argmax = np.argmax(overlaps, axis=1)
    positive_ids = np.where(iou_max >= iou)[0]
    matched_gt_boxes = iou_argmax[positive_ids]
    recall = len(set(matched_gt_boxes)) / gt_boxes.shape[0]
    return recall, positive_ids

"""Data-plane target building: load_image_gt + build_rpn_targets.

Replicates reference mrcnn/model.py:~1210-1560 (load_image_gt,
build_rpn_targets). Augmentation is a callable (image, masks) ->
(image, masks) or the string "fliplr" (imgaug is unavailable; horizontal
flip was the reference's default COCO augmentation anyway).
"""

from __future__ import annotations

import numpy as np

from ..utils.boxes import compute_overlaps, extract_bboxes
from ..utils.image import resize_image
from ..utils.masks import minimize_mask, resize_mask
from ..utils.meta import compose_image_meta


def load_image_gt(dataset, config, image_id, augmentation=None,
                  use_mini_mask=None, rng=None):
    """Load and prepare GT for one image.

    Returns (image, image_meta, class_ids, bbox, mask).
    """
    if use_mini_mask is None:
        use_mini_mask = config.USE_MINI_MASK
    rng = rng or np.random
    image = dataset.load_image(image_id)
    mask, class_ids = dataset.load_mask(image_id)
    original_shape = image.shape
    image, window, scale, padding, crop = resize_image(
        image,
        min_dim=config.IMAGE_MIN_DIM,
        min_scale=config.IMAGE_MIN_SCALE,
        max_dim=config.IMAGE_MAX_DIM,
        mode=config.IMAGE_RESIZE_MODE)
    mask = resize_mask(mask, scale, padding, crop)

    if augmentation == "fliplr":
        if rng.rand() < 0.5:
            image = np.fliplr(image).copy()
            mask = np.fliplr(mask).copy()
    elif callable(augmentation):
        try:
            image, mask = augmentation(image, mask, rng)
        except TypeError:  # simple (image, mask) callables
            image, mask = augmentation(image, mask)

    # Exclude instances whose mask vanished in resize/crop.
    _idx = np.sum(mask, axis=(0, 1)) > 0
    mask = mask[:, :, _idx]
    class_ids = class_ids[_idx]
    bbox = extract_bboxes(mask)

    # Active classes of this image's source dataset.
    active_class_ids = np.zeros([dataset.num_classes], dtype=np.int32)
    source_class_ids = dataset.source_class_ids[
        dataset.image_info[image_id]["source"]]
    active_class_ids[source_class_ids] = 1

    if use_mini_mask:
        mask = minimize_mask(bbox, mask, tuple(config.MINI_MASK_SHAPE))

    image_meta = compose_image_meta(image_id, original_shape, image.shape,
                                    window, scale, active_class_ids)
    return image, image_meta, class_ids, bbox, mask


def generate_random_rois(image_shape, count, gt_boxes, rng=None):
    """Random roi proposals for RPN-free head training (reference
    generate_random_rois, model.py:~1460): 90% placed around GT boxes
    (2x-box neighborhoods), 10% anywhere. Returns [count, 4] pixel rois.
    """
    rng = rng or np.random
    rois = np.zeros((count, 4), dtype=np.int32)
    if gt_boxes is not None and len(gt_boxes):
        rois_per_box = int(0.9 * count / gt_boxes.shape[0])
        for i in range(gt_boxes.shape[0]):
            gy1, gx1, gy2, gx2 = gt_boxes[i][:4]
            h = gy2 - gy1
            w = gx2 - gx1
            r_y1 = max(gy1 - h, 0)
            r_y2 = min(gy2 + h, image_shape[0])
            r_x1 = max(gx1 - w, 0)
            r_x2 = min(gx2 + w, image_shape[1])
            while True:
                y1y2 = rng.randint(r_y1, r_y2, (rois_per_box, 2))
                x1x2 = rng.randint(r_x1, r_x2, (rois_per_box, 2))
                threshold = 1
                y1y2 = y1y2[np.abs(y1y2[:, 0] - y1y2[:, 1]) >=
                            threshold][:rois_per_box]
                x1x2 = x1x2[np.abs(x1x2[:, 0] - x1x2[:, 1]) >=
                            threshold][:rois_per_box]
                if y1y2.shape[0] == rois_per_box and \
                        x1x2.shape[0] == rois_per_box:
                    break
            y1, y2 = np.split(np.sort(y1y2, axis=1), 2, axis=1)
            x1, x2 = np.split(np.sort(x1x2, axis=1), 2, axis=1)
            box_rois = np.hstack([y1, x1, y2, x2])
            rois[rois_per_box * i:rois_per_box * (i + 1)] = box_rois
        remaining_count = count - (rois_per_box * gt_boxes.shape[0])
    else:
        remaining_count = count
    # The rest anywhere in the image.
    while True:
        y1y2 = rng.randint(0, image_shape[0], (remaining_count, 2))
        x1x2 = rng.randint(0, image_shape[1], (remaining_count, 2))
        threshold = 1
        y1y2 = y1y2[np.abs(y1y2[:, 0] - y1y2[:, 1]) >=
                    threshold][:remaining_count]
        x1x2 = x1x2[np.abs(x1x2[:, 0] - x1x2[:, 1]) >=
                    threshold][:remaining_count]
        if y1y2.shape[0] == remaining_count and \
                x1x2.shape[0] == remaining_count:
            break
    y1, y2 = np.split(np.sort(y1y2, axis=1), 2, axis=1)
    x1, x2 = np.split(np.sort(x1x2, axis=1), 2, axis=1)
    rois[-remaining_count:] = np.hstack([y1, x1, y2, x2])
    return rois


def build_rpn_targets(image_shape, anchors, gt_class_ids, gt_boxes, config,
                      rng=None):
    """RPN anchor matching and target deltas (reference :~1400).

    anchors, gt_boxes: pixel coords.
    Returns (rpn_match [A] int32 in {-1,0,1},
             rpn_bbox [RPN_TRAIN_ANCHORS_PER_IMAGE, 4]).
    """
    rng = rng or np.random
    rpn_match = np.zeros([anchors.shape[0]], dtype=np.int32)
    rpn_bbox = np.zeros((config.RPN_TRAIN_ANCHORS_PER_IMAGE, 4))

    if gt_boxes.shape[0] == 0:
        return rpn_match, rpn_bbox

    # Crowd handling (reference :~1410).
    crowd_ix = np.where(gt_class_ids < 0)[0]
    if crowd_ix.shape[0] > 0:
        non_crowd_ix = np.where(gt_class_ids > 0)[0]
        crowd_boxes = gt_boxes[crowd_ix]
        gt_class_ids = gt_class_ids[non_crowd_ix]
        gt_boxes = gt_boxes[non_crowd_ix]
        crowd_overlaps = compute_overlaps(anchors, crowd_boxes)
        crowd_iou_max = np.amax(crowd_overlaps, axis=1)
        no_crowd_bool = (crowd_iou_max < 0.001)
    else:
        no_crowd_bool = np.ones([anchors.shape[0]], dtype=bool)

    if gt_boxes.shape[0] == 0:
        return rpn_match, rpn_bbox

    overlaps = compute_overlaps(anchors, gt_boxes)
    anchor_iou_argmax = np.argmax(overlaps, axis=1)
    anchor_iou_max = overlaps[np.arange(overlaps.shape[0]),
                              anchor_iou_argmax]
    # Negatives: IoU < 0.3 and not touching a crowd.
    rpn_match[(anchor_iou_max < 0.3) & no_crowd_bool] = -1
    # Per-GT best anchor is positive regardless of IoU (ties included).
    gt_iou_argmax = np.argwhere(
        overlaps == np.max(overlaps, axis=0))[:, 0]
    rpn_match[gt_iou_argmax] = 1
    # High-IoU anchors positive.
    rpn_match[anchor_iou_max >= 0.7] = 1

    # Subsample to balance (reference :~1460).
    ids = np.where(rpn_match == 1)[0]
    extra = len(ids) - (config.RPN_TRAIN_ANCHORS_PER_IMAGE // 2)
    if extra > 0:
        ids = rng.choice(ids, extra, replace=False)
        rpn_match[ids] = 0
    ids = np.where(rpn_match == -1)[0]
    extra = len(ids) - (config.RPN_TRAIN_ANCHORS_PER_IMAGE -
                        np.sum(rpn_match == 1))
    if extra > 0:
        ids = rng.choice(ids, extra, replace=False)
        rpn_match[ids] = 0

    # Deltas for positives, packed.
    ids = np.where(rpn_match == 1)[0]
    ix = 0
    for i, a in zip(ids, anchors[ids]):
        gt = gt_boxes[anchor_iou_argmax[i]]
        gt_h = gt[2] - gt[0]
        gt_w = gt[3] - gt[1]
        gt_center_y = gt[0] + 0.5 * gt_h
        gt_center_x = gt[1] + 0.5 * gt_w
        a_h = a[2] - a[0]
        a_w = a[3] - a[1]
        a_center_y = a[0] + 0.5 * a_h
        a_center_x = a[1] + 0.5 * a_w
        rpn_bbox[ix] = [
            (gt_center_y - a_center_y) / a_h,
            (gt_center_x - a_center_x) / a_w,
            np.log(gt_h / a_h),
            np.log(gt_w / a_w),
        ]
        rpn_bbox[ix] /= np.array(config.RPN_BBOX_STD_DEV)
        ix += 1
    return rpn_match, rpn_bbox

"""DetectionTargetLayer: in-graph training target sampling.

Replicates reference detection_targets_graph / DetectionTargetLayer
(mrcnn/model.py:~440-660): trim padding, exclude crowds, IoU matching,
positive/negative subsampling (cap TRAIN_ROIS_PER_IMAGE * ROI_POSITIVE_RATIO
positives, 1:2 ratio), per-roi deltas (÷ BBOX_STD_DEV) and 28x28 mask
targets cropped from GT (mini-)masks, all padded to TRAIN_ROIS_PER_IMAGE.

Runs under no_grad (targets carry no gradient, as in the reference where
the layer's outputs are treated as constants by the losses).
"""

from __future__ import annotations

import torch
import torch.nn as nn

from ..ops import crop_and_resize
from ..utils.boxes import box_refinement_t, overlaps_t


class DetectionTargetLayer(nn.Module):
    def __init__(self, config):
        super().__init__()
        self.config = config

    @torch.no_grad()
    def forward(self, proposals, gt_class_ids, gt_boxes, gt_masks):
        """
        proposals:    [B, N, 4] normalized, zero-padded
        gt_class_ids: [B, MAX_GT] (zero-padded; negative = crowd)
        gt_boxes:     [B, MAX_GT, 4] normalized, zero-padded
        gt_masks:     [B, H_m, W_m, MAX_GT] bool/float (mini or full masks)
        Returns (rois [B,T,4], target_class_ids [B,T],
                 target_deltas [B,T,4], target_masks [B,T,mh,mw]).
        """
        cfg = self.config
        B = proposals.shape[0]
        T = cfg.TRAIN_ROIS_PER_IMAGE
        mh, mw = cfg.MASK_SHAPE
        device = proposals.device

        out_rois = proposals.new_zeros((B, T, 4))
        out_cls = torch.zeros((B, T), dtype=torch.long, device=device)
        out_deltas = proposals.new_zeros((B, T, 4))
        out_masks = proposals.new_zeros((B, T, mh, mw))

        for b in range(B):
            rois, cls, deltas, masks = self._one_image(
                proposals[b], gt_class_ids[b], gt_boxes[b], gt_masks[b])
            n = rois.shape[0]
            if n:
                out_rois[b, :n] = rois
                out_cls[b, :n] = cls
                out_deltas[b, :n] = deltas
                out_masks[b, :n] = masks
        return out_rois, out_cls, out_deltas, out_masks

    def _one_image(self, proposals, gt_class_ids, gt_boxes, gt_masks):
        cfg = self.config
        device = proposals.device
        mh, mw = cfg.MASK_SHAPE

        # Trim zero padding.
        valid_props = proposals.abs().sum(dim=1) > 0
        proposals = proposals[valid_props].float()
        valid_gt = gt_class_ids != 0
        gt_class_ids = gt_class_ids[valid_gt].long()
        gt_boxes_all = gt_boxes[valid_gt].float()
        gt_masks_all = gt_masks[..., valid_gt]

        empty = (proposals.new_zeros((0, 4)),
                 torch.zeros((0,), dtype=torch.long, device=device),
                 proposals.new_zeros((0, 4)),
                 proposals.new_zeros((0, mh, mw)))
        if proposals.shape[0] == 0 or gt_class_ids.shape[0] == 0:
            return empty

        # Separate crowds (negative class ids; reference :~470).
        crowd_ix = (gt_class_ids < 0).nonzero(as_tuple=True)[0]
        non_crowd_ix = (gt_class_ids > 0).nonzero(as_tuple=True)[0]
        crowd_boxes = gt_boxes_all[crowd_ix]
        gt_class_ids = gt_class_ids[non_crowd_ix]
        gt_boxes = gt_boxes_all[non_crowd_ix]
        gt_masks_sel = gt_masks_all[..., non_crowd_ix]
        if gt_class_ids.shape[0] == 0:
            return empty

        overlaps = overlaps_t(proposals, gt_boxes)     # [P, G]
        roi_iou_max = overlaps.max(dim=1).values

        if crowd_ix.numel():
            crowd_overlaps = overlaps_t(proposals, crowd_boxes)
            crowd_iou_max = crowd_overlaps.max(dim=1).values
            no_crowd = crowd_iou_max < 0.001
        else:
            no_crowd = torch.ones_like(roi_iou_max, dtype=torch.bool)

        positive_roi = roi_iou_max >= 0.5
        positive_ix = positive_roi.nonzero(as_tuple=True)[0]
        negative_ix = ((roi_iou_max < 0.5) & no_crowd) \
            .nonzero(as_tuple=True)[0]

        # Subsample (reference :~520): positives capped, negatives at ratio.
        positive_count = int(cfg.TRAIN_ROIS_PER_IMAGE *
                             cfg.ROI_POSITIVE_RATIO)
        if positive_ix.numel() > positive_count:
            perm = torch.randperm(positive_ix.numel(), device=device)
            positive_ix = positive_ix[perm[:positive_count]]
        positive_count = positive_ix.numel()
        r = 1.0 / cfg.ROI_POSITIVE_RATIO
        negative_count = int(r * positive_count) - positive_count
        if negative_ix.numel() > negative_count:
            perm = torch.randperm(negative_ix.numel(), device=device)
            negative_ix = negative_ix[perm[:negative_count]]

        positive_rois = proposals[positive_ix]
        negative_rois = proposals[negative_ix]

        # Assign positives to GT.
        if positive_count:
            pos_overlaps = overlaps[positive_ix]
            roi_gt_box_assignment = pos_overlaps.argmax(dim=1)
            roi_gt_boxes = gt_boxes[roi_gt_box_assignment]
            roi_gt_class_ids = gt_class_ids[roi_gt_box_assignment]
            deltas = box_refinement_t(positive_rois, roi_gt_boxes)
            deltas = deltas / torch.tensor(cfg.BBOX_STD_DEV, device=device)

            # Mask targets: crop GT mask to roi, resize to MASK_SHAPE.
            # gt_masks_sel: [H_m, W_m, G] -> [n_pos, 1, H_m, W_m] gather.
            masks_f = gt_masks_sel.permute(2, 0, 1).float().unsqueeze(1)
            masks_f = masks_f[roi_gt_box_assignment]
            boxes = positive_rois
            if cfg.USE_MINI_MASK:
                # Transform roi coords into mini-mask (gt-box-relative)
                # normalized space (reference :~600).
                y1, x1, y2, x2 = positive_rois.unbind(dim=1)
                gy1, gx1, gy2, gx2 = roi_gt_boxes.unbind(dim=1)
                gh = (gy2 - gy1).clamp(min=1e-6)
                gw = (gx2 - gx1).clamp(min=1e-6)
                boxes = torch.stack([(y1 - gy1) / gh, (x1 - gx1) / gw,
                                     (y2 - gy1) / gh, (x2 - gx1) / gw],
                                    dim=1)
            box_ind = torch.arange(positive_count, device=device)
            # Per-roi feature "batch": each mask is its own image.
            mask_targets = crop_and_resize(masks_f, boxes, box_ind,
                                           (mh, mw)).squeeze(1)
            mask_targets = torch.round(mask_targets)
        else:
            roi_gt_class_ids = torch.zeros((0,), dtype=torch.long,
                                           device=device)
            deltas = proposals.new_zeros((0, 4))
            mask_targets = proposals.new_zeros((0, mh, mw))

        rois = torch.cat([positive_rois, negative_rois], dim=0)
        n_neg = negative_rois.shape[0]
        cls = torch.cat([roi_gt_class_ids,
                         torch.zeros(n_neg, dtype=torch.long,
                                     device=device)])
        deltas = torch.cat([deltas, proposals.new_zeros((n_neg, 4))], dim=0)
        mask_targets = torch.cat(
            [mask_targets, proposals.new_zeros((n_neg, mh, mw))], dim=0)
        return rois, cls, deltas, mask_targets

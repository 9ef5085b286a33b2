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
        # Fully-tensorized sampler: identical selection DISTRIBUTION
        # with fixed shapes and zero host syncs (the loop path's
        # nonzero/randperm cost ~15 syncs/step at B=2). Default ON since
        # the round-2 block2 A/B: 45.41 vs 46.86 ms/step baseline.
        # MASKRCNN_VECTORIZED_TARGETS=0 opts back into the loop path.
        import os
        self.vectorized = \
            os.environ.get("MASKRCNN_VECTORIZED_TARGETS", "1") != "0"

    @torch.no_grad()
    def forward(self, proposals, gt_class_ids, gt_boxes, gt_masks):
        if self.vectorized:
            return self._forward_vectorized(proposals, gt_class_ids,
                                            gt_boxes, gt_masks)
        return self._forward_loop(proposals, gt_class_ids, gt_boxes,
                                  gt_masks)

    # ------------------------------------------------------ loop variant
    def _forward_loop(self, proposals, gt_class_ids, gt_boxes, gt_masks):
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

    # ------------------------------------------- vectorized variant
    @torch.no_grad()
    def _forward_vectorized(self, proposals, gt_class_ids, gt_boxes,
                            gt_masks):
        """Same sampling distribution as the loop variant with FIXED
        shapes and no host syncs: random subsets are drawn with the
        rand*mask top-k reservoir trick; counts/quotas stay tensors.

        Layout difference (semantics-neutral): positives occupy slots
        [0, cap_pos) with invalid slots zeroed, negatives
        [cap_pos, T) — every consumer (losses, heads) is
        padding-agnostic via target_class_ids > 0.
        """
        cfg = self.config
        B, P, _ = proposals.shape
        T = cfg.TRAIN_ROIS_PER_IMAGE
        cap_pos = int(T * cfg.ROI_POSITIVE_RATIO)
        neg_slots = T - cap_pos
        mh, mw = cfg.MASK_SHAPE
        device = proposals.device
        std = torch.tensor(cfg.BBOX_STD_DEV, device=device)
        r = 1.0 / cfg.ROI_POSITIVE_RATIO

        out_rois = proposals.new_zeros((B, T, 4))
        out_cls = torch.zeros((B, T), dtype=torch.long, device=device)
        out_deltas = proposals.new_zeros((B, T, 4))
        out_masks = proposals.new_zeros((B, T, mh, mw))

        for b in range(B):
            prop = proposals[b].float()                       # [P, 4]
            ids = gt_class_ids[b].long()                      # [G]
            boxes = gt_boxes[b].float()                       # [G, 4]
            valid_prop = prop.abs().sum(dim=1) > 0
            non_crowd = (ids > 0).float()
            crowd = (ids < 0).float()

            ov = overlaps_t(prop, boxes)                      # [P, G]
            ov_nc = ov * non_crowd.unsqueeze(0)
            roi_iou_max = ov_nc.max(dim=1).values
            crowd_iou = (ov * crowd.unsqueeze(0)).max(dim=1).values
            pos_mask = (roi_iou_max >= 0.5) & valid_prop
            neg_mask = (roi_iou_max < 0.5) & (crowd_iou < 0.001) & \
                valid_prop

            # Uniform random subset via rand*mask top-k.
            pos_idx = (torch.rand(P, device=device) *
                       pos_mask).topk(cap_pos).indices
            pos_ok = pos_mask[pos_idx]                        # [cap_pos]
            n_pos = pos_ok.sum()
            quota = (n_pos.double() * (r - 1.0)).floor()
            neg_idx = (torch.rand(P, device=device) *
                       neg_mask).topk(neg_slots).indices
            neg_ok = neg_mask[neg_idx] & \
                (torch.arange(neg_slots, device=device) < quota)

            pos_rois = prop[pos_idx] * pos_ok.unsqueeze(1)
            assign = ov_nc[pos_idx].argmax(dim=1)             # [cap_pos]
            roi_gt_boxes = boxes[assign]
            roi_cls = ids[assign].clamp(min=0) * pos_ok.long()

            # Safe refinement (invalid rows later zeroed; clamp avoids
            # inf*0 = nan from zero-area padding rows).
            def safe_refine(box, gt):
                h = (box[:, 2] - box[:, 0]).clamp(min=1e-6)
                w = (box[:, 3] - box[:, 1]).clamp(min=1e-6)
                cy = box[:, 0] + 0.5 * h
                cx = box[:, 1] + 0.5 * w
                gh = (gt[:, 2] - gt[:, 0]).clamp(min=1e-6)
                gw = (gt[:, 3] - gt[:, 1]).clamp(min=1e-6)
                gcy = gt[:, 0] + 0.5 * gh
                gcx = gt[:, 1] + 0.5 * gw
                return torch.stack([(gcy - cy) / h, (gcx - cx) / w,
                                    torch.log(gh / h),
                                    torch.log(gw / w)], dim=1)

            deltas = safe_refine(prop[pos_idx], roi_gt_boxes) / std
            deltas = deltas * pos_ok.unsqueeze(1)

            masks_f = gt_masks[b].permute(2, 0, 1).float() \
                .unsqueeze(1)[assign]                # [cap_pos, 1, H, W]
            crop_boxes = prop[pos_idx]
            if cfg.USE_MINI_MASK:
                y1, x1, y2, x2 = crop_boxes.unbind(dim=1)
                gy1, gx1, gy2, gx2 = roi_gt_boxes.unbind(dim=1)
                gh = (gy2 - gy1).clamp(min=1e-6)
                gw = (gx2 - gx1).clamp(min=1e-6)
                crop_boxes = torch.stack(
                    [(y1 - gy1) / gh, (x1 - gx1) / gw,
                     (y2 - gy1) / gh, (x2 - gx1) / gw], dim=1)
            box_ind = torch.arange(cap_pos, device=device)
            mask_t = crop_and_resize(masks_f, crop_boxes, box_ind,
                                     (mh, mw)).squeeze(1)
            mask_t = torch.round(mask_t) * pos_ok.unsqueeze(1) \
                .unsqueeze(2)

            out_rois[b, :cap_pos] = pos_rois
            out_rois[b, cap_pos:] = prop[neg_idx] * neg_ok.unsqueeze(1)
            out_cls[b, :cap_pos] = roi_cls
            out_deltas[b, :cap_pos] = deltas
            out_masks[b, :cap_pos] = mask_t
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

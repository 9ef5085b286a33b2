"""Proposal generation (RPN outputs -> RoIs).

Replicates reference ProposalLayer (mrcnn/model.py:~250-335): per image,
top-k PRE_NMS_LIMIT fg scores, apply deltas (x RPN_BBOX_STD_DEV) to
anchors, clip to [0,1], NMS at RPN_NMS_THRESHOLD, pad to proposal_count.

Batched torch implementation: top-k and delta application run batched on
the full [B, A] tensors (A=261,888 @1024^2); only the NMS loops over the
(small) batch, using the HIP bitmask NMS on GPU.
"""

from __future__ import annotations

import torch
import torch.nn as nn

from ..ops import batched_nms_padded
from ..utils.boxes import apply_box_deltas_t


class ProposalLayer(nn.Module):
    def __init__(self, config):
        super().__init__()
        self.config = config
        self.register_buffer(
            "rpn_bbox_std_dev",
            torch.tensor(config.RPN_BBOX_STD_DEV, dtype=torch.float32),
            persistent=False)

    def forward(self, rpn_probs, rpn_bbox, anchors, proposal_count):
        """
        rpn_probs: [B, A, 2] (bg, fg)
        rpn_bbox:  [B, A, 4] deltas
        anchors:   [A, 4] normalized
        Returns rois [B, proposal_count, 4] normalized, zero-padded.
        """
        cfg = self.config
        scores = rpn_probs[:, :, 1].float()
        deltas = rpn_bbox.float() * self.rpn_bbox_std_dev.view(1, 1, 4)
        B, A = scores.shape

        pre_nms_limit = min(cfg.PRE_NMS_LIMIT, A)
        top_scores, ix = torch.topk(scores, pre_nms_limit, dim=1,
                                    sorted=True)
        # Gather deltas and anchors per image.
        deltas = torch.gather(deltas, 1,
                              ix.unsqueeze(-1).expand(-1, -1, 4))
        anchors_b = anchors.unsqueeze(0).expand(B, A, 4).float()
        anchors_g = torch.gather(anchors_b, 1,
                                 ix.unsqueeze(-1).expand(-1, -1, 4))

        boxes = apply_box_deltas_t(anchors_g, deltas)
        # Clip to the unit window. clamp (not clip_boxes_t with a fresh
        # torch.tensor) — creating a tensor from host data is a H2D copy,
        # which aborts hipGraph stream capture of the serving forward.
        boxes = boxes.clamp(0.0, 1.0)

        # NMS for a GROUP of images per launch (image-index offset
        # trick): per-image NMS serializes at serving batch 32, but one
        # launch over ALL images makes the bitmask workspace quadratic in
        # boxes/launch (4.6 GB at 32x6000) — chunk to <= ~16k boxes.
        # Padded NMS + masked scatter keeps the path free of host syncs
        # (capture-safe; SURVEY §7 hard-part 3).
        P = boxes.shape[1]
        group = max(1, 16384 // P)
        device = boxes.device
        # Dummy row/col absorbs invalid scatter targets.
        rois_buf = boxes.new_zeros((B + 1, proposal_count + 1, 4))
        for g0 in range(0, B, group):
            g1 = min(g0 + group, B)
            nb = g1 - g0
            flat_boxes = boxes[g0:g1].reshape(-1, 4)
            flat_scores = top_scores[g0:g1].reshape(-1)
            img_ix = torch.arange(nb, device=device) \
                .repeat_interleave(P)
            # Full-size keep: a global-score cap could starve one image
            # of its proposal_count while another image's over-quota
            # survivors (rank >= proposal_count, discarded) fill it.
            max_keep = nb * P
            keep, _count = batched_nms_padded(
                flat_boxes, flat_scores, img_ix, cfg.RPN_NMS_THRESHOLD,
                max_keep)
            valid = keep >= 0                          # [K]
            keep_c = keep.clamp(min=0)
            img = img_ix[keep_c]                       # [K], score-desc
            # Per-image rank of each kept box (global order preserves
            # per-image score order); invalid rows rank out of range.
            onehot = (img.unsqueeze(1) ==
                      torch.arange(nb, device=device).unsqueeze(0)) \
                & valid.unsqueeze(1)
            rank = onehot.long().cumsum(dim=0)[
                torch.arange(img.numel(), device=device), img] - 1
            sel = valid & (rank < proposal_count)
            tgt_img = torch.where(sel, g0 + img, torch.full_like(img, B))
            tgt_rank = torch.where(
                sel, rank, torch.full_like(rank, proposal_count))
            rois_buf.index_put_((tgt_img, tgt_rank), flat_boxes[keep_c])
        return rois_buf[:B, :proposal_count]

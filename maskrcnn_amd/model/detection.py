"""DetectionLayer: inference-time box refinement + per-class NMS.

Replicates reference refine_detections_graph / DetectionLayer
(mrcnn/model.py:~665-825): argmax class per roi, class-specific delta
(x BBOX_STD_DEV), clip to the image window, drop background and low scores,
per-class NMS at DETECTION_NMS_THRESHOLD, keep top
DETECTION_MAX_INSTANCES, zero-pad. Output [B, D, 6] =
(y1, x1, y2, x2, class_id, score), boxes normalized.

The reference's per-class map_fn NMS serializes over 80 classes (N9 note);
here per-class NMS is one batched_nms call (coordinate-offset trick) feeding
the HIP bitmask NMS on GPU.
"""

from __future__ import annotations

import torch
import torch.nn as nn

from ..ops import batched_nms, batched_nms_padded
from ..utils.boxes import apply_box_deltas_t
from ..utils.meta import parse_image_meta_t


class DetectionLayer(nn.Module):
    def __init__(self, config):
        super().__init__()
        self.config = config
        self.register_buffer(
            "bbox_std_dev",
            torch.tensor(config.BBOX_STD_DEV, dtype=torch.float32),
            persistent=False)
        self.register_buffer(
            "_norm_shift",
            torch.tensor([0., 0., 1., 1.], dtype=torch.float32),
            persistent=False)

    @torch.no_grad()
    def forward(self, rois, probs, deltas, image_meta):
        """
        rois:   [B, N, 4] normalized
        probs:  [B, N, C]
        deltas: [B, N, C, 4]
        image_meta: [B, META]
        Returns detections [B, DETECTION_MAX_INSTANCES, 6].

        Batched over images AND classes in ONE NMS launch (the reference
        serializes an 80-class map_fn per image; a python per-image loop
        would likewise serialize at serving batch sizes).
        """
        cfg = self.config
        B, N, C = probs.shape
        device = rois.device
        meta = parse_image_meta_t(image_meta)
        # Windows in normalized coords (reference :~700), per image.
        # All-tensor math: float(shape_tensor[i]) is a host sync, and
        # torch.tensor(...)/list-index gathers are H2D copies — any of
        # them aborts hipGraph stream capture of the serving forward.
        hw = meta["image_shape"][:, :2].float()            # [B, 2]
        scale = hw.repeat(1, 2) - 1.0                      # [B,4] h-1,w-1
        windows = (meta["window"].float() - self._norm_shift) / scale

        if _use_syncfree(rois):
            # Fused HIP prep (one detection_prep launch replaces the
            # ~8 argmax/gather/refine/clip launches below) when the
            # extension is present; torch prep otherwise.
            from ..ops._ext import load_ext
            ext = load_ext()
            if rois.is_cuda and ext is not None:
                dt = probs.dtype if probs.dtype in (
                    torch.float32, torch.bfloat16) else torch.float32
                refined, class_ids, scores_m = ext.detection_prep(
                    rois.float(), probs.to(dt), deltas.to(dt),
                    windows, self.bbox_std_dev,
                    float(cfg.DETECTION_MIN_CONFIDENCE or 0.0))
                return self._nms_scatter(refined, class_ids.long(),
                                         scores_m, B, N, C)

        probs = probs.float()
        class_ids = probs.argmax(dim=2)                        # [B, N]
        class_scores = probs.gather(
            2, class_ids.unsqueeze(-1)).squeeze(-1)            # [B, N]
        deltas_specific = deltas.float().gather(
            2, class_ids[..., None, None].expand(-1, -1, 1, 4)) \
            .squeeze(2)                                        # [B, N, 4]

        refined = apply_box_deltas_t(
            rois.float(), deltas_specific * self.bbox_std_dev)
        # Clip each image to its own window ([B,1] bounds broadcast).
        wy1, wx1, wy2, wx2 = windows.unsqueeze(2).unbind(dim=1)
        y1 = torch.clamp(refined[..., 0], wy1, wy2)
        x1 = torch.clamp(refined[..., 1], wx1, wx2)
        y2 = torch.clamp(refined[..., 2], wy1, wy2)
        x2 = torch.clamp(refined[..., 3], wx1, wx2)
        refined = torch.stack([y1, x1, y2, x2], dim=-1)

        keep = class_ids > 0
        if cfg.DETECTION_MIN_CONFIDENCE:
            keep &= class_scores >= cfg.DETECTION_MIN_CONFIDENCE
        if _use_syncfree(rois):
            return self._forward_syncfree(refined, class_ids,
                                          class_scores, keep, B, N, C)
        flat_ix = keep.reshape(-1).nonzero(as_tuple=True)[0]
        D = cfg.DETECTION_MAX_INSTANCES
        out = rois.new_zeros((B, D, 6), dtype=torch.float32)
        if flat_ix.numel() == 0:
            return out

        img = flat_ix // N
        boxes_f = refined.reshape(-1, 4)[flat_ix]
        scores_f = class_scores.reshape(-1)[flat_ix]
        cls_f = class_ids.reshape(-1)[flat_ix]
        # NMS group = (image, class).
        group = img * (C + 1) + cls_f
        nms_keep = batched_nms(boxes_f, scores_f, group,
                               cfg.DETECTION_NMS_THRESHOLD)
        if nms_keep.numel() == 0:
            return out
        kimg = img[nms_keep]                   # global score-desc order
        onehot = (kimg.unsqueeze(1) ==
                  torch.arange(B, device=device).unsqueeze(0))
        rank = onehot.long().cumsum(dim=0)[
            torch.arange(kimg.numel(), device=device), kimg] - 1
        sel = rank < D
        rows = torch.cat([boxes_f[nms_keep], cls_f[nms_keep, None].float(),
                          scores_f[nms_keep, None]], dim=1)
        out[kimg[sel], rank[sel]] = rows[sel]
        return out

    def _forward_syncfree(self, refined, class_ids, class_scores, keep,
                          B, N, C):
        """Fixed-shape, zero-host-sync detection post-process
        (hipGraph-capture path; auto-selected while capturing or via
        MASKRCNN_SYNCFREE_DETECTION=1).

        Dropped boxes keep their slot with score 0 (a zero-score box can
        never suppress a positive-score one, so NMS results match the
        gathered eager path); padded NMS + masked index_put_ scatter
        into a dummy-row buffer replaces nonzero()/boolean gathers —
        the same pattern as ProposalLayer."""
        scores_m = torch.where(keep, class_scores,
                               torch.zeros_like(class_scores))
        return self._nms_scatter(refined, class_ids, scores_m, B, N, C)

    def _nms_scatter(self, refined, class_ids, scores_m, B, N, C):
        """Chunked grouped padded-NMS + masked scatter (shared by the
        torch and HIP-prep sync-free paths). scores_m: dropped rois
        already zeroed."""
        cfg = self.config
        device = refined.device
        D = cfg.DETECTION_MAX_INSTANCES
        out_buf = refined.new_zeros((B + 1, D + 1, 6))
        group_sz = max(1, 16384 // N)
        for g0 in range(0, B, group_sz):
            g1 = min(g0 + group_sz, B)
            nb = g1 - g0
            flat_boxes = refined[g0:g1].reshape(-1, 4)
            flat_scores = scores_m[g0:g1].reshape(-1)
            flat_cls = class_ids[g0:g1].reshape(-1)
            img_ix = torch.arange(nb, device=device) \
                .repeat_interleave(N)
            grp = img_ix * (C + 1) + flat_cls
            keep_idx, _count = batched_nms_padded(
                flat_boxes, flat_scores, grp,
                cfg.DETECTION_NMS_THRESHOLD, nb * N)
            valid = keep_idx >= 0
            kc = keep_idx.clamp(min=0)
            valid &= flat_scores[kc] > 0      # drop zero-score slots
            img = img_ix[kc]                  # score-desc global order
            onehot = (img.unsqueeze(1) ==
                      torch.arange(nb, device=device).unsqueeze(0)) \
                & valid.unsqueeze(1)
            rank = onehot.long().cumsum(dim=0)[
                torch.arange(img.numel(), device=device), img] - 1
            sel = valid & (rank < D)
            tgt_img = torch.where(sel, g0 + img, torch.full_like(img, B))
            tgt_rank = torch.where(sel, rank, torch.full_like(rank, D))
            rows = torch.cat(
                [flat_boxes[kc], flat_cls[kc, None].float(),
                 flat_scores[kc, None]], dim=1)
            out_buf.index_put_((tgt_img, tgt_rank), rows)
        return out_buf[:B, :D]


def _use_syncfree(t: torch.Tensor) -> bool:
    import os
    flag = os.environ.get("MASKRCNN_SYNCFREE_DETECTION", "")
    if flag == "1":
        return True
    if flag == "0":
        return False
    return t.is_cuda and torch.cuda.is_current_stream_capturing()

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

from ..ops import batched_nms
from ..utils.boxes import apply_box_deltas_t, clip_boxes_t, norm_boxes_t
from ..utils.meta import parse_image_meta_t


class DetectionLayer(nn.Module):
    def __init__(self, config):
        super().__init__()
        self.config = config
        self.register_buffer(
            "bbox_std_dev",
            torch.tensor(config.BBOX_STD_DEV, dtype=torch.float32),
            persistent=False)

    @torch.no_grad()
    def forward(self, rois, probs, deltas, image_meta):
        """
        rois:   [B, N, 4] normalized
        probs:  [B, N, C]
        deltas: [B, N, C, 4]
        image_meta: [B, META]
        Returns detections [B, DETECTION_MAX_INSTANCES, 6].
        """
        cfg = self.config
        B, N, C = probs.shape
        meta = parse_image_meta_t(image_meta)
        image_shape = meta["image_shape"][0]
        # Window in normalized coords (reference :~700).
        windows = norm_boxes_t(meta["window"].float(),
                               (float(image_shape[0]),
                                float(image_shape[1])))

        out = rois.new_zeros((B, cfg.DETECTION_MAX_INSTANCES, 6))
        for b in range(B):
            det = self._one_image(rois[b].float(), probs[b].float(),
                                  deltas[b].float(), windows[b])
            out[b, :det.shape[0]] = det
        return out

    def _one_image(self, rois, probs, deltas, window):
        cfg = self.config
        device = rois.device
        N = rois.shape[0]
        class_ids = probs.argmax(dim=1)                       # [N]
        idx = torch.arange(N, device=device)
        class_scores = probs[idx, class_ids]
        deltas_specific = deltas[idx, class_ids]              # [N, 4]

        refined = apply_box_deltas_t(
            rois, deltas_specific * self.bbox_std_dev)
        refined = clip_boxes_t(refined, window)

        keep = class_ids > 0
        if cfg.DETECTION_MIN_CONFIDENCE:
            keep &= class_scores >= cfg.DETECTION_MIN_CONFIDENCE
        keep_ix = keep.nonzero(as_tuple=True)[0]
        if keep_ix.numel() == 0:
            return rois.new_zeros((0, 6))

        pre_boxes = refined[keep_ix]
        pre_scores = class_scores[keep_ix]
        pre_class = class_ids[keep_ix]

        nms_keep = batched_nms(pre_boxes, pre_scores, pre_class,
                               cfg.DETECTION_NMS_THRESHOLD)
        # Reference also caps per-class keep at DETECTION_MAX_INSTANCES
        # before the global top-k; the global cap below subsumes it.
        nms_keep = nms_keep[:cfg.DETECTION_MAX_INSTANCES]
        boxes = pre_boxes[nms_keep]
        scores = pre_scores[nms_keep]
        cls = pre_class[nms_keep]
        order = scores.argsort(descending=True)
        boxes, scores, cls = boxes[order], scores[order], cls[order]
        return torch.cat([boxes, cls[:, None].float(),
                          scores[:, None]], dim=1)

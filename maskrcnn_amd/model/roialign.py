"""PyramidROIAlign: multi-level RoIAlign over the FPN.

Replicates reference PyramidROIAlign (mrcnn/model.py:~340-430): each roi is
assigned to pyramid level k = 4 + round(log2(sqrt(h*w) / (224/sqrt(1024))))
(tf.round in the reference) clipped to [2,5]; pooling runs per level and
results are scattered back to the original box order. Unlike the reference's crop_and_resize, the op mode
is configurable (true RoIAlign by default for better gradients; see
ops/roi_align.py docstring and SURVEY.md §7 hard-part 6).
"""

from __future__ import annotations

import torch
import torch.nn as nn

from ..ops import roi_align


class PyramidROIAlign(nn.Module):
    def __init__(self, pool_size, image_shape, mode="crop_and_resize",
                 sampling_ratio=2):
        super().__init__()
        self.pool_size = pool_size
        self.image_shape = image_shape  # [H, W, C]
        self.mode = mode
        self.sampling_ratio = sampling_ratio

    def forward(self, rois, feature_maps):
        """
        rois: [B, N, 4] normalized (zero rows = padding)
        feature_maps: [P2, P3, P4, P5], each [B, C, H_l, W_l]
        Returns [B*N, C, pool, pool] (flat roi-major; channels_last on GPU
        so the head convs hit the HIP conv path).
        """
        B, N, _ = rois.shape
        C = feature_maps[0].shape[1]
        boxes = rois.reshape(-1, 4)
        box_ind = torch.arange(B, device=rois.device) \
            .repeat_interleave(N)

        y1, x1, y2, x2 = boxes.unbind(dim=1)
        h = y2 - y1
        w = x2 - x1
        # Level assignment (reference log2_graph section, model.py:~360).
        image_area = float(self.image_shape[0] * self.image_shape[1])
        scale = torch.sqrt(h * w) * (image_area ** 0.5) / 224.0
        levels = torch.round(4.0 + torch.log2(scale.clamp(min=1e-6)))
        levels = levels.clamp(2, 5).to(torch.long)
        # Padding rows (zero area) -> level 2; output stays zero via mask.
        valid = (h > 0) & (w > 0)

        # channels_last output feeds the HIP conv path in the heads.
        # Default ON in training too since the round-2 block2 A/B
        # (45.79 vs 46.86 ms/step baseline); MASKRCNN_TRAIN_CL_ROI=0
        # opts out.
        import os
        cl = (feature_maps[0].is_cuda and
              (not self.training or
               os.environ.get("MASKRCNN_TRAIN_CL_ROI", "1") != "0") and
              feature_maps[0].is_contiguous(
                  memory_format=torch.channels_last))
        out = torch.empty(
            (B * N, C, self.pool_size, self.pool_size),
            dtype=feature_maps[0].dtype, device=feature_maps[0].device,
            memory_format=(torch.channels_last if cl
                           else torch.contiguous_format)).zero_()
        if _use_syncfree(rois):
            # Fixed-shape path (hipGraph capture): pool EVERY roi at
            # every level and mask-combine — 4x the roi_align work (a
            # small fraction of the forward) for zero host syncs and
            # static shapes. nonzero()/boolean gathers would abort a
            # stream capture.
            for i, level in enumerate(range(2, 6)):
                pooled = roi_align(feature_maps[i], boxes, box_ind,
                                   (self.pool_size, self.pool_size),
                                   sampling_ratio=self.sampling_ratio,
                                   mode=self.mode).to(out.dtype)
                m = ((levels == level) & valid).to(out.dtype) \
                    .view(-1, 1, 1, 1)
                out += pooled * m
            return out
        for i, level in enumerate(range(2, 6)):
            sel = ((levels == level) & valid).nonzero(as_tuple=True)[0]
            if sel.numel() == 0:
                continue
            pooled = roi_align(feature_maps[i], boxes[sel], box_ind[sel],
                               (self.pool_size, self.pool_size),
                               sampling_ratio=self.sampling_ratio,
                               mode=self.mode)
            out[sel] = pooled.to(out.dtype)
        return out


def _use_syncfree(t: torch.Tensor) -> bool:
    import os
    flag = os.environ.get("MASKRCNN_SYNCFREE_ROI", "")
    if flag == "1":
        return True
    if flag == "0":
        return False
    return t.is_cuda and torch.cuda.is_current_stream_capturing()

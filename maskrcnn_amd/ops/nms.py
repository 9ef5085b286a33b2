"""Non-maximum suppression.

Replaces `tf.image.non_max_suppression` (N3 in SURVEY.md §2.3; used by
ProposalLayer mrcnn/model.py:~300 and DetectionLayer's per-class NMS :~780).

GPU: bitmask NMS HIP kernel (csrc/hip/nms.hip) — triangular IoU matrix,
64-boxes-per-wavefront ballot, batched over (image, class) in one launch.
CPU: vectorized torch greedy NMS (the numerics oracle).

Boxes are [y1, x1, y2, x2] (any consistent scale; IoU is scale-free).
"""

from __future__ import annotations

import torch

from ._ext import require_ext


def _nms_torch(boxes: torch.Tensor, scores: torch.Tensor,
               iou_threshold: float) -> torch.Tensor:
    """Greedy NMS. Returns kept indices sorted by descending score."""
    if boxes.numel() == 0:
        return torch.zeros((0,), dtype=torch.long, device=boxes.device)
    order = scores.argsort(descending=True)
    boxes = boxes[order]
    y1, x1, y2, x2 = boxes.unbind(dim=1)
    areas = (y2 - y1).clamp(min=0) * (x2 - x1).clamp(min=0)
    n = boxes.shape[0]
    suppressed = torch.zeros(n, dtype=torch.bool, device=boxes.device)
    keep = []
    for i in range(n):
        if suppressed[i]:
            continue
        keep.append(order[i].item())
        if i + 1 >= n:
            break
        yy1 = torch.max(y1[i], y1[i + 1:])
        xx1 = torch.max(x1[i], x1[i + 1:])
        yy2 = torch.min(y2[i], y2[i + 1:])
        xx2 = torch.min(x2[i], x2[i + 1:])
        inter = (yy2 - yy1).clamp(min=0) * (xx2 - xx1).clamp(min=0)
        iou = inter / (areas[i] + areas[i + 1:] - inter).clamp(min=1e-12)
        suppressed[i + 1:] |= iou > iou_threshold
    return torch.tensor(keep, dtype=torch.long, device=boxes.device)


def nms(boxes: torch.Tensor, scores: torch.Tensor, iou_threshold: float,
        max_out: int = -1) -> torch.Tensor:
    """Single-class NMS. Returns indices of kept boxes (score-descending).

    max_out > 0 truncates the kept list.
    """
    if boxes.is_cuda:
        ext = require_ext()
        keep = ext.nms(boxes.contiguous().float(), scores.contiguous().float(),
                       float(iou_threshold))
    else:
        keep = _nms_torch(boxes.float(), scores.float(), iou_threshold)
    if max_out > 0:
        keep = keep[:max_out]
    return keep


def nms_padded(boxes: torch.Tensor, scores: torch.Tensor,
               iou_threshold: float, max_out: int):
    """Fixed-shape NMS with ZERO host syncs.

    Returns (keep [max_out] int64, count [1] int32-on-device): keep holds
    original indices in score-descending order, -1-padded. The GPU path
    never touches the host (capture-safe); CPU falls back to greedy NMS
    + padding.
    """
    if boxes.is_cuda:
        ext = require_ext()
        return ext.nms_padded(boxes.contiguous().float(),
                              scores.contiguous().float(),
                              float(iou_threshold), int(max_out))
    keep = _nms_torch(boxes.float(), scores.float(), iou_threshold)
    keep = keep[:max_out]
    out = torch.full((max_out,), -1, dtype=torch.long)
    out[:keep.shape[0]] = keep
    count = torch.tensor([keep.shape[0]], dtype=torch.int32)
    return out, count


def batched_nms_padded(boxes: torch.Tensor, scores: torch.Tensor,
                       idxs: torch.Tensor, iou_threshold: float,
                       max_out: int):
    """Sync-free grouped NMS (coordinate-offset trick + nms_padded)."""
    if boxes.numel() == 0:
        return (torch.full((max_out,), -1, dtype=torch.long,
                           device=boxes.device),
                torch.zeros(1, dtype=torch.int32, device=boxes.device))
    max_coord = boxes.max()
    offsets = idxs.to(boxes) * (max_coord + 1)
    return nms_padded(boxes + offsets[:, None], scores, iou_threshold,
                      max_out)


def batched_nms(boxes: torch.Tensor, scores: torch.Tensor,
                idxs: torch.Tensor, iou_threshold: float) -> torch.Tensor:
    """NMS over multiple groups (e.g. classes) in one call.

    Implemented with the coordinate-offset trick: boxes of different groups
    are shifted apart so they never overlap, then one single NMS runs.
    Returns kept indices (score-descending within the merged list).
    """
    if boxes.numel() == 0:
        return torch.zeros((0,), dtype=torch.long, device=boxes.device)
    max_coord = boxes.max()
    offsets = idxs.to(boxes) * (max_coord + 1)
    shifted = boxes + offsets[:, None]
    return nms(shifted, scores, iou_threshold)

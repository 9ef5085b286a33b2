"""Box utilities — numpy (data plane) and torch (graph plane) variants.

Replicates the behavior of reference mrcnn/utils.py:~15-240 and the graph
helpers in mrcnn/model.py (apply_box_deltas_graph:~210, clip_boxes_graph:~240,
overlaps_graph:~445, box_refinement_graph). Boxes are [y1, x1, y2, x2]
throughout, matching the reference convention.
"""

from __future__ import annotations

import numpy as np
import torch


# --------------------------------------------------------------------------
# numpy variants (data plane)
# --------------------------------------------------------------------------

def extract_bboxes(mask: np.ndarray) -> np.ndarray:
    """Compute bounding boxes from masks. mask: [H, W, N]. Returns [N, 4]."""
    boxes = np.zeros([mask.shape[-1], 4], dtype=np.int32)
    for i in range(mask.shape[-1]):
        m = mask[:, :, i]
        horizontal = np.where(np.any(m, axis=0))[0]
        vertical = np.where(np.any(m, axis=1))[0]
        if horizontal.shape[0]:
            x1, x2 = horizontal[[0, -1]]
            y1, y2 = vertical[[0, -1]]
            # x2/y2 should not be part of the box (exclusive).
            x2 += 1
            y2 += 1
        else:
            x1, x2, y1, y2 = 0, 0, 0, 0
        boxes[i] = np.array([y1, x1, y2, x2])
    return boxes.astype(np.int32)


def compute_iou(box, boxes, box_area, boxes_area):
    """IoU of one box against an array of boxes (all [y1,x1,y2,x2])."""
    y1 = np.maximum(box[0], boxes[:, 0])
    y2 = np.minimum(box[2], boxes[:, 2])
    x1 = np.maximum(box[1], boxes[:, 1])
    x2 = np.minimum(box[3], boxes[:, 3])
    intersection = np.maximum(x2 - x1, 0) * np.maximum(y2 - y1, 0)
    union = box_area + boxes_area[:] - intersection[:]
    return intersection / union


def compute_overlaps(boxes1: np.ndarray, boxes2: np.ndarray) -> np.ndarray:
    """IoU matrix [N1, N2] between two box sets.

    Fully broadcast (the reference's per-column compute_iou loop was
    320 ms/image for RPN targets at 261,888 anchors — this is ~35 ms,
    the input-pipeline hot spot for real-data training)."""
    boxes1 = np.asarray(boxes1, dtype=np.float32)
    boxes2 = np.asarray(boxes2, dtype=np.float32)
    area1 = (boxes1[:, 2] - boxes1[:, 0]) * (boxes1[:, 3] - boxes1[:, 1])
    area2 = (boxes2[:, 2] - boxes2[:, 0]) * (boxes2[:, 3] - boxes2[:, 1])
    y1 = np.maximum(boxes1[:, None, 0], boxes2[None, :, 0])
    y2 = np.minimum(boxes1[:, None, 2], boxes2[None, :, 2])
    np.subtract(y2, y1, out=y2)
    np.maximum(y2, 0, out=y2)
    x1 = np.maximum(boxes1[:, None, 1], boxes2[None, :, 1])
    x2 = np.minimum(boxes1[:, None, 3], boxes2[None, :, 3])
    np.subtract(x2, x1, out=x2)
    np.maximum(x2, 0, out=x2)
    np.multiply(x2, y2, out=x2)  # intersection
    np.add(area1[:, None], area2[None, :] - x2, out=y2)  # union
    np.divide(x2, y2, out=x2)
    return x2


def compute_overlaps_masks(masks1: np.ndarray, masks2: np.ndarray):
    """IoU overlaps between two sets of masks [H, W, N]."""
    if masks1.shape[-1] == 0 or masks2.shape[-1] == 0:
        return np.zeros((masks1.shape[-1], masks2.shape[-1]))
    m1 = np.reshape(masks1 > .5, (-1, masks1.shape[-1])).astype(np.float32)
    m2 = np.reshape(masks2 > .5, (-1, masks2.shape[-1])).astype(np.float32)
    area1 = np.sum(m1, axis=0)
    area2 = np.sum(m2, axis=0)
    intersections = np.dot(m1.T, m2)
    union = area1[:, None] + area2[None, :] - intersections
    return intersections / np.maximum(union, 1e-8)


def non_max_suppression(boxes, scores, threshold):
    """Greedy NMS (numpy). Returns indices of kept boxes."""
    assert boxes.shape[0] > 0
    if boxes.dtype.kind != "f":
        boxes = boxes.astype(np.float32)
    y1, x1, y2, x2 = boxes[:, 0], boxes[:, 1], boxes[:, 2], boxes[:, 3]
    area = (y2 - y1) * (x2 - x1)
    ixs = scores.argsort()[::-1]
    pick = []
    while len(ixs) > 0:
        i = ixs[0]
        pick.append(i)
        iou = compute_iou(boxes[i], boxes[ixs[1:]], area[i], area[ixs[1:]])
        remove_ixs = np.where(iou > threshold)[0] + 1
        ixs = np.delete(ixs, remove_ixs)
        ixs = np.delete(ixs, 0)
    return np.array(pick, dtype=np.int32)


def apply_box_deltas(boxes: np.ndarray, deltas: np.ndarray) -> np.ndarray:
    """Apply (dy, dx, log(dh), log(dw)) deltas to [N,4] boxes (numpy)."""
    boxes = boxes.astype(np.float32)
    height = boxes[:, 2] - boxes[:, 0]
    width = boxes[:, 3] - boxes[:, 1]
    center_y = boxes[:, 0] + 0.5 * height
    center_x = boxes[:, 1] + 0.5 * width
    center_y += deltas[:, 0] * height
    center_x += deltas[:, 1] * width
    height *= np.exp(deltas[:, 2])
    width *= np.exp(deltas[:, 3])
    y1 = center_y - 0.5 * height
    x1 = center_x - 0.5 * width
    y2 = y1 + height
    x2 = x1 + width
    return np.stack([y1, x1, y2, x2], axis=1)


def box_refinement(box: np.ndarray, gt_box: np.ndarray) -> np.ndarray:
    """Refinement deltas transforming box -> gt_box (numpy)."""
    box = box.astype(np.float32)
    gt_box = gt_box.astype(np.float32)
    height = box[:, 2] - box[:, 0]
    width = box[:, 3] - box[:, 1]
    center_y = box[:, 0] + 0.5 * height
    center_x = box[:, 1] + 0.5 * width
    gt_height = gt_box[:, 2] - gt_box[:, 0]
    gt_width = gt_box[:, 3] - gt_box[:, 1]
    gt_center_y = gt_box[:, 0] + 0.5 * gt_height
    gt_center_x = gt_box[:, 1] + 0.5 * gt_width
    dy = (gt_center_y - center_y) / height
    dx = (gt_center_x - center_x) / width
    dh = np.log(gt_height / height)
    dw = np.log(gt_width / width)
    return np.stack([dy, dx, dh, dw], axis=1)


def norm_boxes(boxes: np.ndarray, shape) -> np.ndarray:
    """Pixel coords -> normalized [0,1] coords (reference norm_boxes).

    Note the reference's convention: divide by (dim - 1) after shifting
    (y2,x2) down by 1 pixel so that normalized boxes are inclusive-exclusive
    compatible.
    """
    h, w = shape[:2]
    scale = np.array([h - 1, w - 1, h - 1, w - 1])
    shift = np.array([0, 0, 1, 1])
    return np.divide((boxes - shift), scale).astype(np.float32)


def denorm_boxes(boxes: np.ndarray, shape) -> np.ndarray:
    """Normalized [0,1] coords -> pixel coords."""
    h, w = shape[:2]
    scale = np.array([h - 1, w - 1, h - 1, w - 1])
    shift = np.array([0, 0, 1, 1])
    return np.around(np.multiply(boxes, scale) + shift).astype(np.int32)


# --------------------------------------------------------------------------
# torch variants (graph plane) — suffix `_t`
# --------------------------------------------------------------------------

def apply_box_deltas_t(boxes: torch.Tensor, deltas: torch.Tensor):
    """Apply deltas to boxes. boxes/deltas: [..., 4]. Torch, differentiable.

    Reference: mrcnn/model.py apply_box_deltas_graph:~210.
    """
    height = boxes[..., 2] - boxes[..., 0]
    width = boxes[..., 3] - boxes[..., 1]
    center_y = boxes[..., 0] + 0.5 * height
    center_x = boxes[..., 1] + 0.5 * width
    center_y = center_y + deltas[..., 0] * height
    center_x = center_x + deltas[..., 1] * width
    height = height * torch.exp(deltas[..., 2])
    width = width * torch.exp(deltas[..., 3])
    y1 = center_y - 0.5 * height
    x1 = center_x - 0.5 * width
    y2 = y1 + height
    x2 = x1 + width
    return torch.stack([y1, x1, y2, x2], dim=-1)


def clip_boxes_t(boxes: torch.Tensor, window: torch.Tensor):
    """Clip boxes to window [y1, x1, y2, x2].

    Reference: mrcnn/model.py clip_boxes_graph:~240.
    """
    wy1, wx1, wy2, wx2 = window.unbind(-1)
    y1 = boxes[..., 0].clamp(min=float(wy1), max=float(wy2)) \
        if window.dim() == 1 else torch.clamp(boxes[..., 0], wy1, wy2)
    if window.dim() == 1:
        x1 = boxes[..., 1].clamp(min=float(wx1), max=float(wx2))
        y2 = boxes[..., 2].clamp(min=float(wy1), max=float(wy2))
        x2 = boxes[..., 3].clamp(min=float(wx1), max=float(wx2))
    else:
        x1 = torch.clamp(boxes[..., 1], wx1, wx2)
        y2 = torch.clamp(boxes[..., 2], wy1, wy2)
        x2 = torch.clamp(boxes[..., 3], wx1, wx2)
    return torch.stack([y1, x1, y2, x2], dim=-1)


def box_refinement_t(box: torch.Tensor, gt_box: torch.Tensor):
    """Torch box refinement (reference box_refinement_graph)."""
    box = box.float()
    gt_box = gt_box.float()
    height = box[:, 2] - box[:, 0]
    width = box[:, 3] - box[:, 1]
    center_y = box[:, 0] + 0.5 * height
    center_x = box[:, 1] + 0.5 * width
    gt_height = gt_box[:, 2] - gt_box[:, 0]
    gt_width = gt_box[:, 3] - gt_box[:, 1]
    gt_center_y = gt_box[:, 0] + 0.5 * gt_height
    gt_center_x = gt_box[:, 1] + 0.5 * gt_width
    dy = (gt_center_y - center_y) / height
    dx = (gt_center_x - center_x) / width
    dh = torch.log(gt_height / height)
    dw = torch.log(gt_width / width)
    return torch.stack([dy, dx, dh, dw], dim=1)


def overlaps_t(boxes1: torch.Tensor, boxes2: torch.Tensor):
    """IoU matrix [N1, N2] (reference overlaps_graph:~445), vectorized."""
    area1 = (boxes1[:, 2] - boxes1[:, 0]) * (boxes1[:, 3] - boxes1[:, 1])
    area2 = (boxes2[:, 2] - boxes2[:, 0]) * (boxes2[:, 3] - boxes2[:, 1])
    y1 = torch.max(boxes1[:, None, 0], boxes2[None, :, 0])
    x1 = torch.max(boxes1[:, None, 1], boxes2[None, :, 1])
    y2 = torch.min(boxes1[:, None, 2], boxes2[None, :, 2])
    x2 = torch.min(boxes1[:, None, 3], boxes2[None, :, 3])
    inter = (y2 - y1).clamp(min=0) * (x2 - x1).clamp(min=0)
    union = area1[:, None] + area2[None, :] - inter
    return inter / union.clamp(min=1e-8)


def norm_boxes_t(boxes: torch.Tensor, shape) -> torch.Tensor:
    """Torch version of norm_boxes (reference norm_boxes_graph)."""
    h, w = float(shape[0]), float(shape[1])
    scale = torch.tensor([h - 1, w - 1, h - 1, w - 1],
                         dtype=boxes.dtype, device=boxes.device)
    shift = torch.tensor([0., 0., 1., 1.],
                         dtype=boxes.dtype, device=boxes.device)
    return (boxes - shift) / scale


def denorm_boxes_t(boxes: torch.Tensor, shape) -> torch.Tensor:
    """Torch version of denorm_boxes (reference denorm_boxes_graph)."""
    h, w = float(shape[0]), float(shape[1])
    scale = torch.tensor([h - 1, w - 1, h - 1, w - 1],
                         dtype=boxes.dtype, device=boxes.device)
    shift = torch.tensor([0., 0., 1., 1.],
                         dtype=boxes.dtype, device=boxes.device)
    return boxes * scale + shift


def batch_slice(inputs, graph_fn, batch_size, names=None):
    """Apply graph_fn per batch item and restack (reference
    utils.batch_slice, mrcnn/utils.py:~820 — a TF workaround for ops
    without batch support; kept for API parity, though the MI355X model
    graph is natively batched and never calls it).

    inputs: tensor or list of tensors with leading batch dim.
    graph_fn: callable over per-item slices returning tensor(s).
    """
    if not isinstance(inputs, list):
        inputs = [inputs]
    outputs = []
    for i in range(batch_size):
        out = graph_fn(*[x[i] for x in inputs])
        if not isinstance(out, (list, tuple)):
            out = [out]
        outputs.append(out)
    # [item][output] -> [output][item], stacked.
    outputs = list(zip(*outputs))
    result = [torch.stack(list(o), dim=0) for o in outputs]
    return result[0] if len(result) == 1 else result

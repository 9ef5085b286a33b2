"""The five Mask R-CNN losses.

Replicates reference mrcnn/model.py:~1020-1200 (smooth_l1_loss,
rpn_class_loss_graph, rpn_bbox_loss_graph, mrcnn_class_loss_graph,
mrcnn_bbox_loss_graph, mrcnn_mask_loss_graph). Semantics preserved:
- RPN class: sparse CE on anchors with match ±1 (0 = neutral, ignored).
- RPN bbox: smooth-L1 on positive anchors; targets are packed per image
  (first n_pos rows of input_rpn_bbox).
- MRCNN class: sparse CE weighted by active_class_ids (multi-dataset).
- MRCNN bbox: smooth-L1 on positive rois, class-specific delta.
- MRCNN mask: binary CE on the GT-class 28x28 mask only.
All mean-reduced over contributing elements (0 when no contributors).
"""

from __future__ import annotations

import torch
import torch.nn.functional as F


def smooth_l1_loss(y_true, y_pred):
    """Elementwise smooth-L1 (reference smooth_l1_loss)."""
    diff = torch.abs(y_true - y_pred)
    less_than_one = (diff < 1.0).to(diff.dtype)
    return less_than_one * 0.5 * diff ** 2 + (1 - less_than_one) * (diff - 0.5)


def rpn_class_loss(rpn_match, rpn_class_logits):
    """rpn_match: [B, A, 1] in {-1, 0, +1}; logits: [B, A, 2]."""
    rpn_match = rpn_match.squeeze(-1)
    anchor_class = (rpn_match == 1).long()
    use = rpn_match != 0
    if use.sum() == 0:
        return rpn_class_logits.sum() * 0.0
    logits = rpn_class_logits[use].float()
    target = anchor_class[use]
    return F.cross_entropy(logits, target)


def rpn_bbox_loss(config, target_bbox, rpn_match, rpn_bbox):
    """target_bbox: [B, MAX_POS, 4] packed per image; rpn_bbox: [B, A, 4]."""
    rpn_match = rpn_match.squeeze(-1)
    # Reference batch_pack_graph packs ALL positive anchors across the
    # batch into one tensor and takes a single global mean, so images
    # with few positives are not over-weighted (mrcnn/model.py:~1060).
    elems = []
    B = rpn_bbox.shape[0]
    for b in range(B):
        pos = (rpn_match[b] == 1).nonzero(as_tuple=True)[0]
        n = pos.numel()
        if n == 0:
            continue
        pred = rpn_bbox[b, pos].float()
        tgt = target_bbox[b, :n].float()
        elems.append(smooth_l1_loss(tgt, pred))
    if not elems:
        return rpn_bbox.sum() * 0.0
    return torch.cat(elems, dim=0).mean()


def mrcnn_class_loss(target_class_ids, pred_class_logits, active_class_ids):
    """target: [B, T]; logits: [B, T, C]; active_class_ids: [B, C]."""
    B, T, C = pred_class_logits.shape
    logits = pred_class_logits.reshape(-1, C).float()
    target = target_class_ids.reshape(-1).long()
    loss = F.cross_entropy(logits, target, reduction="none")
    # Erase losses of predictions of classes not active in the image's
    # dataset (reference mrcnn_class_loss_graph :~1100).
    pred_class = logits.argmax(dim=1)
    batch_ix = torch.arange(B, device=logits.device) \
        .repeat_interleave(T)
    pred_active = active_class_ids[batch_ix, pred_class].float()
    loss = loss * pred_active
    denom = pred_active.sum()
    if denom == 0:
        return logits.sum() * 0.0
    return loss.sum() / denom


def mrcnn_bbox_loss(target_bbox, target_class_ids, pred_bbox):
    """target_bbox: [B,T,4]; ids: [B,T]; pred_bbox: [B,T,C,4]."""
    target_class_ids = target_class_ids.reshape(-1)
    target_bbox = target_bbox.reshape(-1, 4).float()
    B, T, C, _ = pred_bbox.shape
    pred_bbox = pred_bbox.reshape(-1, C, 4).float()
    pos = (target_class_ids > 0).nonzero(as_tuple=True)[0]
    if pos.numel() == 0:
        return pred_bbox.sum() * 0.0
    cls = target_class_ids[pos]
    pred = pred_bbox[pos, cls]
    tgt = target_bbox[pos]
    return smooth_l1_loss(tgt, pred).mean()


def mrcnn_mask_loss(target_masks, target_class_ids, pred_mask_logits):
    """target_masks: [B,T,mh,mw]; ids: [B,T]; pred: [B,T,C,mh,mw] LOGITS."""
    B, T, C, mh, mw = pred_mask_logits.shape
    target_class_ids = target_class_ids.reshape(-1)
    target_masks = target_masks.reshape(-1, mh, mw).float()
    pred = pred_mask_logits.reshape(-1, C, mh, mw).float()
    pos = (target_class_ids > 0).nonzero(as_tuple=True)[0]
    if pos.numel() == 0:
        return pred.sum() * 0.0
    cls = target_class_ids[pos]
    y_pred = pred[pos, cls]
    y_true = target_masks[pos]
    return F.binary_cross_entropy_with_logits(y_pred, y_true)


def compute_losses(config, inputs, outputs):
    """All five losses, weighted by LOSS_WEIGHTS. Returns dict incl. total."""
    lw = config.LOSS_WEIGHTS
    losses = {
        "rpn_class_loss": rpn_class_loss(
            inputs["rpn_match"], outputs["rpn_class_logits"]),
        "rpn_bbox_loss": rpn_bbox_loss(
            config, inputs["rpn_bbox"], inputs["rpn_match"],
            outputs["rpn_bbox"]),
        "mrcnn_class_loss": mrcnn_class_loss(
            outputs["target_class_ids"], outputs["mrcnn_class_logits"],
            inputs["active_class_ids"]),
        "mrcnn_bbox_loss": mrcnn_bbox_loss(
            outputs["target_deltas"], outputs["target_class_ids"],
            outputs["mrcnn_bbox"]),
        "mrcnn_mask_loss": mrcnn_mask_loss(
            outputs["target_masks"], outputs["target_class_ids"],
            outputs["mrcnn_mask"]),
    }
    total = sum(losses[k] * lw.get(k, 1.0) for k in losses)
    losses["total_loss"] = total
    return losses

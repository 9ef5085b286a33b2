"""RoIAlign / crop_and_resize.

Replaces the reference's `tf.image.crop_and_resize` (N2 in SURVEY.md §2.3;
used by PyramidROIAlign mrcnn/model.py:~340 and DetectionTargetLayer's mask
cropping :~620). Two modes:

- ``crop_and_resize``: TF-parity — bilinearly sample a pool_size grid with
  corner alignment (no sampling-bin averaging). This is what the reference
  model computes.
- ``aligned`` (true RoIAlign): sampling_ratio samples per bin, averaged —
  the paper's RoIAlign, better numerics for training.

Boxes are NORMALIZED [y1, x1, y2, x2] (matterport convention: pixel
coords = normalized * (dim-1)).

GPU: hand-written CDNA4 HIP kernels (csrc/hip/roi_align.hip) via the in-tree
extension; fwd + bwd (atomics into feature grads). CPU: pure-torch
grid_sample-based reference below (differentiable; the numerics oracle).
"""

from __future__ import annotations

import torch
import torch.nn.functional as F

from ._ext import require_ext


class _RoiAlignHIP(torch.autograd.Function):
    @staticmethod
    def forward(ctx, features, boxes, box_ind, pool_h, pool_w, sampling_ratio,
                aligned):
        ext = require_ext()
        out = ext.roi_align_forward(features, boxes, box_ind.int(), pool_h,
                                    pool_w, sampling_ratio, aligned)
        ctx.save_for_backward(boxes, box_ind)
        ctx.input_shape = features.shape
        ctx.params = (pool_h, pool_w, sampling_ratio, aligned)
        ctx.in_dtype = features.dtype
        return out

    @staticmethod
    def backward(ctx, grad_out):
        import os
        ext = require_ext()
        boxes, box_ind = ctx.saved_tensors
        pool_h, pool_w, sampling_ratio, aligned = ctx.params
        B, C, H, W = ctx.input_shape
        # MASKRCNN_DETERMINISTIC=1: fixed-point i64 reduction — grads are
        # bitwise-stable across runs (fp atomicAdd order is not). Costs
        # one host sync + an i64 workspace; test/debug mode.
        fn = (ext.roi_align_backward_det
              if os.environ.get("MASKRCNN_DETERMINISTIC") == "1"
              else ext.roi_align_backward)
        grad_in = fn(
            grad_out.contiguous(), boxes, box_ind.int(), B, C, H, W,
            pool_h, pool_w, sampling_ratio, aligned)
        return grad_in.to(ctx.in_dtype), None, None, None, None, None, None


def _torch_sample_grid(features, boxes, box_ind, out_h, out_w, centered):
    """Bilinear-sample an out_h x out_w grid inside each normalized box.

    centered=False: corner-aligned grid (tf.image.crop_and_resize semantics).
    centered=True: samples at cell centers (for bin-averaged RoIAlign).
    Returns [N, C, out_h, out_w].
    """
    B, C, H, W = features.shape
    N = boxes.shape[0]
    device = features.device
    dtype = features.dtype
    if N == 0:
        return features.new_zeros((0, C, out_h, out_w))
    y1, x1, y2, x2 = boxes.unbind(dim=1)
    if centered:
        iy = (torch.arange(out_h, device=device, dtype=dtype) + 0.5) / out_h
        ix = (torch.arange(out_w, device=device, dtype=dtype) + 0.5) / out_w
    else:
        iy = torch.linspace(0, 1, out_h, device=device, dtype=dtype) \
            if out_h > 1 else torch.full((1,), 0.5, device=device, dtype=dtype)
        ix = torch.linspace(0, 1, out_w, device=device, dtype=dtype) \
            if out_w > 1 else torch.full((1,), 0.5, device=device, dtype=dtype)
    # Normalized sample coords in [0,1] image space, per box.
    ys = y1[:, None] + (y2 - y1)[:, None] * iy[None, :]   # [N, out_h]
    xs = x1[:, None] + (x2 - x1)[:, None] * ix[None, :]   # [N, out_w]
    # grid_sample with align_corners=True: -1 -> pixel 0, +1 -> pixel dim-1.
    gy = ys * 2 - 1
    gx = xs * 2 - 1
    grid = torch.stack(
        [gx[:, None, :].expand(N, out_h, out_w),
         gy[:, :, None].expand(N, out_h, out_w)], dim=-1)  # [N, h, w, 2]
    out = features.new_zeros((N, C, out_h, out_w))
    for b in range(B):
        sel = (box_ind == b).nonzero(as_tuple=True)[0]
        if sel.numel() == 0:
            continue
        feat = features[b:b + 1].expand(sel.numel(), C, H, W)
        out[sel] = F.grid_sample(feat, grid[sel], mode="bilinear",
                                 align_corners=True, padding_mode="zeros")
    return out


def crop_and_resize(features: torch.Tensor, boxes: torch.Tensor,
                    box_ind: torch.Tensor, crop_size) -> torch.Tensor:
    """TF-parity crop_and_resize. features [B,C,H,W], boxes [N,4] normalized.

    Returns [N, C, crop_h, crop_w].
    """
    crop_h, crop_w = crop_size
    if features.is_cuda:
        return _RoiAlignHIP.apply(features.contiguous(), boxes.contiguous(),
                                  box_ind, crop_h, crop_w, 0, False)
    return _torch_sample_grid(features, boxes, box_ind, crop_h, crop_w,
                              centered=False)


def roi_align(features: torch.Tensor, boxes: torch.Tensor,
              box_ind: torch.Tensor, output_size, sampling_ratio: int = 2,
              mode: str = "crop_and_resize") -> torch.Tensor:
    """RoIAlign over one feature map.

    mode="crop_and_resize": reference-parity bilinear grid (N2).
    mode="aligned": true sampling-bin-averaged RoIAlign.
    """
    pool_h, pool_w = output_size
    if mode == "crop_and_resize":
        return crop_and_resize(features, boxes, box_ind, (pool_h, pool_w))
    assert mode == "aligned", mode
    s = max(int(sampling_ratio), 1)
    if features.is_cuda:
        return _RoiAlignHIP.apply(features.contiguous(), boxes.contiguous(),
                                  box_ind, pool_h, pool_w, s, True)
    dense = _torch_sample_grid(features, boxes, box_ind, pool_h * s,
                               pool_w * s, centered=True)
    return F.avg_pool2d(dense, kernel_size=s)

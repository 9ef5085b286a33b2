"""Fused FPN lateral + top-down op (HIP MFMA forward, torch backward).

P_i = Conv1x1(C_i; W, b) + NearestUp2(P_{i+1})  — SURVEY.md N7.

Forward runs the hand-written CDNA4 kernel (csrc/hip/fpn_fuse.hip) when
on GPU with bf16 channels-last tensors; the CPU/odd-dtype fallback and the
backward are composed from torch ops (the backward of a 1x1 conv is two
GEMMs — hipBLASLt territory, no custom kernel needed).
"""

from __future__ import annotations

import torch
import torch.nn.functional as F

from ._ext import load_ext


def _fallback(x, w, b, up):
    out = F.conv2d(x, w.view(w.size(0), -1, 1, 1), b.to(x.dtype))
    if up is not None:
        out = out + F.interpolate(up, scale_factor=2, mode="nearest")
    return out


class _FpnFuseFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, w, b, up):
        ext = load_ext()
        if x.is_cuda and ext is None:
            # Never silently degrade to eager on a GPU box.
            from ._ext import require_ext
            require_ext()
        use_hip = (ext is not None and x.is_cuda and
                   x.dtype == torch.bfloat16 and
                   x.is_contiguous(memory_format=torch.channels_last) and
                   x.size(1) % 32 == 0 and
                   # P5 (2048-ch lateral, up=None) measured 0.92-0.96x
                   # vs the plain conv (r2_conv_ab2) — hipBLASLt wins
                   # the pure 1x1 GEMM; the fuse pays only when the
                   # upsample+add round trip is eliminated too.
                   not (up is None and x.size(1) >= 2048))
        ctx.save_for_backward(x, w)
        ctx.has_up = up is not None
        ctx.up_dtype = up.dtype if up is not None else None
        if use_hip:
            out = ext.fpn_fuse(
                x, w.to(torch.bfloat16), b.float(),
                up.to(torch.bfloat16).contiguous(
                    memory_format=torch.channels_last)
                if up is not None else None)
            return out
        with torch.no_grad():
            return _fallback(x, w, b, up)

    @staticmethod
    def backward(ctx, grad_out):
        x, w = ctx.saved_tensors
        go = grad_out
        Cout = w.size(0)
        w4 = w.view(Cout, -1, 1, 1).to(go.dtype)
        # aten.convolution_backward = the fused MIOpen 1x1-conv backward.
        gx, gw, _ = torch.ops.aten.convolution_backward(
            go.contiguous(
                memory_format=torch.channels_last if go.is_cuda
                else torch.contiguous_format),
            x.to(go.dtype), w4, None, [1, 1], [0, 0], [1, 1], False,
            [0, 0], 1,
            [ctx.needs_input_grad[0], ctx.needs_input_grad[1], False])
        grad_x = gx.to(x.dtype) if ctx.needs_input_grad[0] else None
        grad_w = gw.view_as(w).to(w.dtype) \
            if ctx.needs_input_grad[1] else None
        grad_b = go.float().sum(dim=(0, 2, 3)) \
            if ctx.needs_input_grad[2] else None
        grad_up = None
        if ctx.has_up and ctx.needs_input_grad[3]:
            # Nearest-up2 backward = 2x2 sum pooling.
            grad_up = (F.avg_pool2d(go.float(), 2) * 4.0) \
                .to(ctx.up_dtype)
        return grad_x, grad_w, grad_b, grad_up


def fpn_fuse(x: torch.Tensor, weight: torch.Tensor, bias: torch.Tensor,
             up: torch.Tensor = None) -> torch.Tensor:
    """x: [B,Cin,H,W]; weight: [Cout,Cin,1,1] or [Cout,Cin];
    bias: [Cout]; up: [B,Cout,H/2,W/2] or None."""
    return _FpnFuseFn.apply(x, weight, bias, up)

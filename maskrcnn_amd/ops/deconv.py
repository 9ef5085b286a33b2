"""2x2 stride-2 transposed conv (mask-head upsample) — HIP MFMA forward.

SURVEY.md N6: the reference's `Conv2DTranspose(256, (2,2), strides=2)`
(mrcnn/model.py:~1000, build_fpn_mask_graph) upsamples the mask head
14^2 -> 28^2. A 2x2/2 deconv has non-overlapping taps, so the forward is
one MFMA GEMM [M, Cin] x [Cin, 4*Cout] with a pixel-scatter epilogue
(csrc/hip/deconv2x2.hip). Backward reuses aten/MIOpen
convolution_backward (transposed=True) — training-grade, measured
negligible vs the conv stack.
"""

from __future__ import annotations

import torch
import torch.nn.functional as F

from ._ext import load_ext


def pack_weight_deconv(w: torch.Tensor) -> torch.Tensor:
    """ConvTranspose2d weight [Cin, Cout, 2, 2] -> [4, Cout, Cin] bf16
    tap-major (tap = di*2 + dj), contiguous — the kernel's B matrix."""
    Cin, Cout, _, _ = w.shape
    return w.permute(2, 3, 1, 0).reshape(4, Cout, Cin) \
        .contiguous().to(torch.bfloat16)


class _Deconv2x2Fn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, w, b, relu):
        ctx.save_for_backward(x, w)
        ctx.relu = relu
        ext = load_ext()
        use_hip = (ext is not None and x.is_cuda and
                   x.dtype == torch.bfloat16 and
                   x.is_contiguous(memory_format=torch.channels_last) and
                   x.size(1) % 32 == 0)
        if use_hip:
            out = ext.deconv2x2(x, pack_weight_deconv(w),
                                b.float() if b is not None
                                else torch.zeros(w.size(1),
                                                 device=x.device),
                                relu)
        else:
            with torch.no_grad():
                out = F.conv_transpose2d(x, w.to(x.dtype),
                                         b.to(x.dtype) if b is not None
                                         else None, stride=2)
                if relu:
                    out = F.relu(out)
        ctx.save_out = out if relu else None
        return out

    @staticmethod
    def backward(ctx, grad_out):
        x, w = ctx.saved_tensors
        if ctx.relu:
            grad_out = grad_out * (ctx.save_out > 0).to(grad_out.dtype)
        go = grad_out.contiguous(
            memory_format=torch.channels_last if grad_out.is_cuda
            else torch.contiguous_format)
        need_gx, need_gw, need_gb = (ctx.needs_input_grad[0],
                                     ctx.needs_input_grad[1],
                                     ctx.needs_input_grad[2])
        gx, gw, gb = torch.ops.aten.convolution_backward(
            go, x, w.to(go.dtype), [w.size(1)], [2, 2], [0, 0], [1, 1],
            True, [0, 0], 1, [need_gx, need_gw, need_gb])
        return (gx.to(x.dtype) if need_gx else None,
                gw.to(w.dtype) if need_gw else None,
                gb.to(w.dtype) if need_gb else None, None)


def deconv2x2_bias_relu(x, weight, bias, relu=True):
    """Transposed 2x2/2 conv + bias (+ReLU). weight [Cin, Cout, 2, 2]
    (nn.ConvTranspose2d layout)."""
    return _Deconv2x2Fn.apply(x, weight, bias, relu)

"""Fused 3x3 conv + frozen-BN + ReLU (HIP MFMA forward, torch backward).

The SURVEY.md N1 op: frozen BatchNorm folds into the conv (w_eff = w *
scale_per_outchannel; bias_eff = shift + scale * conv_bias), computed
differentiably in torch so training still updates w/bias/BN-affine; the
folded conv+ReLU then runs as one hand-written CDNA4 implicit-GEMM MFMA
kernel (csrc/hip/conv3x3.hip). Backward: dgrad can reuse the forward
kernel on rotated weights; wrw has its own split-K MFMA kernel
(csrc/hip/conv3x3_wrw.hip, MASKRCNN_HIP_WRW=1); aten/MIOpen otherwise.
"""

from __future__ import annotations

import torch
import torch.nn.functional as F

from ._ext import load_ext


def pack_weight(w: torch.Tensor) -> torch.Tensor:
    """[Cout, Cin, 3, 3] -> contiguous [9, Cout, Cin] bf16 (tap-major)."""
    Cout, Cin, _, _ = w.shape
    return w.permute(2, 3, 0, 1).reshape(9, Cout, Cin) \
        .contiguous().to(torch.bfloat16)


def pack_weight_fp8(w: torch.Tensor):
    """[Cout, Cin, 3, 3] -> ([9, Cout, Cin] uint8 e4m3, [Cout] f32
    scale) for the fp8-compute kernel (absmax/448 per out-channel,
    utils/fp8.py convention)."""
    from ..utils.fp8 import quantize_per_channel
    Cout, Cin, _, _ = w.shape
    q, scale = quantize_per_channel(w, dim=0)
    qp = q.view(torch.uint8).permute(2, 3, 0, 1) \
        .reshape(9, Cout, Cin).contiguous()
    return qp, scale.reshape(-1).float()


def quantize_activations_fp8(x: torch.Tensor):
    """Per-tensor e4m3 quant of a channels_last activation — all on
    GPU, no host sync. Returns (uint8 view NHWC, [1] f32 scale)."""
    from ..utils.fp8 import FP8_DTYPE, FP8_MAX
    xf = x.float()
    scale = (xf.abs().amax() / FP8_MAX).clamp(min=1e-12).reshape(1)
    q = (xf / scale).clamp(-FP8_MAX, FP8_MAX).to(FP8_DTYPE) \
        .contiguous(memory_format=torch.channels_last)
    return q.view(torch.uint8), scale


def conv3x3_fp8(x, w_packed_u8, w_scale, bias, relu=True):
    """fp8-compute 3x3 conv: quantize activations per-tensor on the
    fly, run e4m3 x e4m3 MFMA, dequant in the epilogue. Inference-only
    (no autograd). x bf16/f32 channels_last; weights from
    pack_weight_fp8."""
    from ._ext import require_ext
    ext = require_ext()
    xq, a_scale = quantize_activations_fp8(x)
    zb = bias.float() if bias is not None else \
        torch.zeros(w_packed_u8.shape[1], device=x.device)
    return ext.conv3x3_fp8(xq, w_packed_u8, zb, w_scale, a_scale, relu)


def pack_weight_mx(w: torch.Tensor):
    """[Cout, Cin, 3, 3] -> ([9, Cout, Cin] u8 e4m3, [9, Cout, Cin/64]
    u8 e8m0) tap-major for the MX block-scaled kernel: one power-of-two
    scale per (tap, out-channel, 64-in-channel block) — the gfx950
    mfma_scale 16x16x128 scale granularity at OPSEL=0
    (probe-measured, utils/fp8.py mx_scale_src)."""
    from ..utils.fp8 import mx_quantize
    Cout, Cin, _, _ = w.shape
    wt = w.float().permute(2, 3, 0, 1).reshape(9, Cout, Cin)
    q, exp = mx_quantize(wt, dim=-1, block=64)
    return (q.view(torch.uint8).contiguous(), exp.contiguous())


def quantize_activations_mx(x: torch.Tensor):
    """Per-(pixel, 64-channel-block) MX quantization of a channels_last
    activation. Returns (xq [B,H,W,C] u8 contiguous, xs [B,H,W,C/64]
    u8 e8m0). All on GPU, no host sync. One fused HIP pass
    (csrc/hip/mx_quant.hip) when eligible — the eager-torch version is
    ~6 kernels and cost 2-3x the MX conv itself (r13_conv_ab)."""
    ext = load_ext()
    if (ext is not None and x.is_cuda and x.dtype == torch.bfloat16 and
            x.is_contiguous(memory_format=torch.channels_last) and
            x.size(1) % 64 == 0):
        return ext.mx_quant(x)
    from ..utils.fp8 import mx_quantize
    xn = x.permute(0, 2, 3, 1).float()        # NHWC values
    q, exp = mx_quantize(xn, dim=-1, block=64)
    return q.view(torch.uint8).contiguous(), exp.contiguous()


def conv3x3_mx(x, wq, ws, bias, relu=True):
    """MX block-scaled e4m3 3x3 conv at the 2x fp8 MFMA rate
    (mfma_scale_f32_16x16x128_f8f6f4). Inference-only. x bf16/f32
    channels_last [B,C,H,W]; (wq, ws) from pack_weight_mx. Returns
    bf16 channels_last [B,Cout,H,W]."""
    from ._ext import require_ext
    ext = require_ext()
    xq, xs = quantize_activations_mx(x)
    zb = bias.float() if bias is not None else \
        torch.zeros(wq.shape[1], device=x.device)
    return ext.conv3x3_mx(xq, xs, wq, ws, zb, relu)


def rotate_for_dgrad(w: torch.Tensor) -> torch.Tensor:
    """Weight transform making bwd-data a FORWARD conv:
    dgrad(x) = conv3x3(grad_out, w') with
    w'[ci, co, r, s] = w[co, ci, 2-r, 2-s]  (s1p1, 3x3)."""
    return w.flip(2, 3).permute(1, 0, 2, 3)


class _Conv3x3Fn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, w, b, relu):
        ctx.save_for_backward(x, w)
        ctx.relu = relu
        ext = load_ext()
        if x.is_cuda and ext is None:
            # Never silently degrade to eager on a GPU box.
            from ._ext import require_ext
            require_ext()
        use_hip = (ext is not None and x.is_cuda and
                   x.dtype == torch.bfloat16 and
                   x.is_contiguous(memory_format=torch.channels_last) and
                   x.size(1) % 32 == 0)
        if use_hip:
            out = ext.conv3x3_fwd(x, pack_weight(w), b.float(), relu)
        else:
            with torch.no_grad():
                out = F.conv2d(x, w.to(x.dtype), b.to(x.dtype), padding=1)
                if relu:
                    out = F.relu(out)
        ctx.save_out = out if relu else None
        return out

    @staticmethod
    def backward(ctx, grad_out):
        x, w = ctx.saved_tensors
        if ctx.relu:
            grad_out = grad_out * (ctx.save_out > 0).to(grad_out.dtype)
        grad_x = grad_w = grad_b = None
        wd = w.to(grad_out.dtype)
        go_cl = grad_out.contiguous(
            memory_format=torch.channels_last if grad_out.is_cuda
            else torch.contiguous_format)
        # bwd-data of a 3x3 s1p1 conv IS a forward 3x3 conv with
        # rotated-transposed weights — reuse the hand MFMA kernel where
        # its policy wins (note the role swap: the dgrad's "Cin" is this
        # conv's Cout).
        ext = load_ext()
        use_hip_dgrad = (
            ctx.needs_input_grad[0] and ext is not None and
            grad_out.is_cuda and go_cl.dtype == torch.bfloat16 and
            go_cl.is_contiguous(memory_format=torch.channels_last) and
            w.size(0) % 32 == 0)
        if use_hip_dgrad:
            from .conv_policy import use_hip_conv3x3
            use_hip_dgrad = use_hip_conv3x3(
                go_cl.size(0), w.size(0), go_cl.size(2), go_cl.size(3),
                w.size(1))
        # Weight gradient: HIP split-K MFMA kernel (csrc/hip/
        # conv3x3_wrw.hip — wrw was 47% of the original MIOpen step
        # profile) when opted in; aten/MIOpen otherwise. Off by default
        # until GPU-validated (round-2 block1; MASKRCNN_HIP_WRW=1).
        import os
        use_hip_wrw = (
            ctx.needs_input_grad[1] and ext is not None and
            grad_out.is_cuda and go_cl.dtype == torch.bfloat16 and
            x.dtype == torch.bfloat16 and
            x.is_contiguous(memory_format=torch.channels_last) and
            x.size(1) % 8 == 0 and w.size(0) % 8 == 0 and
            os.environ.get("MASKRCNN_HIP_WRW", "0") == "1")
        need_aten_gx = ctx.needs_input_grad[0] and not use_hip_dgrad
        need_aten_gw = ctx.needs_input_grad[1] and not use_hip_wrw
        gx = gw = None
        if need_aten_gx or need_aten_gw:
            gx, gw, _ = torch.ops.aten.convolution_backward(
                go_cl, x, wd, None, [1, 1], [1, 1], [1, 1], False,
                [0, 0], 1, [need_aten_gx, need_aten_gw, False])
        if use_hip_wrw:
            # [9, Cout, Cin] tap-major fp32 -> [Cout, Cin, 3, 3].
            gw = ext.conv3x3_wrw(x, go_cl) \
                .reshape(3, 3, w.size(0), w.size(1)).permute(2, 3, 0, 1)
        if use_hip_dgrad:
            wr = pack_weight(rotate_for_dgrad(w))
            zb = torch.zeros(w.size(1), device=w.device,
                             dtype=torch.float32)
            grad_x = ext.conv3x3_fwd(go_cl, wr, zb, False).to(x.dtype)
        elif ctx.needs_input_grad[0]:
            grad_x = gx.to(x.dtype)
        if ctx.needs_input_grad[1]:
            grad_w = gw.to(w.dtype)
        if ctx.needs_input_grad[2]:
            grad_b = grad_out.float().sum(dim=(0, 2, 3))
        return grad_x, grad_w, grad_b, None


def conv3x3_bias_relu(x, weight, bias, relu=True):
    """3x3 stride-1 pad-1 conv + bias (+ReLU). weight [Cout,Cin,3,3]."""
    return _Conv3x3Fn.apply(x, weight, bias, relu)


def conv3x3_bn_relu(x, weight, conv_bias, bn_weight, bn_bias,
                    running_mean, running_var, eps=1e-5, relu=True):
    """Fold frozen BN into the conv, then run the fused kernel.

    y = BN(conv(x, w) + cb) = conv(x, w * s) + (cb - mean) * s + bn_b,
    with s = bn_w / sqrt(var + eps). The fold is differentiable, so
    training updates every parameter while the fwd runs one kernel.
    """
    scale = bn_weight * (running_var + eps).rsqrt()
    w_eff = weight * scale.view(-1, 1, 1, 1)
    b_eff = bn_bias + (conv_bias - running_mean) * scale \
        if conv_bias is not None else bn_bias - running_mean * scale
    return _Conv3x3Fn.apply(x, w_eff, b_eff, relu)

"""fp8 (OCP e4m3fn) weight quantization for inference.

BASELINE config #5 is the fp8-weight R101-FPN inference path: conv /
linear weights are stored as e4m3fn with a per-output-channel fp32 scale
(absmax calibration) — halving weight bytes and HBM traffic — and
dequantized to bf16 at load. gfx950's fp8 is OCP e4m3fn (NOT the MI300X
fnuz variant — cdna_hip_programming.md §4); torch.float8_e4m3fn matches.

The MX-block-scaled fp8 MFMA compute path (the 2x-rate
mfma_scale_f32_*_f8f6f4 kernels) is the round-2 follow-up; this module
establishes the format, the quantization error budget, and the
SavedModel round trip.
"""

from __future__ import annotations

import torch

FP8_DTYPE = torch.float8_e4m3fn
FP8_MAX = 448.0  # e4m3fn max normal


def quantize_per_channel(w: torch.Tensor, dim: int = 0):
    """Quantize to e4m3fn with a per-channel (dim) scale.

    Returns (w_fp8, scale) with w ~= w_fp8.float() * scale (broadcast
    over dim).
    """
    wf = w.float()
    red_dims = [d for d in range(w.dim()) if d != dim]
    absmax = wf.abs().amax(dim=red_dims, keepdim=True).clamp(min=1e-12)
    scale = (absmax / FP8_MAX)
    q = (wf / scale).clamp(-FP8_MAX, FP8_MAX).to(FP8_DTYPE)
    return q, scale


def dequantize(q: torch.Tensor, scale: torch.Tensor,
               dtype=torch.bfloat16) -> torch.Tensor:
    return (q.float() * scale).to(dtype)


def quantize_state_dict(state: dict, min_numel: int = 4096):
    """Quantize conv/linear weights of a state dict to fp8.

    Returns a new dict where each eligible 'w' becomes two entries
    'w::fp8' (uint8 view) and 'w::scale'; everything else passes through.
    BN/bias/small tensors stay fp32 (negligible bytes, high sensitivity).
    """
    out = {}
    for k, v in state.items():
        if (k.endswith(".weight") and v.dim() in (2, 4) and
                v.numel() >= min_numel and v.is_floating_point()):
            q, scale = quantize_per_channel(v, dim=0)
            out[k + "::fp8"] = q.view(torch.uint8)
            out[k + "::scale"] = scale.float()
        else:
            out[k] = v
    return out


def dequantize_state_dict(state: dict, dtype=torch.float32):
    """Inverse of quantize_state_dict (weights come back in `dtype`)."""
    out = {}
    for k, v in state.items():
        if k.endswith("::fp8"):
            base = k[:-len("::fp8")]
            scale = state[base + "::scale"]
            out[base] = dequantize(v.view(FP8_DTYPE), scale, dtype)
        elif k.endswith("::scale"):
            continue
        else:
            out[k] = v
    return out


def fp8_bytes_saved(state: dict) -> int:
    saved = 0
    for k, v in state.items():
        if k.endswith("::fp8"):
            saved += v.numel() * 3  # fp32 -> 1 byte
    return saved


# --------------------------------------------------------------- MX fp8
# OCP MX block scaling for the gfx950 mfma_scale_*_f8f6f4 kernels
# (round-2 compute path): 32 consecutive K elements share one e8m0
# scale (a biased power-of-two exponent byte). The intrinsic probe
# (NOTES_ROUND2.md) confirmed __builtin_amdgcn_mfma_scale_f32_
# 16x16x128_f8f6f4 takes i32x8 operands (32 e4m3 bytes/lane) plus
# per-lane scale ints.

MX_BLOCK = 32
_E4M3_EMAX = 8  # floor(log2(448))


def mx_quantize(x: torch.Tensor, dim: int = -1, block: int = MX_BLOCK):
    """Block-scaled MX quantization along `dim` (default block 32; the
    gfx950 mfma_scale 16x16x128 instruction applies ONE e8m0 per 64
    consecutive k at OPSEL=0 — measured by the r10 scale-mapping probe —
    so the conv kernel quantizes with block=64).

    Returns (q e4m3fn same shape, exp uint8 with dim shrunk by block):
    x ~= q.float() * 2^(exp - 127) broadcast over each block.
    """
    dim = dim % x.dim()
    K = x.shape[dim]
    assert K % block == 0, f"dim size {K} not a multiple of {block}"
    xf = x.float().movedim(dim, -1)
    shape = xf.shape
    xb = xf.reshape(*shape[:-1], K // block, block)
    amax = xb.abs().amax(dim=-1, keepdim=True).clamp(min=2.0 ** -126)
    e = torch.floor(torch.log2(amax)) - _E4M3_EMAX
    e = e.clamp(-127, 127)
    scale = torch.pow(2.0, e)
    q = (xb / scale).clamp(-FP8_MAX, FP8_MAX).to(FP8_DTYPE)
    q = q.reshape(shape).movedim(-1, dim)
    exp = (e.squeeze(-1) + 127).to(torch.uint8).movedim(-1, dim)
    return q, exp


def mx_dequantize(q: torch.Tensor, exp: torch.Tensor, dim: int = -1,
                  dtype=torch.float32,
                  block: int = MX_BLOCK) -> torch.Tensor:
    dim = dim % q.dim()
    qf = q.float().movedim(dim, -1)
    shape = qf.shape
    qb = qf.reshape(*shape[:-1], shape[-1] // block, block)
    e = exp.movedim(dim, -1).float().unsqueeze(-1) - 127.0
    out = qb * torch.pow(2.0, e)
    return out.reshape(shape).movedim(-1, dim).to(dtype)


def mx_scale_src(row: int, kblock: int):
    """(row, 32-element k-block) -> (source lane, byte) of the
    mfma_scale_f32_16x16x128 A-scale operand at OPSEL=(0,0), as
    MEASURED by the binary-encoded probe on MI355X (gpurun_out/
    r10_probe.log): blocks {0,1} read byte 0 of lane `row`, blocks
    {2,3} byte 0 of lane `16+row` — one e8m0 per 64 consecutive k."""
    return (16 * (kblock // 2) + row, 0)

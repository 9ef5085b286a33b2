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

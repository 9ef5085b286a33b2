"""CPU simulation of the conv3x3_wrw kernel's index math.

The kernel (csrc/hip/conv3x3_wrw.hip) can only execute on gfx950; this
test transcribes its exact staging loops (thread -> (k, channel-octet)
maps, halo columns, bounds guards) and tile GEMM per the documented
MFMA fragment semantics (A[m][k] row-major from lds_dy, B[k][n] =
lds_x col k+dc) and checks the result against autograd's weight
gradient. It verifies the DESIGN — seg decomposition, transposed
staging, dc/halo alignment, tap->(dr,dc) mapping, GEMM orientation,
split-K accumulation — leaving only the (forward-kernel-proven) MFMA
lane plumbing to the round-2 hardware pass.
"""

import numpy as np
import pytest
import torch
import torch.nn.functional as F

CO_T = 64
CI_T = 64
SW = 32
XCOLS = SW + 2
KP = 40


def _sim_wrw(x, dy, split=3):
    """x: [B,H,W,Cin] f32 (NHWC values), dy: [B,H,W,Cout]."""
    B, H, W, Cin = x.shape
    Cout = dy.shape[3]
    dw = np.zeros((9, Cout, Cin), np.float32)
    co_tiles = (Cout + CO_T - 1) // CO_T
    ci_tiles = (Cin + CI_T - 1) // CI_T
    spr = (W + SW - 1) // SW
    total_segs = B * H * spr

    for bx in range(co_tiles):
        for by in range(ci_tiles * 3):
            for bz in range(split):
                co0 = bx * CO_T
                ci_tile, dr = by // 3, by % 3
                ci0 = ci_tile * CI_T
                for seg in range(bz, total_segs, split):
                    w0 = (seg % spr) * SW
                    h = (seg // spr) % H
                    n = seg // (spr * H)
                    gy = h + dr - 1
                    if gy < 0 or gy >= H:
                        continue
                    # --- dy staging: tid -> (k = tid>>3, c8 = (tid&7)*8)
                    lds_dy = np.zeros((CO_T, KP), np.float32)
                    for tid in range(256):
                        k = tid >> 3
                        c8 = (tid & 7) * 8
                        gw = w0 + k
                        if gw < W and co0 + c8 < Cout:
                            lds_dy[c8:c8 + 8, k] = dy[n, h, gw,
                                                      co0 + c8:co0 + c8 + 8]
                    # --- x halo staging: u -> (c8 = (u&7)*8, col = u>>3)
                    lds_x = np.zeros((CI_T, KP), np.float32)
                    for u in range(XCOLS * 8):
                        c8 = (u & 7) * 8
                        col = u >> 3
                        gx = w0 - 1 + col
                        if 0 <= gx < W and ci0 + c8 < Cin:
                            lds_x[c8:c8 + 8, col] = x[n, gy, gx,
                                                      ci0 + c8:ci0 + c8 + 8]
                    # --- tile GEMM per dc (MFMA fragment semantics):
                    # D[m][n] += sum_k lds_dy[m][k] * lds_x[n][k + dc]
                    for dc in range(3):
                        tap = dr * 3 + dc
                        a = lds_dy[:, :SW]                       # [64, 32]
                        b = lds_x[:, dc:dc + SW]                 # [64, 32]
                        d = a @ b.T                              # [co, ci]
                        co1 = min(co0 + CO_T, Cout)
                        ci1 = min(ci0 + CI_T, Cin)
                        dw[tap, co0:co1, ci0:ci1] += \
                            d[:co1 - co0, :ci1 - ci0]
    return dw


@pytest.mark.parametrize("b,h,w,cin,cout,split", [
    (1, 5, 7, 8, 8, 1),       # W < one segment (zero-padded k)
    (2, 6, 40, 16, 8, 3),     # two segments per row + split-K
    (1, 4, 33, 8, 16, 2),     # ragged second segment (1 pixel)
])
def test_wrw_simulation_matches_autograd(b, h, w, cin, cout, split):
    torch.manual_seed(0)
    x = torch.randn(b, cin, h, w)
    dy = torch.randn(b, cout, h, w)
    xr = x.requires_grad_(True)
    wt = torch.zeros(cout, cin, 3, 3, requires_grad=True)
    F.conv2d(xr, wt, padding=1).backward(dy)

    x_nhwc = x.detach().permute(0, 2, 3, 1).contiguous().numpy()
    dy_nhwc = dy.permute(0, 2, 3, 1).contiguous().numpy()
    dw = _sim_wrw(x_nhwc, dy_nhwc, split=split)
    got = torch.from_numpy(dw).reshape(3, 3, cout, cin) \
        .permute(2, 3, 0, 1)
    torch.testing.assert_close(got, wt.grad, atol=1e-3, rtol=1e-4)


def test_wrw_sim_channel_tile_guards():
    """Cout/Cin beyond one 64-tile exercise the co0/ci0 guards."""
    torch.manual_seed(1)
    b, h, w, cin, cout = 1, 3, 9, 72, 80
    x = torch.randn(b, cin, h, w)
    dy = torch.randn(b, cout, h, w)
    wt = torch.zeros(cout, cin, 3, 3, requires_grad=True)
    F.conv2d(x, wt, padding=1).backward(dy)
    dw = _sim_wrw(x.permute(0, 2, 3, 1).contiguous().numpy(),
                  dy.permute(0, 2, 3, 1).contiguous().numpy(), split=2)
    got = torch.from_numpy(dw).reshape(3, 3, cout, cin) \
        .permute(2, 3, 0, 1)
    torch.testing.assert_close(got, wt.grad, atol=1e-3, rtol=1e-4)


def test_xcd_swizzle_bijective():
    """The v3 kernel's XCD-aware blockIdx remap (conv3x3_v3.hip) must be
    a bijection for EVERY workgroup count — the naive (orig%8)*ceil+...
    form silently drops/duplicates tiles when nwg%8 != 0 (guide §5)."""
    NXCD = 8
    for nwg in list(range(1, 100)) + [256, 768, 1000, 4095, 4096]:
        seen = set()
        for bid in range(nwg):
            q, r = divmod(nwg, NXCD)
            xcd, pos = bid % NXCD, bid // NXCD
            out = (xcd * (q + 1) if xcd < r
                   else r * (q + 1) + (xcd - r) * q) + pos
            assert 0 <= out < nwg, (nwg, bid, out)
            seen.add(out)
        assert len(seen) == nwg, f"not bijective at nwg={nwg}"


def _sim_conv3x3_fp8(xq, wq, bias, w_scale, a_scale, relu=True):
    """Transcription of conv3x3_fp8.hip: implicit GEMM over 9 taps x
    Cin/32 chunks with the kernel's exact staging maps (thread ->
    (pixel row, 8-byte channel octet)) and the dequant epilogue
    acc * a_scale * w_scale[co] + bias.

    xq: [B,H,W,Cin] int-coded e4m3 values (float here), wq: [9,Cout,
    Cin] tap-major. Returns [B,H,W,Cout] float."""
    B, H, W, Cin = xq.shape
    _, Cout, _ = wq.shape
    BM, BN, BK = 64, 64, 32
    M = B * H * W
    out = np.zeros((M, Cout), np.float32)
    for m0 in range(0, M, BM):
        for n0 in range(0, Cout, BN):
            acc = np.zeros((BM, BN), np.float32)
            for tap in range(9):
                dy, dx = tap // 3 - 1, tap % 3 - 1
                for k0 in range(0, Cin, BK):
                    a_tile = np.zeros((BM, BK), np.float32)
                    b_tile = np.zeros((BN, BK), np.float32)
                    for tid in range(256):
                        lrow, lcol = tid >> 2, (tid & 3) * 8
                        m = m0 + lrow
                        if m < M:
                            px, rest = m % W, m // W
                            py, pb = rest % H, rest // H
                            sy, sx = py + dy, px + dx
                            if 0 <= sy < H and 0 <= sx < W:
                                a_tile[lrow, lcol:lcol + 8] = \
                                    xq[pb, sy, sx, k0 + lcol:k0 + lcol + 8]
                        n = n0 + lrow
                        if n < Cout:
                            b_tile[lrow, lcol:lcol + 8] = \
                                wq[tap, n, k0 + lcol:k0 + lcol + 8]
                    acc += a_tile @ b_tile.T
            for r in range(min(BM, M - m0)):
                for c in range(min(BN, Cout - n0)):
                    v = acc[r, c] * a_scale * w_scale[n0 + c] + bias[n0 + c]
                    out[m0 + r, n0 + c] = max(v, 0.0) if relu else v
    return out.reshape(B, H, W, Cout)


def test_fp8_kernel_simulation_matches_reference():
    """Design check of conv3x3_fp8.hip against the quantized-math
    reference (exact: both sides use the same e4m3-coded operands)."""
    torch.manual_seed(0)
    B, H, W, Cin, Cout = 1, 5, 6, 32, 16
    from maskrcnn_amd.ops.conv3x3 import pack_weight_fp8
    from maskrcnn_amd.utils.fp8 import FP8_DTYPE, FP8_MAX

    x = torch.randn(B, Cin, H, W)
    w = torch.randn(Cout, Cin, 3, 3) * 0.1
    bias = torch.randn(Cout)
    qp, w_scale = pack_weight_fp8(w)
    amax = x.abs().amax().clamp(min=1e-12)
    a_scale = float(amax / FP8_MAX)
    xq = (x / a_scale).clamp(-FP8_MAX, FP8_MAX).to(FP8_DTYPE)

    sim = _sim_conv3x3_fp8(
        xq.float().permute(0, 2, 3, 1).contiguous().numpy(),
        qp.view(torch.float8_e4m3fn).float().numpy(),
        bias.numpy(), w_scale.numpy(), a_scale, relu=True)

    # Reference: same dequantized operands through torch conv.
    x_deq = xq.float() * a_scale
    w_deq = (qp.view(torch.float8_e4m3fn).float() *
             w_scale.view(1, -1, 1)) \
        .reshape(3, 3, Cout, Cin).permute(2, 3, 0, 1)
    ref = F.relu(F.conv2d(x_deq, w_deq, bias, padding=1))
    got = torch.from_numpy(sim).permute(0, 3, 1, 2)
    torch.testing.assert_close(got, ref, atol=1e-3, rtol=1e-3)


def test_detection_prep_simulation_matches_torch():
    """Transcription of detection_prep.hip's per-roi math vs the
    DetectionLayer torch prep (argmax, class delta, apply, clip,
    zero-score masking)."""
    from maskrcnn_amd.data.synthetic import ShapesConfig
    from maskrcnn_amd.utils.boxes import apply_box_deltas_t

    class C(ShapesConfig):
        DETECTION_MIN_CONFIDENCE = 0.4

    cfg = C()
    torch.manual_seed(0)
    B, N, Ccls = 2, 30, cfg.NUM_CLASSES
    rois = torch.rand(B, N, 4)
    rois[..., 2:] = rois[..., :2] * 0.5 + 0.4
    probs = torch.softmax(torch.randn(B, N, Ccls) * 2, -1)
    deltas = torch.randn(B, N, Ccls, 4) * 0.1
    windows = torch.tensor([[0.0, 0.0, 1.0, 1.0],
                            [0.1, 0.05, 0.9, 0.95]])
    std = torch.tensor(cfg.BBOX_STD_DEV, dtype=torch.float32)

    # --- kernel math, one "thread" per (b, n)
    ref_boxes = torch.zeros(B, N, 4)
    ref_cls = torch.zeros(B, N, dtype=torch.long)
    ref_scores = torch.zeros(B, N)
    import math
    for b in range(B):
        for n in range(N):
            p = probs[b, n]
            best = int(p.argmax())
            best_v = float(p[best])
            d = deltas[b, n, best] * std
            y1, x1, y2, x2 = rois[b, n].tolist()
            h, w = y2 - y1, x2 - x1
            cy = y1 + 0.5 * h + float(d[0]) * h
            cx = x1 + 0.5 * w + float(d[1]) * w
            h *= math.exp(float(d[2]))
            w *= math.exp(float(d[3]))
            y1, x1 = cy - 0.5 * h, cx - 0.5 * w
            y2, x2 = y1 + h, x1 + w
            wy1, wx1, wy2, wx2 = windows[b].tolist()
            box = [min(max(y1, wy1), wy2), min(max(x1, wx1), wx2),
                   min(max(y2, wy1), wy2), min(max(x2, wx1), wx2)]
            ref_boxes[b, n] = torch.tensor(box)
            ref_cls[b, n] = best
            keep = best > 0 and best_v >= cfg.DETECTION_MIN_CONFIDENCE
            ref_scores[b, n] = best_v if keep else 0.0

    # --- the layer's torch prep (the eager path's block)
    class_ids = probs.argmax(2)
    class_scores = probs.gather(2, class_ids[..., None]).squeeze(-1)
    ds = deltas.gather(2, class_ids[..., None, None]
                       .expand(-1, -1, 1, 4)).squeeze(2)
    refined = apply_box_deltas_t(rois, ds * std)
    wy1, wx1, wy2, wx2 = windows.unsqueeze(2).unbind(dim=1)
    refined = torch.stack(
        [refined[..., 0].clamp(wy1, wy2), refined[..., 1].clamp(wx1, wx2),
         refined[..., 2].clamp(wy1, wy2), refined[..., 3].clamp(wx1, wx2)],
        dim=-1)
    keep = (class_ids > 0) & (class_scores >=
                              cfg.DETECTION_MIN_CONFIDENCE)
    scores_m = torch.where(keep, class_scores,
                           torch.zeros_like(class_scores))

    torch.testing.assert_close(ref_boxes, refined, atol=1e-5, rtol=1e-5)
    assert torch.equal(ref_cls, class_ids)
    torch.testing.assert_close(ref_scores, scores_m, atol=1e-6,
                               rtol=1e-6)

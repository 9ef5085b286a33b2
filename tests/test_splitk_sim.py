"""CPU simulation of conv3x3_splitk.hip's NEW index math.

The tile/staging/MFMA structure is byte-identical to conv3x3.hip v1
(hardware-validated round 1); what split-K adds is (a) the flattened
(tap, k-chunk) -> grid.z partition, (b) per-z fp32 slab accumulation,
(c) the fixed-order reduce + bias + ReLU epilogue. This test
transcribes exactly those loops (same c ≡ z (mod split) striding, same
tap/k0 decomposition, same shifted-window zero-fill semantics per
BK-chunk) and checks the reduced result against torch's conv2d.

GPU execution of the kernel itself is gated behind
MASKRCNN_HIP_CONV_SPLITK=1 (see test_conv3x3.py) — this round's GPU
budget closed before it could run on hardware.
"""

import numpy as np
import pytest
import torch
import torch.nn.functional as F

BK = 32
SPLIT = 8


def _sim_splitk(x, w, bias, relu=True, split=SPLIT):
    """x: [B,H,W,Cin] f32; w: [9,Cout,Cin] f32 (tap-major, the packed
    layout the kernel receives); returns [B,H,W,Cout]."""
    B, H, W, Cin = x.shape
    Cout = w.shape[1]
    M = B * H * W
    KC = Cin // BK
    chunks = 9 * KC
    slabs = np.zeros((split, M, Cout), np.float32)

    xs = x.reshape(M, Cin)
    for z in range(split):
        for c in range(z, chunks, split):   # the kernel's z-stride
            tap, k0 = c // KC, (c % KC) * BK
            dy, dx = tap // 3 - 1, tap % 3 - 1
            # a_tile staging semantics: per output pixel m, the
            # shifted source pixel's BK channels, zero outside image.
            a = np.zeros((M, BK), np.float32)
            for m in range(M):
                px = m % W
                py = (m // W) % H
                pb = m // (W * H)
                sy, sx = py + dy, px + dx
                if 0 <= sy < H and 0 <= sx < W:
                    a[m] = x[pb, sy, sx, k0:k0 + BK]
            slabs[z] += a @ w[tap, :, k0:k0 + BK].T
    out = slabs[0].copy()
    for z in range(1, split):               # fixed-order reduce
        out += slabs[z]
    out += bias[None, :]
    if relu:
        out = np.maximum(out, 0.0)
    return out.reshape(B, H, W, Cout)


def test_chunk_partition_covers_all_exactly_once():
    for cin in (64, 256, 512):
        chunks = 9 * (cin // BK)
        seen = []
        for z in range(SPLIT):
            seen += list(range(z, chunks, SPLIT))
        assert sorted(seen) == list(range(chunks))


def test_splitk_sim_matches_conv2d():
    rng = np.random.default_rng(11)
    B, H, W, Cin, Cout = 1, 6, 5, 64, 32
    x = rng.standard_normal((B, H, W, Cin), dtype=np.float32)
    wk = rng.standard_normal((3, 3, Cin, Cout), dtype=np.float32) * 0.1
    bias = rng.standard_normal(Cout).astype(np.float32)
    # the kernel's packed layout: [tap, Cout, Cin]
    w_packed = wk.transpose(0, 1, 3, 2).reshape(9, Cout, Cin)

    got = _sim_splitk(x, w_packed, bias, relu=True)

    xt = torch.from_numpy(x).permute(0, 3, 1, 2)
    wt = torch.from_numpy(wk).permute(3, 2, 0, 1)  # [Cout,Cin,3,3]
    ref = F.conv2d(xt, wt, torch.from_numpy(bias), padding=1)
    ref = torch.relu(ref).permute(0, 2, 3, 1).numpy()
    np.testing.assert_allclose(got, ref, rtol=1e-4, atol=1e-4)


def test_splitk_sim_split_invariance():
    """Any split degree must produce the same reduced result (the
    partition only reorders the fp32 slab sums)."""
    rng = np.random.default_rng(3)
    x = rng.standard_normal((1, 4, 4, 32), dtype=np.float32)
    w = rng.standard_normal((9, 16, 32), dtype=np.float32)
    b = np.zeros(16, np.float32)
    r1 = _sim_splitk(x, w, b, relu=False, split=1)
    r4 = _sim_splitk(x, w, b, relu=False, split=4)
    r8 = _sim_splitk(x, w, b, relu=False, split=8)
    # fp32 summation-order noise only (different slab order)
    np.testing.assert_allclose(r1, r4, rtol=1e-4, atol=1e-5)
    np.testing.assert_allclose(r1, r8, rtol=1e-4, atol=1e-5)


def _sim_splitk_lanes(x, w, bias, split=SPLIT, relu=True):
    """LANE-LEVEL transcription of conv3x3_splitk.hip: per-thread LDS
    staging (lrow = tid>>2, lcol = (tid&3)*8, two short4s = 16
    channels), per-lane MFMA fragment reads (arow/bcol/kfrag) with the
    v_mfma_f32_16x16x32_bf16 semantics (acc lane l row r covers output
    row 16*?+..., see fragment map), and the exact epilogue index map
    (m = m0 + wave_m*32 + mi*16 + (lane>>4)*4 + r). Numerically fp32
    (the HW multiplies bf16 inputs; tolerance covers that)."""
    B, H, W, Cin = x.shape
    Cout = w.shape[1]
    M = B * H * W
    BM = BN = 64
    BKc = 32
    KC = Cin // BKc
    CHUNKS = 9 * KC
    grid_x = (M + BM - 1) // BM
    grid_y = (Cout + BN - 1) // BN
    slabs = np.zeros((split, M, Cout), np.float32)

    for bz in range(split):
        for bx in range(grid_x):
            for by in range(grid_y):
                m0 = bx * BM
                n0 = by * BN
                # acc[wave][mi][ni][lane][r]
                acc = np.zeros((4, 2, 2, 64, 4), np.float32)
                for c in range(bz, CHUNKS, split):
                    tap, k0 = c // KC, (c % KC) * BKc
                    dy, dx = tap // 3 - 1, tap % 3 - 1
                    a_tile = np.zeros((BM, BKc), np.float32)
                    b_tile = np.zeros((BN, BKc), np.float32)
                    for tid in range(256):
                        lrow = tid >> 2
                        lcol = (tid & 3) * 8
                        m_load = m0 + lrow
                        if m_load < M:
                            px = m_load % W
                            py = (m_load // W) % H
                            pb = m_load // (W * H)
                            sy, sx = py + dy, px + dx
                            if 0 <= sy < H and 0 <= sx < W:
                                a_tile[lrow, lcol:lcol + 8] = \
                                    x[pb, sy, sx, k0 + lcol:k0 + lcol + 8]
                        n = n0 + lrow
                        if n < Cout:
                            b_tile[lrow, lcol:lcol + 8] = \
                                w[tap, n, k0 + lcol:k0 + lcol + 8]
                    # Standard mfma_16x16x32 per-fragment math: for the
                    # 16x16 output block (mi, ni) of wave (wave_m,
                    # wave_n): OUT = Afrag @ Bfrag^T where Afrag is the
                    # 16x32 block rows [wave_m*32+mi*16, +16) cols
                    # [0, 32) of a_tile, Bfrag likewise of b_tile.
                    for wave in range(4):
                        wave_m, wave_n = wave >> 1, wave & 1
                        for mi in range(2):
                            r0 = wave_m * 32 + mi * 16
                            A = a_tile[r0:r0 + 16, :]
                            for ni in range(2):
                                c0 = wave_n * 32 + ni * 16
                                Bf = b_tile[c0:c0 + 16, :]
                                out16 = A @ Bf.T
                                # distribute into the acc lane map:
                                # out16[row, col]: col = lane & 15,
                                # row = (lane>>4)*4 + r
                                for lane in range(64):
                                    col = lane & 15
                                    base = (lane >> 4) * 4
                                    for r in range(4):
                                        acc[wave, mi, ni, lane, r] += \
                                            out16[base + r, col]
                # epilogue: the kernel's exact index map
                slab = slabs[bz]
                for wave in range(4):
                    wave_m, wave_n = wave >> 1, wave & 1
                    for mi in range(2):
                        for ni in range(2):
                            for lane in range(64):
                                ncol = n0 + wave_n * 32 + ni * 16 \
                                    + (lane & 15)
                                if ncol >= Cout:
                                    continue
                                for r in range(4):
                                    m = m0 + wave_m * 32 + mi * 16 \
                                        + (lane >> 4) * 4 + r
                                    if m >= M:
                                        continue
                                    slab[m, ncol] = \
                                        acc[wave, mi, ni, lane, r]
    out = slabs.sum(axis=0) + bias[None, :]
    if relu:
        out = np.maximum(out, 0.0)
    return out.reshape(B, H, W, Cout)


def test_splitk_lane_level_sim_matches_conv2d():
    """Full lane-level transcription (staging maps, fragment blocks,
    acc lane map, epilogue index map) vs torch conv2d — the same
    methodology that HW-validated conv3x3_wrw first try."""
    rng = np.random.default_rng(5)
    B, H, W, Cin, Cout = 1, 4, 5, 32, 48  # ragged N (48 < 64 tile)
    x = rng.standard_normal((B, H, W, Cin), dtype=np.float32)
    wk = rng.standard_normal((3, 3, Cin, Cout), dtype=np.float32) * 0.1
    bias = rng.standard_normal(Cout).astype(np.float32)
    w_packed = wk.transpose(0, 1, 3, 2).reshape(9, Cout, Cin)
    got = _sim_splitk_lanes(x, w_packed, bias, split=4)
    xt = torch.from_numpy(x).permute(0, 3, 1, 2)
    wt = torch.from_numpy(wk).permute(3, 2, 0, 1)
    ref = F.conv2d(xt, wt, torch.from_numpy(bias), padding=1)
    ref = torch.relu(ref).permute(0, 2, 3, 1).numpy()
    np.testing.assert_allclose(got, ref, rtol=1e-4, atol=1e-4)


def test_splitk_lane_level_multi_tile():
    """M > 64 (two M-tiles) + 64-channel K (two k-chunks per tap)."""
    rng = np.random.default_rng(8)
    B, H, W, Cin, Cout = 2, 6, 6, 64, 32   # M = 72 -> grid_x = 2
    x = rng.standard_normal((B, H, W, Cin), dtype=np.float32)
    wk = rng.standard_normal((3, 3, Cin, Cout), dtype=np.float32) * 0.1
    bias = np.zeros(Cout, np.float32)
    w_packed = wk.transpose(0, 1, 3, 2).reshape(9, Cout, Cin)
    got = _sim_splitk_lanes(x, w_packed, bias, split=8, relu=False)
    xt = torch.from_numpy(x).permute(0, 3, 1, 2)
    wt = torch.from_numpy(wk).permute(3, 2, 0, 1)
    ref = F.conv2d(xt, wt, torch.from_numpy(bias), padding=1)
    ref = ref.permute(0, 2, 3, 1).numpy()
    np.testing.assert_allclose(got, ref, rtol=1e-4, atol=1e-4)

"""Shape-based dispatch between the hand-written conv3x3 MFMA kernels and
MIOpen, from A/B measurements on MI355X (profiles/r01_conv_ab.md round 1;
gpurun_out/r2_conv_ab2.log round 2 — v4 = 256² tile 8-wave ring, v3 =
128² counted-vmcnt, v2 = 128² glds, v1 = 64² single-buffered):

    shape (B,Cin,H,W,Cout)       best-ours         MIOpen   ratio
    (2, 64, 256,256, 64)        v1 0.036ms         0.046ms  1.26x  <- ours
    (2, 128,128,128,128)        v1/v2 0.037        0.037    1.00x  <- ours (+fusion)
    (2, 256, 64, 64,256)        v1 0.039           0.039    1.00x  <- ours (+fusion)
    (400,256, 14, 14,256)       v3 0.154           0.172    1.12x  <- ours
    (2, 256,256,256,256)        v4 0.221           0.226    1.02x  <- ours
    (2, 256,256,256,512)        v4 0.438           0.426    0.97x  <- MIOpen
    (2, 512, 32, 32,512)        v1 0.075           0.055    0.73x  <- MIOpen

Policy: HIP kernel when Cin <= 128, or Cin == 256 with small maps
(H*W <= 4096: the 64² bottleneck stage + head convs), or Cin == 256 on
big maps with Cout <= 256 (v4 wins the fpn_p2-style shape; the
Cout=512 RPN shape stays MIOpen at 0.97x). The version split lives in
the binding: M >= 131072 -> v4, M >= 32768 -> v3, else v1.
"""

from __future__ import annotations


def use_hip_conv3x3(batch: int, cin: int, h: int, w: int,
                    cout: int | None = None) -> bool:
    if cin <= 128:
        return True
    if cin == 256 and h * w <= 4096:
        return True
    if (cin == 256 and batch * h * w >= 131072 and
            cout is not None and cout <= 256):
        return True
    return False

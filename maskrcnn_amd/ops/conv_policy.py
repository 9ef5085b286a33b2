"""Shape-based dispatch between the hand-written conv3x3 MFMA kernels and
MIOpen, from A/B measurements on MI355X (profiles/r01_conv_ab.md; v2 =
128x128-tile glds kernel, v1 = 64x64 single-buffered):

    shape (B,Cin,H,W,Cout)       best-ours   MIOpen   ratio
    (2, 64, 256,256, 64)        v1 0.036ms   0.046ms  1.26x  <- ours
    (2, 128,128,128,128)        v1 0.037     0.037    1.00x  <- ours (+BN/ReLU fusion)
    (2, 256, 64, 64,256)        v1 0.037     0.038    1.03x  <- ours (+fusion)
    (400,256, 14, 14,256)       v2 0.155     0.184    1.19x  <- ours
    (2, 512, 32, 32,512)        v1 0.074     0.053    0.72x  <- MIOpen
    (2, 256,256,256,256)        v2 0.248     0.222    0.89x  <- MIOpen
    (2, 256,256,256,512)        v2 0.480     0.418    0.87x  <- MIOpen

Policy: HIP kernel when Cin <= 128, or Cin == 256 with small maps
(H*W <= 4096, which covers the 64x64 bottleneck stage and the 14x14
head convs at any roi batch). The version split (v1 small-M / v2
large-M) lives in the binding (M >= 32768 -> v2).
"""

from __future__ import annotations


def use_hip_conv3x3(batch: int, cin: int, h: int, w: int) -> bool:
    if cin <= 128:
        return True
    if cin == 256 and h * w <= 4096:
        return True
    return False

"""Shape-based dispatch between the hand-written conv3x3 MFMA kernel and
MIOpen, from A/B measurements on MI355X (profiles/r01_conv_ab.md):

    shape (B,Cin,H,W,Cout)        ours    MIOpen   ratio
    (2, 64, 256,256, 64)        0.036ms   0.047ms  1.30x  <- ours
    (2, 128,128,128,128)        0.037     0.041    1.11x  <- ours
    (2, 256, 64, 64,256)        0.038     0.039    1.03x  <- ours (+fused
                                                            BN/ReLU saves 2
                                                            elementwise passes)
    (2, 512, 32, 32,512)        0.074     0.054    0.74x  <- MIOpen
    (2, 256,256,256,256)        0.450     0.223    0.49x  <- MIOpen
    (400,256, 14, 14,256)       0.266     0.181    0.68x  <- MIOpen

Policy: use the HIP kernel when Cin <= 128, or Cin == 256 with
M = B*H*W <= 16384. Revisit when the 128x128-tile glds version lands.
"""

from __future__ import annotations


def use_hip_conv3x3(batch: int, cin: int, h: int, w: int) -> bool:
    m = batch * h * w
    if cin <= 128:
        return True
    if cin == 256 and m <= 16384:
        return True
    return False

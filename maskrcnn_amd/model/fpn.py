"""Feature Pyramid Network neck.

Replicates the FPN section of reference MaskRCNN.build (mrcnn/model.py:
~1900-1950): lateral 1x1 convs (256ch) + nearest x2 upsample + add,
3x3 output smoothing convs, P6 = stride-2 maxpool of P5. RPN consumes
[P2..P6]; the heads consume [P2..P5].

MI355X-native: on GPU the lateral 1x1 conv + upsample + add runs as ONE
hand-written CDNA4 MFMA kernel per level (csrc/hip/fpn_fuse.hip, N7),
replacing three kernels and a full P-map HBM round trip; the 3x3
smoothing convs stay on MIOpen. Numerics are identical either way
(ops/fpn_fuse.py falls back to torch off-GPU).
"""

from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..ops.fpn_fuse import fpn_fuse


class FPN(nn.Module):
    def __init__(self, in_channels=(256, 512, 1024, 2048), out_channels=256,
                 use_fused=True):
        super().__init__()
        c2, c3, c4, c5 = in_channels
        self.fpn_c5p5 = nn.Conv2d(c5, out_channels, 1)
        self.fpn_c4p4 = nn.Conv2d(c4, out_channels, 1)
        self.fpn_c3p3 = nn.Conv2d(c3, out_channels, 1)
        self.fpn_c2p2 = nn.Conv2d(c2, out_channels, 1)
        self.fpn_p2 = nn.Conv2d(out_channels, out_channels, 3, padding=1)
        self.fpn_p3 = nn.Conv2d(out_channels, out_channels, 3, padding=1)
        self.fpn_p4 = nn.Conv2d(out_channels, out_channels, 3, padding=1)
        self.fpn_p5 = nn.Conv2d(out_channels, out_channels, 3, padding=1)
        self.out_channels = out_channels
        self.use_fused = use_fused

    def forward(self, c2, c3, c4, c5):
        # Fused path (measured 1.4-1.9x per level, profiles/
        # r01_conv_ab.md). Default ON in training too since the round-2
        # block2 A/B: 45.84 vs 46.86 ms/step baseline (same backward
        # either way). MASKRCNN_TRAIN_FUSED_FPN=0 opts out.
        import os
        if self.use_fused and (not self.training or
                               os.environ.get("MASKRCNN_TRAIN_FUSED_FPN",
                                              "1") != "0"):
            p5 = fpn_fuse(c5, self.fpn_c5p5.weight, self.fpn_c5p5.bias)
            p4 = fpn_fuse(c4, self.fpn_c4p4.weight, self.fpn_c4p4.bias, p5)
            p3 = fpn_fuse(c3, self.fpn_c3p3.weight, self.fpn_c3p3.bias, p4)
            p2 = fpn_fuse(c2, self.fpn_c2p2.weight, self.fpn_c2p2.bias, p3)
        else:
            p5 = self.fpn_c5p5(c5)
            p4 = self.fpn_c4p4(c4) + F.interpolate(p5, scale_factor=2,
                                                   mode="nearest")
            p3 = self.fpn_c3p3(c3) + F.interpolate(p4, scale_factor=2,
                                                   mode="nearest")
            p2 = self.fpn_c2p2(c2) + F.interpolate(p3, scale_factor=2,
                                                   mode="nearest")
        p2 = self._smooth(self.fpn_p2, p2)
        p3 = self._smooth(self.fpn_p3, p3)
        p4 = self._smooth(self.fpn_p4, p4)
        p5 = self._smooth(self.fpn_p5, p5)
        p6 = F.max_pool2d(p5, kernel_size=1, stride=2)
        return [p2, p3, p4, p5, p6]

    def _smooth(self, conv, x):
        """3x3 output conv; the P2 shape (256ch @ quarter-res) is the
        biggest single conv in the forward and the v4 256²-tile kernel
        wins it vs MIOpen (r2_conv_ab2: 0.221 vs 0.226 ms) — route
        through the HIP kernel where the measured policy says so."""
        if (x.is_cuda and x.dtype == torch.bfloat16 and
                x.is_contiguous(memory_format=torch.channels_last)):
            from ..ops.conv_policy import use_hip_conv3x3
            if use_hip_conv3x3(x.shape[0], x.shape[1], x.shape[2],
                               x.shape[3], conv.out_channels):
                from ..ops.conv3x3 import conv3x3_bias_relu
                return conv3x3_bias_relu(x, conv.weight, conv.bias,
                                         relu=False)
        return conv(x)

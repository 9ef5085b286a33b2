"""Region Proposal Network head.

Replicates reference rpn_graph / build_rpn_model (mrcnn/model.py:~830-905):
shared 3x3 conv (512ch) -> 1x1 conv 2A logits + softmax, 1x1 conv 4A deltas.
One shared module applied to every pyramid level; per-level outputs are
flattened anchor-major and concatenated across levels (level order = anchor
generation order, so anchors[i] corresponds to row i).
"""

from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..ops.conv_policy import use_hip_conv3x3


class RPN(nn.Module):
    def __init__(self, in_channels=256, anchors_per_location=3,
                 anchor_stride=1, conv_channels=512):
        super().__init__()
        self.rpn_conv_shared = nn.Conv2d(in_channels, conv_channels, 3,
                                         stride=anchor_stride, padding=1)
        self.rpn_class_raw = nn.Conv2d(conv_channels,
                                       2 * anchors_per_location, 1)
        self.rpn_bbox_pred = nn.Conv2d(conv_channels,
                                       4 * anchors_per_location, 1)
        self.anchors_per_location = anchors_per_location

    def forward_level(self, x):
        """One pyramid level -> (logits [B,A_l,2], probs [B,A_l,2],
        deltas [B,A_l,4])."""
        conv = self.rpn_conv_shared
        if (x.is_cuda and not self.training and conv.stride[0] == 1 and
                use_hip_conv3x3(x.shape[0], x.shape[1], x.shape[2],
                                x.shape[3], conv.out_channels)):
            from ..ops.conv3x3 import conv3x3_bias_relu
            shared = conv3x3_bias_relu(x, conv.weight, conv.bias,
                                       relu=True)
        else:
            shared = F.relu(conv(x))
        # [B, 2A, H, W] -> [B, H, W, 2A] -> [B, H*W*A, 2]
        cls = self.rpn_class_raw(shared)
        B, _, H, W = cls.shape
        cls = cls.permute(0, 2, 3, 1).reshape(B, -1, 2)
        probs = F.softmax(cls, dim=-1)
        bbox = self.rpn_bbox_pred(shared)
        bbox = bbox.permute(0, 2, 3, 1).reshape(B, -1, 4)
        return cls, probs, bbox

    def forward(self, feature_maps):
        """Apply to [P2..P6]; concatenate anchor-major across levels."""
        logits, probs, bbox = [], [], []
        for f in feature_maps:
            lg, pr, bb = self.forward_level(f)
            logits.append(lg)
            probs.append(pr)
            bbox.append(bb)
        return (torch.cat(logits, dim=1), torch.cat(probs, dim=1),
                torch.cat(bbox, dim=1))

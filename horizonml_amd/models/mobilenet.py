"""MobileNetV2 on the fused MI355X blocks.

Capability parity with the reference's ``train.py:60-68`` (``MODEL_TYPE``
selects mobilenet_v2 or resnet18 for the Docker "edge node" simulation) —
rebuilt on the framework's fused modules so the GPU path runs the
hand-written gfx950 kernels end to end: ``ConvBNAct(act="relu6")`` for the
standard/pointwise convs, ``DepthwiseConvBNAct`` (c-blocked stencil
kernels) for the depthwise 3×3s, and the inverted-residual add fused into
the projection conv's BN epilogue.  CPU falls back to the modules' torch
reference paths.
"""
from __future__ import annotations

import torch.nn as nn

from .layers import ConvBNAct, DepthwiseConvBNAct, GlobalAvgPool, Linear


class InvertedResidual(nn.Module):
    def __init__(self, in_ch, out_ch, stride, expand):
        super().__init__()
        hidden = in_ch * expand
        self.use_res = stride == 1 and in_ch == out_ch
        self.expand = (ConvBNAct(in_ch, hidden, 1, act="relu6")
                       if expand != 1 else None)
        self.dw = DepthwiseConvBNAct(hidden, 3, stride=stride, act="relu6")
        # projection: linear bottleneck (no activation); the residual add
        # rides its BN epilogue
        self.project = ConvBNAct(hidden, out_ch, 1, act="none")

    def forward(self, x):
        h = self.expand(x) if self.expand is not None else x
        h = self.dw(h)
        return self.project(h, residual=x if self.use_res else None)


class MobileNetV2(nn.Module):
    # (expand, out_ch, n, stride) — the standard V2 schedule
    cfg = [(1, 16, 1, 1), (6, 24, 2, 2), (6, 32, 3, 2), (6, 64, 4, 2),
           (6, 96, 3, 1), (6, 160, 3, 2), (6, 320, 1, 1)]

    def __init__(self, num_classes: int = 10):
        super().__init__()
        features = [ConvBNAct(3, 32, 3, stride=2, act="relu6")]
        in_ch = 32
        for expand, out_ch, n, stride in self.cfg:
            for i in range(n):
                features.append(InvertedResidual(
                    in_ch, out_ch, stride if i == 0 else 1, expand))
                in_ch = out_ch
        features.append(ConvBNAct(in_ch, 1280, 1, act="relu6"))
        self.features = nn.Sequential(*features)
        self.pool = GlobalAvgPool()
        self.dropout = nn.Dropout(0.2)
        self.fc = Linear(1280, num_classes)

    def forward(self, x):
        x = self.features(x)
        x = self.pool(x)
        x = self.dropout(x)
        return self.fc(x)


def mobilenet_v2(num_classes: int = 10) -> MobileNetV2:
    return MobileNetV2(num_classes)

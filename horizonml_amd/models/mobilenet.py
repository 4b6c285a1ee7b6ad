"""MobileNetV2 — capability parity for the legacy Docker/env-var entry.

The reference's ``train.py:60-68`` selects mobilenet_v2 or resnet18 via the
``MODEL_TYPE`` env var for the CPU "edge node" simulation.  That path is
CPU-only in the reference, so this implementation uses plain torch modules
(inverted residuals with depthwise convs); the MI355X kernel surface targets
the ResNet hot path.
"""
from __future__ import annotations

import torch
import torch.nn as nn


def _cbr(in_ch, out_ch, k=3, s=1, groups=1):
    return nn.Sequential(
        nn.Conv2d(in_ch, out_ch, k, s, k // 2, groups=groups, bias=False),
        nn.BatchNorm2d(out_ch),
        nn.ReLU6(inplace=True),
    )


class InvertedResidual(nn.Module):
    def __init__(self, in_ch, out_ch, stride, expand):
        super().__init__()
        hidden = in_ch * expand
        self.use_res = stride == 1 and in_ch == out_ch
        layers = []
        if expand != 1:
            layers.append(_cbr(in_ch, hidden, k=1))
        layers += [
            _cbr(hidden, hidden, k=3, s=stride, groups=hidden),  # depthwise
            nn.Conv2d(hidden, out_ch, 1, 1, 0, bias=False),
            nn.BatchNorm2d(out_ch),
        ]
        self.conv = nn.Sequential(*layers)

    def forward(self, x):
        out = self.conv(x)
        return x + out if self.use_res else out


class MobileNetV2(nn.Module):
    # (expand, out_ch, n, stride) — the standard V2 schedule
    cfg = [(1, 16, 1, 1), (6, 24, 2, 2), (6, 32, 3, 2), (6, 64, 4, 2),
           (6, 96, 3, 1), (6, 160, 3, 2), (6, 320, 1, 1)]

    def __init__(self, num_classes: int = 10):
        super().__init__()
        features = [_cbr(3, 32, k=3, s=2)]
        in_ch = 32
        for expand, out_ch, n, stride in self.cfg:
            for i in range(n):
                features.append(InvertedResidual(
                    in_ch, out_ch, stride if i == 0 else 1, expand))
                in_ch = out_ch
        features.append(_cbr(in_ch, 1280, k=1))
        self.features = nn.Sequential(*features)
        self.classifier = nn.Sequential(
            nn.Dropout(0.2), nn.Linear(1280, num_classes))

    def forward(self, x):
        x = self.features(x)
        x = torch.flatten(nn.functional.adaptive_avg_pool2d(x, 1), 1)
        return self.classifier(x)


def mobilenet_v2(num_classes: int = 10) -> MobileNetV2:
    return MobileNetV2(num_classes)

"""ResNet family, built on the fused MI355X blocks in ``layers.py``.

Capability parity with the torchvision models the reference instantiates
(``resnet18`` at reference ``data_parallel_train.py:198``,
``layer_model_parallel_train.py:30``, ``tensor_parallel_train.py:74``;
``resnet50`` for the hybrid DP×PP config in BASELINE.json), but implemented
from scratch against our fused ConvBNAct blocks so the GPU path runs the
hand-written gfx950 kernels end to end.

Stems: ``imagenet`` = 7x7/2 conv + 3x3/2 maxpool (what the reference uses
even on CIFAR 32x32 inputs); ``cifar`` = 3x3/1 conv, no pool (offered as a
deliberate improvement, not the parity default).
"""
from __future__ import annotations

from typing import List, Tuple

import torch
import torch.nn as nn

from .layers import ConvBNAct, GlobalAvgPool, Linear, MaxPool2d3x3s2


class BasicBlock(nn.Module):
    expansion = 1

    def __init__(self, in_ch: int, ch: int, stride: int = 1):
        super().__init__()
        self.conv1 = ConvBNAct(in_ch, ch, 3, stride=stride, act=True)
        # conv2 fuses BN + residual add + final ReLU into one epilogue:
        # out = relu(bn(conv(x)) + identity)
        self.conv2 = ConvBNAct(ch, ch, 3, stride=1, act=True)
        self.downsample = (ConvBNAct(in_ch, ch, 1, stride=stride, act=False)
                           if (stride != 1 or in_ch != ch) else None)

    def _fusable(self) -> bool:
        return (type(self.conv1) is ConvBNAct and type(self.conv2) is ConvBNAct
                and (self.downsample is None
                     or type(self.downsample) is ConvBNAct))

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        if x.is_cuda and self._fusable():
            from ._functional_gpu import res_block
            return res_block(x, (self.conv1, self.conv2), self.downsample)
        identity = self.downsample(x) if self.downsample is not None else x
        out = self.conv1(x)
        return self.conv2(out, residual=identity)


class Bottleneck(nn.Module):
    expansion = 4

    def __init__(self, in_ch: int, ch: int, stride: int = 1):
        super().__init__()
        out_ch = ch * self.expansion
        self.conv1 = ConvBNAct(in_ch, ch, 1, stride=1, act=True)
        self.conv2 = ConvBNAct(ch, ch, 3, stride=stride, act=True)
        self.conv3 = ConvBNAct(ch, out_ch, 1, stride=1, act=True)
        self.downsample = (ConvBNAct(in_ch, out_ch, 1, stride=stride, act=False)
                           if (stride != 1 or in_ch != out_ch) else None)

    def _fusable(self) -> bool:
        return (type(self.conv1) is ConvBNAct and type(self.conv2) is ConvBNAct
                and type(self.conv3) is ConvBNAct
                and (self.downsample is None
                     or type(self.downsample) is ConvBNAct))

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        if x.is_cuda and self._fusable():
            from ._functional_gpu import res_block
            return res_block(x, (self.conv1, self.conv2, self.conv3),
                             self.downsample)
        identity = self.downsample(x) if self.downsample is not None else x
        out = self.conv1(x)
        out = self.conv2(out)
        return self.conv3(out, residual=identity)


class Stem(nn.Module):
    def __init__(self, kind: str = "imagenet", out_ch: int = 64):
        super().__init__()
        self.kind = kind
        if kind == "imagenet":
            self.conv = ConvBNAct(3, out_ch, 7, stride=2, padding=3, act=True)
            self.pool = MaxPool2d3x3s2()
        elif kind == "cifar":
            self.conv = ConvBNAct(3, out_ch, 3, stride=1, padding=1, act=True)
            self.pool = nn.Identity()
        else:
            raise ValueError(f"unknown stem kind {kind!r}")

    def forward(self, x):
        return self.pool(self.conv(x))


class Tail(nn.Module):
    """avgpool + flatten + fc — the reference's last layer group
    (``layer_model_parallel_train.py:47-52``)."""

    def __init__(self, in_features: int, num_classes: int):
        super().__init__()
        self.pool = GlobalAvgPool()
        self.fc = Linear(in_features, num_classes)

    def forward(self, x):
        return self.fc(self.pool(x))


class ResNet(nn.Module):
    def __init__(self, block, layers: List[int], num_classes: int = 10,
                 stem: str = "imagenet", width: int = 64):
        super().__init__()
        self.block_type = block
        self.stem = Stem(stem, width)
        self.in_ch = width
        self.layer1 = self._make_layer(block, width, layers[0], 1)
        self.layer2 = self._make_layer(block, width * 2, layers[1], 2)
        self.layer3 = self._make_layer(block, width * 4, layers[2], 2)
        self.layer4 = self._make_layer(block, width * 8, layers[3], 2)
        self.tail = Tail(width * 8 * block.expansion, num_classes)

    def _make_layer(self, block, ch: int, n: int, stride: int) -> nn.Sequential:
        blocks = [block(self.in_ch, ch, stride)]
        self.in_ch = ch * block.expansion
        for _ in range(n - 1):
            blocks.append(block(self.in_ch, ch, 1))
        return nn.Sequential(*blocks)

    def forward(self, x):
        x = self.stem(x)
        x = self.layer1(x)
        x = self.layer2(x)
        x = self.layer3(x)
        x = self.layer4(x)
        return self.tail(x)

    # ---- partitioning support (SURVEY.md §2.2 SplitResNet parity) -------
    def layer_units(self, granularity: str = "group") -> List[Tuple[str, nn.Module]]:
        """Ordered pipeline units.

        ``group``  → 5 units: stem, layer1..layer3, (layer4+tail) — the
                     reference's layer-group decomposition
                     (``layer_model_parallel_train.py:37-52``).
        ``block``  → stem, every residual block, tail — supports up to
                     2 + Σlayers stages (10 for resnet18 → 8-stage pipelines
                     per BASELINE.json config #3).
        """
        if granularity == "group":
            tail = nn.Sequential(self.layer4, self.tail)
            return [("stem", self.stem), ("layer1", self.layer1),
                    ("layer2", self.layer2), ("layer3", self.layer3),
                    ("tail", tail)]
        elif granularity == "block":
            units: List[Tuple[str, nn.Module]] = [("stem", self.stem)]
            for li, layer in enumerate(
                    (self.layer1, self.layer2, self.layer3, self.layer4), 1):
                for bi, blk in enumerate(layer):
                    units.append((f"layer{li}.{bi}", blk))
            units.append(("tail", self.tail))
            return units
        raise ValueError(f"unknown granularity {granularity!r}")


def resnet18(num_classes: int = 10, stem: str = "imagenet") -> ResNet:
    return ResNet(BasicBlock, [2, 2, 2, 2], num_classes, stem)


def resnet34(num_classes: int = 10, stem: str = "imagenet") -> ResNet:
    return ResNet(BasicBlock, [3, 4, 6, 3], num_classes, stem)


def resnet50(num_classes: int = 1000, stem: str = "imagenet") -> ResNet:
    return ResNet(Bottleneck, [3, 4, 6, 3], num_classes, stem)


def build_model(name: str, num_classes: int = 10, stem: str = "imagenet") -> nn.Module:
    name = name.lower()
    if name in ("resnet18", "resnet"):
        return resnet18(num_classes, stem)
    if name == "resnet34":
        return resnet34(num_classes, stem)
    if name == "resnet50":
        return resnet50(num_classes, stem)
    if name in ("mobilenet", "mobilenet_v2"):
        from .mobilenet import mobilenet_v2
        return mobilenet_v2(num_classes)
    raise ValueError(f"unknown model {name!r}")

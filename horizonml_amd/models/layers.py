"""Fused building-block modules for the MI355X hot path.

Design notes (MI355X-first):

* Conv weights are stored **KRSC** (out-ch, kh, kw, in-ch) f32 — the layout
  the gfx950 implicit-GEMM MFMA kernel consumes (channels innermost →
  coalesced K-dim loads).  The CPU reference path permutes to torch's OIHW
  on the fly.
* Activations on GPU travel as **bf16, channels_last** (NHWC memory);
  each fused op reads/writes NHWC directly.
* One module = one fused kernel chain on GPU:
  ``ConvBNAct`` = implicit-GEMM conv (+stats epilogue) → BN-apply+residual
  +ReLU; training-mode batch stats are accumulated by the conv epilogue via
  per-channel atomics, so the whole block is 2 kernel launches instead of
  torch's ~5.

Reference parity: replaces the torchvision resnet18 blocks used at reference
``data_parallel_train.py:198-199``, ``layer_model_parallel_train.py:30``,
``tensor_parallel_train.py:74-88`` (SURVEY.md §2.4 K1–K7).
"""
from __future__ import annotations

import math
from typing import Optional

import torch
import torch.nn as nn
import torch.nn.functional as F



def _use_hip(x: torch.Tensor) -> bool:
    return x.is_cuda


class ConvBNAct(nn.Module):
    """Fused Conv2d + BatchNorm2d [+ residual add] [+ ReLU].

    ``forward(x, residual=None)`` — the residual (if given) is added after BN
    and before the activation, matching the ResNet basic-block epilogue
    (fused into the conv epilogue on GPU).
    """

    ACT = {False: 0, "none": 0, True: 1, "relu": 1, "relu6": 2}

    def __init__(self, in_ch: int, out_ch: int, kernel_size: int = 3,
                 stride: int = 1, padding: Optional[int] = None,
                 act=True, eps: float = 1e-5, momentum: float = 0.1):
        super().__init__()
        if padding is None:
            padding = kernel_size // 2
        self.in_ch, self.out_ch = in_ch, out_ch
        self.kernel_size, self.stride, self.padding = kernel_size, stride, padding
        # activation kind: 0 none, 1 ReLU, 2 ReLU6 (mobilenet)
        self.act = self.ACT[act] if not isinstance(act, int) else act
        self.eps, self.momentum = eps, momentum
        # KRSC layout (out, kh, kw, in)
        self.weight = nn.Parameter(
            torch.empty(out_ch, kernel_size, kernel_size, in_ch))
        self.bn_weight = nn.Parameter(torch.ones(out_ch))
        self.bn_bias = nn.Parameter(torch.zeros(out_ch))
        self.register_buffer("running_mean", torch.zeros(out_ch))
        self.register_buffer("running_var", torch.ones(out_ch))
        self.register_buffer("num_batches_tracked",
                             torch.tensor(0, dtype=torch.long))
        # bf16 KRSC shadow for the GPU kernels; refreshed by the engine
        # (or lazily) after optimizer steps.
        self.register_buffer("weight_bf16", torch.empty(0), persistent=False)
        self.reset_parameters()

    def reset_parameters(self):
        # Kaiming-normal fan_out on the conv weight (ResNet convention).
        fan_out = self.kernel_size * self.kernel_size * self.out_ch
        with torch.no_grad():
            self.weight.normal_(0, math.sqrt(2.0 / fan_out))

    def extra_repr(self):
        return (f"{self.in_ch}, {self.out_ch}, k={self.kernel_size}, "
                f"s={self.stride}, p={self.padding}, act={self.act}")

    # -- GPU shadow management -------------------------------------------
    def refresh_shadow(self):
        """(Re)materialize the bf16 KRSC weight shadow on the weight's device.
        When a FlatParamManager owns the shadow (``_managed``) the fused
        optimizer kernel keeps it current — nothing to do here."""
        if getattr(self, "_managed", False):
            return
        with torch.no_grad():
            if (self.weight_bf16.shape == self.weight.shape
                    and self.weight_bf16.device == self.weight.device):
                self.weight_bf16.copy_(self.weight.detach())
            else:
                self.weight_bf16 = self.weight.detach() \
                    .to(torch.bfloat16).contiguous()

    def _shadow(self) -> torch.Tensor:
        if getattr(self, "_managed", False):
            return self.weight_bf16
        if (self.weight_bf16.numel() != self.weight.numel()
                or self.weight_bf16.device != self.weight.device):
            self.refresh_shadow()
        return self.weight_bf16

    # -- forward ----------------------------------------------------------
    def forward(self, x: torch.Tensor,
                residual: Optional[torch.Tensor] = None) -> torch.Tensor:
        if _use_hip(x):
            from . import _functional_gpu as FG
            return FG.conv_bn_act(x, self, residual)
        # CPU reference path: torch ops, full autograd.
        w = self.weight.permute(0, 3, 1, 2)  # KRSC -> OIHW
        y = F.conv2d(x, w, None, stride=self.stride, padding=self.padding)
        y = F.batch_norm(y, self.running_mean, self.running_var,
                         self.bn_weight, self.bn_bias,
                         self.training, self.momentum, self.eps)
        if residual is not None:
            y = y + residual
        if self.act == 1:
            y = F.relu(y)
        elif self.act == 2:
            y = F.hardtanh(y, 0.0, 6.0)  # ReLU6
        return y


class DepthwiseConvBNAct(nn.Module):
    """Fused depthwise Conv2d (groups=C) + BatchNorm2d [+ act] — the
    MobileNet inverted-residual middle op, on the c-blocked gfx950 stencil
    kernels (no MFMA: depthwise is bandwidth-bound).  Weight layout
    [R, S, C] (channels innermost, matching NHWC activations)."""

    def __init__(self, ch: int, kernel_size: int = 3, stride: int = 1,
                 padding: Optional[int] = None, act="relu6",
                 eps: float = 1e-5, momentum: float = 0.1):
        super().__init__()
        if padding is None:
            padding = kernel_size // 2
        self.ch = ch
        self.kernel_size, self.stride, self.padding = (kernel_size, stride,
                                                       padding)
        self.act = ConvBNAct.ACT[act] if not isinstance(act, int) else act
        self.eps, self.momentum = eps, momentum
        self.weight = nn.Parameter(
            torch.empty(kernel_size, kernel_size, ch))
        self.bn_weight = nn.Parameter(torch.ones(ch))
        self.bn_bias = nn.Parameter(torch.zeros(ch))
        self.register_buffer("running_mean", torch.zeros(ch))
        self.register_buffer("running_var", torch.ones(ch))
        self.register_buffer("weight_bf16", torch.empty(0), persistent=False)
        with torch.no_grad():
            fan = kernel_size * kernel_size
            self.weight.normal_(0, math.sqrt(2.0 / fan))

    def _shadow(self) -> torch.Tensor:
        if getattr(self, "_managed", False):
            return self.weight_bf16
        if (self.weight_bf16.numel() != self.weight.numel()
                or self.weight_bf16.device != self.weight.device):
            with torch.no_grad():
                self.weight_bf16 = self.weight.detach() \
                    .to(torch.bfloat16).contiguous()
        return self.weight_bf16

    def refresh_shadow(self):
        if not getattr(self, "_managed", False):
            with torch.no_grad():
                self.weight_bf16 = self.weight.detach() \
                    .to(torch.bfloat16).contiguous()

    def extra_repr(self):
        return (f"{self.ch}, k={self.kernel_size}, s={self.stride}, "
                f"act={self.act}")

    def forward(self, x: torch.Tensor,
                residual: Optional[torch.Tensor] = None) -> torch.Tensor:
        if _use_hip(x):
            from . import _functional_gpu as FG
            return FG.dw_conv_bn_act(x, self, residual)
        # CPU reference path: [R,S,C] -> (C,1,R,S) grouped conv
        w = self.weight.permute(2, 0, 1).unsqueeze(1)
        y = F.conv2d(x, w, None, stride=self.stride, padding=self.padding,
                     groups=self.ch)
        y = F.batch_norm(y, self.running_mean, self.running_var,
                         self.bn_weight, self.bn_bias, self.training,
                         self.momentum, self.eps)
        if residual is not None:
            y = y + residual
        if self.act == 1:
            y = F.relu(y)
        elif self.act == 2:
            y = F.hardtanh(y, 0.0, 6.0)
        return y


class MaxPool2d3x3s2(nn.Module):
    """3x3/2 max-pool with padding 1 (the ResNet stem pool)."""

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        if _use_hip(x):
            from . import _functional_gpu as FG
            return FG.maxpool2d(x)
        return F.max_pool2d(x, kernel_size=3, stride=2, padding=1)


class GlobalAvgPool(nn.Module):
    """Adaptive average pool to 1x1 + flatten."""

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        if x.shape[2] == 1 and x.shape[3] == 1:
            # CIFAR-shaped nets reach the pool at 1x1 spatial: the "pool"
            # is a zero-copy reshape (saves a kernel in each direction)
            return x.reshape(x.shape[0], x.shape[1])
        if _use_hip(x):
            from . import _functional_gpu as FG
            return FG.global_avgpool(x)
        return torch.flatten(F.adaptive_avg_pool2d(x, 1), 1)


class Linear(nn.Module):
    """Final classifier linear. Weight [out, in] f32 +

    bf16 shadow for the MFMA small-GEMM kernel."""

    def __init__(self, in_features: int, out_features: int, bias: bool = True):
        super().__init__()
        self.in_features, self.out_features = in_features, out_features
        self.weight = nn.Parameter(torch.empty(out_features, in_features))
        self.bias = nn.Parameter(torch.zeros(out_features)) if bias else None
        self.register_buffer("weight_bf16", torch.empty(0), persistent=False)
        self.reset_parameters()

    def reset_parameters(self):
        bound = 1.0 / math.sqrt(self.in_features)
        with torch.no_grad():
            self.weight.uniform_(-bound, bound)
            if self.bias is not None:
                self.bias.uniform_(-bound, bound)

    def refresh_shadow(self):
        if getattr(self, "_managed", False):
            return
        with torch.no_grad():
            self.weight_bf16 = self.weight.detach().to(torch.bfloat16).contiguous()

    def _shadow(self) -> torch.Tensor:
        if getattr(self, "_managed", False):
            return self.weight_bf16
        if (self.weight_bf16.numel() != self.weight.numel()
                or self.weight_bf16.device != self.weight.device):
            self.refresh_shadow()
        return self.weight_bf16

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        if _use_hip(x):
            from . import _functional_gpu as FG
            return FG.linear(x, self)
        return F.linear(x, self.weight, self.bias)


def refresh_all_shadows(model: nn.Module):
    """Refresh every bf16 weight shadow in `model`.

    Called after EVERY optimizer step by the eager engines; in-place
    shadows + one ``_foreach_copy_`` keep the per-step cost to a couple of
    multi-tensor kernels."""
    dsts, srcs = [], []
    for m in model.modules():
        if isinstance(m, (ConvBNAct, Linear, DepthwiseConvBNAct)) \
                and not getattr(m, "_managed", False):
            if (m.weight_bf16.shape != m.weight.shape
                    or m.weight_bf16.device != m.weight.device):
                m.refresh_shadow()  # (re)allocate
            else:
                dsts.append(m.weight_bf16)
                srcs.append(m.weight.detach())
    if not dsts:
        return
    with torch.no_grad():
        try:
            torch._foreach_copy_(dsts, srcs)
        except (RuntimeError, AttributeError):
            for d, sr in zip(dsts, srcs):
                d.copy_(sr)

"""GPU dispatch: autograd Functions over the gfx950 HIP extension.

These are the only code paths taken for CUDA (= ROCm/HIP) tensors — there is
no eager-torch fallback on GPU; a missing extension raises at the first op
(see ``horizonml_amd.ops.extension``).

Tensor conventions (see ``layers.py``): activations bf16 channels_last,
conv weights bf16 KRSC shadows of the f32 master parameters, BN params and
classifier logits f32.
"""
from __future__ import annotations

from typing import Optional

import torch

from .. import ops as _ops


def _C():
    return _ops.extension()


def _cl(t: torch.Tensor) -> torch.Tensor:
    if t.dim() == 4 and not t.is_contiguous(memory_format=torch.channels_last):
        return t.contiguous(memory_format=torch.channels_last)
    if t.dim() != 4 and not t.is_contiguous():
        return t.contiguous()
    return t


class ConvBNActFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, bn_weight, bn_bias, residual, mod):
        x = _cl(x)
        w_bf16 = mod._shadow()
        if residual is not None:
            residual = _cl(residual)
        stats_buf = (getattr(mod, "_stats_buf", None)
                     if mod.training else None)
        y, convout, smean, sinvstd = _C().conv_bn_act_fwd(
            x, w_bf16, bn_weight, bn_bias, mod.running_mean, mod.running_var,
            mod.stride, mod.padding, mod.momentum, mod.eps, mod.training,
            mod.act, residual, stats_buf)
        ctx.save_for_backward(x, y, convout, w_bf16, bn_weight, smean,
                              sinvstd)
        ctx.mod = mod
        ctx.has_res = residual is not None
        return y

    @staticmethod
    def backward(ctx, dy):
        x, y, convout, w_bf16, gamma, smean, sinvstd = ctx.saved_tensors
        mod = ctx.mod
        need_dx = ctx.needs_input_grad[0]
        # RSCK weight image for the dgrad implicit GEMM (B-fragment wants
        # contiguous out-channels at fixed (r,s,c)); the engine pre-builds
        # this per step via the batched permute kernel, else permute here.
        w_rsck = getattr(mod, "_w_rsck", None)
        if need_dx and (w_rsck is None or w_rsck.numel() != w_bf16.numel()):
            w_rsck = w_bf16.permute(1, 2, 3, 0).contiguous()
        dy = _cl(dy.to(torch.bfloat16) if dy.dtype != torch.bfloat16 else dy)
        # Direct-grad mode (FlatParamManager): kernels accumulate straight
        # into the pre-zeroed flat .grad views and we return None so autograd
        # skips its per-parameter accumulate kernels (~60 launches/step).
        direct = (getattr(mod, "_managed", False)
                  and mod.weight.grad is not None)
        dw_out = mod.weight.grad if direct else None
        dg_out = mod.bn_weight.grad if direct else None
        db_out = mod.bn_bias.grad if direct else None
        dx, dw, dgamma, dbeta, dres = _C().conv_bn_act_bwd(
            dy, y, x, w_bf16, w_rsck if need_dx else w_bf16, convout, gamma,
            smean, sinvstd, mod.stride, mod.padding, mod.act, need_dx,
            ctx.has_res, dw_out, dg_out, db_out)
        if direct:
            return (dx if need_dx else None, None, None, None,
                    dres if ctx.has_res else None, None)
        return (dx if need_dx else None, dw, dgamma, dbeta,
                dres if ctx.has_res else None, None)


def conv_bn_act(x, mod, residual=None):
    return ConvBNActFn.apply(x, mod.weight, mod.bn_weight, mod.bn_bias,
                             residual, mod)


class MaxPoolFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x):
        x = _cl(x)
        y, idx = _C().maxpool_fwd(x)
        ctx.save_for_backward(idx)
        ctx.hw = (x.shape[2], x.shape[3])
        return y

    @staticmethod
    def backward(ctx, dy):
        (idx,) = ctx.saved_tensors
        dy = _cl(dy.to(torch.bfloat16) if dy.dtype != torch.bfloat16 else dy)
        return _C().maxpool_bwd(dy, idx, *ctx.hw)


def maxpool2d(x):
    return MaxPoolFn.apply(x)


class AvgPoolFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x):
        x = _cl(x)
        ctx.hw = (x.shape[2], x.shape[3])
        return _C().avgpool_fwd(x)

    @staticmethod
    def backward(ctx, dy):
        dy = dy.to(torch.bfloat16).contiguous()
        return _C().avgpool_bwd(dy, *ctx.hw)


def global_avgpool(x):
    return AvgPoolFn.apply(x)


class LinearFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, bias, mod):
        x = x.contiguous()
        w_bf16 = mod._shadow()
        y = _C().linear_fwd(x, w_bf16, bias)
        ctx.save_for_backward(x, w_bf16)
        ctx.need_db = bias is not None
        ctx.mod = mod
        return y

    @staticmethod
    def backward(ctx, dy):
        x, w_bf16 = ctx.saved_tensors
        mod = ctx.mod
        dy = dy.float().contiguous()
        need_dx = ctx.needs_input_grad[0]
        direct = (getattr(mod, "_managed", False)
                  and mod.weight.grad is not None)
        dw_out = mod.weight.grad if direct else None
        db_out = (mod.bias.grad if (direct and mod.bias is not None)
                  else None)
        dx, dw, db = _C().linear_bwd(dy, x, w_bf16, need_dx, ctx.need_db,
                                     dw_out, db_out)
        if direct:
            return (dx if need_dx else None, None, None, None)
        return (dx if need_dx else None, dw,
                db if ctx.need_db else None, None)


def linear(x, mod):
    return LinearFn.apply(x, mod.weight, mod.bias, mod)


class CrossEntropyFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, logits, target):
        logits = logits.float().contiguous()
        loss, dlogits = _C().cross_entropy_fwd_bwd(logits, target)
        ctx.save_for_backward(dlogits)
        return loss

    @staticmethod
    def backward(ctx, gout):
        (dlogits,) = ctx.saved_tensors
        return dlogits * gout, None


def cross_entropy(logits: torch.Tensor, target: torch.Tensor) -> torch.Tensor:
    """Fused log-softmax + NLL + grad on GPU; torch on CPU."""
    if logits.is_cuda:
        return CrossEntropyFn.apply(logits, target)
    return torch.nn.functional.cross_entropy(logits.float(), target)

"""GPU dispatch: autograd Functions over the gfx950 HIP extension.

These are the only code paths taken for CUDA (= ROCm/HIP) tensors — there is
no eager-torch fallback on GPU; a missing extension raises at the first op
(see ``horizonml_amd.ops.extension``).

Tensor conventions (see ``layers.py``): activations bf16 channels_last,
conv weights bf16 KRSC shadows of the f32 master parameters, BN params and
classifier logits f32.
"""
from __future__ import annotations

import torch

from .. import ops as _ops


def _C():
    return _ops.extension()


def _bwd_done(mod):
    """Backward-complete notification for the bucketed DP reducer
    (``parallel/flat_reducer.py``): by the time a unit's autograd backward
    returns, its BN/bias grads are launched and its deferred wgrad tasks
    registered, so the unit's flat-gradient range may be flushed + reduced.
    No-op unless a scheduler installed ``_bwd_done_cb`` on the module."""
    cb = getattr(mod, "_bwd_done_cb", None)
    if cb is not None:
        cb()


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
        ctx.save_for_backward(x, y, convout, w_bf16, bn_weight, bn_bias,
                              smean, sinvstd)
        ctx.mod = mod
        ctx.has_res = residual is not None
        return y

    @staticmethod
    def backward(ctx, dy):
        x, y, convout, w_bf16, gamma, beta, smean, sinvstd = \
            ctx.saved_tensors
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
            beta, smean, sinvstd, mod.stride, mod.padding, mod.act, need_dx,
            ctx.has_res, dw_out, dg_out, db_out, None, None, 0, False, None,
            None)
        _bwd_done(mod)
        if direct:
            return (dx if need_dx else None, None, None, None,
                    dres if ctx.has_res else None, None)
        return (dx if need_dx else None, dw, dgamma, dbeta,
                dres if ctx.has_res else None, None)


def conv_bn_act(x, mod, residual=None):
    return ConvBNActFn.apply(x, mod.weight, mod.bn_weight, mod.bn_bias,
                             residual, mod)


class DwConvBNActFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, bn_weight, bn_bias, residual, mod):
        x = _cl(x)
        w_bf16 = mod._shadow()
        if residual is not None:
            residual = _cl(residual)
        y, convout, smean, sinvstd = _C().dw_conv_bn_fwd(
            x, w_bf16, bn_weight, bn_bias, mod.running_mean,
            mod.running_var, mod.stride, mod.padding, mod.momentum, mod.eps,
            mod.training, mod.act, residual)
        ctx.save_for_backward(x, y, convout, w_bf16, bn_weight, bn_bias,
                              smean, sinvstd)
        ctx.mod = mod
        ctx.has_res = residual is not None
        return y

    @staticmethod
    def backward(ctx, dy):
        x, y, convout, w_bf16, gamma, beta, smean, sinvstd = \
            ctx.saved_tensors
        mod = ctx.mod
        need_dx = ctx.needs_input_grad[0]
        dy = _cl(dy.to(torch.bfloat16) if dy.dtype != torch.bfloat16 else dy)
        dx, dw, dgamma, dbeta, dres = _C().dw_conv_bn_bwd(
            dy, y, x, w_bf16, convout, gamma, beta, smean, sinvstd,
            mod.stride, mod.padding, mod.act, need_dx, ctx.has_res)
        _bwd_done(mod)
        return (dx if need_dx else None, dw, dgamma, dbeta,
                dres if ctx.has_res else None, None)


def dw_conv_bn_act(x, mod, residual=None):
    return DwConvBNActFn.apply(x, mod.weight, mod.bn_weight, mod.bn_bias,
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
        _bwd_done(mod)
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


def _rsck(mod, w_bf16):
    w = getattr(mod, "_w_rsck", None)
    if w is None or w.numel() != w_bf16.numel():
        w = w_bf16.permute(1, 2, 3, 0).contiguous()
    return w


class ResBlockFn(torch.autograd.Function):
    """Whole residual block (BasicBlock / Bottleneck) as one autograd node.

    Besides saving Python/autograd overhead, this fuses the residual
    gradient junction: the reference graph produces dx(main path) + d(res
    path) as two tensors summed by an autograd add kernel per block — here
    the last backward dgrad ACCUMULATES into the other path's buffer
    (``dx_accum`` in ``conv_bn_act_bwd``), eliminating one elementwise
    kernel + one tensor round-trip per block (8 per ResNet18 step).
    """

    @staticmethod
    def forward(ctx, x, mods, ds_mod, *params):
        x = _cl(x)
        training = mods[0].training
        if ds_mod is not None:
            y_ds, convout_ds, smean_ds, sinvstd_ds = _C().conv_bn_act_fwd(
                x, ds_mod._shadow(), ds_mod.bn_weight, ds_mod.bn_bias,
                ds_mod.running_mean, ds_mod.running_var, ds_mod.stride,
                ds_mod.padding, ds_mod.momentum, ds_mod.eps, training,
                ds_mod.act, None,
                getattr(ds_mod, "_stats_buf", None) if training else None)
            identity = y_ds
        else:
            identity = x
        h = x
        saved = [x]
        n = len(mods)
        for i, m in enumerate(mods):
            res = identity if i == n - 1 else None
            y, convout, smean, sinvstd = _C().conv_bn_act_fwd(
                h, m._shadow(), m.bn_weight, m.bn_bias, m.running_mean,
                m.running_var, m.stride, m.padding, m.momentum, m.eps,
                training, m.act, res,
                getattr(m, "_stats_buf", None) if training else None)
            saved += [y, convout, smean, sinvstd, m._shadow(),
                      m.bn_weight, m.bn_bias]
            h = y
        if ds_mod is not None:
            saved += [y_ds, convout_ds, smean_ds, sinvstd_ds,
                      ds_mod._shadow(), ds_mod.bn_weight, ds_mod.bn_bias]
        ctx.save_for_backward(*saved)
        ctx.mods = mods
        ctx.ds_mod = ds_mod
        return h

    @staticmethod
    def backward(ctx, dy):
        saved = ctx.saved_tensors
        mods, ds_mod = ctx.mods, ctx.ds_mod
        n = len(mods)
        x = saved[0]
        per = [saved[1 + 7 * i: 1 + 7 * (i + 1)] for i in range(n)]
        ds = (saved[1 + 7 * n: 1 + 7 * (n + 1)]
              if ds_mod is not None else None)

        dy = _cl(dy.to(torch.bfloat16) if dy.dtype != torch.bfloat16 else dy)
        cur = dy
        dres = None
        grads = [None] * (3 * n + (3 if ds_mod is not None else 0))

        next_sums = None  # (sum_dz, sum_dzx) computed by conv i+1's bwd
        for i in range(n - 1, -1, -1):
            m = mods[i]
            y, convout, smean, sinvstd, w_bf16, gamma, beta = per[i]
            inp = per[i - 1][0] if i > 0 else x
            direct = (getattr(m, "_managed", False)
                      and m.weight.grad is not None)
            has_res = i == n - 1
            # fuse the residual junction: for identity blocks the first
            # conv's dgrad accumulates into dres; for downsample blocks the
            # downsample dgrad accumulates into the main-path dx below
            dx_accum = dres if (i == 0 and ds_mod is None) else None
            # intra-block edge fusion: while producing dx (= conv i-1's
            # dy), also complete conv i-1's BN-backward channel sums
            fuse_up, up_mask, sums = None, 0, None
            if i >= 1:
                m_up = mods[i - 1]
                (y_u, convout_u, smean_u, sinvstd_u, _w_u, gamma_u,
                 beta_u) = per[i - 1]
                if (getattr(m_up, "_managed", False)
                        and m_up.weight.grad is not None):
                    sums = (m_up.bn_bias.grad, m_up.bn_weight.grad)
                else:
                    sums = (torch.zeros_like(m_up.bn_bias),
                            torch.zeros_like(m_up.bn_weight))
                up_mask = (2 if m_up.act == 1
                           else 4 if m_up.act == 2 else 0)
                fuse_up = [convout_u, y_u, smean_u, sinvstd_u, gamma_u,
                           beta_u, sums[0], sums[1]]
            sums_ready = next_sums is not None
            dx, dw, dgamma, dbeta, dres_i = _C().conv_bn_act_bwd(
                cur, y, inp, w_bf16, _rsck(m, w_bf16), convout, gamma, beta,
                smean, sinvstd, m.stride, m.padding, m.act, True, has_res,
                m.weight.grad if direct else None,
                m.bn_weight.grad if direct else None,
                m.bn_bias.grad if direct else None, dx_accum,
                fuse_up, up_mask, sums_ready,
                next_sums[0] if sums_ready else None,
                next_sums[1] if sums_ready else None)
            next_sums = sums
            if has_res:
                dres = dres_i
            if not direct:
                grads[3 * i: 3 * i + 3] = [dw, dgamma, dbeta]
            cur = dx

        if ds_mod is not None:
            (y_ds, convout_ds, smean_ds, sinvstd_ds, w_ds, gamma_ds,
             beta_ds) = ds
            direct = (getattr(ds_mod, "_managed", False)
                      and ds_mod.weight.grad is not None)
            dxds, dw, dgamma, dbeta, _ = _C().conv_bn_act_bwd(
                dres, y_ds, x, w_ds, _rsck(ds_mod, w_ds), convout_ds,
                gamma_ds, beta_ds, smean_ds, sinvstd_ds, ds_mod.stride,
                ds_mod.padding, ds_mod.act, True, False,
                ds_mod.weight.grad if direct else None,
                ds_mod.bn_weight.grad if direct else None,
                ds_mod.bn_bias.grad if direct else None, cur, None, 0,
                False, None, None)
            if not direct:
                grads[3 * n: 3 * n + 3] = [dw, dgamma, dbeta]
            cur = dxds  # == cur buffer, accumulated

        _bwd_done(mods[0])  # whole-block unit: callback lives on conv1
        return (cur, None, None, *grads)


def res_block(x, mods, ds_mod):
    params = []
    for m in mods:
        params += [m.weight, m.bn_weight, m.bn_bias]
    if ds_mod is not None:
        params += [ds_mod.weight, ds_mod.bn_weight, ds_mod.bn_bias]
    return ResBlockFn.apply(x, mods, ds_mod, *params)

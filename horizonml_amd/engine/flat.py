"""Flat parameter/gradient management for the MI355X fast path.

MI355X-first memory design (288 GB HBM3E: keep everything resident, few big
buffers, single-kernel updates):

* ONE flat f32 master buffer holds every trainable parameter (modules' params
  are re-pointed to views), ONE flat f32 gradient buffer (``.grad`` views are
  pre-assigned so autograd accumulates in place — stable addresses under
  hipGraph capture), ONE flat bf16 shadow the conv/linear kernels consume.
* The fused Adam/SGD kernel updates master + emits the bf16
  shadow + zeroes the gradient buffer in a single pass over the flat range.
* A second flat bf16 buffer holds the RSCK (dgrad) weight images, refreshed
  by one batched permute kernel per step.
* DP gradient sync becomes ONE bf16 all-reduce of the flat buffer.
"""
from __future__ import annotations

from typing import List

import torch
import torch.nn as nn

from .. import ops as _ops
from ..models.layers import ConvBNAct, DepthwiseConvBNAct, Linear


class FlatParamManager:
    def __init__(self, model: nn.Module, device: torch.device):
        # managed convs defer their weight-grad GEMMs: flush_wgrad() (called
        # by the fused optimizer step) runs them as one batched kernel
        _ops.extension().set_wgrad_defer(True)
        params = [p for p in model.parameters() if p.requires_grad]
        total = sum(p.numel() for p in params)
        self.params = params
        self.numel = total
        self.master = torch.zeros(total, device=device)
        self.grad = torch.zeros(total, device=device)
        self.shadow = torch.zeros(total, device=device, dtype=torch.bfloat16)
        self.slices = {}
        off = 0
        for p in params:
            n = p.numel()
            with torch.no_grad():
                self.master[off:off + n].copy_(p.data.flatten())
            p.data = self.master[off:off + n].view(p.shape)
            p.grad = self.grad[off:off + n].view(p.shape)
            self.slices[id(p)] = (off, n)
            off += n
        with torch.no_grad():
            self.shadow.copy_(self.master)

        # wire bf16 shadow views into the modules; collect conv RSCK metadata
        convs = []
        for m in model.modules():
            if isinstance(m, (ConvBNAct, Linear, DepthwiseConvBNAct)):
                woff, wn = self.slices[id(m.weight)]
                m.weight_bf16 = self.shadow[woff:woff + wn].view(
                    m.weight.shape)
                m._managed = True
                if isinstance(m, ConvBNAct):
                    convs.append((m, woff, wn))
        # BN-stats arena: one pre-zeroed f32 slab per conv ([2K] each),
        # re-zeroed by the fused optimizer kernel each step (no per-conv
        # fill kernels in the forward).
        stats_total = sum(2 * m.out_ch for m, _, _ in convs)
        self.stats_arena = torch.zeros(max(1, stats_total), device=device)
        soff = 0
        for m, _, _ in convs:
            m._stats_buf = self.stats_arena[soff:soff + 2 * m.out_ch]
            soff += 2 * m.out_ch

        rsck_total = sum(wn for _, _, wn in convs)
        self.rsck = torch.zeros(rsck_total, device=device,
                                dtype=torch.bfloat16)
        meta: List[List[int]] = []
        doff = 0
        self.max_elem = 1
        for m, woff, wn in convs:
            K, R, S, C = m.weight.shape
            meta.append([woff, doff, wn, (K << 16) | C])
            m._w_rsck = self.rsck[doff:doff + wn].view(R, S, C, K)
            doff += wn
            self.max_elem = max(self.max_elem, wn)
        self.meta = (torch.tensor(meta, dtype=torch.int32, device=device)
                     if meta else None)
        self.refresh_rsck()

    def refresh_rsck(self):
        if self.meta is not None:
            _ops.extension().permute_krsc_rsck(self.shadow, self.rsck,
                                               self.meta, self.max_elem)

    def zero_grad(self):
        self.grad.zero_()


class HorizonAdam:
    """Fused flat-buffer Adam (K9): 1 elementwise kernel + step increment +
    batched RSCK permute per step; zero_grad folded in."""

    def __init__(self, mgr: FlatParamManager, lr: float = 1e-3,
                 betas=(0.9, 0.999), eps: float = 1e-8,
                 weight_decay: float = 0.0):
        self.mgr = mgr
        self.lr, self.betas, self.eps, self.wd = lr, betas, eps, weight_decay
        dev = mgr.master.device
        self.m = torch.zeros_like(mgr.master)
        self.v = torch.zeros_like(mgr.master)
        self.step_t = torch.zeros(1, device=dev)

    def step(self, zero_grad: bool = True, grad_bf16=None,
             grad_scale: float = 1.0, probe=None):
        """``grad_bf16``: consume an all-reduced bf16 gradient buffer
        directly (× ``grad_scale``) — skips the DP unpack pass.
        ``probe``: optional ``(prev, sumsq, out)`` f32 tensors — the
        grad-divergence probe fused into the Adam pass (the standalone
        kernel costs a 3×|grad| extra sweep per step)."""
        _ops.extension().flush_wgrad()  # batched deferred weight grads
        p0, p1, p2 = probe if probe is not None else (None, None, None)
        _ops.extension().adam_step(self.mgr.master, self.mgr.grad, self.m,
                                   self.v, self.mgr.shadow, self.step_t,
                                   self.lr, self.betas[0], self.betas[1],
                                   self.eps, self.wd, zero_grad,
                                   self.mgr.stats_arena, grad_bf16,
                                   grad_scale, p0, p1, p2)
        self.mgr.refresh_rsck()

    # -- checkpoint round-trip (utils/checkpoint.py) ----------------------
    def state_dict(self):
        return {"kind": "horizon_adam", "lr": self.lr, "betas": self.betas,
                "eps": self.eps, "weight_decay": self.wd,
                "m": self.m.detach().cpu(), "v": self.v.detach().cpu(),
                "step_t": self.step_t.detach().cpu()}

    def load_state_dict(self, state):
        if state.get("kind") != "horizon_adam":
            raise ValueError("checkpoint optimizer state is not HorizonAdam")
        with torch.no_grad():
            self.m.copy_(state["m"].to(self.m.device))
            self.v.copy_(state["v"].to(self.v.device))
            self.step_t.copy_(state["step_t"].to(self.step_t.device))


class HorizonSGD:
    def __init__(self, mgr: FlatParamManager, lr: float = 0.1,
                 momentum: float = 0.9, weight_decay: float = 0.0):
        self.mgr = mgr
        self.lr, self.mu, self.wd = lr, momentum, weight_decay
        self.mom = (torch.zeros_like(mgr.master) if momentum > 0 else None)

    def step(self, zero_grad: bool = True, grad_bf16=None,
             grad_scale: float = 1.0, probe=None):
        """``probe``: optional ``(prev, sumsq, out)`` — the divergence
        probe fused into the SGD pass, like HorizonAdam's."""
        _ops.extension().flush_wgrad()  # batched deferred weight grads
        p0, p1, p2 = probe if probe is not None else (None, None, None)
        _ops.extension().sgd_step(self.mgr.master, self.mgr.grad, self.mom,
                                  self.mgr.shadow, self.lr, self.mu, self.wd,
                                  zero_grad, grad_bf16, grad_scale,
                                  p0, p1, p2)
        self.mgr.stats_arena.zero_()
        self.mgr.refresh_rsck()

    def state_dict(self):
        return {"kind": "horizon_sgd", "lr": self.lr, "momentum": self.mu,
                "weight_decay": self.wd,
                "mom": (self.mom.detach().cpu()
                        if self.mom is not None else None)}

    def load_state_dict(self, state):
        if state.get("kind") != "horizon_sgd":
            raise ValueError("checkpoint optimizer state is not HorizonSGD")
        if (state.get("mom") is None) != (self.mom is None):
            raise ValueError("momentum buffer mismatch with checkpoint")
        if self.mom is not None:
            with torch.no_grad():
                self.mom.copy_(state["mom"].to(self.mom.device))

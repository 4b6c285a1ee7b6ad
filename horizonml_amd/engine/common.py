"""Shared training-loop machinery for all three strategy engines."""
from __future__ import annotations

from typing import Iterable, List, Optional

import torch


class Meters:
    """Loss/accuracy accumulators that stay on-device until epoch end (the
    reference's per-step ``.item()`` reads would force a hipStreamSynchronize
    every step on GPU)."""

    def __init__(self, device: Optional[torch.device]):
        dev = device if device is not None else torch.device("cpu")
        self.loss_sum = torch.zeros((), dtype=torch.float32, device=dev)
        self.correct = torch.zeros((), dtype=torch.float32, device=dev)
        self.count = 0

    def update(self, loss: torch.Tensor, logits: torch.Tensor,
               labels: torch.Tensor):
        bs = labels.shape[0]
        self.loss_sum += loss.detach() * bs
        self.correct += (logits.detach().argmax(1) == labels).sum()
        self.count += bs

    def epoch_values(self):
        n = max(1, self.count)
        loss = (self.loss_sum / n).item()
        acc = 100.0 * (self.correct / n).item()
        self.loss_sum.zero_()
        self.correct.zero_()
        self.count = 0
        return loss, acc


class GradDivergenceProbe:
    """‖g_t − g_{t−1}‖₂ over all gradients (reference
    ``data_parallel_train.py:132-145``; kernel K11 in SURVEY.md §2.4).

    Keeps the previous flat gradient on-device; per-step results accumulate
    on-device and are read once per epoch (no per-step sync).
    """

    def __init__(self, params: Iterable[torch.Tensor]):
        self.params: List[torch.Tensor] = [p for p in params if p.requires_grad]
        if not self.params:
            raise ValueError("no trainable params for divergence probe")
        dev = self.params[0].device
        self.prev_list = [torch.zeros_like(p, dtype=torch.float32)
                          for p in self.params]
        self.sum = torch.zeros((), dtype=torch.float32, device=dev)
        self.n = 0
        self.first = True

    @torch.no_grad()
    def step(self):
        grads = [(p.grad if p.grad is not None else torch.zeros_like(p))
                 for p in self.params]
        if not self.first:
            diffs = torch._foreach_sub(grads, self.prev_list)
            norms = torch._foreach_norm(diffs)
            self.sum += torch.stack([n.float() for n in norms]) \
                .square().sum().sqrt()
            self.n += 1
        torch._foreach_copy_(self.prev_list, grads)
        self.first = False

    def epoch_value(self) -> float:
        v = (self.sum / max(1, self.n)).item()
        self.sum.zero_()
        self.n = 0
        return v


class FlatEagerOptimizer:
    """The flat fast-path machinery (FlatParamManager: direct grads into
    one flat buffer, batched deferred wgrads, fused Adam/SGD + bf16 shadow
    refresh) behind a torch-optimizer interface, WITHOUT graph capture —
    the eager PP/TP engines' GPU optimizer (r02).  Replaces torch Adam's
    multi-tensor passes + per-conv wgrad launches + per-param zero fills.

    Ordering contract: deferred weight-gradient GEMMs complete at
    ``flush_wgrad()`` (or inside ``step()``); anything reading ``p.grad``
    before the step — DP finalize packs, divergence probes — must flush
    first.  Gradients are zeroed at ``zero_grad()`` (loop start), never by
    the step, so probes may read them after stepping.
    """

    def __init__(self, module, name: str, lr: float, device):
        from .flat import FlatParamManager, HorizonAdam, HorizonSGD
        self.mgr = FlatParamManager(module, device)
        self.opt = (HorizonAdam(self.mgr, lr=lr) if name == "adam"
                    else HorizonSGD(self.mgr, lr=lr))

    def zero_grad(self, set_to_none: bool = False):
        # set_to_none would detach the flat views — always zero in place
        self.mgr.zero_grad()

    def flush_wgrad(self):
        from .. import ops as _ops
        _ops.extension().flush_wgrad()

    def step(self):
        self.opt.step(zero_grad=False)

    def state_dict(self):
        return self.opt.state_dict()

    def load_state_dict(self, state):
        self.opt.load_state_dict(state)


def build_engine_optimizer(module, params, name: str, lr: float, device):
    """Fused flat-eager optimizer on GPU (extension present), torch
    optimizer otherwise."""
    if device is not None and device.type == "cuda":
        from .. import ops as _ops
        if _ops.has_extension():
            return FlatEagerOptimizer(module, name, lr, device)
    return build_optimizer(params, name, lr=lr)


def build_optimizer(params, name: str = "adam", lr: float = 1e-3,
                    momentum: float = 0.9, weight_decay: float = 0.0):
    """Reference default: Adam(lr=1e-3) (``data_parallel_train.py:205``).
    The north star also names the SGD step — both supported."""
    name = name.lower()
    # foreach=True batches the per-parameter update into multi-tensor
    # kernels — the eager engine paths are launch-bound otherwise
    if name == "adam":
        return torch.optim.Adam(params, lr=lr, weight_decay=weight_decay,
                                foreach=True)
    if name == "sgd":
        return torch.optim.SGD(params, lr=lr, momentum=momentum,
                               weight_decay=weight_decay, foreach=True)
    raise ValueError(f"unknown optimizer {name!r}")


def progress_iter(loader, desc: str, enabled: bool = True):
    """tqdm per-worker progress bar (reference ``data_parallel_train.py:102``
    console parity); auto-disabled on non-TTY output so logs/CI stay clean."""
    if not enabled:
        return loader
    try:
        from tqdm import tqdm
        return tqdm(loader, desc=desc, leave=False, disable=None)
    except ImportError:
        return loader

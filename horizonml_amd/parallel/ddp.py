"""Custom data-parallel engine: bucketed gradient all-reduce over RCCL/xGMI.

Replaces the reference's ``DDP(model)`` wrap (``data_parallel_train.py:202``,
``train.py:68``; collective C3 in SURVEY.md §2.5) with an explicit engine:

* parameters are broadcast from rank 0 at construction (replica consistency);
* gradients are grouped into **buckets in reverse parameter order** (the
  order backward produces them), each bucket backed by a preallocated flat
  communication buffer;
* a per-parameter ``post_accumulate_grad_hook`` packs the grad into its
  bucket; when the bucket is full its all-reduce is launched **async** so
  communication overlaps the rest of backward (RCCL schedules on its own
  stream; ``work.wait()`` only fences the compute stream);
* on GPU the comm dtype is **bf16** (half the xGMI bytes of the reference's
  fp32 gloo payload); CPU/gloo keeps fp32;
* averaging (÷world_size) is fused into the unpack pass.

xGMI note: each MI355X has 7 point-to-point links (~153 GB/s each); RCCL's
ring all-reduce is per-link bound, so the default bucket size is chosen large
(whole-model 22 MB bf16 fits ~2 buckets) to amortize launch latency while
still overlapping with backward.
"""
from __future__ import annotations

from typing import List, Optional

import torch
import torch.distributed as dist
import torch.nn as nn


class _Bucket:
    def __init__(self, params: List[torch.Tensor], comm_dtype: torch.dtype,
                 device: torch.device):
        self.params = params
        numel = sum(p.numel() for p in params)
        self.flat = torch.zeros(numel, dtype=comm_dtype, device=device)
        self.views = []
        off = 0
        for p in params:
            n = p.numel()
            self.views.append(self.flat[off:off + n].view(p.shape))
            off += n
        self.pending = 0
        self.work = None

    def reset(self):
        self.pending = len(self.params)
        self.work = None


class BucketedDataParallel(nn.Module):
    def __init__(self, module: nn.Module, bucket_cap_mb: float = 25.0,
                 comm_dtype: Optional[torch.dtype] = None,
                 process_group=None, profiler=None,
                 broadcast_params: bool = True, parameters=None,
                 defer_reduction: bool = False):
        """``defer_reduction``: do NOT launch bucket all-reduces from the
        grad hooks; pack everything from the fully-accumulated ``p.grad`` in
        ``finalize_backward`` instead.  Required whenever one optimizer step
        spans SEVERAL ``backward()`` calls (pipeline microbatching in hybrid
        DP×PP): hook-triggered launches would all-reduce first-microbatch
        partial gradients and race with later microbatches packing into the
        same flat buffer."""
        super().__init__()
        self.module = module
        self.group = process_group
        self.profiler = profiler
        self.defer_reduction = defer_reduction
        self.world_size = (dist.get_world_size(process_group)
                           if dist.is_initialized() else 1)
        params = (list(parameters) if parameters is not None
                  else [p for p in module.parameters() if p.requires_grad])
        if not params:
            raise ValueError("module has no trainable parameters")
        self.device = params[0].device
        if comm_dtype is None:
            comm_dtype = (torch.bfloat16 if self.device.type == "cuda"
                          else torch.float32)
        self.comm_dtype = comm_dtype

        if broadcast_params and self.world_size > 1:
            with torch.no_grad():
                for p in params:
                    dist.broadcast(p.data, src=self._group_src(), group=process_group)
                for b in module.buffers():
                    if b.dtype.is_floating_point or b.dtype in (torch.int64,):
                        dist.broadcast(b.data, src=self._group_src(), group=process_group)

        # Buckets in reverse parameter order ≈ gradient-ready order.
        self.buckets: List[_Bucket] = []
        self._param_bucket = {}
        cap = int(bucket_cap_mb * 1024 * 1024)
        cur: List[torch.Tensor] = []
        cur_bytes = 0
        esize = torch.tensor([], dtype=comm_dtype).element_size()
        for p in reversed(params):
            cur.append(p)
            cur_bytes += p.numel() * esize
            if cur_bytes >= cap:
                self._seal_bucket(cur)
                cur, cur_bytes = [], 0
        if cur:
            self._seal_bucket(cur)

        self._hooks = []
        for p in params:
            self._hooks.append(
                p.register_post_accumulate_grad_hook(self._grad_ready))
        self._reset_buckets()

    def _group_src(self) -> int:
        if self.group is None:
            return 0
        return dist.get_global_rank(self.group, 0)

    def _seal_bucket(self, params: List[torch.Tensor]):
        b = _Bucket(params, self.comm_dtype, self.device)
        for i, p in enumerate(params):
            self._param_bucket[id(p)] = (b, i)
        self.buckets.append(b)

    def _reset_buckets(self):
        for b in self.buckets:
            b.reset()

    # -- backward-hook machinery ------------------------------------------
    def _grad_ready(self, p: torch.Tensor):
        if self.world_size <= 1 or self.defer_reduction:
            return
        b, i = self._param_bucket[id(p)]
        b.views[i].copy_(p.grad.detach())  # cast into comm dtype
        b.pending -= 1
        if b.pending == 0:
            b.work = dist.all_reduce(b.flat, op=dist.ReduceOp.SUM,
                                     group=self.group, async_op=True)
            if self.profiler is not None:
                self.profiler.add_bytes(b.flat.numel()
                                        * b.flat.element_size())

    def finalize_backward(self):
        """Wait for in-flight all-reduces, average, and scatter back into
        ``param.grad``.  Call after ``loss.backward()``; the wait time here is
        the *exposed* (non-overlapped) communication time."""
        if self.world_size <= 1:
            return
        inv = 1.0 / self.world_size
        for b in self.buckets:
            if b.pending != 0:
                # deferred mode, or grads for some params never materialized
                # (frozen subgraph this step) — (re)pack from the accumulated
                # ``p.grad`` and reduce, staying collective across ranks
                for v, p in zip(b.views, b.params):
                    if p.grad is None:
                        v.zero_()
                    else:
                        v.copy_(p.grad.detach())
                b.work = dist.all_reduce(b.flat, op=dist.ReduceOp.SUM,
                                         group=self.group, async_op=True)
                if self.profiler is not None:
                    self.profiler.add_bytes(b.flat.numel()
                                            * b.flat.element_size())
            if b.work is not None:
                b.work.wait()
        for b in self.buckets:
            for v, p in zip(b.views, b.params):
                if p.grad is not None:
                    p.grad.detach().copy_(v).mul_(inv)
        self._reset_buckets()

    def forward(self, *args, **kwargs):
        return self.module(*args, **kwargs)

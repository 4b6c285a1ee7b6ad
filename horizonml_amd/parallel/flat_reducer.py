"""Bucketed all-reduce of the flat gradient buffer, overlapped with the
tail of backward (SURVEY.md §2.5 C3 "overlap with remaining backward";
VERDICT r01 next-round item 2).

The flat fast path (``engine/flat.py``) holds every gradient in ONE flat f32
buffer, with conv weight-gradient GEMMs deferred to a batched flush.  The
single-buffer DP sync (one bf16 all-reduce after backward) leaves the whole
payload exposed on the step's critical path — ~250 µs for ResNet18's 22 MB
bf16 at 8 ranks on a ring over 153 GB/s xGMI links, fully serial after a
~1 ms backward.  This module splits the flat buffer into a few contiguous
**buckets in reverse-layer order** (the order backward completes them) and
launches each bucket's pipeline as soon as its layers' backward is done:

    flush that bucket's deferred wgrads   (batched kernel, this range only)
    pack f32→bf16 into the bucket's slice of the comm buffer
    RCCL all-reduce of the slice (async — RCCL's stream overlaps the
    remaining backward / the next bucket's wgrad flush)

Readiness is signalled by per-unit backward callbacks (``_bwd_done_cb`` on
the unit's root module, invoked at the end of the unit's autograd backward —
see ``models/_functional_gpu.py``): a bucket fires when ALL its units have
completed backward.  Everything is stream-ordered, so the schedule is
hipGraph-capture-safe: callbacks run at capture time and the recorded
launch order replays identically.

Correctness contract (tested under gloo, ws=2, f32): bucketed reduction of
disjoint slices is bit-for-bit equal to one all-reduce of the whole buffer.
"""
from __future__ import annotations

from typing import Callable, List, Optional, Sequence, Tuple

import torch
import torch.distributed as dist


class FlatBucketReducer:
    """All-reduce a flat gradient buffer in bucket slices.

    ``buckets``: ordered ``(lo, hi)`` element ranges (expected ready order —
    reverse layer order).  Ranges must be disjoint; together they must cover
    exactly ``[0, grad.numel())`` (validated).
    ``flush_range_fn(lo, hi)``: optional hook run before packing a bucket —
    the flat path passes the deferred-wgrad range flush here.
    """

    def __init__(self, grad: torch.Tensor, buckets: Sequence[Tuple[int, int]],
                 group=None, comm_dtype: Optional[torch.dtype] = None,
                 flush_range_fn: Optional[Callable[[int, int], None]] = None,
                 profiler=None, comm_buf: Optional[torch.Tensor] = None):
        cover = 0
        for lo, hi in sorted(buckets):
            if lo != cover or hi <= lo:
                raise ValueError(f"buckets must tile [0,{grad.numel()}) "
                                 f"contiguously, got {list(buckets)}")
            cover = hi
        if cover != grad.numel():
            raise ValueError(f"buckets cover {cover} != {grad.numel()}")
        self.grad = grad
        self.buckets: List[Tuple[int, int]] = list(buckets)
        self.group = group
        self.flush_range_fn = flush_range_fn
        self.profiler = profiler
        if comm_dtype is None:
            comm_dtype = (torch.bfloat16 if grad.is_cuda else torch.float32)
        self.comm = (comm_buf if comm_buf is not None else
                     torch.zeros(grad.numel(), device=grad.device,
                                 dtype=comm_dtype))
        self._works: List[Optional[object]] = [None] * len(self.buckets)
        self._reduced = [False] * len(self.buckets)

    @property
    def world_size(self) -> int:
        return (dist.get_world_size(self.group)
                if dist.is_available() and dist.is_initialized() else 1)

    def begin_step(self):
        self._works = [None] * len(self.buckets)
        self._reduced = [False] * len(self.buckets)

    def reduce_bucket(self, i: int):
        """Flush + pack + launch the async all-reduce for bucket ``i``."""
        if self._reduced[i]:
            return
        self._reduced[i] = True
        lo, hi = self.buckets[i]
        if self.flush_range_fn is not None:
            self.flush_range_fn(lo, hi)
        seg = self.comm[lo:hi]
        seg.copy_(self.grad[lo:hi])
        if self.world_size > 1:
            self._works[i] = dist.all_reduce(seg, op=dist.ReduceOp.SUM,
                                             group=self.group, async_op=True)
            if self.profiler is not None:
                self.profiler.add_bytes(seg.numel() * seg.element_size())

    def reduce_all(self):
        """Single-shot fallback path: reduce every not-yet-reduced bucket."""
        for i in range(len(self.buckets)):
            self.reduce_bucket(i)

    def wait(self):
        """Fence the issuing stream on every in-flight all-reduce.  After
        this, ``self.comm`` holds the SUM over ranks (scale by 1/world in
        the consumer — the fused optimizer's ``grad_scale``)."""
        for i, w in enumerate(self._works):
            if w is not None:
                w.wait()
                self._works[i] = None


def partition_unit_sizes(unit_sizes: Sequence[int],
                         n_buckets: int) -> List[int]:
    """Split ordered units into ``n_buckets`` contiguous non-empty groups
    with near-equal total size (greedy by cumulative share).  Returns the
    unit count per bucket."""
    n_units = len(unit_sizes)
    n_buckets = max(1, min(n_buckets, n_units))
    total = max(1, sum(unit_sizes))
    counts: List[int] = []
    start = 0
    acc = 0
    for b in range(n_buckets):
        target = total * (b + 1) / n_buckets
        end = max(start + 1, start)  # every bucket takes ≥1 unit
        acc += unit_sizes[start]
        while (end < n_units and acc < target
               and n_units - end > n_buckets - b - 1):
            acc += unit_sizes[end]
            end += 1
        counts.append(end - start)
        start = end
    counts[-1] += n_units - start
    return counts


def grad_units(model: torch.nn.Module):
    """Ordered gradient-producing units of a model: ``(root_module,
    params)`` where ``root_module`` is the module whose backward completion
    means every param in ``params`` has its gradient written (the module
    carrying the ``_bwd_done_cb`` hook in ``models/_functional_gpu.py``).

    Units follow ``model.parameters()`` order, so each unit's params are a
    contiguous run of the flat buffer.
    """
    from ..models.layers import ConvBNAct, DepthwiseConvBNAct, Linear
    from ..models.resnet import BasicBlock, Bottleneck

    units = []

    def walk(m):
        if isinstance(m, (BasicBlock, Bottleneck)):
            # fused res_block: one autograd node; callback lands on conv1
            units.append((m.conv1, list(m.parameters())))
            return
        if isinstance(m, (ConvBNAct, DepthwiseConvBNAct, Linear)):
            units.append((m, list(m.parameters())))
            return
        for c in m.children():
            walk(c)

    walk(model)
    # sanity: units must cover all trainable params exactly once, in order
    unit_params = [p for _, ps in units for p in ps]
    model_params = [p for p in model.parameters() if p.requires_grad]
    if [id(p) for p in unit_params] != [id(p) for p in model_params]:
        raise ValueError("grad_units does not tile model.parameters(); "
                         "model unsupported for bucketed reduction")
    return units


def build_bucket_schedule(model: torch.nn.Module, slices,
                          n_buckets: int):
    """Plan the reverse-layer bucket schedule for a flat-managed model.

    ``slices``: ``FlatParamManager.slices`` (id(param) -> (offset, numel)).
    Returns ``(ranges, bucket_mods)`` where ``ranges[i]`` is the flat
    ``(lo, hi)`` of bucket ``i`` and ``bucket_mods[i]`` the root modules
    whose backward completion readies it — both listed in REVERSE layer
    order (bucket 0 = the model tail, ready first during backward).
    """
    units = grad_units(model)
    sizes = [sum(slices[id(p)][1] for p in ps) for _, ps in units]
    counts = partition_unit_sizes(sizes, n_buckets)
    ranges: List[Tuple[int, int]] = []
    bucket_mods: List[List[torch.nn.Module]] = []
    start = 0
    for c in counts:
        group = units[start:start + c]
        los = [slices[id(ps[0])][0] for _, ps in group]
        his = [slices[id(ps[-1])][0] + slices[id(ps[-1])][1]
               for _, ps in group]
        ranges.append((min(los), max(his)))
        bucket_mods.append([m for m, _ in group])
        start += c
    # reverse: backward readies the tail of the model first
    return ranges[::-1], bucket_mods[::-1]


class BackwardBucketScheduler:
    """Wire per-unit backward callbacks to a :class:`FlatBucketReducer`.

    Installs ``_bwd_done_cb`` on each bucket's root modules; when the last
    unit of a bucket reports backward-done, the bucket's flush+pack+
    all-reduce launches — overlapping the remaining backward.
    """

    def __init__(self, reducer: FlatBucketReducer,
                 bucket_mods: Sequence[Sequence[torch.nn.Module]]):
        self.reducer = reducer
        self._counts = [len(ms) for ms in bucket_mods]
        self._remaining = list(self._counts)
        for bi, mods in enumerate(bucket_mods):
            for m in mods:
                m._bwd_done_cb = self._make_cb(bi)

    def _make_cb(self, bi: int):
        def cb():
            self._remaining[bi] -= 1
            if self._remaining[bi] == 0:
                self.reducer.reduce_bucket(bi)
        return cb

    def begin_step(self):
        self._remaining = list(self._counts)
        self.reducer.begin_step()

    def remove(self, bucket_mods):
        for mods in bucket_mods:
            for m in mods:
                if hasattr(m, "_bwd_done_cb"):
                    del m._bwd_done_cb

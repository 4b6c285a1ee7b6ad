"""Distributed runtime: RCCL-over-xGMI first, gloo for CPU plumbing.

Replaces the reference's per-script ``setup_distributed`` (gloo, 300 s
timeout, entry barrier — ``data_parallel_train.py:28-40``,
``layer_model_parallel_train.py:88-100``, ``tensor_parallel_train.py:108-120``,
env-var variant ``train.py:15-41``) with one module:

* backend ``nccl`` (= RCCL on ROCm) when CUDA/HIP devices are visible — one
  process per GPU, rank→device binding via ``torch.cuda.set_device``;
* backend ``gloo`` on CPU-only hosts (BASELINE.json config #1 plumbing);
* sub-communicators for hybrid DP×PP (``new_group`` per DP ring / PP chain).
"""
from __future__ import annotations

import datetime
import os
from dataclasses import dataclass, field
from typing import Dict, Optional, Tuple

import torch
import torch.distributed as dist

DEFAULT_TIMEOUT_S = 300  # reference parity: 300 s gloo op timeout (C1)


def auto_backend(prefer: Optional[str] = None,
                 world_size: Optional[int] = None,
                 verbose: bool = True) -> str:
    """nccl (RCCL) on GPU hosts, gloo otherwise — and gloo whenever this
    node's rank count oversubscribes its visible GPUs (RCCL rejects two
    ranks on one device, e.g. the reference's default world_size=5 on a
    1-GPU box).

    Oversubscription is judged on the ranks *per node*: under a multi-node
    launcher ``LOCAL_WORLD_SIZE`` is authoritative (a 16-rank job over two
    8-GPU nodes must stay on RCCL); in the single-node spawn path there is
    no env and ``world_size`` == ranks on this node.  The fallback decision
    is printed once so an accidental CPU run is never silent.
    """
    if prefer in ("nccl", "rccl"):
        return "nccl"
    if prefer == "gloo":
        return "gloo"
    if not torch.cuda.is_available():
        return "gloo"
    local_ranks = os.environ.get("LOCAL_WORLD_SIZE")
    local_ranks = int(local_ranks) if local_ranks is not None else world_size
    if local_ranks is not None and local_ranks > torch.cuda.device_count():
        if verbose:
            print(f"[horizonml] {local_ranks} ranks on this node > "
                  f"{torch.cuda.device_count()} visible GPU(s) — falling "
                  "back to gloo/CPU (pass --backend nccl to force RCCL)",
                  flush=True)
        return "gloo"
    return "nccl"


def bind_gpu(rank: int) -> Optional[torch.device]:
    """Pin this process to its GPU (LOCAL_RANK if set, else rank % ngpus)."""
    if not torch.cuda.is_available():
        return None
    local = int(os.environ.get("LOCAL_RANK", rank % torch.cuda.device_count()))
    torch.cuda.set_device(local)
    return torch.device("cuda", local)


@dataclass
class DistContext:
    rank: int
    world_size: int
    backend: str
    device: Optional[torch.device]
    groups: Dict[str, "dist.ProcessGroup"] = field(default_factory=dict)

    @property
    def is_gpu(self) -> bool:
        return self.device is not None and self.device.type == "cuda"


def setup_distributed(rank: int, world_size: int, port: int,
                      backend: Optional[str] = None,
                      master_addr: str = "127.0.0.1",
                      timeout_s: int = DEFAULT_TIMEOUT_S) -> DistContext:
    """Init process group and barrier (entry barrier parity with
    ``data_parallel_train.py:40``)."""
    os.environ.setdefault("MASTER_ADDR", master_addr)
    os.environ.setdefault("MASTER_PORT", str(port))
    # keep explicit values authoritative for spawned workers
    os.environ["MASTER_ADDR"] = master_addr
    os.environ["MASTER_PORT"] = str(port)
    be = auto_backend(backend, world_size)
    device = bind_gpu(rank) if be == "nccl" else None
    dist.init_process_group(backend=be, rank=rank, world_size=world_size,
                            timeout=datetime.timedelta(seconds=timeout_s))
    ctx = DistContext(rank=rank, world_size=world_size, backend=be,
                      device=device)
    tune_cpu_threads(ctx)
    barrier(ctx)
    return ctx


def setup_from_env(backend: Optional[str] = None,
                   timeout_s: int = DEFAULT_TIMEOUT_S) -> DistContext:
    """Env-var rendezvous (RANK/WORLD_SIZE/MASTER_ADDR/MASTER_PORT) — the
    legacy Docker entry mode (``train.py:15-41``) and torchrun."""
    rank = int(os.environ.get("RANK", 0))
    world = int(os.environ.get("WORLD_SIZE", 1))
    be = auto_backend(backend, world)
    device = bind_gpu(rank) if be == "nccl" else None
    dist.init_process_group(backend=be, rank=rank, world_size=world,
                            timeout=datetime.timedelta(seconds=timeout_s))
    ctx = DistContext(rank=rank, world_size=world, backend=be, device=device)
    tune_cpu_threads(ctx)
    barrier(ctx)
    return ctx


def tune_cpu_threads(ctx: "DistContext"):
    """GPU workers do only tiny CPU tensor work (batch gathers, metric
    scalars): the default 128-thread intra-op pool turns each such op into
    a thread wake-storm (~30 ms per batch measured on MI355X boxes).  Cap
    the pool; CPU-only workers keep the full pool for real compute."""
    if ctx.is_gpu:
        try:
            torch.set_num_threads(min(8, torch.get_num_threads()))
        except RuntimeError:
            pass


def barrier(ctx: Optional[DistContext] = None, group=None):
    if not (dist.is_available() and dist.is_initialized()):
        return
    if ctx is not None and ctx.is_gpu:
        dist.barrier(group=group, device_ids=[ctx.device.index])
    else:
        dist.barrier(group=group)


def make_hybrid_groups(ctx: DistContext, dp_size: int, pp_size: int
                       ) -> Tuple["dist.ProcessGroup", "dist.ProcessGroup", int, int]:
    """Sub-communicators for hybrid DP×PP (BASELINE.json config #5).

    Rank layout: rank = dp_rank * pp_size + pp_stage  (PP chains are
    contiguous rank runs → adjacent pipeline stages sit on xGMI-adjacent
    GPUs for the p2p activation relay; DP rings stride across chains).
    Returns (dp_group, pp_group, dp_rank, pp_stage).
    """
    if dp_size * pp_size != ctx.world_size:
        raise ValueError(f"dp_size*pp_size={dp_size * pp_size} != "
                         f"world_size={ctx.world_size}")
    dp_rank, pp_stage = divmod(ctx.rank, pp_size)
    dp_group = pp_group = None
    # all ranks must call new_group for every group, in the same order
    for stage in range(pp_size):
        ranks = [d * pp_size + stage for d in range(dp_size)]
        g = dist.new_group(ranks=ranks)
        if pp_stage == stage:
            dp_group = g
    for d in range(dp_size):
        ranks = [d * pp_size + s for s in range(pp_size)]
        g = dist.new_group(ranks=ranks)
        if dp_rank == d:
            pp_group = g
    ctx.groups["dp"] = dp_group
    ctx.groups["pp"] = pp_group
    return dp_group, pp_group, dp_rank, pp_stage


def teardown_distributed(ctx: Optional[DistContext] = None,
                         final_barrier_timeout_s: int = 5):
    """Clean teardown.

    The reference ends with a best-effort completion-tensor fan-out + 5 s
    barrier (Q6, ``data_parallel_train.py:211-230``).  RCCL group teardown +
    launcher-side join replaces the completion tensors; the
    best-effort barrier semantics are preserved.
    """
    if not (dist.is_available() and dist.is_initialized()):
        return
    try:
        barrier(ctx)
    except Exception:  # noqa: BLE001 — best-effort, like the reference
        pass
    try:
        dist.destroy_process_group()
    except Exception:  # noqa: BLE001
        pass

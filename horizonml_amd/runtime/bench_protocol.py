"""Rank-coordination + record contract for bench.py (the driver contract).

bench.py's multi-rank execution has three pieces of coordination logic that
must be right the FIRST time they run on an 8-GPU node:

1. **capture agreement** — all ranks must agree whether the hipGraph replay
   path is used *before* any replay, else a rank-divergent capture failure
   deadlocks captured collectives;
2. **MAX-elapsed reduction** — the whole-job timing is the slowest rank's
   (barrier + sync bracketed by the caller);
3. **the one-line JSON record** — the driver parses exactly one line from
   rank 0 with the BASELINE.json metric/config contract.

They are factored here, device-agnostic, so a 2-rank CPU/gloo test
(`tests/test_bench_protocol_cpu.py`) can rehearse the exact code bench.py
runs over RCCL.
"""
from __future__ import annotations

import json
from typing import Optional

import torch
import torch.distributed as dist

# reference's best DP number: ≈19.6 s per 1000-sample epoch on CPU/gloo
#  = 51.0 images/sec
BASELINE_IMAGES_PER_SEC = 1000.0 / 19.6


def agree_all_ranks(ok: bool, world: int,
                    device: Optional[torch.device] = None) -> bool:
    """True only if EVERY rank passed ``ok=True`` (all-reduce MIN).  Used to
    gate graph-replay mode: a rank whose capture failed forces every rank
    to the eager path so collective sequences stay aligned."""
    if world <= 1 or not (dist.is_available() and dist.is_initialized()):
        return ok
    t = torch.tensor([1.0 if ok else 0.0],
                     device=device if device is not None else "cpu")
    dist.all_reduce(t, op=dist.ReduceOp.MIN)
    return bool(t.item() >= 1.0)


def max_elapsed_over_ranks(elapsed_s: float, world: int,
                           device: Optional[torch.device] = None) -> float:
    """Whole-job elapsed = MAX over ranks (the slowest rank defines the
    step time; reporting rank 0's own clock would overstate throughput)."""
    if world <= 1 or not (dist.is_available() and dist.is_initialized()):
        return elapsed_s
    t = torch.tensor([elapsed_s],
                     device=device if device is not None else "cpu")
    dist.all_reduce(t, op=dist.ReduceOp.MAX)
    return float(t.item())


def build_record(*, elapsed_s: float, steps: int, warmup: int, world: int,
                 batch_size: int, model: str, optimizer: str, exec_mode: str,
                 final_loss: float, infer: bool = False,
                 comm_mode: str = "none", image_size: int = 32) -> dict:
    """Assemble the driver-contract JSON record (one line, rank 0 only).

    ``value`` is the WHOLE-JOB aggregate images/sec over all ranks (weak
    scaling: per-GPU batch fixed), ``vs_baseline`` = value / the reference's
    51.0 img/s.
    """
    if elapsed_s <= 0 or steps <= 0:
        raise ValueError("timed region must cover >=1 step with >0 elapsed")
    ms_per_step = elapsed_s / steps * 1000.0
    global_batch = batch_size * world
    ips = global_batch * steps / elapsed_s
    return {
        "metric": "images/sec",
        "value": round(ips, 2),
        "unit": "images/sec",
        "n_gpus": world,
        "steps": steps,
        "warmup": warmup,
        "ms_per_step": round(ms_per_step, 4),
        "higher_is_better": True,
        "scaling": "weak",
        "vs_baseline": round(ips / BASELINE_IMAGES_PER_SEC, 2),
        "dtype": "bf16",
        "data": "synthetic",
        "config": {
            "model": model + ("_infer" if infer
                              else ("_cifar10" if image_size == 32
                                    else f"_synthetic{image_size}")),
            "global_batch": global_batch,
            "seq_len": None,
            "image": f"3x{image_size}x{image_size}",
            "parallelism": f"dp{world}",
            "optimizer": optimizer,
            "exec": exec_mode,
            "comm": comm_mode,
            "epoch_time_s_1000_samples": round(1000.0 / ips, 6),
            "final_loss": round(final_loss, 4),
        },
    }


def emit_record(record: dict) -> str:
    """Serialize to the single line the driver parses."""
    return json.dumps(record)

"""Process launcher + timeout supervision + CSV aggregation.

Reference parity (``run_data_parallel`` at ``data_parallel_train.py:233-291``
and the MP/TP twins):

* ``mp.set_start_method('spawn')``, one process per worker, shared free-port
  rendezvous on 127.0.0.1;
* join deadline scaled by sample size (DP/MP ``max(120, 120·n/1000)`` s, TP
  ``max(400, 400·n/1000)`` s — ``data_parallel_train.py:252``,
  ``tensor_parallel_train.py:346``), stragglers force-terminated;
* wall-clock ``total_training_time`` measured around the spawn/join;
* per-worker CSVs re-read and concatenated into
  ``combined_results_{sample_size}.csv`` with ``worker`` and
  ``total_training_time`` columns (``data_parallel_train.py:276-291``).

On a GPU host each worker binds one MI355X (LOCAL_RANK = rank).
"""
from __future__ import annotations

import os
import time
from typing import Callable, Optional

import pandas as pd
import torch.multiprocessing as mp

from ..utils.ports import find_free_port


def timeout_for(sample_size: int, base: int = 120) -> float:
    """Reference deadline: ``max(base, base * n / 1000)`` seconds."""
    return max(base, base * sample_size / 1000)


def aggregate_worker_csvs(logs_dir: str, world_size: int, sample_size: int,
                          total_training_time: float) -> Optional[pd.DataFrame]:
    """Concat per-worker CSVs into combined_results_{n}.csv (schema parity)."""
    frames = []
    for rank in range(world_size):
        path = os.path.join(logs_dir, f"worker_{rank}_samples_{sample_size}.csv")
        if os.path.isfile(path):
            df = pd.read_csv(path)
            df["worker"] = rank
            frames.append(df)
    if not frames:
        return None
    combined = pd.concat(frames, ignore_index=True)
    combined["total_training_time"] = total_training_time
    out = os.path.join(logs_dir, f"combined_results_{sample_size}.csv")
    combined.to_csv(out, index=False)
    return combined


def run_workers(worker_fn: Callable, world_size: int, epochs: int,
                sample_size: int, logs_dir: str,
                timeout_base: int = 120,
                extra_args: tuple = ()) -> Optional[pd.DataFrame]:
    """Spawn ``world_size`` worker processes and supervise them.

    ``worker_fn(rank, world_size, epochs, sample_size, port, logs_dir,
    *extra_args)`` runs in each spawned process.
    Returns the combined DataFrame (or None if no worker produced a CSV).
    """
    os.makedirs(logs_dir, exist_ok=True)
    try:
        mp.set_start_method("spawn", force=True)
    except RuntimeError:
        pass
    port = find_free_port()

    start = time.time()
    procs = []
    for rank in range(world_size):
        p = mp.Process(target=worker_fn,
                       args=(rank, world_size, epochs, sample_size, port,
                             logs_dir, *extra_args))
        p.start()
        procs.append(p)

    deadline = start + timeout_for(sample_size, timeout_base) * max(1, epochs)
    for p in procs:
        remaining = max(0.0, deadline - time.time())
        p.join(timeout=remaining)
    for p in procs:
        if p.is_alive():
            print(f"[launcher] force-terminating straggler pid={p.pid}")
            p.terminate()
            p.join(timeout=10)
            if p.is_alive():
                p.kill()
    total_training_time = time.time() - start
    return aggregate_worker_csvs(logs_dir, world_size, sample_size,
                                 total_training_time)

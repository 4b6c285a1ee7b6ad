"""Per-worker CSV metrics — byte-parity with the reference schema.

Schema (``data_parallel_train.py:161-180``): per-worker file
``{logs_dir}/worker_{rank}_samples_{n}.csv`` with columns

    epoch, loss, accuracy, epoch_time, avg_step_time, compute_time,
    comm_time, idle_time, avg_cpu, avg_memory, grad_divergence
    [, avg_bandwidth]            # MP/TP only, like the reference

flushed **every epoch** (the reference's crash-resilience property —
SURVEY.md §5.3).  On GPU hosts two extra columns (``gpu_memory_mb``,
``gpu_util``) are appended after the reference set; name-based parsers of the
reference layout keep working.

The legacy Docker entry writes ``training_logs_worker_{rank}.csv`` with
``Worker,Epoch,Loss,Accuracy,Time`` (``train.py:115-116``) — see
``write_legacy_row``.
"""
from __future__ import annotations

import csv
import os
from dataclasses import dataclass
from typing import List, Optional

import psutil
import torch

REFERENCE_COLUMNS = ["epoch", "loss", "accuracy", "epoch_time",
                     "avg_step_time", "compute_time", "comm_time",
                     "idle_time", "avg_cpu", "avg_memory", "grad_divergence"]


@dataclass
class EpochMetrics:
    epoch: int
    loss: float
    accuracy: float
    epoch_time: float
    avg_step_time: float
    compute_time: float
    comm_time: float
    idle_time: float
    avg_cpu: float
    avg_memory: float
    grad_divergence: float
    avg_bandwidth: Optional[float] = None
    gpu_memory_mb: Optional[float] = None
    gpu_util: Optional[float] = None


class MetricsWriter:
    def __init__(self, logs_dir: str, rank: int, sample_size: int,
                 with_bandwidth: bool = False, with_gpu: bool = False):
        os.makedirs(logs_dir, exist_ok=True)
        self.path = os.path.join(logs_dir,
                                 f"worker_{rank}_samples_{sample_size}.csv")
        self.columns = list(REFERENCE_COLUMNS)
        if with_bandwidth:
            self.columns.append("avg_bandwidth")
        self.with_gpu = with_gpu and torch.cuda.is_available()
        if self.with_gpu:
            self.columns += ["gpu_memory_mb", "gpu_util"]
        self.rows: List[EpochMetrics] = []

    def append(self, m: EpochMetrics):
        self.rows.append(m)
        self.flush()

    def flush(self):
        """Rewrite the whole CSV (reference behavior: per-epoch rewrite,
        ``data_parallel_train.py:179-180``)."""
        with open(self.path, "w", newline="") as f:
            w = csv.writer(f)
            w.writerow(self.columns)
            for m in self.rows:
                row = [getattr(m, c) for c in self.columns]
                w.writerow(["" if v is None else v for v in row])


def sample_host_resources(proc: Optional[psutil.Process] = None):
    """psutil CPU% + RSS MB (reference ``data_parallel_train.py:105-106``)."""
    p = proc or psutil.Process()
    return p.cpu_percent(interval=None), p.memory_info().rss / (1024 * 1024)


def sample_gpu_resources(device: Optional[torch.device]):
    """GPU memory (MB) + utilization via torch/amdsmi (north-star §5.1)."""
    if device is None or device.type != "cuda":
        return None, None
    mem_mb = torch.cuda.memory_allocated(device) / (1024 * 1024)
    util = None
    try:
        import amdsmi  # noqa: PLC0415
        amdsmi.amdsmi_init()
        handles = amdsmi.amdsmi_get_processor_handles()
        h = handles[device.index]
        util = float(amdsmi.amdsmi_get_gpu_activity(h)["gfx_activity"])
    except Exception:  # noqa: BLE001 — amdsmi optional / may lack permission
        util = None
    return mem_mb, util


def write_legacy_row(logs_dir: str, rank: int, rows: List[dict]):
    """Legacy Docker-entry CSV (``train.py:115-116``)."""
    os.makedirs(logs_dir, exist_ok=True)
    path = os.path.join(logs_dir, f"training_logs_worker_{rank}.csv")
    with open(path, "w", newline="") as f:
        w = csv.DictWriter(f, fieldnames=["Worker", "Epoch", "Loss",
                                          "Accuracy", "Time"])
        w.writeheader()
        for r in rows:
            w.writerow(r)
    return path

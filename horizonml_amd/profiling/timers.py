"""Compute/comm/idle timing on hipEvents.

The reference times with ``time.time()`` around host-blocking CPU ops and
mislabels backward+step as "comm" in DP/TP (SURVEY.md Q4,
``data_parallel_train.py:111-124``).  Here the categories are honest:

* ``compute`` — forward + backward + optimizer kernels,
* ``comm``    — collectives / p2p (RCCL over xGMI),
* ``idle``    — barrier / peer-wait time,

measured with ``torch.cuda.Event`` (hipEventRecord/hipEventElapsedTime) pairs
on the issuing stream when on GPU, and ``time.perf_counter`` on CPU.  GPU
event pairs are resolved lazily at ``epoch_end()`` (one sync per epoch), so
timing never serializes the hot loop.

The CSV *columns* stay the reference's (§5.5); the semantic difference is
documented here and in README.
"""
from __future__ import annotations

import time
from contextlib import contextmanager
from typing import List, Optional, Tuple

import torch


class StepProfiler:
    CATEGORIES = ("compute", "comm", "idle")

    def __init__(self, device: Optional[torch.device] = None):
        self.use_cuda = device is not None and device.type == "cuda"
        self.device = device
        # set True when comm() sections are nested inside compute() (the
        # pipeline relay): epoch_end then reports compute − comm.
        self.subtract_comm_from_compute = False
        self.reset_epoch()
        self.bytes_sent = 0.0
        self.step_times: List[float] = []

    def reset_epoch(self):
        self._cpu_totals = {c: 0.0 for c in self.CATEGORIES}
        self._event_pairs: List[Tuple[str, torch.cuda.Event, torch.cuda.Event]] = []
        self.bytes_sent = 0.0
        self.step_times = []
        self._step_start: Optional[float] = None

    # -- step bracketing (host wall-clock, parity with avg_step_time) -----
    def step_begin(self):
        self._step_start = time.perf_counter()

    def step_end(self):
        if self._step_start is not None:
            self.step_times.append(time.perf_counter() - self._step_start)
            self._step_start = None

    # -- category timing --------------------------------------------------
    @contextmanager
    def track(self, category: str):
        if self.use_cuda:
            start = torch.cuda.Event(enable_timing=True)
            end = torch.cuda.Event(enable_timing=True)
            start.record()
            try:
                yield
            finally:
                end.record()
                self._event_pairs.append((category, start, end))
        else:
            t0 = time.perf_counter()
            try:
                yield
            finally:
                self._cpu_totals[category] += time.perf_counter() - t0

    def compute(self):
        return self.track("compute")

    def comm(self):
        return self.track("comm")

    def idle(self):
        return self.track("idle")

    def add_bytes(self, n: float):
        """Account bytes moved by a collective/p2p op (bandwidth column)."""
        self.bytes_sent += float(n)

    # -- epoch rollup ------------------------------------------------------
    def epoch_end(self) -> dict:
        totals = dict(self._cpu_totals)
        if self.use_cuda and self._event_pairs:
            torch.cuda.synchronize(self.device)
            for cat, s, e in self._event_pairs:
                totals[cat] += s.elapsed_time(e) / 1000.0  # ms -> s
        if self.subtract_comm_from_compute:
            totals["compute"] = max(0.0, totals["compute"] - totals["comm"])
        n_steps = max(1, len(self.step_times))
        out = {
            "compute_time": totals["compute"],
            "comm_time": totals["comm"],
            "idle_time": totals["idle"],
            "avg_step_time": sum(self.step_times) / n_steps,
            "bytes_sent": self.bytes_sent,
            "avg_bandwidth": (self.bytes_sent / totals["comm"]
                              if totals["comm"] > 0 else 0.0),
        }
        self.reset_epoch()
        return out

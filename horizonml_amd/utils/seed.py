"""Seed discipline.

The reference subsamples its dataset with an *unseeded* ``torch.randperm`` in
every spawned worker (reference ``data_parallel_train.py:60``,
``layer_model_parallel_train.py:120``, ``tensor_parallel_train.py:140``), so
each rank trains on a different random subset.  This module
fixes that by construction: every rank derives the same subset from the same
seed, and per-rank generator state is derived as ``base_seed + rank`` so
non-shared randomness (e.g. dropout) still differs across ranks.
"""
from __future__ import annotations

import random

import numpy as np
import torch

DEFAULT_SEED = 1234


def seed_everything(seed: int = DEFAULT_SEED, rank: int = 0) -> None:
    """Seed python / numpy / torch. Rank-dependent so per-rank streams differ."""
    s = int(seed) + int(rank)
    random.seed(s)
    np.random.seed(s % (2**32 - 1))
    torch.manual_seed(s)
    if torch.cuda.is_available():
        torch.cuda.manual_seed_all(s)
    import os
    if os.environ.get("HZ_DETERMINISTIC") == "1":
        enable_deterministic()


def shared_subset_indices(dataset_len: int, sample_size: int,
                          seed: int = DEFAULT_SEED) -> torch.Tensor:
    """Deterministic random subset, identical on every rank.

    Replaces the reference's per-rank unseeded
    ``torch.randperm(len(dataset))[:sample_size]``; required for the pipeline
    (images and labels must come from the same subset on every stage) and for
    tensor parallelism (all ranks must see identical batches).
    """
    g = torch.Generator().manual_seed(int(seed))
    n = min(int(sample_size), int(dataset_len))
    return torch.randperm(dataset_len, generator=g)[:n]


def enable_deterministic(warn_only: bool = True) -> None:
    """Deterministic mode (the rebuild's answer to the
    reference's absent seed discipline).

    Besides ``torch.use_deterministic_algorithms``, this switches the
    gfx950 kernels to fixed-order reductions (BN batch stats, backward
    channel sums, CE loss) in place of their atomic accumulations — GPU
    training becomes bitwise run-to-run reproducible.  Measured cost at
    batch 64: ~9x step time (the single-block serial reductions dominate
    at small batches) — a debugging/reproducibility tool, not a production
    mode.  Weight gradients are atomic-free in both modes.
    """
    import os
    os.environ.setdefault("CUBLAS_WORKSPACE_CONFIG", ":4096:8")
    torch.use_deterministic_algorithms(True, warn_only=warn_only)
    if torch.cuda.is_available():
        try:
            from .. import ops as _ops
            _ops.extension().set_deterministic(True)
        except Exception:  # extension absent on pure-CPU installs
            pass

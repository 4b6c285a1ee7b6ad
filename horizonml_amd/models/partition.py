"""Per-rank segment partitioner for the layer-wise pipeline.

Same contiguous floor/remainder distribution the reference's ``SplitResNet``
packs its 5 layer groups with (``layer_model_parallel_train.py:57-71``), but:

* generalized to any ordered unit list (``ResNet.layer_units``), so 8-stage
  pipelines (BASELINE.json config #3) partition at residual-block granularity
  instead of raising like the reference's ``num_workers > 5`` guard
  (``layer_model_parallel_train.py:54-55``);
* empty stages become ``nn.Identity()`` exactly like the reference.
"""
from __future__ import annotations

from typing import List, Sequence, Tuple

import torch.nn as nn


def split_counts(n_units: int, n_stages: int) -> List[int]:
    """Contiguous balanced split: floor division, remainder spread over the
    first ``n_units % n_stages`` stages."""
    if n_stages <= 0:
        raise ValueError("n_stages must be positive")
    base, rem = divmod(n_units, n_stages)
    return [base + (1 if i < rem else 0) for i in range(n_stages)]


def partition_units(units: Sequence[Tuple[str, nn.Module]],
                    n_stages: int) -> List[nn.Sequential]:
    """Pack ordered (name, module) units into ``n_stages`` nn.Sequential
    segments. A stage with no units gets ``nn.Identity()``."""
    counts = split_counts(len(units), n_stages)
    segments: List[nn.Sequential] = []
    it = iter(units)
    for c in counts:
        mods = [next(it)[1] for _ in range(c)]
        segments.append(nn.Sequential(*mods) if mods
                        else nn.Sequential(nn.Identity()))
    return segments


def partition_model(model, n_stages: int) -> List[nn.Sequential]:
    """Partition a model exposing ``layer_units(granularity)``.

    Uses the reference's 5 layer groups when ``n_stages <= 5`` and
    per-block granularity beyond that.
    """
    gran = "group" if n_stages <= 5 else "block"
    units = model.layer_units(gran)
    if n_stages > len(units):
        raise ValueError(
            f"n_stages={n_stages} exceeds available units ({len(units)}) "
            f"at granularity {gran!r}")
    return partition_units(units, n_stages)

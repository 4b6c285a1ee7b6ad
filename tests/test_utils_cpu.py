"""Unit tests for the small runtime/util layers (CPU-only)."""
import os

import pytest
import torch

from horizonml_amd.models.partition import partition_units, split_counts
from horizonml_amd.runtime.distributed import auto_backend
from horizonml_amd.runtime.launcher import timeout_for
from horizonml_amd.utils.ports import find_free_port
from horizonml_amd.utils.seed import shared_subset_indices


def test_split_counts_balanced_contiguous():
    assert split_counts(5, 5) == [1, 1, 1, 1, 1]
    assert split_counts(5, 3) == [2, 2, 1]      # remainder spread first
    assert split_counts(10, 8) == [2, 2, 1, 1, 1, 1, 1, 1]
    assert split_counts(3, 5) == [1, 1, 1, 0, 0]  # empty tail stages
    with pytest.raises(ValueError):
        split_counts(4, 0)


def test_partition_units_empty_stage_is_identity():
    import torch.nn as nn
    units = [(f"u{i}", nn.Linear(2, 2)) for i in range(3)]
    segs = partition_units(units, 5)
    assert len(segs) == 5
    out = torch.randn(1, 2)
    for s in segs[3:]:
        assert torch.equal(s(out), out)  # identity stages pass through


def test_timeout_for_reference_scaling():
    # reference: max(base, base * n / 1000) — data_parallel_train.py:252
    assert timeout_for(1000) == 120
    assert timeout_for(500) == 120
    assert timeout_for(5000) == 600
    assert timeout_for(1000, base=400) == 400


def test_find_free_port_binds():
    import socket
    p = find_free_port()
    s = socket.socket()
    s.bind(("127.0.0.1", p))  # must be bindable right after
    s.close()


def test_auto_backend_rules(monkeypatch, capsys):
    assert auto_backend("gloo") == "gloo"
    assert auto_backend("nccl") == "nccl"
    assert auto_backend("rccl") == "nccl"
    if not torch.cuda.is_available():
        assert auto_backend(None) == "gloo"
        assert auto_backend(None, world_size=8) == "gloo"
    # oversubscription is judged per NODE: a multi-node job (global world 16,
    # LOCAL_WORLD_SIZE 8 on an 8-GPU node) must stay on RCCL
    monkeypatch.setattr(torch.cuda, "is_available", lambda: True)
    monkeypatch.setattr(torch.cuda, "device_count", lambda: 8)
    monkeypatch.setenv("LOCAL_WORLD_SIZE", "8")
    assert auto_backend(None, world_size=16) == "nccl"
    # single-node spawn path: world_size IS the per-node rank count, and the
    # gloo/CPU fallback is announced, never silent
    monkeypatch.delenv("LOCAL_WORLD_SIZE", raising=False)
    capsys.readouterr()
    assert auto_backend(None, world_size=16) == "gloo"
    assert "falling back to gloo" in capsys.readouterr().out


def test_select_engine_rules():
    from horizonml_amd.engine.dp import select_engine
    from horizonml_amd.runtime.distributed import DistContext
    cpu = DistContext(0, 1, "gloo", None)
    assert select_engine("auto", cpu, "resnet18") == "eager"
    assert select_engine("eager", cpu, "resnet18") == "eager"
    with pytest.raises(RuntimeError):
        select_engine("flat", cpu, "resnet18")  # GPU-only path
    gpu = DistContext(0, 1, "nccl", torch.device("cuda", 0))
    assert select_engine("auto", gpu, "resnet18") == "flat"
    assert select_engine("auto", gpu, "mobilenet_v2") == "flat"
    assert select_engine("eager", gpu, "resnet18") == "eager"


def test_shared_subset_deterministic_and_shared():
    a = shared_subset_indices(50000, 1000, seed=7)
    b = shared_subset_indices(50000, 1000, seed=7)
    assert torch.equal(a, b)            # Q1 fix: every rank derives the same
    c = shared_subset_indices(50000, 1000, seed=8)
    assert not torch.equal(a, c)
    assert len(set(a.tolist())) == 1000  # no duplicates
    assert shared_subset_indices(10, 50).numel() == 10  # clamped


def test_checkpoint_atomic_tmp(tmp_path):
    """save_checkpoint writes tmp + rename — no partial file left behind."""
    from horizonml_amd.models import resnet18
    from horizonml_amd.utils.checkpoint import save_checkpoint
    m = resnet18(num_classes=10)
    path = str(tmp_path / "c.pt")
    save_checkpoint(path, m, epoch=1)
    assert os.path.isfile(path)
    assert not os.path.exists(path + ".tmp")


def test_split_counts_invariants_fuzz():
    from hypothesis import given, settings
    from hypothesis import strategies as st

    from horizonml_amd.models.partition import split_counts

    @settings(max_examples=200, deadline=None)
    @given(st.integers(min_value=0, max_value=512),
           st.integers(min_value=1, max_value=64))
    def check(n_units, n_stages):
        counts = split_counts(n_units, n_stages)
        assert sum(counts) == n_units
        assert len(counts) == n_stages
        # balanced: max-min <= 1, and the remainder loads the front
        assert max(counts) - min(counts) <= 1
        assert counts == sorted(counts, reverse=True)

    check()

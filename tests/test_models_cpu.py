"""CPU tests: models, partitioner, data layer, metrics schema."""
import os

import pytest
import torch

from horizonml_amd.data import SyntheticCIFAR10, build_dataset, get_dataloader
from horizonml_amd.models import (build_model, mobilenet_v2, partition_model,
                                  resnet18, resnet50, split_counts)
from horizonml_amd.profiling.metrics import (REFERENCE_COLUMNS, EpochMetrics,
                                             MetricsWriter)
from horizonml_amd.utils.seed import shared_subset_indices


def test_resnet18_shapes_and_params():
    m = resnet18(num_classes=10)
    y = m(torch.randn(2, 3, 32, 32))
    assert y.shape == (2, 10)
    # exact parity with torchvision resnet18(fc->10): SURVEY.md §2.4
    assert sum(p.numel() for p in m.parameters()) == 11_181_642


def test_resnet50_runs():
    m = resnet50(num_classes=10)
    y = m(torch.randn(2, 3, 32, 32))
    assert y.shape == (2, 10)


def test_mobilenet_v2():
    m = mobilenet_v2(num_classes=10)
    y = m(torch.randn(2, 3, 32, 32))
    assert y.shape == (2, 10)


def test_backward_flows():
    m = resnet18()
    loss = m(torch.randn(2, 3, 32, 32)).sum()
    loss.backward()
    assert all(p.grad is not None for p in m.parameters())


def test_split_counts():
    assert split_counts(5, 5) == [1, 1, 1, 1, 1]
    assert split_counts(5, 3) == [2, 2, 1]
    assert split_counts(5, 2) == [3, 2]
    assert split_counts(10, 8) == [2, 2, 1, 1, 1, 1, 1, 1]


@pytest.mark.parametrize("n_stages", [1, 2, 3, 5, 8])
def test_partition_preserves_forward(n_stages):
    torch.manual_seed(0)
    m = resnet18().eval()
    segs = partition_model(m, n_stages)
    assert len(segs) == n_stages
    x = torch.randn(2, 3, 32, 32)
    with torch.no_grad():
        ref = m(x)
        z = x
        for s in segs:
            z = s(z)
    assert torch.allclose(z, ref, atol=1e-5)


def test_shared_subset_identical_across_calls():
    a = shared_subset_indices(50000, 1000, seed=7)
    b = shared_subset_indices(50000, 1000, seed=7)
    assert torch.equal(a, b)
    assert len(set(a.tolist())) == 1000


def test_synthetic_dataset_deterministic():
    d1 = SyntheticCIFAR10(64, seed=3)
    d2 = SyntheticCIFAR10(64, seed=3)
    x1, y1 = d1[5]
    x2, y2 = d2[5]
    assert torch.equal(x1, x2) and y1 == y2
    assert x1.shape == (3, 32, 32)


def test_dp_dataloader_shards(tmp_path):
    l0, s0 = get_dataloader(0, 2, batch_size=16, sample_size=64,
                            strategy="dp", synthetic=True)
    l1, s1 = get_dataloader(1, 2, batch_size=16, sample_size=64,
                            strategy="dp", synthetic=True)
    assert len(l0.dataset) == 64
    assert s0 is not None and s1 is not None
    assert len(list(iter(s0))) == 32  # half each


def test_mp_dataloader_same_order():
    l0, _ = get_dataloader(0, 3, batch_size=16, sample_size=48,
                           strategy="mp", synthetic=True)
    l1, _ = get_dataloader(2, 3, batch_size=16, sample_size=48,
                           strategy="mp", synthetic=True)
    x0, y0 = next(iter(l0))
    x1, y1 = next(iter(l1))
    # Q1 fix: every rank must see identical data in identical order
    assert torch.equal(x0, x1) and torch.equal(y0, y1)


def test_metrics_schema(tmp_path):
    w = MetricsWriter(str(tmp_path), rank=3, sample_size=500,
                      with_bandwidth=True)
    w.append(EpochMetrics(1, 2.0, 10.0, 1.0, 0.1, 0.5, 0.2, 0.05, 50.0,
                          300.0, 0.7, avg_bandwidth=1e6))
    path = os.path.join(str(tmp_path), "worker_3_samples_500.csv")
    assert os.path.isfile(path)
    with open(path) as f:
        header = f.readline().strip().split(",")
    assert header == REFERENCE_COLUMNS + ["avg_bandwidth"]


def test_build_model_names():
    for name in ["resnet18", "resnet34", "resnet50", "mobilenet_v2"]:
        assert build_model(name, 10) is not None
    with pytest.raises(ValueError):
        build_model("nope")


def test_partition_resnet50_8_stages():
    """BASELINE config #3/#5 shapes: resnet50 split into 8 pipeline stages
    must preserve the forward exactly."""
    import torch
    from horizonml_amd.models import build_model, partition_model
    torch.manual_seed(0)
    model = build_model("resnet50", num_classes=10)
    model.eval()
    segs = partition_model(model, 8)
    assert len(segs) == 8
    x = torch.randn(2, 3, 32, 32)
    with torch.no_grad():
        ref = model(x)
        h = x
        for s in segs:
            h = s(h)
    assert torch.allclose(h, ref, atol=1e-5), (h - ref).abs().max()


def test_raw_loader_matches_normalized():
    """raw=True (GPU engines: uint8 batches + on-device normalize) must
    yield exactly the normalized loader's values."""
    import torch
    from horizonml_amd.data import get_dataloader
    from horizonml_amd.data.cifar import normalize_uint8
    a, _ = get_dataloader(0, 1, 16, 64, strategy="mp", synthetic=True)
    b, _ = get_dataloader(0, 1, 16, 64, strategy="mp", synthetic=True,
                          raw=True)
    for (xa, ya), (xb, yb) in zip(a, b):
        assert xb.dtype == torch.uint8
        assert torch.equal(ya, yb)
        assert torch.allclose(xa, normalize_uint8(xb), atol=1e-6)

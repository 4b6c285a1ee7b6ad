"""Checkpoint/resume round-trip (an addition over the reference, which has
no persistence — SURVEY.md §5.4)."""
import torch

from horizonml_amd.models import resnet18
from horizonml_amd.utils.checkpoint import load_checkpoint, save_checkpoint


def test_checkpoint_roundtrip(tmp_path):
    torch.manual_seed(0)
    m1 = resnet18(num_classes=10)
    opt = torch.optim.Adam(m1.parameters(), lr=1e-3)
    x = torch.randn(4, 3, 32, 32)
    y = torch.randint(0, 10, (4,))
    loss = torch.nn.functional.cross_entropy(m1(x), y)
    loss.backward()
    opt.step()
    path = str(tmp_path / "ckpt.pt")
    save_checkpoint(path, m1, opt, epoch=3, extra={"note": "t"})

    torch.manual_seed(123)
    m2 = resnet18(num_classes=10)
    opt2 = torch.optim.Adam(m2.parameters(), lr=1e-3)
    state = load_checkpoint(path, m2, opt2)
    assert state["epoch"] == 3
    assert state["extra"]["note"] == "t"
    for (n1, p1), (n2, p2) in zip(m1.named_parameters(),
                                  m2.named_parameters()):
        assert torch.equal(p1, p2), n1
    # optimizer moments restored
    s1 = opt.state_dict()["state"]
    s2 = opt2.state_dict()["state"]
    assert set(s1.keys()) == set(s2.keys())
    k = next(iter(s1))
    assert torch.equal(s1[k]["exp_avg"], s2[k]["exp_avg"])
    # training continues identically from the restored state
    l1 = torch.nn.functional.cross_entropy(m1(x), y)
    l2 = torch.nn.functional.cross_entropy(m2(x), y)
    assert torch.allclose(l1, l2)


def test_dp_entrypoint_checkpoint_resume(tmp_path):
    """--checkpoint on the DP entrypoint: a killed-and-relaunched run
    resumes at the next epoch instead of restarting (per-epoch save on
    rank 0, auto-resume when the file exists)."""
    import pandas as pd

    from data_parallel_train import run_data_parallel

    logs = str(tmp_path / "logs")
    ckpt = str(tmp_path / "dp.ckpt")
    run_data_parallel(world_size=1, epochs=2, sample_size=64,
                      logs_dir=logs, batch_size=32, backend="gloo",
                      synthetic=True, checkpoint_path=ckpt)
    df1 = pd.read_csv(f"{logs}/worker_0_samples_64.csv")
    assert list(df1["epoch"]) == [1, 2]

    # relaunch asking for 4 epochs total: must run ONLY epochs 3 and 4
    logs2 = str(tmp_path / "logs2")
    run_data_parallel(world_size=1, epochs=4, sample_size=64,
                      logs_dir=logs2, batch_size=32, backend="gloo",
                      synthetic=True, checkpoint_path=ckpt)
    df2 = pd.read_csv(f"{logs2}/worker_0_samples_64.csv")
    assert list(df2["epoch"]) == [3, 4], \
        f"resume did not skip completed epochs: {list(df2['epoch'])}"
    # training actually continued (loss keeps falling across the restart)
    assert df2["loss"].iloc[-1] < df1["loss"].iloc[0]


def test_pp_entrypoint_checkpoint_resume(tmp_path):
    """Pipeline checkpoint: per-RANK files (each stage owns distinct
    params) with MIN-epoch agreement; a relaunch resumes every stage at
    the same epoch."""
    import pandas as pd

    from layer_model_parallel_train import run_model_parallel

    logs = str(tmp_path / "logs")
    ckpt = str(tmp_path / "pp.ckpt")
    run_model_parallel(world_size=2, epochs=1, sample_size=32,
                       logs_dir=logs, batch_size=16, backend="gloo",
                       synthetic=True, checkpoint_path=ckpt)
    import os
    assert os.path.isfile(ckpt + ".rank0") and os.path.isfile(
        ckpt + ".rank1"), "per-rank checkpoint files missing"
    logs2 = str(tmp_path / "logs2")
    run_model_parallel(world_size=2, epochs=3, sample_size=32,
                       logs_dir=logs2, batch_size=16, backend="gloo",
                       synthetic=True, checkpoint_path=ckpt)
    df = pd.read_csv(f"{logs2}/worker_1_samples_32.csv")
    assert list(df["epoch"]) == [2, 3], \
        f"pipeline resume wrong epochs: {list(df['epoch'])}"


def test_fused_optimizer_state_roundtrip(tmp_path):
    """HorizonAdam/HorizonSGD moments survive save/load (they
    were silently dropped).  Uses a stand-in manager so the state-dict path
    is testable without the HIP extension."""
    import types

    from horizonml_amd.engine.flat import HorizonAdam, HorizonSGD

    mgr = types.SimpleNamespace(master=torch.arange(8, dtype=torch.float32))
    adam = HorizonAdam(mgr, lr=2e-3)
    with torch.no_grad():
        adam.m.copy_(torch.randn(8))
        adam.v.copy_(torch.rand(8))
        adam.step_t.fill_(17.0)
    model = torch.nn.Linear(2, 2)
    path = str(tmp_path / "fused.pt")
    save_checkpoint(path, model, adam, epoch=1)
    adam2 = HorizonAdam(mgr, lr=2e-3)
    state = load_checkpoint(path, model, adam2)
    assert "optimizer" in state
    assert torch.equal(adam2.m, adam.m)
    assert torch.equal(adam2.v, adam.v)
    assert float(adam2.step_t) == 17.0

    sgd = HorizonSGD(mgr, lr=0.1, momentum=0.9)
    with torch.no_grad():
        sgd.mom.copy_(torch.randn(8))
    save_checkpoint(path, model, sgd, epoch=2)
    sgd2 = HorizonSGD(mgr, lr=0.1, momentum=0.9)
    load_checkpoint(path, model, sgd2)
    assert torch.equal(sgd2.mom, sgd.mom)
    # kind mismatch fails loudly instead of silently skipping state
    import pytest
    with pytest.raises(ValueError):
        HorizonAdam(mgr).load_state_dict(sgd.state_dict())


def test_hybrid_entrypoint_checkpoint_resume(tmp_path):
    """Hybrid DPxPP checkpoint: per-rank files across both groups (each
    stage owns distinct params, each DP replica its own file) and the
    MIN-epoch agreement spans the WHOLE world, so a relaunch resumes all
    four ranks on the same epoch."""
    import os

    import pandas as pd

    from hybrid_parallel_train import run_hybrid_parallel

    logs = str(tmp_path / "logs")
    ckpt = str(tmp_path / "hy.ckpt")
    run_hybrid_parallel(dp_size=2, pp_size=2, epochs=1, sample_size=32,
                        logs_dir=logs, batch_size=16, model_name="resnet18",
                        backend="gloo", synthetic=True,
                        checkpoint_path=ckpt)
    for r in range(4):
        assert os.path.isfile(f"{ckpt}.rank{r}"), f"missing rank{r} file"
    logs2 = str(tmp_path / "logs2")
    run_hybrid_parallel(dp_size=2, pp_size=2, epochs=3, sample_size=32,
                        logs_dir=logs2, batch_size=16,
                        model_name="resnet18", backend="gloo",
                        synthetic=True, checkpoint_path=ckpt)
    for r in range(4):
        df = pd.read_csv(f"{logs2}/worker_{r}_samples_32.csv")
        assert list(df["epoch"]) == [2, 3], \
            f"rank {r} resumed wrong epochs: {list(df['epoch'])}"


def test_tp_entrypoint_checkpoint_resume(tmp_path):
    """Tensor-parallel checkpoint: per-rank files (shard params differ by
    rank); resume continues the epoch sequence on both ranks."""
    import os

    import pandas as pd

    from tensor_parallel_train import run_tensor_parallel

    logs = str(tmp_path / "logs")
    ckpt = str(tmp_path / "tp.ckpt")
    run_tensor_parallel(world_size=2, epochs=1, sample_size=32,
                        logs_dir=logs, batch_size=16, backend="gloo",
                        synthetic=True, checkpoint_path=ckpt)
    for r in range(2):
        assert os.path.isfile(f"{ckpt}.rank{r}"), f"missing rank{r} file"
    logs2 = str(tmp_path / "logs2")
    run_tensor_parallel(world_size=2, epochs=3, sample_size=32,
                        logs_dir=logs2, batch_size=16, backend="gloo",
                        synthetic=True, checkpoint_path=ckpt)
    for r in range(2):
        df = pd.read_csv(f"{logs2}/worker_{r}_samples_32.csv")
        assert list(df["epoch"]) == [2, 3], \
            f"rank {r} resumed wrong epochs: {list(df['epoch'])}"

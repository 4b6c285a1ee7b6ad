"""Pipeline + tensor-parallel correctness over 2-process CPU/gloo."""
import os

import torch
import torch.multiprocessing as mp
import torch.nn as nn

from horizonml_amd.runtime.distributed import (setup_distributed,
                                               teardown_distributed)
from horizonml_amd.utils.ports import find_free_port


def _spawn(fn, world, args=()):
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    port = find_free_port()
    procs = [ctx.Process(target=fn, args=(r, world, port, q, *args))
             for r in range(world)]
    for p in procs:
        p.start()
    out = {}
    for _ in range(world):
        r, v = q.get(timeout=180)
        out[r] = v
    for p in procs:
        p.join(timeout=60)
        if p.is_alive():
            p.terminate()
    return out


# --------------------------------------------------------------- pipeline --
def _pp_worker(rank, world, port, q):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    ctx = setup_distributed(rank, world, port, backend="gloo")
    from horizonml_amd.parallel.pipeline import PipelineStage
    torch.manual_seed(0)
    full = nn.Sequential(nn.Linear(6, 12), nn.Tanh(), nn.Linear(12, 8),
                         nn.Tanh(), nn.Linear(8, 4))
    seg = (nn.Sequential(full[0], full[1], full[2])
           if rank == 0 else nn.Sequential(full[3], full[4]))
    stage = PipelineStage(seg, rank, world)
    opt = torch.optim.SGD(seg.parameters(), lr=0.1)
    g = torch.Generator().manual_seed(7)
    X = torch.randn(4, 6, generator=g)
    Y = torch.randint(0, 4, (4,), generator=g)
    loss_fn = lambda logits, y: nn.functional.cross_entropy(logits, y)  # noqa
    for _ in range(3):
        opt.zero_grad()
        stage.forward_backward(X if rank == 0 else None,
                               Y if rank == world - 1 else None,
                               loss_fn=loss_fn, microbatches=2)
        opt.step()
    flat = torch.cat([p.detach().flatten() for p in seg.parameters()])
    q.put((rank, flat.tolist()))
    teardown_distributed(ctx)


def test_pipeline_matches_single_process():
    """2-stage pipeline with true backward relay ≡ single-process training
    (the Q2 fix: EVERY stage must receive exact gradients)."""
    out = _spawn(_pp_worker, 2)
    # single-process reference
    torch.manual_seed(0)
    full = nn.Sequential(nn.Linear(6, 12), nn.Tanh(), nn.Linear(12, 8),
                         nn.Tanh(), nn.Linear(8, 4))
    opt = torch.optim.SGD(full.parameters(), lr=0.1)
    g = torch.Generator().manual_seed(7)
    X = torch.randn(4, 6, generator=g)
    Y = torch.randint(0, 4, (4,), generator=g)
    for _ in range(3):
        opt.zero_grad()
        # microbatches=2, each loss is mean over its chunk and backprops
        # independently -> grads are SUMS of per-chunk mean losses
        for xc, yc in zip(X.chunk(2), Y.chunk(2)):
            nn.functional.cross_entropy(full(xc), yc).backward()
        opt.step()
    ref0 = torch.cat([p.detach().flatten()
                      for p in list(full.parameters())[:4]])
    ref1 = torch.cat([p.detach().flatten()
                      for p in list(full.parameters())[4:]])
    got0 = torch.tensor(out[0])
    got1 = torch.tensor(out[1])
    assert torch.allclose(got0, ref0, atol=1e-5), \
        f"stage0 max diff {(got0 - ref0).abs().max()}"
    assert torch.allclose(got1, ref1, atol=1e-5)


# ----------------------------------------------------------------- TP ------
def _gather_nhwc_worker(rank, world, port, q):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    ctx = setup_distributed(rank, world, port, backend="gloo")
    from horizonml_amd.parallel.tensor_parallel import (_GatherChannelsNHWC,
                                                        gather_from_parallel)
    g = torch.Generator().manual_seed(20 + rank)
    x1 = torch.randn(2, 3, 4, 5, generator=g).requires_grad_(True)
    x2 = x1.detach().clone().requires_grad_(True)
    y1 = _GatherChannelsNHWC.apply(x1, None)
    y2 = gather_from_parallel(x2, dim=1, group=None)
    gy = torch.randn(2, 6, 4, 5, generator=g)
    y1.backward(gy)
    y2.backward(gy)
    q.put((rank, (torch.allclose(y1, y2, atol=1e-6),
                  torch.allclose(x1.grad, x2.grad, atol=1e-6))))
    teardown_distributed(ctx)


def test_gather_channels_nhwc_matches_dim1_gather():
    """The layout-preserving NHWC channel gather (ShardedConvBNAct GPU
    path, VERDICT r01 weak-7 fix) must be numerically identical to the
    generic dim=1 gather, forward and backward."""
    out = _spawn(_gather_nhwc_worker, 2)
    for r in (0, 1):
        fwd_ok, bwd_ok = out[r]
        assert fwd_ok, f"rank {r} forward mismatch"
        assert bwd_ok, f"rank {r} backward mismatch"


def _tp_worker(rank, world, port, q):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    ctx = setup_distributed(rank, world, port, backend="gloo")
    from horizonml_amd.parallel.tensor_parallel import ColumnParallelLinear
    torch.manual_seed(0)
    full_w = torch.randn(4, 8)   # [out, in]
    full_b = torch.randn(4)
    lin = ColumnParallelLinear(8, 4, world, rank, bias=True)
    with torch.no_grad():
        lin.local.weight.copy_(full_w[rank * 2:(rank + 1) * 2])
        lin.local.bias.copy_(full_b[rank * 2:(rank + 1) * 2])
    g = torch.Generator().manual_seed(3)
    x = torch.randn(5, 8, generator=g).requires_grad_(True)
    y = lin(x)
    loss = (y * torch.arange(20, dtype=torch.float32).reshape(5, 4)).sum()
    loss.backward()
    q.put((rank, {
        "y": y.detach().tolist(),
        "dx": x.grad.tolist(),
        "dw": lin.local.weight.grad.tolist(),
        "db": lin.local.bias.grad.tolist(),
    }))
    teardown_distributed(ctx)


def test_column_parallel_linear_correct():
    out = _spawn(_tp_worker, 2)
    torch.manual_seed(0)
    full_w = torch.randn(4, 8)
    full_b = torch.randn(4)
    g = torch.Generator().manual_seed(3)
    x = torch.randn(5, 8, generator=g).requires_grad_(True)
    w = full_w.clone().requires_grad_(True)
    b = full_b.clone().requires_grad_(True)
    y = x @ w.T + b
    (y * torch.arange(20, dtype=torch.float32).reshape(5, 4)).sum().backward()
    for r in (0, 1):
        assert torch.allclose(torch.tensor(out[r]["y"]), y.detach(),
                              atol=1e-5)
        assert torch.allclose(torch.tensor(out[r]["dx"]), x.grad, atol=1e-5)
        assert torch.allclose(torch.tensor(out[r]["dw"]),
                              w.grad[r * 2:(r + 1) * 2], atol=1e-5)
        assert torch.allclose(torch.tensor(out[r]["db"]),
                              b.grad[r * 2:(r + 1) * 2], atol=1e-5)


def _rs_worker(rank, world, port, q):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    ctx = setup_distributed(rank, world, port, backend="gloo")
    from horizonml_amd.parallel.tensor_parallel import (
        reduce_scatter_to_parallel)
    g = torch.Generator().manual_seed(40 + rank)
    x = torch.randn(4, 8, generator=g).requires_grad_(True)
    y = reduce_scatter_to_parallel(x, dim=-1)   # rank's 4-wide shard of sum
    gy = torch.full((4, 4), float(rank + 1))
    y.backward(gy)
    q.put((rank, (y.detach().tolist(), x.grad.tolist())))
    teardown_distributed(ctx)


def test_reduce_scatter_autograd():
    """reduce-scatter forward = sum-then-shard; backward = all-gather of
    the shard grads (the north-star C5/C6 collective pairing)."""
    out = _spawn(_rs_worker, 2)
    xs = [torch.randn(4, 8, generator=torch.Generator().manual_seed(40 + r))
          for r in (0, 1)]
    total = xs[0] + xs[1]
    for r in (0, 1):
        got_y = torch.tensor(out[r][0])
        assert torch.allclose(got_y, total[:, r * 4:(r + 1) * 4], atol=1e-6)
        # backward: grad of x = concat of both ranks' shard grads (1s, 2s)
        got_gx = torch.tensor(out[r][1])
        expect = torch.cat([torch.full((4, 4), 1.0),
                            torch.full((4, 4), 2.0)], dim=1)
        assert torch.equal(got_gx, expect)


def _tp_resnet_worker(rank, world, port, q):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    ctx = setup_distributed(rank, world, port, backend="gloo")
    from horizonml_amd.parallel.tp_models import build_tp_resnet18
    torch.manual_seed(0)
    model = build_tp_resnet18(world, rank, mode="full")
    g = torch.Generator().manual_seed(5)
    x = torch.randn(2, 3, 32, 32, generator=g)
    y = model(x)
    loss = y.square().mean()
    loss.backward()
    has_shard_grads = all(
        p.grad is not None for p in model.parameters()
        if getattr(p, "tensor_parallel", False))
    q.put((rank, {"y_shape": list(y.shape), "loss": float(loss),
                  "shard_grads": has_shard_grads}))
    teardown_distributed(ctx)


def test_sharded_conv_resnet_runs():
    out = _spawn(_tp_resnet_worker, 2)
    for r in (0, 1):
        assert out[r]["y_shape"] == [2, 10]
        assert out[r]["shard_grads"]
    # identical input + gathered activations => identical loss on all ranks
    assert abs(out[0]["loss"] - out[1]["loss"]) < 1e-5


def _pp_odd_mb_worker(rank, world, port, q):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    ctx = setup_distributed(rank, world, port, backend="gloo")
    from horizonml_amd.parallel.pipeline import PipelineStage
    torch.manual_seed(0)
    full = nn.Sequential(nn.Linear(6, 12), nn.Tanh(), nn.Linear(12, 4))
    seg = (nn.Sequential(full[0], full[1]) if rank == 0
           else nn.Sequential(full[2]))
    stage = PipelineStage(seg, rank, world)
    g = torch.Generator().manual_seed(7)
    # batch 10 with 3 microbatches -> uneven chunks (4,3,3): the relay's
    # size-header protocol must carry the varying shapes
    X = torch.randn(10, 6, generator=g)
    Y = torch.randint(0, 4, (10,), generator=g)
    loss_fn = lambda logits, y: nn.functional.cross_entropy(logits, y)  # noqa
    total, n, _ = stage.forward_backward(X if rank == 0 else None,
                                         Y if rank == world - 1 else None,
                                         loss_fn=loss_fn, microbatches=3)
    grads_ok = all(p.grad is not None and torch.isfinite(p.grad).all()
                   for p in seg.parameters())
    q.put((rank, (float(total) if total is not None else -1.0,
                  n, bool(grads_ok))))
    teardown_distributed(ctx)


def test_pipeline_uneven_microbatches():
    out = _spawn(_pp_odd_mb_worker, 2)
    assert out[0][2] and out[1][2], "missing/non-finite grads"
    assert out[1][1] == 10, "sample count wrong across uneven microbatches"


def test_tp_full_ws4_end_to_end(tmp_path):
    """BASELINE config #4 shape: tensor-parallel world_size=4 with sharded
    conv2d (--tp_mode full) — all ranks train, identical data, identical
    loss trajectory across ranks (gathered activations)."""
    import pandas as pd

    from tensor_parallel_train import run_tensor_parallel
    df = run_tensor_parallel(world_size=4, epochs=1, sample_size=32,
                             logs_dir=str(tmp_path), batch_size=16,
                             backend="gloo", synthetic=True, tp_mode="full")
    assert df is not None
    assert sorted(df["worker"].unique().tolist()) == list(range(4))
    # identical inputs + gathered activations => same loss on every rank
    losses = df.groupby("worker")["loss"].last()
    assert losses.max() - losses.min() < 1e-4, losses.tolist()


def test_pipeline_8_stages_end_to_end(tmp_path):
    """BASELINE config #3 shape: ResNet18 split into 8 pipeline stages
    (block granularity — the reference's 5-group cap lifted), full
    backward relay, every stage training."""
    import pandas as pd

    from layer_model_parallel_train import run_model_parallel
    df = run_model_parallel(world_size=8, epochs=1, sample_size=32,
                            logs_dir=str(tmp_path), batch_size=16,
                            backend="gloo", synthetic=True)
    assert df is not None
    assert sorted(df["worker"].unique().tolist()) == list(range(8))
    last = df[df["worker"] == 7]
    assert (last["loss"] > 0).all()   # real loss on the last stage
    assert (last["accuracy"] >= 0).all()


def _pp_negotiated_worker(rank, world, port, q, header_mode):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    if header_mode:
        os.environ["HZ_PP_HEADER"] = "1"
    else:
        os.environ.pop("HZ_PP_HEADER", None)
    ctx = setup_distributed(rank, world, port, backend="gloo")
    from horizonml_amd.parallel.pipeline import PipelineStage
    torch.manual_seed(0)
    full = nn.Sequential(nn.Linear(6, 12), nn.Tanh(), nn.Linear(12, 8),
                         nn.Tanh(), nn.Linear(8, 4))
    seg = (nn.Sequential(full[0], full[1], full[2])
           if rank == 0 else nn.Sequential(full[3], full[4]))
    stage = PipelineStage(seg, rank, world)
    opt = torch.optim.SGD(seg.parameters(), lr=0.1)
    g = torch.Generator().manual_seed(11)
    X = torch.randn(21, 6, generator=g)   # steps of batch 8, 8, 5 (ragged)
    Y = torch.randint(0, 4, (21,), generator=g)
    loss_fn = lambda logits, y: nn.functional.cross_entropy(logits, y)  # noqa
    for lo, hi in ((0, 8), (8, 16), (16, 21)):
        opt.zero_grad()
        stage.forward_backward(X[lo:hi] if rank == 0 else None,
                               Y[lo:hi] if rank == world - 1 else None,
                               loss_fn=loss_fn, microbatches=2,
                               batch_hint=hi - lo)
    flat = torch.cat([p.detach().flatten() for p in seg.parameters()])
    gr = torch.cat([p.grad.flatten() for p in seg.parameters()])
    q.put((rank, (flat.tolist(), gr.tolist())))
    teardown_distributed(ctx)


def test_pipeline_static_shape_negotiation_matches_header_mode():
    """Steps of varying batch (8, 8, 5 with microbatches=2) through the
    negotiated relay: the header-skip decisions must agree across ranks for
    EVERY (peer, chunk-size), and the result must be identical to the
    reference per-hop header protocol (HZ_PP_HEADER=1 compat flag).
    VERDICT r01 item 5."""
    fast = _spawn(_pp_negotiated_worker, 2, args=(False,))
    hdr = _spawn(_pp_negotiated_worker, 2, args=(True,))
    for r in (0, 1):
        assert torch.equal(torch.tensor(fast[r][0]),
                           torch.tensor(hdr[r][0])), f"params differ r{r}"
        assert torch.equal(torch.tensor(fast[r][1]),
                           torch.tensor(hdr[r][1])), f"grads differ r{r}"

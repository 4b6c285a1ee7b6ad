#!/usr/bin/env python3
"""Flagship benchmark — the driver contract.

Measures the BASELINE.json headline: ResNet18 / CIFAR-10-shaped synthetic
data, data-parallel training, images/sec + epoch time, on N MI355X GPUs
(weak scaling: per-GPU batch 64 like the reference's per-worker batch,
``data_parallel_train.py:196``).

Fast path: bf16 channels_last activations through the gfx950 kernels, flat
f32 master / bf16 shadow parameters, fused Adam, single bf16 RCCL all-reduce
of the flat gradient, the whole training step captured in a hipGraph
(launch-bound workload: ~130 kernels/step at batch 64).

`vs_baseline`: the reference's best DP number is ≈19.6 s per 1000-sample
epoch on CPU/gloo = 51.0 images/sec; `vs_baseline` =
our aggregate images/sec ÷ 51.0.

Usage: python bench.py --gpus N --steps K --warmup W
(N>1 is launched by the driver via torch.distributed.run, one rank per GPU;
standalone multi-GPU invocation re-execs torchrun itself.)
"""
from __future__ import annotations

import argparse
import os
import sys
import time


def parse_args():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    # long enough (~0.3 s timed region at 1 ms/step) that driver-side GPU
    # sampling and clock cross-checks see a busy device
    ap.add_argument("--steps", type=int, default=300)
    ap.add_argument("--warmup", type=int, default=20)
    ap.add_argument("--batch-size", type=int, default=64,
                    help="per-GPU batch (reference parity: 64)")
    ap.add_argument("--image-size", type=int, default=32,
                    help="input H=W (224 for the ImageNet-shaped configs)")
    ap.add_argument("--model", type=str, default="resnet18")
    ap.add_argument("--optimizer", type=str, default="adam")
    ap.add_argument("--no-graph", action="store_true",
                    help="disable hipGraph capture (eager fallback)")
    ap.add_argument("--infer", action="store_true",
                    help="serving mode: eval-only forward throughput")
    return ap.parse_args()


def maybe_reexec_torchrun(args):
    if args.gpus > 1 and "WORLD_SIZE" not in os.environ:
        import subprocess
        cmd = [sys.executable, "-m", "torch.distributed.run",
               "--nnodes=1", f"--nproc-per-node={args.gpus}",
               "--master-addr", "127.0.0.1", "--master-port", "29517",
               os.path.abspath(__file__)] + sys.argv[1:]
        os.environ.setdefault("HSA_ENABLE_IPC_MODE_LEGACY", "0")
        raise SystemExit(subprocess.call(cmd))


def main():
    args = parse_args()
    maybe_reexec_torchrun(args)

    import torch
    import torch.distributed as dist

    rank = int(os.environ.get("RANK", 0))
    world = int(os.environ.get("WORLD_SIZE", 1))
    local_rank = int(os.environ.get("LOCAL_RANK", 0))
    assert torch.cuda.is_available(), "bench.py needs an MI355X"
    torch.set_num_threads(8)  # tiny host-side work; avoid pool wake-storms
    torch.cuda.set_device(local_rank)
    dev = torch.device("cuda", local_rank)
    # HZ_FORCE_COMM=1: exercise the full RCCL comm path (all-reduce inside
    # the captured graph) on a single-rank communicator — 1-GPU validation
    # of the multi-GPU code path
    force_comm = os.environ.get("HZ_FORCE_COMM") == "1" and world == 1
    if world > 1 or force_comm:
        dist.init_process_group("nccl", rank=rank, world_size=world)
    use_comm = world > 1 or force_comm

    from horizonml_amd import ops as _ops
    from horizonml_amd.engine.flat import (FlatParamManager, HorizonAdam,
                                           HorizonSGD)
    from horizonml_amd.models import build_model
    from horizonml_amd.models._functional_gpu import cross_entropy
    _ops.extension()  # no silent fallback

    torch.manual_seed(1234)  # identical replicas on every rank
    model = build_model(args.model, num_classes=10).to(dev)
    if args.infer:
        model.eval()
    else:
        model.train()
    mgr = FlatParamManager(model, dev)
    opt = (HorizonAdam(mgr, lr=1e-3) if args.optimizer == "adam"
           else HorizonSGD(mgr, lr=0.1))

    bs = args.batch_size
    torch.manual_seed(1234 + rank)  # different data per rank (DP semantics)
    pool_n = 8
    pool_x = [torch.randn(bs, 3, args.image_size, args.image_size,
                          device=dev)
              .to(memory_format=torch.channels_last).to(torch.bfloat16)
              for _ in range(pool_n)]
    pool_y = [torch.randint(0, 10, (bs,), device=dev) for _ in range(pool_n)]
    comm_buf = (torch.zeros_like(mgr.grad, dtype=torch.bfloat16)
                if use_comm else None)
    inv_world = 1.0 / world

    # Overlapped bucketed all-reduce (default at world>1): the flat grad is
    # reduced in reverse-layer buckets launched DURING backward (per-unit
    # callbacks), each bucket's deferred wgrads flushed just before its
    # pack, so xGMI communication overlaps the remaining backward + the
    # next bucket's wgrad GEMMs.  HZ_BUCKETS=1 keeps the single-buffer
    # post-backward fallback.
    n_buckets = int(os.environ.get("HZ_BUCKETS", "4")) if use_comm else 1
    sched = None
    if use_comm and n_buckets > 1 and not args.infer:
        from horizonml_amd.parallel.flat_reducer import (
            BackwardBucketScheduler, FlatBucketReducer, build_bucket_schedule)
        ranges, bucket_mods = build_bucket_schedule(model, mgr.slices,
                                                    n_buckets)
        reducer = FlatBucketReducer(
            mgr.grad, ranges, comm_buf=comm_buf,
            flush_range_fn=lambda lo, hi:
                _ops.extension().flush_wgrad_range(mgr.grad, lo, hi))
        sched = BackwardBucketScheduler(reducer, bucket_mods)

    def train_step(xb, yb):
        if args.infer:  # serving: eval-mode forward only
            with torch.no_grad():
                logits = model(xb)
            return logits.float().sum()
        if sched is not None:
            sched.begin_step()
        logits = model(xb)
        loss = cross_entropy(logits, yb)
        loss.backward()
        if use_comm:
            if sched is not None:
                # buckets launched from backward callbacks; fence them
                sched.reducer.reduce_all()  # safety net: fire stragglers
                sched.reducer.wait()
            else:
                _ops.extension().flush_wgrad()  # wgrads complete before sync
                comm_buf.copy_(mgr.grad)        # pack f32 -> bf16
                dist.all_reduce(comm_buf)       # RCCL over xGMI
            # fused optimizer consumes the reduced bf16 buffer directly
            opt.step(grad_bf16=comm_buf, grad_scale=inv_world)
        else:
            opt.step()                        # fused adam + zero_grad + rsck
        return loss

    # ---- warmup + graph capture -----------------------------------------
    # hipGraph capture recipe (torch "whole-network capture"): warm up on a
    # SIDE stream so AccumulateGrad nodes are not bound to the default
    # stream, drop every reference to the warmup autograd graph, then
    # capture one graph PER POOL ENTRY (each binds its own input tensors —
    # no per-step staging copies at replay time).
    mode = "eager" if args.no_graph else "graph"
    side = torch.cuda.Stream()
    side.wait_stream(torch.cuda.current_stream())
    with torch.cuda.stream(side):
        for i in range(3):
            loss = train_step(pool_x[i % pool_n], pool_y[i % pool_n])
    torch.cuda.current_stream().wait_stream(side)
    torch.cuda.synchronize()
    del loss
    losses_static = []

    graphs = None
    if mode == "graph":
        try:
            graphs = []
            for i in range(pool_n):
                g = torch.cuda.CUDAGraph()
                with torch.cuda.graph(g):
                    losses_static.append(train_step(pool_x[i], pool_y[i]))
                graphs.append(g)
        except Exception as e:  # noqa: BLE001
            if rank == 0:
                print(f"[bench] graph capture failed ({e!r}); eager fallback",
                      file=sys.stderr)
            graphs = None
            mode = "eager"
        if world > 1:
            # all ranks must agree on the execution mode BEFORE any replay:
            # a rank-divergent capture failure would otherwise deadlock the
            # captured collectives (replay on one side only)
            from horizonml_amd.runtime.bench_protocol import agree_all_ranks
            if not agree_all_ranks(graphs is not None, world, dev):
                graphs = None
                mode = "eager"
        if graphs is not None:
            graphs[0].replay()
            torch.cuda.synchronize()

    def run_step(i):
        if graphs is not None:
            graphs[i % pool_n].replay()
        else:
            train_step(pool_x[i % pool_n], pool_y[i % pool_n])

    for i in range(args.warmup):
        run_step(i)

    if world > 1:
        dist.barrier(device_ids=[local_rank])
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for i in range(args.steps):
        run_step(i)
    torch.cuda.synchronize()
    if world > 1:
        dist.barrier(device_ids=[local_rank])
    t1 = time.perf_counter()

    from horizonml_amd.runtime.bench_protocol import (build_record,
                                                      emit_record,
                                                      max_elapsed_over_ranks)
    elapsed_s = max_elapsed_over_ranks(t1 - t0, world, dev)
    if graphs is not None:
        final_loss = float(losses_static[-1].detach().float().cpu())
    else:
        final_loss = float(train_step(pool_x[0], pool_y[0])
                           .detach().float().cpu())

    if rank == 0:
        out = build_record(
            elapsed_s=elapsed_s, steps=args.steps, warmup=args.warmup,
            world=world, batch_size=bs, model=args.model,
            optimizer=args.optimizer, exec_mode=mode,
            final_loss=final_loss, infer=args.infer,
            image_size=args.image_size,
            comm_mode=(f"bucketed{n_buckets}" if sched is not None
                       else ("flat1" if use_comm else "none")))
        print(emit_record(out), flush=True)
    if world > 1 or force_comm:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()

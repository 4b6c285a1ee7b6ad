#!/usr/bin/env python3
"""Data-parallel training entrypoint (strategy 1).

CLI/CSV parity with the reference's ``data_parallel_train.py`` (flags
``--world_size --epochs --sample_size``, per-worker CSVs + combined CSV in
``data_parallel_logs/``), rebuilt on the MI355X-native engine:
one process per GPU, bf16 bucketed RCCL all-reduce over xGMI, gfx950 HIP
kernels for the ResNet hot path.  On CPU-only hosts it runs the gloo
plumbing configuration (BASELINE.json config #1).
"""
from __future__ import annotations

import argparse

from horizonml_amd.engine.dp import dp_worker
from horizonml_amd.runtime.launcher import run_workers


def run_data_parallel(world_size: int, epochs: int, sample_size: int,
                      logs_dir: str = "data_parallel_logs",
                      batch_size: int = 64, model_name: str = "resnet18",
                      backend=None, synthetic=None, lr: float = 1e-3,
                      optimizer_name: str = "adam", engine: str = "auto",
                      checkpoint_path=None, per_step_barrier: bool = False):
    """Launcher parity with reference ``run_data_parallel``
    (``data_parallel_train.py:233-291``). Returns the combined DataFrame."""
    return run_workers(dp_worker, world_size, epochs, sample_size, logs_dir,
                       timeout_base=120,
                       extra_args=(batch_size, model_name, backend, synthetic,
                                   lr, optimizer_name, engine,
                                   checkpoint_path, per_step_barrier))


def main():
    ap = argparse.ArgumentParser(description="Data-parallel training")
    ap.add_argument("--world_size", type=int, default=5)
    ap.add_argument("--epochs", type=int, default=5)
    ap.add_argument("--sample_size", type=int, default=1000)
    ap.add_argument("--logs_dir", type=str, default="data_parallel_logs")
    ap.add_argument("--batch_size", type=int, default=64)
    ap.add_argument("--model", type=str, default="resnet18")
    ap.add_argument("--backend", type=str, default=None,
                    choices=[None, "nccl", "gloo"], nargs="?")
    ap.add_argument("--synthetic", action="store_true", default=None,
                    help="force synthetic CIFAR-shaped data")
    ap.add_argument("--lr", type=float, default=1e-3)
    ap.add_argument("--deterministic", action="store_true",
                    help="torch.use_deterministic_algorithms (GPU kernels "
                         "remain reproducible only to bf16/atomic rounding)")
    ap.add_argument("--optimizer", type=str, default="adam",
                    choices=["adam", "sgd"])
    ap.add_argument("--engine", type=str, default="auto",
                    choices=["auto", "eager", "flat"],
                    help="flat = hipGraph/fused fast path (GPU default); "
                         "eager = bucketed-DDP torch-optimizer loop")
    ap.add_argument("--checkpoint", type=str, default=None,
                    help="checkpoint file: saved per epoch (rank 0), "
                         "resumed from when it exists")
    ap.add_argument("--per_step_barrier", action="store_true",
                    help="restore the reference's full-world barrier after "
                         "every step (exact idle_time semantics; the "
                         "default barriers per epoch)")
    args = ap.parse_args()
    if args.deterministic:
        import os
        os.environ["HZ_DETERMINISTIC"] = "1"
    df = run_data_parallel(args.world_size, args.epochs, args.sample_size,
                           args.logs_dir, args.batch_size, args.model,
                           args.backend, args.synthetic, args.lr,
                           args.optimizer, args.engine, args.checkpoint,
                           args.per_step_barrier)
    if df is not None:
        print(df.tail(args.world_size).to_string(index=False))


if __name__ == "__main__":
    main()

#!/usr/bin/env python3
"""Hybrid DP×PP training entrypoint (BASELINE.json config #5).

The reference frames HorizonML as a "hybrid framework" (README/deck) but
ships no combined strategy; this entrypoint provides it
MI355X-native: ``world_size = dp_size × pp_size`` processes, one per GPU.
Rank layout ``rank = dp_rank * pp_size + pp_stage`` — each pipeline chain
occupies contiguous ranks so adjacent stages relay activations/gradients
over their direct xGMI link (RCCL send/recv), while each stage's parameter
replicas sync with a bucketed bf16 all-reduce over the per-stage DP
sub-communicator.

Default configuration is the north-star hybrid: ResNet50 on ImageNet-shaped
synthetic data, 2×4 on 8 GPUs (``--dp_size 2 --pp_size 4 --model resnet50
--image_size 224 --num_classes 1000``).  CSV/log layout matches the other
strategies (per-worker CSVs + combined_results_{n}.csv in
``hybrid_parallel_logs/``).
"""
from __future__ import annotations

import argparse

from horizonml_amd.engine.pp import hybrid_worker
from horizonml_amd.runtime.launcher import run_workers


def run_hybrid_parallel(dp_size: int, pp_size: int, epochs: int,
                        sample_size: int,
                        logs_dir: str = "hybrid_parallel_logs",
                        batch_size: int = 64, model_name: str = "resnet50",
                        backend=None, synthetic=None, lr: float = 1e-3,
                        optimizer_name: str = "adam", microbatches: int = 1,
                        num_classes: int = 10, image_size: int = 32,
                        checkpoint_path=None):
    world_size = dp_size * pp_size
    return run_workers(hybrid_worker, world_size, epochs, sample_size,
                       logs_dir, timeout_base=240,
                       extra_args=(batch_size, model_name, backend, synthetic,
                                   lr, optimizer_name, microbatches,
                                   dp_size, pp_size, num_classes, image_size,
                                   checkpoint_path))


def main():
    ap = argparse.ArgumentParser(description="Hybrid DP×PP training")
    ap.add_argument("--dp_size", type=int, default=2)
    ap.add_argument("--pp_size", type=int, default=4)
    ap.add_argument("--epochs", type=int, default=5)
    ap.add_argument("--sample_size", type=int, default=1000)
    ap.add_argument("--logs_dir", type=str, default="hybrid_parallel_logs")
    ap.add_argument("--batch_size", type=int, default=64)
    ap.add_argument("--model", type=str, default="resnet50")
    ap.add_argument("--backend", type=str, default=None, nargs="?",
                    choices=[None, "nccl", "gloo"])
    ap.add_argument("--synthetic", action="store_true", default=None)
    ap.add_argument("--lr", type=float, default=1e-3)
    ap.add_argument("--deterministic", action="store_true",
                    help="torch.use_deterministic_algorithms (GPU kernels "
                         "remain reproducible only to bf16/atomic rounding)")
    ap.add_argument("--optimizer", type=str, default="adam",
                    choices=["adam", "sgd"])
    ap.add_argument("--microbatches", type=int, default=1)
    ap.add_argument("--num_classes", type=int, default=10)
    ap.add_argument("--image_size", type=int, default=32,
                    help="224 for the ImageNet-shaped hybrid config")
    ap.add_argument("--checkpoint", type=str, default=None,
                    help="checkpoint base path (per-rank files)")
    args = ap.parse_args()
    if args.deterministic:
        import os
        os.environ["HZ_DETERMINISTIC"] = "1"
    df = run_hybrid_parallel(args.dp_size, args.pp_size, args.epochs,
                             args.sample_size, args.logs_dir,
                             args.batch_size, args.model, args.backend,
                             args.synthetic, args.lr, args.optimizer,
                             args.microbatches, args.num_classes,
                             args.image_size, args.checkpoint)
    if df is not None:
        print(df.tail(args.dp_size * args.pp_size).to_string(index=False))


if __name__ == "__main__":
    main()

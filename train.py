#!/usr/bin/env python3
"""Legacy env-var DDP entrypoint (Docker "edge node" simulation parity).

Reference ``train.py``: rendezvous from ``RANK/WORLD_SIZE/MASTER_ADDR/
MASTER_PORT`` env vars, ``MODEL_TYPE`` selects resnet18 or mobilenet_v2,
batch size 2, per-epoch CSV ``training_logs_worker_{rank}.csv`` with
``Worker,Epoch,Loss,Accuracy,Time`` (``train.py:15-126``).  On a GPU host
this binds one MI355X per rank over RCCL instead of gloo.
"""
from __future__ import annotations

import os
import time

import torch
import torch.nn.functional as F

from horizonml_amd.data import get_dataloader
from horizonml_amd.engine.common import build_optimizer
from horizonml_amd.models import build_model
from horizonml_amd.parallel import BucketedDataParallel
from horizonml_amd.profiling.metrics import write_legacy_row
from horizonml_amd.runtime.distributed import (barrier, setup_from_env,
                                               teardown_distributed)
from horizonml_amd.utils.seed import seed_everything


def main():
    epochs = int(os.environ.get("EPOCHS", 5))
    sample_size = int(os.environ.get("SAMPLE_SIZE", 1000))
    batch_size = int(os.environ.get("BATCH_SIZE", 2))
    model_type = os.environ.get("MODEL_TYPE", "resnet")
    logs_dir = os.environ.get("LOGS_DIR", ".")

    ctx = setup_from_env()
    seed_everything(rank=ctx.rank)
    try:
        loader, sampler = get_dataloader(ctx.rank, ctx.world_size,
                                         batch_size, sample_size,
                                         strategy="dp", raw=ctx.is_gpu,
                                         synthetic=os.environ.get(
                                             "SYNTHETIC") == "1" or None)
        name = "resnet18" if model_type.startswith("resnet") else "mobilenet_v2"
        model = build_model(name, num_classes=10)
        if ctx.is_gpu:
            model = model.to(ctx.device)
        ddp = BucketedDataParallel(model)
        opt = build_optimizer(model.parameters(), "adam", lr=1e-3)

        rows = []
        for epoch in range(epochs):
            if sampler is not None:
                sampler.set_epoch(epoch)
            start = time.time()
            loss_sum, correct, count = 0.0, 0, 0
            for x, y in loader:
                if ctx.is_gpu:
                    x = x.to(ctx.device)
                    if x.dtype == torch.uint8:
                        from horizonml_amd.data.cifar import normalize_uint8
                        x = normalize_uint8(x)
                    x = x.to(
                        memory_format=torch.channels_last).to(torch.bfloat16)
                    y = y.to(ctx.device)
                opt.zero_grad(set_to_none=True)
                logits = ddp(x)
                if logits.is_cuda:
                    from horizonml_amd.models._functional_gpu import \
                        cross_entropy
                    loss = cross_entropy(logits, y)
                else:
                    loss = F.cross_entropy(logits.float(), y)
                loss.backward()
                ddp.finalize_backward()
                opt.step()
                if ctx.is_gpu:
                    from horizonml_amd.models import refresh_all_shadows
                    refresh_all_shadows(model)
                bs = y.shape[0]
                loss_sum += float(loss.detach()) * bs
                correct += int((logits.detach().argmax(1) == y).sum())
                count += bs
            elapsed = time.time() - start
            rows.append({"Worker": ctx.rank, "Epoch": epoch + 1,
                         "Loss": loss_sum / max(1, count),
                         "Accuracy": 100.0 * correct / max(1, count),
                         "Time": elapsed})
            write_legacy_row(logs_dir, ctx.rank, rows)
            print(f"[worker {ctx.rank}] epoch {epoch + 1}/{epochs} "
                  f"loss={rows[-1]['Loss']:.4f} "
                  f"acc={rows[-1]['Accuracy']:.2f}% time={elapsed:.1f}s",
                  flush=True)
            barrier(ctx)
    finally:
        teardown_distributed(ctx)


if __name__ == "__main__":
    main()

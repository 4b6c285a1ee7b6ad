"""CIFAR-10 data layer.

The reference downloads CIFAR-10 via torchvision with a rank-0 gate
(``data_parallel_train.py:51-55``) and subsamples it with an unseeded
per-rank randperm.  Here:

* default is **synthetic CIFAR-shaped data** (BASELINE.json requires it —
  there is no network on the target boxes): uint8 images generated once from
  a fixed seed, normalized on the fly with the reference's (0.5,0.5,0.5)
  mean/std (``data_parallel_train.py:44-47``);
* if a real CIFAR-10 python-pickle directory exists (``cifar-10-batches-py``)
  it is used instead — no download is ever attempted;
* the random subset is derived from a *shared* seed so every rank sees the
  same subset (Q1 fix), and DP shards it with ``DistributedSampler``.
"""
from __future__ import annotations

import os
import pickle
from typing import Optional, Tuple

import numpy as np
import torch
from torch.utils.data import DataLoader, Dataset, Subset
from torch.utils.data.distributed import DistributedSampler

from ..utils.seed import DEFAULT_SEED, shared_subset_indices

_MEAN = 0.5
_STD = 0.5


class SyntheticCIFAR10(Dataset):
    """Deterministic random CIFAR-shaped dataset (uint8 in memory).

    ``image_size=224`` gives the ImageNet-shaped variant used by the hybrid
    DP×PP ResNet50 configuration (BASELINE.json config #5)."""

    def __init__(self, n: int = 50000, num_classes: int = 10,
                 seed: int = DEFAULT_SEED, image_size: int = 32):
        g = torch.Generator().manual_seed(seed)
        self.images = torch.randint(0, 256, (n, 3, image_size, image_size),
                                    dtype=torch.uint8, generator=g)
        self.labels = torch.randint(0, num_classes, (n,),
                                    dtype=torch.long, generator=g)

    def __len__(self):
        return self.images.shape[0]

    def __getitem__(self, i: int):
        x = self.images[i].to(torch.float32).div_(255.0).sub_(_MEAN).div_(_STD)
        return x, self.labels[i]

    def __getitems__(self, idxs):
        # batched fetch (DataLoader uses this when present): ONE vectorized
        # normalize per batch instead of 4 tiny CPU ops per sample — the
        # per-sample path made the eager engines dataloader-bound
        idx = torch.as_tensor(idxs)
        x = self.images[idx].to(torch.float32).div_(255.0) \
            .sub_(_MEAN).div_(_STD)
        ys = self.labels[idx]
        return list(zip(x.unbind(0), ys.unbind(0)))


class CIFAR10Local(Dataset):
    """Real CIFAR-10 from a local ``cifar-10-batches-py`` directory."""

    def __init__(self, root: str, train: bool = True):
        base = root
        if os.path.isdir(os.path.join(root, "cifar-10-batches-py")):
            base = os.path.join(root, "cifar-10-batches-py")
        files = ([f"data_batch_{i}" for i in range(1, 6)] if train
                 else ["test_batch"])
        imgs, labels = [], []
        for f in files:
            path = os.path.join(base, f)
            with open(path, "rb") as fh:
                d = pickle.load(fh, encoding="bytes")
            imgs.append(np.asarray(d[b"data"], dtype=np.uint8))
            labels.extend(d[b"labels"])
        data = np.concatenate(imgs).reshape(-1, 3, 32, 32)
        self.images = torch.from_numpy(data)
        self.labels = torch.tensor(labels, dtype=torch.long)

    def __len__(self):
        return self.images.shape[0]

    def __getitem__(self, i: int):
        x = self.images[i].to(torch.float32).div_(255.0).sub_(_MEAN).div_(_STD)
        return x, self.labels[i]

    def __getitems__(self, idxs):
        idx = torch.as_tensor(idxs)
        x = self.images[idx].to(torch.float32).div_(255.0) \
            .sub_(_MEAN).div_(_STD)
        ys = self.labels[idx]
        return list(zip(x.unbind(0), ys.unbind(0)))


def build_dataset(data_dir: str = "./data", synthetic: Optional[bool] = None,
                  n: int = 50000, seed: int = DEFAULT_SEED,
                  image_size: int = 32, num_classes: int = 10) -> Dataset:
    """Real CIFAR-10 if present under ``data_dir`` (unless ``synthetic=True``),
    else synthetic."""
    if synthetic is not True and image_size == 32 and num_classes == 10:
        for cand in (data_dir, os.path.join(data_dir, "cifar-10-batches-py")):
            if os.path.isfile(os.path.join(cand, "data_batch_1")):
                return CIFAR10Local(data_dir, train=True)
        if synthetic is False:
            raise FileNotFoundError(
                f"no CIFAR-10 batches under {data_dir!r} and synthetic=False")
    return SyntheticCIFAR10(n=n, seed=seed, image_size=image_size,
                            num_classes=num_classes)


class RawView(Dataset):
    """uint8 view of a dataset: normalization deferred to the consumer
    (GPU engines normalize on-device — 4x fewer host-copy bytes and no CPU
    elementwise work in the hot loop)."""

    def __init__(self, base):
        self.base = base

    def __len__(self):
        return len(self.base)

    def __getitem__(self, i: int):
        return self.base.images[i], self.base.labels[i]

    def __getitems__(self, idxs):
        idx = torch.as_tensor(idxs)
        return list(zip(self.base.images[idx].unbind(0),
                        self.base.labels[idx].unbind(0)))


def normalize_uint8(x: torch.Tensor) -> torch.Tensor:
    """The reference transform ((0.5,0.5,0.5) mean/std) for raw uint8
    batches, applied on whatever device x lives on.

    GPU: ONE fused kernel producing the bf16 channels_last layout the
    gfx950 conv kernels consume (the eager to(f32)/div/sub/div + permute +
    cast chain costs ~6 dispatches per step); callers' subsequent
    ``.to(channels_last).to(bf16)`` become no-ops.  CPU keeps f32."""
    if x.is_cuda and x.dim() == 4 and x.dtype == torch.uint8:
        from .. import ops as _ops
        return _ops.extension().normalize_u8(x.contiguous(), _MEAN, _STD)
    return x.to(torch.float32).div_(255.0).sub_(_MEAN).div_(_STD)


def get_dataloader(rank: int, world_size: int, batch_size: int = 64,
                   sample_size: int = 1000, strategy: str = "dp",
                   data_dir: str = "./data", synthetic: Optional[bool] = None,
                   seed: int = DEFAULT_SEED, drop_last: bool = False,
                   image_size: int = 32, num_classes: int = 10,
                   raw: bool = False
                   ) -> Tuple[DataLoader, Optional[DistributedSampler]]:
    """Reference-parity dataloader.

    * ``dp``: shared random subset + DistributedSampler shard per rank
      (``data_parallel_train.py:59-71``).
    * ``mp``/``tp``: every rank iterates the *same* full subset in the same
      order (the reference intends this but breaks it — Q1; fixed here),
      ``shuffle=False`` like ``layer_model_parallel_train.py:103-131``.
    * ``hybrid``: DP×PP — ``rank``/``world_size`` here are the *DP replica*
      coordinates (dp_rank, dp_size); the shard is deterministic
      (``shuffle=False``) so every pipeline stage of one DP chain iterates
      identical batches without extra synchronization.
    """
    ds = build_dataset(data_dir, synthetic,
                       n=max(50000, sample_size), seed=seed,
                       image_size=image_size, num_classes=num_classes)
    idx = shared_subset_indices(len(ds), sample_size, seed=seed)
    if raw:
        ds = RawView(ds)
    subset = Subset(ds, idx.tolist())
    if strategy in ("dp", "hybrid"):
        sampler = DistributedSampler(subset, num_replicas=world_size,
                                     rank=rank,
                                     shuffle=(strategy == "dp"), seed=seed,
                                     drop_last=drop_last)
        loader = DataLoader(subset, batch_size=batch_size, sampler=sampler,
                            num_workers=0, pin_memory=torch.cuda.is_available(),
                            drop_last=drop_last)
        return loader, sampler
    loader = DataLoader(subset, batch_size=batch_size, shuffle=False,
                        num_workers=0, pin_memory=torch.cuda.is_available(),
                        drop_last=drop_last)
    return loader, None

"""Tensor-parallel model builders.

Two modes, matching SURVEY.md §2.3:

* ``fc``     — reference-parity topology: replicated ResNet backbone +
               column-parallel final fc (the reference's
               ``TensorParallelResNet``, ``tensor_parallel_train.py:67-105``),
               but autograd-correct (Q3 fix).
* ``full``   — every conv out-channel-sharded (``ShardedConvBNAct``) +
               column-parallel fc — BASELINE.json config #4
               ("sharded conv2d + all-gather").
"""
from __future__ import annotations

import torch.nn as nn

from ..models.layers import ConvBNAct
from ..models.resnet import ResNet, resnet18
from .tensor_parallel import ColumnParallelLinear, ShardedConvBNAct


def _shard_convs(module: nn.Module, world_size: int, rank: int, group):
    for name, child in module.named_children():
        if isinstance(child, ConvBNAct):
            if child.out_ch % world_size != 0:
                continue  # leave non-divisible convs replicated
            new = ShardedConvBNAct(child.in_ch, child.out_ch,
                                   child.kernel_size, child.stride,
                                   child.padding, child.act,
                                   world_size=world_size, rank=rank,
                                   group=group)
            setattr(module, name, new)
        else:
            _shard_convs(child, world_size, rank, group)


def build_tp_resnet18(world_size: int, rank: int, group=None,
                      num_classes: int = 10, mode: str = "fc") -> ResNet:
    model = resnet18(num_classes=num_classes)
    if mode == "full":
        _shard_convs(model, world_size, rank, group)
    elif mode != "fc":
        raise ValueError(f"unknown tp mode {mode!r}")
    # column-parallel classifier; out_features 10 is not divisible by
    # typical world sizes, so pad shards like the reference's
    # out_features_per_worker floor-division would lose classes — instead we
    # gather unpadded shards when divisible, else keep the fc replicated on
    # non-divisible worlds and shard only when 10 % ws == 0.
    fc = model.tail.fc
    if fc.out_features % world_size == 0:
        model.tail.fc = ColumnParallelLinear(
            fc.in_features, fc.out_features, world_size, rank, bias=True,
            group=group)
    else:
        # shard along in-features instead (row-parallel keeps all 10 logits)
        from .tensor_parallel import RowParallelLinear
        if fc.in_features % world_size == 0:
            model.tail.fc = RowParallelLinear(
                fc.in_features, fc.out_features, world_size, rank, bias=True,
                group=group, input_is_parallel=False)
    return model

"""Layer-wise pipeline engine over blocking p2p send/recv (xGMI on GPU).

Reference parity + correction:

* the reference relays activations rank→rank with blocking ``dist.send`` /
  ``dist.recv`` and a [4]-int64 size header
  (``layer_model_parallel_train.py:182-227``) but **never sends gradients
  upstream** — only the last rank trains.  This engine keeps
  the blocking-relay structure and the size-header protocol, and adds the
  reverse relay: each batch runs forward rank0→rankN-1, then the
  activation-gradient flows rankN-1→rank0 and **every stage steps its own
  optimizer** (the north-star requirement: "blocking send/recv for the
  layer-wise pipeline activations/gradients").
* On GPU the p2p ops are RCCL ``ncclSend/ncclRecv`` over the direct xGMI
  link between adjacent ranks; activations travel bf16.

``microbatches > 1`` splits each batch into chunks relayed back-to-back
(fill-drain schedule — 1F1B is unnecessary at ResNet scale) to cut the
serial-pipeline bubble the reference suffers from.
"""
from __future__ import annotations

from typing import List, Optional, Tuple

import torch
import torch.distributed as dist

_HDR_LEN = 8  # up to 8 dims; reference used 4 (conv activations only)


class PipelineStage:
    def __init__(self, segment: torch.nn.Module, stage: int, n_stages: int,
                 device: Optional[torch.device] = None, group=None,
                 profiler=None):
        self.seg = segment
        self.stage = stage
        self.n_stages = n_stages
        self.device = device or torch.device("cpu")
        self.group = group
        self.profiler = profiler
        self.is_first = stage == 0
        self.is_last = stage == n_stages - 1
        self._dtype = (torch.bfloat16 if self.device.type == "cuda"
                       else torch.float32)

    # global rank of a pipeline-stage neighbor (identity without a group)
    def _rank_of(self, stage: int) -> int:
        if self.group is None:
            return stage
        return dist.get_global_rank(self.group, stage)

    # -- p2p helpers (size-header protocol, reference C4) -----------------
    def _track_comm(self):
        import contextlib
        if self.profiler is None:
            return contextlib.nullcontext()
        return self.profiler.comm()

    def _send(self, t: torch.Tensor, dst_stage: int):
        t = t.contiguous()
        hdr = torch.full((_HDR_LEN,), -1, dtype=torch.int64,
                         device=self.device)
        hdr[:t.dim()] = torch.tensor(t.shape, dtype=torch.int64,
                                     device=self.device)
        dst = self._rank_of(dst_stage)
        # header+payload are ordered on the same (src,dst) pair — no tags
        # (RCCL p2p does not support them)
        with self._track_comm():
            dist.send(hdr, dst=dst, group=self.group)
            dist.send(t, dst=dst, group=self.group)
        if self.profiler is not None:
            self.profiler.add_bytes(t.numel() * t.element_size())

    def _recv(self, src_stage: int) -> torch.Tensor:
        src = self._rank_of(src_stage)
        hdr = torch.empty(_HDR_LEN, dtype=torch.int64, device=self.device)
        with self._track_comm():
            dist.recv(hdr, src=src, group=self.group)
            shape = [int(d) for d in hdr.cpu().tolist() if d >= 0]
            t = torch.empty(*shape, dtype=self._dtype, device=self.device)
            dist.recv(t, src=src, group=self.group)
        return t

    # -- one training step -------------------------------------------------
    def forward_backward(self, x: Optional[torch.Tensor],
                         labels: Optional[torch.Tensor],
                         loss_fn=None, microbatches: int = 1
                         ) -> Tuple[Optional[torch.Tensor], int]:
        """Run one batch through this stage (both directions).

        First stage passes ``x``; last stage passes ``labels`` + ``loss_fn``
        and gets (total_loss, n_samples) back; others pass nothing.
        """
        chunks_in: List[torch.Tensor] = []
        chunks_out: List[torch.Tensor] = []
        losses: List[torch.Tensor] = []
        n = 0

        if self.is_first:
            xs = x.chunk(microbatches) if microbatches > 1 else [x]
        else:
            xs = [None] * microbatches

        # forward relay
        for mb in range(microbatches):
            if self.is_first:
                inp = xs[mb]
                inp_l = inp
            else:
                inp = self._recv(self.stage - 1)
                inp.requires_grad_(True)
                inp_l = inp
            out = self.seg(inp_l)
            if not self.is_last:
                self._send(out.detach(), self.stage + 1)
                chunks_in.append(inp)
                chunks_out.append(out)
            else:
                chunks_in.append(inp)
                chunks_out.append(out)

        # backward relay (reverse order keeps peer matching simple)
        if self.is_last:
            if microbatches > 1:
                ys = labels.chunk(microbatches)
            else:
                ys = [labels]
        total_loss = None
        correct = 0
        for mb in reversed(range(microbatches)):
            if self.is_last:
                out = chunks_out[mb]
                loss = loss_fn(out, ys[mb])
                losses.append(loss.detach() * ys[mb].shape[0])
                n += ys[mb].shape[0]
                correct += int((out.detach().argmax(1) == ys[mb]).sum())
                loss.backward()
            else:
                gout = self._recv(self.stage + 1)
                chunks_out[mb].backward(gout)
            if not self.is_first:
                g = chunks_in[mb].grad
                self._send(g.to(self._dtype), self.stage - 1)
        if self.is_last and losses:
            total_loss = torch.stack(losses).sum()
        return total_loss, n, correct

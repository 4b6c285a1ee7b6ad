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

**Static-shape negotiation** (VERDICT r01 item 5): the header protocol
costs a host sync per hop per microbatch on GPU (``hdr.cpu()`` parses the
size before the payload recv can be posted).  Shapes are static per
(peer, microbatch-size) in this workload, so each side negotiates a shape
ONCE via the header exchange and caches it — subsequent messages are
payload-only, posted without touching the host.  Both sides derive the
same cache-key sequence (chunk sizes come from the shared loader), so the
skip decisions agree by construction.  ``HZ_PP_HEADER=1`` restores the
reference's per-hop header protocol (C4 parity/debugging), as does calling
``forward_backward`` without a ``batch_hint`` on intermediate stages.
"""
from __future__ import annotations

import os
from typing import List, Optional, Tuple

import torch
import torch.distributed as dist

_HDR_LEN = 8  # up to 8 dims; reference used 4 (conv activations only)


def chunk_sizes(n: int, m: int) -> List[int]:
    """Sizes ``torch.chunk(x, m)`` produces for a length-``n`` batch (may
    be fewer than ``m`` chunks)."""
    if n <= 0:
        return []
    per = -(-n // m)  # ceil
    full, rem = divmod(n, per)
    return [per] * full + ([rem] if rem else [])


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
        # static-shape negotiation caches: (peer_stage, batch) -> shape
        self._hdr_always = os.environ.get("HZ_PP_HEADER") == "1"
        self._sent_shapes = {}
        self._recv_shapes = {}

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

    def _send(self, t: torch.Tensor, dst_stage: int,
              negotiated: bool = False):
        """``negotiated``: the receiver knows this (peer, batch) shape is
        cacheable — send the header only on the first occurrence."""
        t = t.contiguous()
        key = (dst_stage, t.shape[0])
        skip_hdr = (negotiated and not self._hdr_always
                    and self._sent_shapes.get(key) == tuple(t.shape))
        dst = self._rank_of(dst_stage)
        with self._track_comm():
            if not skip_hdr:
                # header+payload are ordered on the same (src,dst) pair —
                # no tags (RCCL p2p does not support them)
                hdr = torch.full((_HDR_LEN,), -1, dtype=torch.int64,
                                 device=self.device)
                hdr[:t.dim()] = torch.tensor(t.shape, dtype=torch.int64,
                                             device=self.device)
                dist.send(hdr, dst=dst, group=self.group)
                if negotiated:
                    self._sent_shapes[key] = tuple(t.shape)
            dist.send(t, dst=dst, group=self.group)
        if self.profiler is not None:
            self.profiler.add_bytes(t.numel() * t.element_size())

    def _recv(self, src_stage: int,
              expect_batch: Optional[int] = None) -> torch.Tensor:
        """``expect_batch``: leading dim of the incoming message (known from
        the shared loader).  On a cache hit the payload recv is posted
        directly — no header, no ``hdr.cpu()`` host sync on the hot path."""
        src = self._rank_of(src_stage)
        key = (src_stage, expect_batch)
        cached = (None if (expect_batch is None or self._hdr_always)
                  else self._recv_shapes.get(key))
        with self._track_comm():
            if cached is not None:
                t = torch.empty(*cached, dtype=self._dtype,
                                device=self.device)
                dist.recv(t, src=src, group=self.group)
                return t
            hdr = torch.empty(_HDR_LEN, dtype=torch.int64,
                              device=self.device)
            dist.recv(hdr, src=src, group=self.group)
            shape = [int(d) for d in hdr.cpu().tolist() if d >= 0]
            if expect_batch is not None:
                if shape[0] != expect_batch:
                    raise RuntimeError(
                        f"pipeline shape negotiation mismatch: expected "
                        f"batch {expect_batch} from stage {src_stage}, got "
                        f"{shape}")
                self._recv_shapes[key] = tuple(shape)
            t = torch.empty(*shape, dtype=self._dtype, device=self.device)
            dist.recv(t, src=src, group=self.group)
        return t

    # -- one training step -------------------------------------------------
    def forward_backward(self, x: Optional[torch.Tensor],
                         labels: Optional[torch.Tensor],
                         loss_fn=None, microbatches: int = 1,
                         batch_hint: Optional[int] = None
                         ) -> Tuple[Optional[torch.Tensor], int]:
        """Run one batch through this stage (both directions).

        First stage passes ``x``; last stage passes ``labels`` + ``loss_fn``
        and gets (total_loss, n_samples) back; others pass nothing.

        ``batch_hint`` (the step's full batch size — known on every rank
        from the shared loader) enables static-shape negotiation: headers
        are exchanged once per (peer, microbatch-size) and skipped after.
        Must be passed on ALL ranks or NONE (the skip decisions must agree).
        """
        if x is not None:
            batch_hint = x.shape[0]
        elif labels is not None and batch_hint is None:
            batch_hint = labels.shape[0]
        negotiated = batch_hint is not None
        if negotiated:
            sizes = chunk_sizes(batch_hint, microbatches)
            m_eff = len(sizes)
        else:
            sizes = [None] * microbatches
            m_eff = microbatches

        chunks_in: List[torch.Tensor] = []
        chunks_out: List[torch.Tensor] = []
        losses: List[torch.Tensor] = []
        n = 0

        if self.is_first:
            xs = list(x.chunk(microbatches)) if microbatches > 1 else [x]
        else:
            xs = [None] * m_eff

        # forward relay
        for mb in range(m_eff):
            if self.is_first:
                inp = xs[mb]
            else:
                inp = self._recv(self.stage - 1, expect_batch=sizes[mb])
                inp.requires_grad_(True)
            out = self.seg(inp)
            if not self.is_last:
                self._send(out.detach(), self.stage + 1,
                           negotiated=negotiated)
            chunks_in.append(inp)
            chunks_out.append(out)

        # backward relay (reverse order keeps peer matching simple)
        if self.is_last:
            ys = (list(labels.chunk(microbatches)) if microbatches > 1
                  else [labels])
        total_loss = None
        correct = 0
        for mb in reversed(range(m_eff)):
            if self.is_last:
                out = chunks_out[mb]
                loss = loss_fn(out, ys[mb])
                losses.append(loss.detach() * ys[mb].shape[0])
                n += ys[mb].shape[0]
                correct += int((out.detach().argmax(1) == ys[mb]).sum())
                loss.backward()
            else:
                gout = self._recv(self.stage + 1, expect_batch=sizes[mb])
                chunks_out[mb].backward(gout)
            if not self.is_first:
                g = chunks_in[mb].grad
                self._send(g.to(self._dtype), self.stage - 1,
                           negotiated=negotiated)
        if self.is_last and losses:
            total_loss = torch.stack(losses).sum()
        return total_loss, n, correct

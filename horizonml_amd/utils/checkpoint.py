"""Checkpoint / resume.

The reference has no persistence at all (SURVEY.md §5.4 — every run is from
random init); this subsystem is an addition, sized for MI355X training runs:
one file per rank holds the model, optimizer, RNG and progress state, and
the flat fast path (FlatParamManager) round-trips its f32 master buffer so
bf16 shadows are rebuilt exactly.
"""
from __future__ import annotations

import os
from typing import Any, Dict, Optional

import torch


def save_checkpoint(path: str, model: torch.nn.Module,
                    optimizer: Optional[Any] = None, epoch: int = 0,
                    mgr: Optional[Any] = None,
                    extra: Optional[Dict[str, Any]] = None) -> str:
    """Atomic save (tmp + rename): a killed run never corrupts the file."""
    state: Dict[str, Any] = {
        "model": {k: v.cpu() for k, v in model.state_dict().items()},
        "epoch": int(epoch),
        "torch_rng": torch.get_rng_state(),
    }
    if torch.cuda.is_available():
        state["cuda_rng"] = torch.cuda.get_rng_state_all()
    if optimizer is not None and hasattr(optimizer, "state_dict"):
        state["optimizer"] = optimizer.state_dict()
    if mgr is not None:  # FlatParamManager: master is the source of truth
        state["flat_master"] = mgr.master.detach().cpu()
    if extra:
        state["extra"] = extra
    os.makedirs(os.path.dirname(os.path.abspath(path)), exist_ok=True)
    tmp = path + ".tmp"
    torch.save(state, tmp)
    os.replace(tmp, path)
    return path


def load_checkpoint(path: str, model: torch.nn.Module,
                    optimizer: Optional[Any] = None,
                    mgr: Optional[Any] = None,
                    restore_rng: bool = True) -> Dict[str, Any]:
    """Returns the raw state dict; ``epoch`` is the last COMPLETED epoch."""
    state = torch.load(path, map_location="cpu", weights_only=False)
    model.load_state_dict(state["model"])
    if optimizer is not None and "optimizer" in state:
        optimizer.load_state_dict(state["optimizer"])
    if mgr is not None and "flat_master" in state:
        with torch.no_grad():
            mgr.master.copy_(state["flat_master"].to(mgr.master.device))
            mgr.shadow.copy_(mgr.master)
        mgr.refresh_rsck()
    if restore_rng:
        torch.set_rng_state(state["torch_rng"])
        if torch.cuda.is_available() and "cuda_rng" in state:
            try:
                torch.cuda.set_rng_state_all(state["cuda_rng"])
            except RuntimeError:
                pass  # different device count than at save time
    return state

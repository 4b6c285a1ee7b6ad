"""horizonml_amd.ops — CDNA4 HIP kernel surface.

The hot ops of the ResNet hot path (SURVEY.md §2.4 K1–K11) are implemented as
hand-written gfx950 HIP kernels in ``csrc/`` and exposed through an in-tree
torch extension ``horizonml_amd/ops/_C*.so`` (built by ``setup.py build_ext
--inplace`` / ``__graft_entry__.build()``).

Dispatch policy:
* CPU tensors → plain PyTorch compositions (used by the CPU/gloo plumbing
  configs and as the numerics reference for the kernels).
* CUDA (= HIP/ROCm) tensors → the extension, **with no silent fallback**: if
  the extension is missing on a GPU box the op raises, per the project rule
  that GPU tests must exercise the native path.
"""
from __future__ import annotations

import importlib

_EXT = None
_EXT_ERR: Exception | None = None


def _load_extension():
    global _EXT, _EXT_ERR
    if _EXT is not None:
        return _EXT
    try:
        _EXT = importlib.import_module("horizonml_amd.ops._C")
    except Exception as e:  # noqa: BLE001
        _EXT_ERR = e
        raise RuntimeError(
            "horizonml_amd HIP extension (horizonml_amd/ops/_C) is not built "
            "or failed to import. Build it in-tree with "
            "`python setup.py build_ext --inplace` (PYTORCH_ROCM_ARCH=gfx950). "
            f"Original error: {e!r}"
        ) from e
    return _EXT


def extension():
    """Return the loaded HIP extension module, raising loudly if absent."""
    return _load_extension()


def has_extension() -> bool:
    try:
        _load_extension()
        return True
    except RuntimeError:
        return False

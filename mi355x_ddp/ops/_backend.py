"""HIP extension loader.

The extension is built IN-TREE (csrc/ -> mi355x_ddp/_C*.so) by
``__graft_entry__.build()`` or ``python csrc/build.py``; a JIT cache is never
used so the .so travels with the repo snapshot to GPU boxes.

Policy (required by the build contract):
- On a machine WITH a GPU, calling a native op when the extension is missing
  raises — no silent eager fallback on the hardware path.
- On CPU-only machines (CI here), ops fall back to reference torch
  implementations so the full test suite runs without a GPU.
- ``MI355X_FORCE_FALLBACK=1`` forces the torch path everywhere (parity A/B).
"""
from __future__ import annotations

import os

import torch

_C = None
_import_error: Exception | None = None
try:  # built in-tree as mi355x_ddp/_C.cpython-*.so
    from mi355x_ddp import _C as _C  # type: ignore
except Exception as e:  # pragma: no cover - exercised only when ext missing
    _import_error = e


def extension_available() -> bool:
    return _C is not None


def force_fallback() -> bool:
    return os.environ.get("MI355X_FORCE_FALLBACK", "0") == "1"


def native_enabled(tensor: torch.Tensor | None = None) -> bool:
    """True when the HIP kernel path should run for this tensor."""
    if force_fallback():
        return False
    on_gpu = tensor.is_cuda if tensor is not None else torch.cuda.is_available()
    if not on_gpu:
        return False
    if _C is None:
        raise RuntimeError(
            "mi355x_ddp native extension (_C) is not built but a GPU op was "
            "requested on a CUDA/HIP device. Build it in-tree with "
            "`python csrc/build.py` (or __graft_entry__.build()); refusing to "
            f"fall back silently. Import error was: {_import_error!r}")
    return True


def C():
    if _C is None:
        raise RuntimeError(
            f"mi355x_ddp._C extension not available: {_import_error!r}")
    return _C

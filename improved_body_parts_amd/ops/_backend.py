"""HIP extension loader + dispatch policy.

The extension ``_ibp_hip`` is built IN-TREE (``python -m improved_body_parts_amd.ops.build``
or ``__graft_entry__.build()``) so the ``.so`` ships with the repo snapshot to a GPU
box. On a CUDA/ROCm device the extension is mandatory: a silent eager fallback
would make GPU tests pass without the native path (the failure mode the build
rules warn about), so ``use_hip_for`` raises when CUDA input arrives and the
extension is absent, unless IBP_AMD_ALLOW_EAGER=1.
"""
from __future__ import annotations

import importlib
import os

import torch

_EXT = None
_EXT_ERR: Exception | None = None
_TRIED = False


def _try_load():
    global _EXT, _EXT_ERR, _TRIED
    if _TRIED:
        return
    _TRIED = True
    try:
        _EXT = importlib.import_module("improved_body_parts_amd.ops._ibp_hip")
    except Exception as e:  # pragma: no cover - exercised only without the built .so
        _EXT = None
        _EXT_ERR = e


def hip_available() -> bool:
    _try_load()
    return _EXT is not None and torch.cuda.is_available()


def hip_extension():
    """Return the loaded extension module, raising with a build hint if absent."""
    _try_load()
    if _EXT is None:
        raise RuntimeError(
            "HIP extension _ibp_hip is not built. Run "
            "`python -m improved_body_parts_amd.ops.build` (gfx950) first."
        ) from _EXT_ERR
    return _EXT


def require_hip():
    if not torch.cuda.is_available():
        raise RuntimeError("require_hip() called without a visible GPU")
    return hip_extension()


def use_hip_for(x: torch.Tensor) -> bool:
    """Dispatch decision for a tensor: HIP kernels for CUDA tensors, eager for CPU."""
    if not x.is_cuda:
        return False
    _try_load()
    if _EXT is None:
        if os.environ.get("IBP_AMD_ALLOW_EAGER") == "1":
            return False
        raise RuntimeError(
            "Input is on GPU but the _ibp_hip extension is not built; the HIP "
            "path is mandatory on MI355X (set IBP_AMD_ALLOW_EAGER=1 to debug "
            "with eager PyTorch). Build with "
            "`python -m improved_body_parts_amd.ops.build`."
        ) from _EXT_ERR
    return True

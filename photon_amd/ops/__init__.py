"""photon_amd.ops — CDNA4 HIP kernels with PyTorch reference fallbacks.

The compiled extension ``_photon_hip`` lives IN-TREE (photon_amd/ops/) so it
travels with the repo snapshot to GPU boxes. On a CUDA/ROCm device the HIP
path is mandatory: if the extension is missing, GPU ops raise loudly instead
of silently falling back to eager PyTorch (set PHOTON_ALLOW_FALLBACK=1 to
override for debugging). On CPU the PyTorch reference implementation runs —
that is the reference's ``attn_impl: torch`` / DeviceCPU plumbing path
(photon/clients/trainer_utils.py:1243-1247).
"""

from __future__ import annotations

import os

import torch

_EXT = None
_EXT_ERR: str | None = None


def _try_load() -> None:
    global _EXT, _EXT_ERR
    if _EXT is not None or _EXT_ERR is not None:
        return
    try:
        import importlib

        _EXT = importlib.import_module("photon_amd.ops._photon_hip")
    except ImportError as e:
        _EXT_ERR = str(e)


def hip_ext():
    """Return the HIP extension module or None (CPU-only environments)."""
    _try_load()
    return _EXT


def require_hip_ext():
    """Return the extension; raise if running on GPU without it."""
    _try_load()
    if _EXT is None and not allow_fallback():
        raise RuntimeError(
            "photon_amd HIP extension (_photon_hip) is not built but a GPU op "
            f"was requested (import error: {_EXT_ERR}). Build it with "
            "`python setup.py build_ext --inplace` (PYTORCH_ROCM_ARCH=gfx950) "
            "or set PHOTON_ALLOW_FALLBACK=1 to use the slow eager path."
        )
    return _EXT


def allow_fallback() -> bool:
    return os.environ.get("PHOTON_ALLOW_FALLBACK", "0") == "1"


def use_hip(t: torch.Tensor) -> bool:
    """True if this tensor should go through the HIP kernel path."""
    if not t.is_cuda:
        return False
    _try_load()
    if _EXT is not None:
        return True
    if allow_fallback():
        return False
    # GPU tensor, no extension, no explicit fallback permission: fail loudly.
    require_hip_ext()
    return False  # unreachable

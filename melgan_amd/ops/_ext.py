"""HIP extension loader for melgan_amd.

The gfx950 kernels live in ``melgan_amd/kernels/*.hip`` and are built
IN-TREE (the resulting .so sits next to this file's package) by
``__graft_entry__.build()`` or ``python setup.py build_ext --inplace``.

Policy (see the round contract): on a machine WITH a GPU the HIP path is
the one that runs — if the extension is missing there we raise loudly
instead of silently falling back to eager PyTorch.  On CPU-only machines
(the CI container) the PyTorch composite path is used and the extension
is optional.
"""

from __future__ import annotations

import importlib
import os

import torch

_EXT = None
_TRIED = False


def _find_ext():
    """Import the in-tree extension module ``melgan_amd._C``."""
    try:
        return importlib.import_module("melgan_amd._C")
    except ImportError:
        return None


def ext():
    """Return the extension module or None (CPU-only fallback allowed)."""
    global _EXT, _TRIED
    if not _TRIED:
        _EXT = _find_ext()
        _TRIED = True
    return _EXT


def have_ext() -> bool:
    return ext() is not None


def require_ext():
    """On a GPU box the native kernels must be present — fail loudly."""
    e = ext()
    if e is None:
        raise RuntimeError(
            "melgan_amd._C (gfx950 HIP kernels) is not built but a GPU is "
            "visible. Build in-tree with `python -m melgan_amd.kernels.build` "
            "or __graft_entry__.build(). Refusing to silently fall back to "
            "eager PyTorch on GPU."
        )
    return e


def use_hip(x: torch.Tensor) -> bool:
    """True when ``x`` lives on a ROCm GPU and the HIP path should run."""
    if not x.is_cuda:
        return False
    if os.environ.get("MELGAN_AMD_DISABLE_HIP", "0") == "1":
        return False
    return True

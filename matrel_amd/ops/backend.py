"""Backend selection: hand-written HIP/CDNA4 extension on GPU, torch on CPU.

Policy (see repo instructions + SURVEY.md §2.2): on a GPU box the HIP
extension IS the compute path - if it is missing, ops raise instead of
silently falling back to PyTorch eager. On CPU-only hosts (the dev
container, gloo multi-process tests) torch CPU kernels are the reference
implementation.
"""

from __future__ import annotations

import torch

_C = None
_tried = False


def _load():
    global _C, _tried
    if _tried:
        return _C
    _tried = True
    try:
        from .. import _C as ext  # in-tree built extension matrel_amd/_C.so
        _C = ext
    except ImportError:
        _C = None
    return _C


def ext():
    """The HIP extension module, or None when not built."""
    return _load()


def require_ext():
    """On a GPU host the HIP extension is mandatory: fail loudly."""
    e = _load()
    if e is None:
        raise RuntimeError(
            "matrel_amd HIP extension (matrel_amd/_C) is not built. "
            "On a GPU host the hand-written CDNA4 kernels are the only "
            "compute path (no PyTorch fallback). Build it with: "
            "python setup.py build_ext --inplace  (PYTORCH_ROCM_ARCH=gfx950)")
    return e


def use_hip(t: torch.Tensor) -> bool:
    return t.is_cuda

"""fengshen_amd.ops — HIP/CDNA4 fused kernels with eager oracles.

Every op has (a) a hand-written gfx950 HIP kernel in csrc/, compiled in-tree
into ``fengshen_amd/ops/_C*.so``, and (b) a plain-PyTorch eager reference used
as the CPU fallback and as the numerics oracle in tests (the reference's
forward_torch_softmax pattern, fused_softmax.py:184-199).

On a GPU box the HIP path is mandatory: if the extension is missing we raise
instead of silently falling back (so "native code not loaded" can't pass).
Set FENGSHEN_AMD_FORCE_EAGER=1 to override (tests / debugging only).
"""
from __future__ import annotations

import importlib
import logging
import os

import torch

logger = logging.getLogger(__name__)

_EXT = None
_EXT_TRIED = False


def _load_extension():
    global _EXT, _EXT_TRIED
    if _EXT_TRIED:
        return _EXT
    _EXT_TRIED = True
    try:
        _EXT = importlib.import_module("fengshen_amd.ops._C")
        logger.info("loaded fengshen_amd HIP extension: %s", _EXT.__file__)
    except ImportError as e:
        _EXT = None
        if torch.cuda.is_available() and os.environ.get("FENGSHEN_AMD_FORCE_EAGER") != "1":
            raise RuntimeError(
                "GPU present but the fengshen_amd HIP extension is not built. "
                "Run `python -m fengshen_amd.ops.build` (or __graft_entry__.build()). "
                f"Import error: {e}"
            ) from e
    return _EXT


def get_ext():
    """The compiled HIP extension module, or None (CPU-only environments)."""
    return _load_extension()


def has_ext() -> bool:
    return _load_extension() is not None


def use_hip(t: torch.Tensor) -> bool:
    """True if the HIP kernel path should run for this tensor."""
    if not t.is_cuda:
        return False
    if os.environ.get("FENGSHEN_AMD_FORCE_EAGER") == "1":
        return False
    return _load_extension() is not None

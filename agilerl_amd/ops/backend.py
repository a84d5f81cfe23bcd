"""HIP extension loading for the hand-written CDNA4 kernels.

The compiled extension lives IN-TREE (``agilerl_amd/ops/_hip_ops*.so``) so it
travels to GPU boxes with the repo snapshot.  Build via
``python setup.py build_ext --inplace`` or ``__graft_entry__.build()``
(hipcc, ``--offload-arch=gfx950``).

Dispatch policy:
- CUDA/ROCm tensors: the HIP extension is REQUIRED.  If it is missing on a
  GPU machine we raise loudly rather than silently falling back to eager
  (a silent eager fallback would invalidate benchmarks).
- CPU tensors: eager PyTorch implementations (also the numerics reference
  the GPU kernels are tested against).

Set ``AGILERL_AMD_FORCE_EAGER=1`` to force eager paths everywhere (debug).
"""

from __future__ import annotations

import importlib
import os
from typing import Optional

import torch

_EXT = None
_EXT_TRIED = False


def _try_load():
    global _EXT, _EXT_TRIED
    if _EXT_TRIED:
        return _EXT
    _EXT_TRIED = True
    try:
        _EXT = importlib.import_module("agilerl_amd.ops._hip_ops")
    except ImportError:
        _EXT = None
    return _EXT


def extension() -> Optional[object]:
    """The loaded HIP extension module, or None."""
    if os.environ.get("AGILERL_AMD_FORCE_EAGER") == "1":
        return None
    return _try_load()


def has_extension() -> bool:
    return extension() is not None


def require_extension():
    ext = extension()
    if ext is None:
        raise RuntimeError(
            "agilerl_amd HIP extension (agilerl_amd.ops._hip_ops) is not built "
            "but a GPU tensor was passed. Build it in-tree with "
            "`python setup.py build_ext --inplace` (requires hipcc, "
            "PYTORCH_ROCM_ARCH=gfx950). Refusing to silently fall back to "
            "eager on GPU."
        )
    return ext


def use_hip(*tensors: torch.Tensor) -> bool:
    """True if the op should dispatch to the HIP kernel for these tensors."""
    if os.environ.get("AGILERL_AMD_FORCE_EAGER") == "1":
        return False
    return any(t.is_cuda for t in tensors if isinstance(t, torch.Tensor))

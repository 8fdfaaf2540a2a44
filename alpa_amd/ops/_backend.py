"""HIP extension loading + anti-silent-fallback guard.

The gfx950 kernels live in ``alpa_amd/ops/csrc/*.hip`` and are compiled
in-tree (``python setup.py build_ext`` or ``__graft_entry__.build()``) into
``alpa_amd/ops/_hip_ops*.so``.  On a GPU box the extension MUST be present:
running eager/PyTorch fallbacks silently would invalidate every measurement,
so :func:`hip_ops` raises if CUDA (ROCm) is available but the extension
failed to import.
"""
from __future__ import annotations

import os

import torch

_EXT = None
_EXT_ERR: Exception | None = None
_TRIED = False


def _try_import():
    global _EXT, _EXT_ERR, _TRIED
    if _TRIED:
        return
    _TRIED = True
    try:
        from alpa_amd.ops import _hip_ops  # type: ignore
        _EXT = _hip_ops
    except ImportError as e:  # extension not built
        _EXT_ERR = e


def hip_ops_available() -> bool:
    _try_import()
    return _EXT is not None


def hip_ops():
    """Return the extension module; raise loudly if on GPU without it."""
    _try_import()
    if _EXT is None:
        from ..global_env import global_config
        if torch.cuda.is_available() and global_config.require_hip_kernels_on_gpu \
                and not os.environ.get("ALPA_AMD_ALLOW_EAGER_FALLBACK"):
            raise RuntimeError(
                "alpa_amd HIP extension (_hip_ops) is not built but a GPU is "
                "present. Build it with `python setup.py build_ext --inplace` "
                "(or __graft_entry__.build()). Refusing to fall back to eager "
                f"PyTorch silently. Import error: {_EXT_ERR}")
        return None
    return _EXT


def use_hip(x: torch.Tensor) -> bool:
    """True when op dispatch should go to the hand-written gfx950 kernel."""
    from ..global_env import global_config
    if not (x.is_cuda and global_config.use_hip_kernels):
        return False
    return hip_ops() is not None

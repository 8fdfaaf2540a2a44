"""Version + extension ABI guard.

Capability analog of the reference's ``alpa/version.py``
(check_alpa_jaxlib_version:10): the pure-Python package and the compiled
HIP extension evolve together; a stale ``_hip_ops.so`` left over from an
older build silently exposes an incompatible call surface.  The guard
asserts the loaded extension's ABI_VERSION is at least what this Python
tree expects — loudly, at first use.
"""
from __future__ import annotations

__version__ = "0.1.0"

#: minimum extension ABI this Python tree can drive
MIN_HIP_OPS_ABI = 2


def check_hip_ops_version() -> int:
    """Return the loaded extension's ABI version; raise if the extension
    is present but predates MIN_HIP_OPS_ABI (rebuild with
    ``python setup.py build_ext --inplace``).  Returns -1 when the
    extension is absent (CPU-only environments)."""
    from .ops._backend import hip_ops_available, hip_ops
    if not hip_ops_available():
        return -1
    ext = hip_ops()
    abi = getattr(ext, "ABI_VERSION", 0)
    if abi < MIN_HIP_OPS_ABI:
        raise RuntimeError(
            f"alpa_amd HIP extension ABI {abi} < required "
            f"{MIN_HIP_OPS_ABI}: the in-tree _hip_ops.so is stale. "
            "Rebuild it with `python setup.py build_ext --inplace`.")
    return abi

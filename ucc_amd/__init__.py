"""ucc_amd — MI355X-native collective communication framework.

A from-scratch implementation of the openucx/ucc capability set for
8x AMD Instinct MI355X over xGMI: UCC-compatible C API (src/api/ucc.h),
static TL stack (self / shm / cdna4 / rccl), HIP gfx950 executor kernels,
and this Python package for testing, benchmarking and torch integration.
"""

from . import dtypes  # noqa: F401

__version__ = "1.3.0"

_core_mod = None


def core():
    """Lazily import the native extension (built by `make` / build())."""
    global _core_mod
    if _core_mod is None:
        import importlib

        _core_mod = importlib.import_module("ucc_amd._core")
    return _core_mod

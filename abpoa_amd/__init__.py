"""abpoa_amd — MI355X-native partial order alignment (abPOA-compatible).

Thin ctypes access to the native library for tests and bench. The compute
path is the HIP/CDNA4 core inside libabpoa_amd.so; this package performs no
alignment math in Python.
"""
import ctypes
import os

_CSRC = os.path.join(os.path.dirname(os.path.abspath(__file__)), "csrc")
LIB_PATH = os.path.join(_CSRC, "libabpoa_amd.so")
CLI_PATH = os.path.join(_CSRC, "abpoa_amd")
CLI_CPUTEST_PATH = os.path.join(_CSRC, "abpoa_amd_cputest")
ORACLE_SO = os.path.join(os.path.dirname(_CSRC), os.pardir, "oracle", "liboracle.so")

_lib = None


def lib():
    """Load (once) and return the native library handle."""
    global _lib
    if _lib is None:
        if not os.path.exists(LIB_PATH):
            raise RuntimeError("libabpoa_amd.so not built; run __graft_entry__.build()")
        _lib = ctypes.CDLL(LIB_PATH)
        _lib.abpoa_amd_get_stats.argtypes = [ctypes.POINTER(ctypes.c_uint64)] * 3
        _lib.abpoa_amd_get_stats.restype = None
        _lib.abpoa_amd_reset_stats.restype = None
    return _lib


def get_stats():
    """(dp_cells, kernel_ns, n_launches) accumulated by the native core."""
    a, b, c = ctypes.c_uint64(), ctypes.c_uint64(), ctypes.c_uint64()
    lib().abpoa_amd_get_stats(ctypes.byref(a), ctypes.byref(b), ctypes.byref(c))
    return a.value, b.value, c.value


def reset_stats():
    lib().abpoa_amd_reset_stats()

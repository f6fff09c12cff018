"""ctypes binding to the in-tree HIP library (flox_amd/libfloxhip.so).

The library implements the C ABI declared in include/floxhip.h — the drop-in
boundary for the reference's engine seam (flox/aggregations.py:60-133).
This module FAILS LOUDLY if the extension or a GPU is missing: the product
path has no CPU fallback (parity claims depend on the HIP kernels being the
code that runs).
"""

from __future__ import annotations

import ctypes
import os

import numpy as np

_LIB_PATH = os.path.join(os.path.dirname(os.path.abspath(__file__)), "libfloxhip.so")

# fh_dtype / fh_ldtype / fh_opset / fh_flags — mirror include/floxhip.h
F32, F64, I64, I32 = 0, 1, 2, 3
L_I64, L_I32 = 0, 1
SET_SUM_COUNT = 0
SET_SUM_COUNT_PRESENT = 1
SET_COUNT = 2
SET_MIN_FULL = 3
SET_MIN_COUNT = 4
SET_MAX_FULL = 5
SET_MAX_COUNT = 6
SET_SSD = 7
SET_PROD = 8
SET_IDXMIN = 9
SET_IDXMAX = 10
SET_WELFORD = 11
SET_ARGMIN_PAIR = 12
SET_ARGMAX_PAIR = 13
FLAG_SKIPNAN = 1
FLAG_FORCE_LDS = 2
FLAG_FORCE_ATOMIC = 4
FLAG_SORTED_LABELS = 8
FLAG_NO_HOST_SYNC = 16

VDTYPE_OF = {
    np.dtype("float32"): F32,
    np.dtype("float64"): F64,
    np.dtype("int64"): I64,
    np.dtype("int32"): I32,
}
LDTYPE_OF = {np.dtype("int64"): L_I64, np.dtype("int32"): L_I32}


class FhCall(ctypes.Structure):
    _fields_ = [
        ("op_set", ctypes.c_int),
        ("vdtype", ctypes.c_int),
        ("ldtype", ctypes.c_int),
        ("flags", ctypes.c_int),
        ("n", ctypes.c_int64),
        ("ngroups", ctypes.c_int64),
        ("values", ctypes.c_void_p),
        ("labels", ctypes.c_void_p),
        ("labels2", ctypes.c_void_p),
        ("g0", ctypes.c_int64),
        ("g1", ctypes.c_int64),
        ("means", ctypes.c_void_p),
        ("out_sum", ctypes.c_void_p),
        ("out_count", ctypes.c_void_p),
        ("out_present", ctypes.c_void_p),
        ("out_min", ctypes.c_void_p),
        ("out_max", ctypes.c_void_p),
        ("out_nanflag", ctypes.c_void_p),
        ("scratch", ctypes.c_void_p),
        ("scratch_bytes", ctypes.c_int64),
        ("stream", ctypes.c_void_p),
        ("path_used", ctypes.c_int),
        # column path (fh_grouped_reduce_cols)
        ("perm", ctypes.c_void_p),
        ("m", ctypes.c_int64),
        ("ldm", ctypes.c_int64),
        ("chunk_offsets", ctypes.c_void_p),
        ("nchunks", ctypes.c_int64),
        ("target", ctypes.c_void_p),
        ("row_offset", ctypes.c_int64),
    ]


_lib = None


def load_library():
    """Load libfloxhip.so; raise (never fall back) if it is absent."""
    global _lib
    if _lib is not None:
        return _lib
    if not os.path.exists(_LIB_PATH):
        raise RuntimeError(
            f"flox_amd HIP extension not built: {_LIB_PATH} missing. "
            "Run __graft_entry__.build() (hipcc --offload-arch=gfx950)."
        )
    lib = ctypes.CDLL(_LIB_PATH)
    lib.fh_grouped_reduce.argtypes = [ctypes.POINTER(FhCall)]
    lib.fh_grouped_reduce.restype = ctypes.c_int
    lib.fh_grouped_reduce_cols.argtypes = [ctypes.POINTER(FhCall)]
    lib.fh_grouped_reduce_cols.restype = ctypes.c_int
    lib.fh_scratch_bytes.argtypes = [ctypes.POINTER(FhCall)]
    lib.fh_scratch_bytes.restype = ctypes.c_int64
    lib.fh_quantile_scratch_bytes.argtypes = [ctypes.POINTER(FhCall)]
    lib.fh_quantile_scratch_bytes.restype = ctypes.c_int64
    lib.fh_grouped_quantile.argtypes = [ctypes.POINTER(FhCall), ctypes.c_int]
    lib.fh_grouped_quantile.restype = ctypes.c_int
    lib.fh_scan_scratch_bytes.argtypes = [ctypes.POINTER(FhCall)]
    lib.fh_scan_scratch_bytes.restype = ctypes.c_int64
    lib.fh_grouped_scan.argtypes = [ctypes.POINTER(FhCall), ctypes.c_int]
    lib.fh_grouped_scan.restype = ctypes.c_int
    lib.fh_pack_argkeys.argtypes = [
        ctypes.c_void_p, ctypes.c_int, ctypes.c_int64, ctypes.c_int64,
        ctypes.c_int, ctypes.c_int, ctypes.c_void_p, ctypes.c_void_p,
    ]
    lib.fh_pack_argkeys.restype = ctypes.c_int
    lib.fh_error_string.argtypes = [ctypes.c_int]
    lib.fh_error_string.restype = ctypes.c_char_p
    lib.fh_version.argtypes = []
    lib.fh_version.restype = ctypes.c_int
    if lib.fh_version() != 1:
        raise RuntimeError("libfloxhip.so ABI version mismatch")
    _lib = lib
    return lib


def check(code: int) -> None:
    if code != 0:
        msg = load_library().fh_error_string(code).decode()
        raise RuntimeError(f"floxhip error {code}: {msg}")

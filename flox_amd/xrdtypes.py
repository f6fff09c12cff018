"""Output-dtype promotion and fill-value rules for flox_amd.

Reproduces the reference's rules so results are dtype-identical:
reference flox/xrdtypes.py:153-209 (_normalize_dtype, _maybe_promote_int,
_get_fill_value) and the per-aggregation final_dtype/final_fill_value choices
of flox/aggregations.py:304-546.
"""

from __future__ import annotations

import numpy as np

# funcs that always produce floating point, preserving a floating input dtype
FLOAT_FUNCS = {
    "mean", "nanmean", "var", "nanvar", "std", "nanstd",
    "quantile", "nanquantile", "median", "nanmedian",
}
# funcs whose output dtype equals the input dtype (preserves_dtype=True)
PRESERVES_DTYPE = {
    "min", "nanmin", "max", "nanmax", "first", "nanfirst", "last", "nanlast",
    "mode", "nanmode",
}
NAN_SKIP = {
    "nansum",
    "nanprod",
    "nanmean",
    "nanvar",
    "nanstd",
    "nanmin",
    "nanmax",
    "count",
    "nanfirst",
    "nanlast",
}


ARG_FUNCS = {"argmax", "argmin", "nanargmax", "nanargmin"}
BOOL_FUNCS = {"any", "all"}


def final_dtype(func: str, array_dtype: np.dtype, dtype=None) -> np.dtype:
    """Output dtype (reference xrdtypes.py:153-186)."""
    array_dtype = np.dtype(array_dtype)
    if dtype is not None:
        return np.dtype(dtype)
    if func == "count" or func in ARG_FUNCS:
        return np.dtype(np.intp)
    if func in BOOL_FUNCS:
        return np.dtype(bool)
    if func in ("quantile", "nanquantile"):
        # quantile's final_dtype is ALWAYS float64 (reference
        # aggregations.py:695-710); median preserves a floating input dtype
        return np.dtype("float64")
    if func in FLOAT_FUNCS:
        if array_dtype.kind in "fc":
            return array_dtype
        return np.dtype("float64")
    if func in PRESERVES_DTYPE:
        return array_dtype
    # sum/prod: promote sub-platform ints (xrdtypes.py:175-185)
    if array_dtype.kind == "i":
        return np.result_type(array_dtype, np.int_)
    if array_dtype.kind == "u":
        return np.result_type(array_dtype, np.uint)
    return array_dtype


def fill_default(func: str, out_dtype: np.dtype):
    """final_fill_value for groups with no members when the user gave none
    (reference aggregations.py:304-546 + xrdtypes.py:188-209: NA sentinel ->
    NaN for floats, iinfo.min for ints; count -> 0; prod -> 1)."""
    out_dtype = np.dtype(out_dtype)
    if func == "count":
        return 0
    if func in ARG_FUNCS:
        return -1  # reference aggregations.py:589/617 final_fill_value=-1
    if func in BOOL_FUNCS:
        return False  # reference aggregations.py:651-676
    if func == "prod":
        return 1
    if out_dtype.kind in "fc":
        return float("nan")
    return np.iinfo(out_dtype).min

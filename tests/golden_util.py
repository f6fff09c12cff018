"""Helpers to iterate the committed golden fixtures (tests/golden/golden_cases.npz)."""

from __future__ import annotations

import os

import numpy as np

GOLDEN = os.path.join(os.path.dirname(__file__), "golden", "golden_cases.npz")


def load_golden_cases():
    """Yield (name, inputs-dict, expected-result, expected-groups-list)."""
    data = np.load(GOLDEN, allow_pickle=False)
    names = sorted({k.split("::")[0] for k in data.files})
    for name in names:
        def get(suffix, default=None):
            key = f"{name}::{suffix}"
            return data[key] if key in data.files else default

        bys = []
        i = 0
        while get(f"by{i}") is not None:
            bys.append(get(f"by{i}"))
            i += 1
        expected = []
        i = 0
        while get(f"expected{i}") is not None:
            e = get(f"expected{i}")
            if get(f"interval{i}") is not None:
                import pandas as pd

                e = pd.IntervalIndex.from_breaks(e)
            expected.append(e)
            i += 1
        groups = []
        i = 0
        while get(f"groups{i}") is not None:
            groups.append(get(f"groups{i}"))
            i += 1
        func = name.split("_")[0]
        kw = {}
        if expected:
            kw["expected_groups"] = tuple(expected) if len(expected) > 1 else expected[0]
        fv = get("fill_value")
        if fv is not None:
            kw["fill_value"] = fv.item()
        mc = get("min_count")
        if mc is not None:
            kw["min_count"] = int(mc)
        ds = get("dtype_s")
        if ds is not None:
            kw["dtype"] = np.dtype(str(ds))
        ddof = get("ddof")
        if ddof is not None:
            kw["finalize_kwargs"] = {"ddof": int(ddof)}
        qv = get("q")
        if qv is not None:
            kw["finalize_kwargs"] = {"q": qv.item() if get("q_scalar") else qv.tolist()}
        if get("isbin") is not None:
            kw["isbin"] = True
        if get("nosort") is not None:
            kw["sort"] = False
        axv = get("axis")
        if axv is not None:
            kw["axis"] = tuple(int(a) for a in np.atleast_1d(axv))
        if get("scan") is not None:
            kw["_scan"] = True
            func = name.split("_")[1]
        if get("customagg") is not None:
            # reconstruct the fixed custom Aggregation the generator pinned
            import flox_amd

            func = flox_amd.CustomAggregation(
                name="custommean", numpy="mean", chunk=("sum", "nanlen"),
                combine=("sum", "sum"), finalize=lambda s, c: s / c,
                fill_value=0, final_fill_value=float("nan"))
        yield name, dict(array=get("array"), by=tuple(bys), func=func, **kw), get("result"), groups


def tolerance_for(name, result_dtype):
    """fp tolerance per case class: engine="flox" (the golden producer)
    accumulates fp32 in fp32 while our builds accumulate in f64 (npg
    semantics, reference tests/test_properties.py:146-151) -> loose rtol for
    f32; tight for f64 (reference tests/__init__.py:96-99 uses
    rtol=1e-15/atol=1e-18 for f64, var/std rtol 1e-13 test_core.py:259)."""
    if result_dtype.kind in "iub":
        return dict(rtol=0, atol=0)
    if result_dtype.kind == "c":
        # complex: per-component precision class (c64 components are f32)
        if result_dtype.itemsize == 8:
            return dict(rtol=2e-6, atol=1e-5)
        return dict(rtol=1e-12, atol=1e-11)
    if result_dtype.itemsize == 2:
        # float16: the reference's intermediate arithmetic is f16 itself
        # (e.g. its 3-pass var rounds each pass); we compute in f32/f64 and
        # cast back — agree to f16 ulp scale
        return dict(rtol=2e-3, atol=1e-3)
    if result_dtype.itemsize == 4:
        if name.startswith("scan_"):
            # the reference computes segment scans as (global cumsum) minus
            # (previous groups' cumsum) in fp32 — cancellation differs from a
            # direct per-segment sum by ~1 ulp of the global running sum
            return dict(rtol=2e-6, atol=1e-5)
        return dict(rtol=2e-6, atol=1e-7)
    if name.startswith("scan_"):
        # f64 scans: the reference forms segment scans as global-cumsum
        # minus prior-group offsets — the cancellation differs from a direct
        # per-segment sum by ~1 ulp of the running sum
        return dict(rtol=1e-11, atol=1e-12)
    if "var" in name or "std" in name:
        return dict(rtol=1e-12, atol=1e-14)
    if "quantile" in name or "median" in name:
        # the lerp t*(b-a)+a vs (1-t)*a+t*b forms differ by ~1 ulp
        return dict(rtol=1e-12, atol=1e-14)
    # atol absorbs atomic-order cancellation on near-zero f64 sums
    # (group sums of O(10) addends cancelling to ~1e-3)
    return dict(rtol=1e-13, atol=1e-11)

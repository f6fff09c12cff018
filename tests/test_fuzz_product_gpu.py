"""Randomized product-vs-oracle sweep on the GPU — the product-side twin of
tools/fuzz_oracle_vs_reference.py (which pins the oracle to the reference in
the build container). Together they close the chain
  HIP product == oracle == reference
over random shapes, dtypes (incl. datetime64/NaT) and kwarg combinations."""

import os
import sys

import numpy as np
import pytest

sys.path.insert(0, os.path.join(os.path.dirname(os.path.dirname(os.path.abspath(__file__))), "tools"))

import flox_amd
from oracle import groupby_reduce as oracle_reduce
from oracle import groupby_scan as oracle_scan

pytestmark = pytest.mark.gpu

N_CASES = int(os.environ.get("FUZZ_CASES", "80"))


def _run_case(i, rng):
    from fuzz_oracle_vs_reference import FUNCS, SCANS, make_case

    # ~20% big cases exercise the partition / packed-arg / sorted-direct /
    # atomic kernel paths (the small default stays on the LDS path)
    arr, by, kw = make_case(rng, big=bool(rng.random() < 0.2))
    is_scan = rng.random() < 0.2 and "axis" not in kw and "min_count" not in kw
    if is_scan:
        func = str(rng.choice(SCANS))
        kw.pop("fill_value", None)
        kw.pop("sort", None)
    else:
        func = str(rng.choice(FUNCS))
        if func in ("quantile", "nanquantile"):
            q = [0.25, 0.9] if rng.random() < 0.5 else float(rng.random())
            kw["finalize_kwargs"] = {"q": q}
        if func in ("var", "nanvar", "std", "nanstd") and rng.random() < 0.3:
            kw["finalize_kwargs"] = {"ddof": 1}
    a = np.asarray(arr)
    if a.dtype.kind in "Mm" and (is_scan and func in ("cumsum", "nancumsum")):
        return None  # datetime cumsum wrap: covered by goldens only
    if a.dtype.kind in "Mm" and not is_scan and func not in (
        "min", "nanmin", "max", "nanmax", "count",
        "median", "nanmedian", "quantile", "nanquantile",
    ):
        return None  # datetime wrap arithmetic: out of parity scope
    if func in ("prod", "nanprod"):
        if a.dtype.kind in "iuMm":
            # unsigned: a -1 bound wraps to the dtype max and products then
            # overflow int64 order-dependently (documented wrap class)
            arr = np.clip(arr, 0 if a.dtype.kind == "u" else -1, 1)
        elif np.asarray(arr).size > 1000:
            # big groups: fp products overflow/underflow at order-dependent
            # points; sign-only values keep them exact
            arr = np.sign(np.asarray(arr))
        else:
            arr = np.clip(arr, -2.0, 2.0)  # keep fp products bounded
    bys = by if isinstance(by, tuple) else (by,)
    try:
        if is_scan:
            want = oracle_scan(arr, *bys, func=func, **kw)
        else:
            want, *_ = oracle_reduce(arr, *bys, func=func, **kw)
    except NotImplementedError:
        return None
    try:
        if is_scan:
            got = flox_amd.groupby_scan(arr, *bys, func=func, **kw)
        else:
            got, *_ = flox_amd.groupby_reduce(arr, *bys, func=func, **kw)
    except NotImplementedError:
        return None
    got, want = np.asarray(got), np.asarray(want)
    ctx = f"[{i}] func={func} scan={is_scan} dt={np.asarray(arr).dtype} shape={np.shape(arr)} kw={sorted(kw)}"
    assert got.shape == want.shape, ctx + f" {got.shape} vs {want.shape}"
    assert got.dtype == want.dtype, ctx + f" {got.dtype} vs {want.dtype}"
    if want.dtype.kind in "Mm":
        wi, gi = want.view("i8").astype("f8"), got.view("i8").astype("f8")
        nat = float(np.iinfo(np.int64).min)
        np.testing.assert_array_equal(wi == nat, gi == nat, err_msg=ctx)
        # NaT-lerp cells (beyond ~30k years) can bracket different pairs at
        # virtual-index rounding edges — loose there, tight on date-scale
        natish = np.abs(wi) > 1e12
        np.testing.assert_allclose(gi[~natish], wi[~natish], rtol=1e-9, atol=1.0, err_msg=ctx)
        np.testing.assert_allclose(gi[natish], wi[natish], rtol=1e-6, atol=1.0, err_msg=ctx)
    elif want.dtype.kind in "iub":
        np.testing.assert_array_equal(got, want, err_msg=ctx)
    else:
        fin = want[np.isfinite(want)]
        scale = 1 + float(np.max(np.abs(fin), initial=0.0))

        def _prec(dtp):
            if dtp.kind == "c":
                return dtp.itemsize // 2  # per-component precision
            return dtp.itemsize if dtp.kind == "f" else 8

        eff = min(_prec(want.dtype), _prec(np.asarray(arr).dtype))
        if eff == 2:
            # f16 precision class: the oracle computes quantile/lerp
            # arithmetic in the INPUT precision (np.quantile preserves f16),
            # the product lerps the promoted f32 values in f64
            tol = dict(rtol=2e-3, atol=1e-3 * scale)
        elif eff == 4:
            tol = dict(rtol=3e-5, atol=1e-4 * scale)
        else:
            tol = dict(rtol=1e-10, atol=1e-10 * scale)
        np.testing.assert_allclose(got, want, equal_nan=True, err_msg=ctx, **tol)
    return True


def test_fuzz_product_vs_oracle():
    rng = np.random.default_rng(int(os.environ.get("FUZZ_SEED", "424242")))
    n_run = 0
    for i in range(N_CASES):
        if _run_case(i, rng) is not None:
            n_run += 1
    assert n_run > N_CASES // 3, f"only {n_run} cases executed"

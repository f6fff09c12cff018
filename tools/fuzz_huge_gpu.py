"""Huge-size randomized GPU fuzz: 2e6..2e7-row cases
through the partition / packed-arg / pair-arg / sorted-direct / atomic /
overflow-fallback paths at realistic group counts (8e3..1.2e7).

Checkers:
  * reductions (sum/nansum/mean/nanmean/count/min/nanmin/max/nanmax/var):
    vectorized numpy oracle (np.bincount / minimum.at with f64 accumulation,
    the npg contract) — no per-group Python loops, so 2e7 rows is seconds;
  * arg-reductions: the product's two INDEPENDENT implementations must agree
    bit-for-bit (partition pair/packed path vs the forced two-pass
    LDS/atomic form), plus the gathered value at each index must equal the
    per-group extremum from the numpy oracle.

Label layouts: uniform, sorted (FH_SORTED_LABELS direct path), heavily
skewed (optimistic-region overflow -> exact fallback).

Usage: FUZZ_HUGE_CASES=N python tools/fuzz_huge_gpu.py [seed]
"""

import os
import sys
import time

sys.path.insert(0, os.environ.get("GRAFT_REPO_ROOT", "/root/repo"))
import numpy as np  # noqa: E402
import torch  # noqa: E402

import flox_amd  # noqa: E402
from flox_amd import core as fa_core  # noqa: E402

N_CASES = int(os.environ.get("FUZZ_HUGE_CASES", "20"))

RED_FUNCS = ["sum", "nansum", "mean", "nanmean", "count", "min", "nanmin",
             "max", "nanmax", "var", "nanvar"]
ARG_FUNCS = ["argmin", "argmax", "nanargmin", "nanargmax"]


def np_reduce(func, vals, labels, ng):
    """Vectorized f64-accumulating reference (npg semantics)."""
    v = vals.astype(np.float64) if vals.dtype.kind == "f" else vals.astype(np.int64)
    nan = np.isnan(v) if vals.dtype.kind == "f" else np.zeros(len(v), bool)
    skip = func.startswith("nan") or func == "count"
    m = ~nan if skip else np.ones(len(v), bool)
    counts = np.bincount(labels[~nan], minlength=ng)
    present = np.bincount(labels, minlength=ng) > 0
    if func == "count":
        return counts.astype(np.intp), counts == 0
    if func in ("sum", "nansum"):
        if vals.dtype.kind == "f":
            s = np.bincount(labels[m], weights=v[m], minlength=ng)
        else:
            s = np.zeros(ng, np.int64)
            np.add.at(s, labels[m], v[m])
        if not skip:
            nansum_mark = np.bincount(labels[nan], minlength=ng) > 0
            if vals.dtype.kind == "f":
                s[nansum_mark] = np.nan
        out = s.astype(vals.dtype if vals.dtype.kind == "f" else np.int64)
        return out, ~present
    if func in ("mean", "nanmean"):
        s = np.bincount(labels[m], weights=v[m], minlength=ng)
        if not skip and vals.dtype.kind == "f":
            s[np.bincount(labels[nan], minlength=ng) > 0] = np.nan
        with np.errstate(invalid="ignore", divide="ignore"):
            out = s / counts
        return out.astype(vals.dtype if vals.dtype.kind == "f" else np.float64), counts == 0
    if func in ("var", "nanvar"):
        s = np.bincount(labels[m], weights=v[m], minlength=ng)
        with np.errstate(invalid="ignore", divide="ignore"):
            mu = s / counts
        d = v - mu[labels]
        ssd = np.bincount(labels[m], weights=(d * d)[m], minlength=ng)
        if not skip and vals.dtype.kind == "f":
            ssd[np.bincount(labels[nan], minlength=ng) > 0] = np.nan
        with np.errstate(invalid="ignore", divide="ignore"):
            out = ssd / counts
        out[counts == 0] = np.nan
        return out.astype(vals.dtype if vals.dtype.kind == "f" else np.float64), counts == 0
    # min / max families
    ismin = "min" in func
    ext = np.full(ng, np.inf if ismin else -np.inf)
    mm = ~nan
    if ismin:
        np.minimum.at(ext, labels[mm], v[mm])
    else:
        np.maximum.at(ext, labels[mm], v[mm])
    if not skip and vals.dtype.kind == "f":
        hasnan = np.bincount(labels[nan], minlength=ng) > 0
        ext[hasnan] = np.nan
    out = ext.astype(vals.dtype) if vals.dtype.kind == "f" else ext
    if vals.dtype.kind != "f":
        out = np.where(counts > 0, ext, 0).astype(np.int64)
    empty = counts == 0 if skip else ~present
    return out, empty


def make_labels(rng, n, ng, layout):
    if layout == "uniform":
        return rng.integers(0, ng, n)
    if layout == "sorted":
        return np.sort(rng.integers(0, ng, n))
    # skew: most rows in a handful of groups (overflow -> exact fallback)
    lab = rng.integers(0, max(ng // 1000, 2), n)
    m = rng.random(n) < 0.05
    lab[m] = rng.integers(0, ng, int(m.sum()))
    return lab


def cols_case(i, rng):
    """Column-path (leading-dims) case at realistic shapes: exercises the
    VC=8/VC=vec/vc=1 kernels and the multi-chunk slab mode (row range split
    + k_combine fold), checked against the oracle (vectorized per lead
    row)."""
    from oracle import groupby_reduce as oracle_reduce

    n_t = int(rng.integers(2_000, 30_001))
    m = int(rng.integers(64, 1_501))
    ng = int(rng.choice([8, 128, 1000]))
    dtype = str(rng.choice(["float32", "float64"]))
    func = str(rng.choice(["mean", "nanmean", "sum", "var", "nanvar",
                           "min", "nanmax", "count"]))
    vals = (rng.standard_normal((m, n_t)) * 10).astype(dtype)
    if rng.random() < 0.6:
        vals[rng.random((m, n_t)) < 0.05] = np.nan
    labels = rng.integers(0, ng, n_t)
    if rng.random() < 0.3:
        labels = np.sort(labels)
    t0 = time.perf_counter()
    want, *_ = oracle_reduce(vals, labels, func=func, expected_groups=np.arange(ng))
    got, *_ = flox_amd.groupby_reduce(vals, labels, func=func, expected_groups=np.arange(ng))
    got = np.asarray(got)
    assert got.shape == want.shape and got.dtype == want.dtype
    if want.dtype.kind == "f":
        fin = want[np.isfinite(want)]
        scale = 1 + float(np.max(np.abs(fin), initial=0.0))
        tol = (dict(rtol=3e-5, atol=1e-4 * scale) if want.dtype.itemsize == 4
               else dict(rtol=1e-10, atol=1e-10 * scale))
        np.testing.assert_allclose(got, want, equal_nan=True,
                                   err_msg=f"[{i}] cols {func} {dtype} ({m}x{n_t}, ng={ng})",
                                   **tol)
    else:
        np.testing.assert_array_equal(got, want,
                                      err_msg=f"[{i}] cols {func} {dtype}")
    dt = time.perf_counter() - t0
    print(f"[{i}] OK cols:{func} {dtype} m={m} n_t={n_t} ng={ng} ({dt:.1f}s)",
          flush=True)


def scan_case(i, rng):
    """Huge-size grouped scans (block-carry chains over thousands of blocks,
    sentinel carry slot, sorted fast path) against the pinned oracle."""
    from oracle import groupby_scan as oracle_scan

    n = int(rng.integers(2_000_000, 8_000_001))
    ng = int(rng.choice([8192, 100_000, 1_000_000]))
    dtype = str(rng.choice(["float32", "float64", "int64"]))
    layout = str(rng.choice(["uniform", "sorted"]))
    labels = rng.integers(0, ng, n)
    if layout == "sorted":
        labels = np.sort(labels)
    if dtype == "int64":
        vals = rng.integers(-1000, 1000, n).astype(np.int64)
        func = str(rng.choice(["cumsum", "ffill", "bfill"]))
    else:
        vals = (rng.standard_normal(n) * 100).astype(dtype)
        if rng.random() < 0.7:
            vals[rng.random(n) < 0.05] = np.nan
        func = str(rng.choice(["cumsum", "nancumsum", "ffill", "bfill"]))
    t0 = time.perf_counter()
    want = oracle_scan(vals, labels, func=func, expected_groups=np.arange(ng))
    got = np.asarray(flox_amd.groupby_scan(vals, labels, func=func,
                                           expected_groups=np.arange(ng)))
    ctx = f"[{i}] scan:{func} {dtype} {layout} n={n} ng={ng}"
    assert got.shape == want.shape and got.dtype == want.dtype, ctx
    if want.dtype.kind == "f":
        fin = np.isfinite(want)
        scale = 1 + float(np.max(np.abs(want[fin]), initial=0.0))
        tol = (dict(rtol=3e-5, atol=1e-4 * scale) if want.dtype.itemsize == 4
               else dict(rtol=1e-10, atol=1e-10 * scale))
        np.testing.assert_allclose(got, want, equal_nan=True, err_msg=ctx, **tol)
    else:
        np.testing.assert_array_equal(got, want, err_msg=ctx)
    dt = time.perf_counter() - t0
    print(f"[{i}] OK scan:{func} {dtype} {layout} n={n:.1e} ng={ng:.0e} ({dt:.1f}s)",
          flush=True)


def order_case(i, rng):
    """Order statistics (rocprim-sorted quantile path) at 1e6+ rows against
    the pinned oracle. Sizes are capped by the oracle's per-group sort cost
    (~5-15 s/case on host cores); mode stays in the 300k product fuzz."""
    from oracle import groupby_reduce as oracle_reduce

    n = int(rng.integers(1_000_000, 3_000_001))
    ng = int(rng.choice([8192, 100_000]))
    dtype = str(rng.choice(["float32", "float64"]))
    vals = (rng.standard_normal(n) * 100).astype(dtype)
    if rng.random() < 0.7:
        vals[rng.random(n) < 0.05] = np.nan
    labels = rng.integers(0, ng, n)
    func = str(rng.choice(["quantile", "nanquantile", "median", "nanmedian"]))
    kw = {}
    if "quantile" in func:
        q = [0.25, 0.5, 0.75] if rng.random() < 0.3 else float(rng.uniform(0.02, 0.98))
        kw["finalize_kwargs"] = {"q": q}
    t0 = time.perf_counter()
    want, *_ = oracle_reduce(vals, labels, func=func,
                             expected_groups=np.arange(ng), **kw)
    got, *_ = flox_amd.groupby_reduce(vals, labels, func=func,
                                      expected_groups=np.arange(ng), **kw)
    got = np.asarray(got)
    ctx = f"[{i}] order:{func} {dtype} n={n} ng={ng}"
    assert got.shape == want.shape, ctx
    fin = np.isfinite(want)
    scale = 1 + float(np.max(np.abs(want[fin]), initial=0.0))
    # tier by INPUT precision: quantile outputs are always f64, but the
    # interpolation endpoints are the f32 inputs, so product-vs-oracle
    # arithmetic-order differences sit at f32 eps of the operands
    tol = (dict(rtol=3e-5, atol=1e-4 * scale) if dtype == "float32"
           else dict(rtol=1e-10, atol=1e-10 * scale))
    np.testing.assert_allclose(got.astype(np.float64), want.astype(np.float64),
                               equal_nan=True, err_msg=ctx, **tol)
    dt = time.perf_counter() - t0
    print(f"[{i}] OK order:{func} {dtype} n={n:.1e} ng={ng:.0e} ({dt:.1f}s)",
          flush=True)


def one_case(i, rng):
    r = rng.random()
    if r < 0.25:
        cols_case(i, rng)
        return
    if r < 0.40:
        scan_case(i, rng)
        return
    if r < 0.48:
        order_case(i, rng)
        return
    n = int(rng.integers(2_000_000, 20_000_001))
    ng = int(rng.choice([8192, 100_000, 1_000_000, 12_000_000]))
    dtype = str(rng.choice(["float32", "float64", "int64"]))
    layout = str(rng.choice(["uniform", "uniform", "sorted", "skew"]))
    labels = make_labels(rng, n, ng, layout)
    if dtype in ("float32", "float64"):
        vals = (rng.standard_normal(n) * 100).astype(dtype)
        if rng.random() < 0.6:
            vals[rng.random(n) < 0.05] = np.nan
    else:
        vals = rng.integers(-1000, 1000, n).astype(np.int64)

    kind = "arg" if rng.random() < 0.35 else "red"
    t0 = time.perf_counter()
    if kind == "red":
        func = str(rng.choice(RED_FUNCS))
        want, empty = np_reduce(func, vals, labels, ng)
        got, *_ = flox_amd.groupby_reduce(vals, labels, func=func,
                                          expected_groups=range(ng))
        got = np.asarray(got)
        ok = ~empty
        if want.dtype.kind == "f":
            scale = 1 + np.nanmax(np.abs(want[ok]), initial=0.0)
            rtol, atol = (3e-5, 1e-4 * scale) if want.dtype.itemsize == 4 else (1e-10, 1e-10 * scale)
            np.testing.assert_allclose(got[ok].astype(np.float64),
                                       want[ok].astype(np.float64),
                                       equal_nan=True, rtol=rtol, atol=atol,
                                       err_msg=f"[{i}] {func} {dtype} {layout} n={n} ng={ng}")
        else:
            np.testing.assert_array_equal(got[ok], want[ok],
                                          err_msg=f"[{i}] {func} {dtype} {layout} n={n} ng={ng}")
    else:
        func = str(rng.choice(ARG_FUNCS))
        if "nan" in func and dtype == "int64":
            func = func.replace("nan", "")
        fast, *_ = flox_amd.groupby_reduce(vals, labels, func=func,
                                           expected_groups=range(ng))
        old = fa_core.PACKED_ARG_THRESHOLD
        fa_core.PACKED_ARG_THRESHOLD = 10**18  # force the two-pass form
        try:
            slow, *_ = flox_amd.groupby_reduce(vals, labels, func=func,
                                               expected_groups=range(ng))
        finally:
            fa_core.PACKED_ARG_THRESHOLD = old
        np.testing.assert_array_equal(
            np.asarray(fast), np.asarray(slow),
            err_msg=f"[{i}] {func} {dtype} {layout} n={n} ng={ng} (cross-path)")
        # value-at-index must equal the per-group extremum
        base = func.replace("arg", "")
        want_ext, empty = np_reduce(base, vals, labels, ng)
        idx = np.asarray(fast)
        ok = ~empty & (idx >= 0) & (idx < n)
        picked = vals[idx[ok]]
        w = want_ext[ok]
        if vals.dtype.kind == "f":
            np.testing.assert_array_equal(
                np.isnan(picked), np.isnan(w.astype(np.float64)),
                err_msg=f"[{i}] {func} nan-pos")
            fin = ~np.isnan(picked)
            np.testing.assert_array_equal(picked[fin], w[fin].astype(vals.dtype),
                                          err_msg=f"[{i}] {func} value@idx")
        else:
            np.testing.assert_array_equal(picked, w, err_msg=f"[{i}] {func} value@idx")
    dt = time.perf_counter() - t0
    print(f"[{i}] OK {kind}:{func} {dtype} {layout} n={n:.1e} ng={ng:.0e} ({dt:.1f}s)",
          flush=True)


def main():
    seed = int(sys.argv[1]) if len(sys.argv) > 1 else 20260915
    rng = np.random.default_rng(seed)
    for i in range(N_CASES):
        one_case(i, rng)
        torch.cuda.empty_cache()
    print(f"huge fuzz: {N_CASES} cases clean (seed {seed})")


if __name__ == "__main__":
    main()

"""Generate golden parity fixtures from the reference implementation.

Runs ONLY in the build container, where the reference (xarray-contrib/flox)
is mounted read-only at /root/reference. The reference cannot be imported
as-is there (Python 3.10 vs 3.11 syntax; missing numpy_groupies/toolz), so
this script loads it with three mechanical in-memory shims (verified
equivalent in SURVEY.md §8c) and module stubs — nothing from the reference is
ever written into this repository; only the resulting input/output vectors
are committed, as tests/golden/golden_cases.npz.

Usage:  python tests/golden/generate.py
"""

from __future__ import annotations

import os
import re
import sys
import tempfile
import types

import numpy as np

REF = "/root/reference/flox"
OUT = os.path.join(os.path.dirname(__file__), "golden_cases.npz")


def load_reference():
    """Import the reference flox package with mechanical Python-3.10 shims."""
    tmp = tempfile.mkdtemp(prefix="floxref_")
    pkg = os.path.join(tmp, "floxref")
    os.makedirs(pkg)
    for fname in os.listdir(REF):
        if not fname.endswith(".py"):
            continue
        with open(os.path.join(REF, fname)) as f:
            src = f.read()
        # shim 1: PEP-646 star-subscript (aggregations.py:400,408) -> tuple()
        src = src.replace("array[*not_last]", "array[tuple(not_last)]")
        src = src.replace("array[*not_first]", "array[tuple(not_first)]")
        # shim 2: starred return annotation (core.py:754)
        src = re.sub(
            r"-> tuple\[DaskArray, \*tuple\[np\.ndarray \| DaskArray, \.\.\.\]\]:", ":", src
        )
        # shim 3: typing.Self (multiarray.py:2)
        src = src.replace("from typing import Self", "from typing_extensions import Self")
        # intra-package imports: flox.X -> floxref.X
        src = re.sub(r"\bfrom flox(\.|\b)", r"from floxref\1", src)
        src = re.sub(r"\bimport flox\b", "import floxref", src)
        with open(os.path.join(pkg, fname), "w") as f:
            f.write(src)

    # stub the absent hard deps; neither is touched by the eager engine="flox" path
    npg_stub = types.ModuleType("numpy_groupies")

    def _unavailable(*a, **k):  # pragma: no cover
        raise RuntimeError("numpy_groupies stub: not available in this container")

    npg_stub.aggregate_numpy = types.SimpleNamespace(aggregate=_unavailable)
    npg_stub.aggregate_numba = types.SimpleNamespace(aggregate=_unavailable)
    npg_stub.aggregate = _unavailable
    sys.modules.setdefault("numpy_groupies", npg_stub)

    toolz_stub = types.ModuleType("toolz")
    toolz_stub.partition_all = _unavailable
    toolz_stub.unique = _unavailable
    toolz_stub.memoize = lambda f=None, **k: (f if f is not None else (lambda g: g))
    sys.modules.setdefault("toolz", toolz_stub)

    sys.path.insert(0, tmp)
    import floxref  # noqa: F401
    import floxref.core as core

    return core


def gen_cases():
    """Yield (name, kwargs-dict) cases. Mirrors the shapes of the reference's
    own golden tests (test_core.py:126-206) plus seeded random sweeps."""
    rng = np.random.default_rng(42)
    labels_basic = np.array([0, 0, 2, 2, 2, 1, 1, 2, 2, 1, 1, 0])
    nan_labels = labels_basic.astype(float).copy()
    nan_labels[[1, 4, 5]] = np.nan

    funcs = [
        "count", "sum", "nansum", "prod", "nanprod", "mean", "nanmean",
        "var", "nanvar", "std", "nanstd", "min", "nanmin", "max", "nanmax",
    ]

    ones = np.ones((12,))
    vals_f8 = rng.standard_normal(12) * 10
    vals_f4 = vals_f8.astype(np.float32)
    vals_nan = vals_f8.copy()
    vals_nan[[2, 5, 11]] = np.nan
    vals_int = rng.integers(-50, 50, 12).astype(np.int64)
    vals_i32 = vals_int.astype(np.int32)

    for func in funcs:
        yield f"{func}_ones_basic", dict(array=ones, by=labels_basic, func=func)
        yield f"{func}_f64_basic", dict(array=vals_f8, by=labels_basic, func=func)
        yield f"{func}_f32_basic", dict(array=vals_f4, by=labels_basic.copy(), func=func)
        yield f"{func}_f64_nanvals", dict(array=vals_nan, by=labels_basic, func=func)
        yield f"{func}_i64_basic", dict(array=vals_int, by=labels_basic, func=func)
        yield f"{func}_i32_basic", dict(array=vals_i32, by=labels_basic, func=func)
        # NaN in by -> those rows are dropped (factorize.py:201-210)
        yield f"{func}_nanby", dict(array=vals_f8, by=nan_labels, func=func)
        # expected_groups superset with a missing group + fill_value
        yield (
            f"{func}_expected_fill",
            dict(
                array=vals_f8,
                by=labels_basic,
                func=func,
                expected_groups=np.array([0, 1, 2, 5]),
                fill_value=123.0,
            ),
        )
        # expected_groups as a range, labels uniform
        big_by = rng.integers(0, 17, 200)
        big_vals = rng.standard_normal(200)
        yield (
            f"{func}_range17",
            dict(array=big_vals, by=big_by, func=func, expected_groups=np.arange(17)),
        )
        # 2-D array, 1-D by (reduce over trailing axis)
        arr2 = rng.standard_normal((3, 40))
        by2 = rng.integers(0, 5, 40)
        yield f"{func}_2d_lead", dict(array=arr2, by=by2, func=func, expected_groups=np.arange(5))
        # ddof for var/std
        if "var" in func or "std" in func:
            yield (
                f"{func}_ddof1",
                dict(array=vals_f8, by=labels_basic, func=func, finalize_kwargs={"ddof": 1}),
            )

    # small integer / unsigned / half dtypes (the reference promotes before
    # reducing: u* sums -> uint64, i8/i16 -> int64, min/max preserve dtype,
    # mean/var -> float64 for ints and float16 for float16)
    sd_by = rng.integers(0, 5, 64)
    for dt in ["uint8", "uint16", "uint32", "uint64", "int8", "int16", "float16"]:
        if np.dtype(dt).kind == "f":
            sdv = (rng.standard_normal(64) * 10).astype(dt)
        else:
            info = np.iinfo(dt)
            sdv = rng.integers(max(info.min, -120), min(info.max, 250), 64).astype(dt)
        for func in ["sum", "min", "max", "mean", "count", "var", "prod",
                     "first", "last", "median"]:
            yield f"{func}_{dt}", dict(
                array=sdv.copy(), by=sd_by, func=func, expected_groups=np.arange(5)
            )

    # complex values: the linear set (componentwise re/im)
    cv = (rng.standard_normal(80) + 1j * rng.standard_normal(80)).astype(np.complex128)
    cv[rng.random(80) < 0.15] = np.nan
    cv.real[rng.random(80) < 0.05] = np.nan  # partial-NaN values (null per isnull)
    c_by = rng.integers(0, 6, 80)
    for func in ["sum", "nansum", "mean", "nanmean", "count"]:
        yield f"{func}_c128", dict(array=cv.copy(), by=c_by, func=func,
                                   expected_groups=np.arange(6))
        yield f"{func}_c64", dict(array=cv.astype(np.complex64), by=c_by, func=func,
                                  expected_groups=np.arange(6))
    # min_count masking and explicit output dtype
    vmc = rng.standard_normal(60)
    vmc[rng.random(60) < 0.5] = np.nan
    bmc = rng.integers(0, 4, 60)
    yield "nansum_min_count10", dict(
        array=vmc, by=bmc, func="nansum", expected_groups=np.arange(6),
        min_count=10, fill_value=np.nan,
    )
    yield "mean_dtype_f32", dict(
        array=vmc, by=bmc, func="mean", expected_groups=np.arange(4),
        dtype=np.float32,
    )
    # timedelta64 values
    tdv = rng.integers(0, 1000, 60).astype("timedelta64[s]")
    for func in ["sum", "min", "max", "count"]:
        yield f"{func}_timedelta", dict(
            array=tdv, by=bmc, func=func, expected_groups=np.arange(4)
        )

    # multi-by (2-D groupby, like BASELINE config 5)
    by_a = rng.integers(0, 4, 300)
    by_b = rng.integers(0, 6, 300)
    vals = rng.standard_normal(300)
    vals[::17] = np.nan
    for func in ["nanmean", "sum", "count", "nanmax", "nanvar"]:
        yield (
            f"{func}_multi_by",
            dict(
                array=vals,
                by=(by_a, by_b),
                func=func,
                expected_groups=(np.arange(4), np.arange(6)),
            ),
        )
    # quantile family (reference aggregate_flox.py:50-130)
    qvals = rng.standard_normal(500) * 10
    qby = rng.integers(0, 9, 500)
    qnan = qvals.copy()
    qnan[rng.random(500) < 0.1] = np.nan
    yield "median_basic", dict(array=qvals, by=qby, func="median", expected_groups=np.arange(9))
    yield "nanmedian_nan", dict(array=qnan, by=qby, func="nanmedian", expected_groups=np.arange(9))
    yield "quantile_q25", dict(
        array=qvals, by=qby, func="quantile", expected_groups=np.arange(9),
        finalize_kwargs={"q": 0.25},
    )
    yield "quantile_multi", dict(
        array=qvals, by=qby, func="quantile", expected_groups=np.arange(9),
        finalize_kwargs={"q": [0.1, 0.5, 0.9]},
    )
    yield "nanquantile_q75", dict(
        array=qnan, by=qby, func="nanquantile", expected_groups=np.arange(9),
        finalize_kwargs={"q": 0.75},
    )
    yield "quantile_withnan", dict(
        array=qnan, by=qby, func="quantile", expected_groups=np.arange(9),
        finalize_kwargs={"q": 0.5},
    )
    yield "median_f32", dict(
        array=qvals.astype(np.float32), by=qby, func="median", expected_groups=np.arange(9)
    )
    yield "median_int", dict(
        array=rng.integers(-100, 100, 500).astype(np.int64), by=qby, func="median",
        expected_groups=np.arange(9),
    )
    # isbin: values grouped into right-closed bins (reference factorize.py:55-82)
    bvals = rng.standard_normal(400)
    bby = rng.standard_normal(400) * 3
    edges = np.array([-4.0, -1.5, 0.0, 0.5, 2.0, 5.0])
    for func in ["sum", "mean", "count", "nanmax", "var"]:
        yield (
            f"{func}_isbin",
            dict(array=bvals, by=bby, func=func, expected_groups=edges, isbin=True),
        )
    yield "mean_isbin_int", dict(
        array=rng.standard_normal(300),
        by=rng.integers(0, 50, 300).astype(np.float64),
        func="mean",
        expected_groups=np.array([0, 10, 20, 30, 40, 50]).astype(np.float64),
        isbin=True,
    )
    # datetime64 values (viewed as int64 inside the reference, core.py:985-1001)
    tvals = (np.datetime64("2020-01-01") + rng.integers(0, 10_000, 60).astype("timedelta64[m]"))
    tby = rng.integers(0, 5, 60)
    for func in ["min", "max", "count", "first", "nanmax", "last"]:
        yield f"{func}_datetime", dict(array=tvals, by=tby, func=func, expected_groups=np.arange(5))
    # NaT values (int64 min through the view) + an empty trailing group
    tnat = tvals.copy()
    tnat[rng.random(60) < 0.25] = np.datetime64("NaT")
    for func in ["min", "max", "nanmin", "nanmax", "count", "median"]:
        yield f"{func}_datetime_nat", dict(
            array=tnat, by=tby, func=func, expected_groups=np.arange(6)
        )
    # nanfirst/nanlast on datetime with leading dims (NaT skipping is per
    # (lead, row)) and with an axis subset of by's dims
    tv2 = (np.datetime64("2022-03-01") + rng.integers(0, 9000, (3, 50)).astype("timedelta64[m]"))
    tv2[rng.random((3, 50)) < 0.3] = np.datetime64("NaT")
    tb2 = rng.integers(0, 6, 50)
    for func in ["nanfirst", "nanlast", "first", "last"]:
        yield f"{func}_datetime_lead", dict(
            array=tv2, by=tb2, func=func, expected_groups=np.arange(6)
        )
    tv3 = (np.datetime64("2022-03-01") + rng.integers(0, 9000, (4, 5, 30)).astype("timedelta64[m]"))
    tv3[rng.random((4, 5, 30)) < 0.25] = np.datetime64("NaT")
    tb3 = rng.integers(0, 6, (4, 5, 30))
    for func in ["nanfirst", "nanlast"]:
        yield f"{func}_datetime_axis_subset", dict(
            array=tv3, by=tb3, func=func, axis=(2,), expected_groups=np.arange(6)
        )
    # grouping BY datetime labels (int64-view factorize; NaT drops rows;
    # expected datetime groups align to the by's unit)
    dby = (np.datetime64("2022-01-01")
           + rng.integers(0, 6, 300).astype("timedelta64[D]")).astype("datetime64[s]")
    dbyn = dby.copy()
    dbyn[rng.random(300) < 0.15] = np.datetime64("NaT")
    dvals = rng.standard_normal(300)
    for func in ["mean", "sum", "count", "nanmax", "var"]:
        yield f"{func}_dtby", dict(array=dvals, by=dbyn, func=func)
    yield "mean_dtby_expected", dict(
        array=dvals, by=dby, func="mean",
        expected_groups=np.datetime64("2022-01-01") + np.arange(6).astype("timedelta64[D]"),
    )
    yield "median_dtby", dict(array=dvals, by=dbyn, func="median")
    # datetime bin edges (isbin on datetime labels)
    dt_edges = np.datetime64("2022-01-01") + (np.arange(5) * 2).astype("timedelta64[D]")
    for func in ["mean", "count", "sum"]:
        yield f"{func}_dt_isbin", dict(
            array=dvals, by=dbyn, func=func, expected_groups=dt_edges, isbin=True
        )
    # grouping BY string labels (the pd.factorize hash path)
    sb = rng.choice(np.array(["north", "south", "east", "west", "up"]), 250)
    sv2 = rng.standard_normal(250)
    for func in ["mean", "sum", "count", "nanmax", "var"]:
        yield f"{func}_strby", dict(array=sv2, by=sb.copy(), func=func)
    yield "mean_strby_expected", dict(
        array=sv2, by=sb.copy(), func="mean",
        expected_groups=np.array(["east", "north", "zz"]), fill_value=-5.0,
    )
    yield "sum_strby_nosort", dict(array=sv2, by=sb.copy(), func="sum", sort=False)
    # mixed multi-by: string x int
    yield "nanmean_str_int_multiby", dict(
        array=sv2, by=(sb.copy(), rng.integers(0, 4, 250)), func="nanmean",
        expected_groups=(np.array(["east", "north", "south", "up", "west"]),
                         np.arange(4)),
    )
    # pd.IntervalIndex expected_groups (binning without isbin=True)
    import pandas as pd
    iv_vals = rng.standard_normal(300)
    iv_by = rng.standard_normal(300) * 2
    yield "mean_intervalindex", dict(
        array=iv_vals, by=iv_by, func="mean",
        expected_groups=pd.IntervalIndex.from_breaks(np.array([-3.0, -1.0, 0.5, 3.0])),
    )
    # axis subset of by's dims (offset-labels, factorize.py:24-39)
    a3 = rng.standard_normal((4, 5, 30))
    b3 = rng.integers(0, 6, (4, 5, 30))
    for func in ["sum", "mean", "count", "nanmax"]:
        yield (
            f"{func}_axis_subset",
            dict(array=a3, by=b3, func=func, axis=(2,),
                 expected_groups=np.arange(6), fill_value=-99.0),
        )
    yield "mean_axis_subset2", dict(
        array=a3, by=b3, func="mean", axis=(1, 2),
        expected_groups=np.arange(6), fill_value=-99.0,
    )
    # axis subset with extra leading array dims (array.ndim > by.ndim)
    a4 = rng.standard_normal((2, 4, 5, 30))
    b4 = rng.integers(0, 6, (4, 5, 30))
    for func, axs in [("mean", (3,)), ("sum", (2, 3)), ("nanmax", (3,))]:
        yield f"{func}_lead_axis_subset{len(axs)}", dict(
            array=a4, by=b4, func=func, axis=axs,
            expected_groups=np.arange(6), fill_value=-99.0,
        )
    # order funcs with subset + extra lead dims (the composed folds)
    a4n = a4.copy()
    a4n[rng.random(a4.shape) < 0.15] = np.nan
    yield "median_lead_axis_subset", dict(
        array=a4, by=b4, func="median", axis=(3,),
        expected_groups=np.arange(6), fill_value=-99.0,
    )
    yield "nanquantile_lead_axis_subset", dict(
        array=a4n, by=b4, func="nanquantile", axis=(3,),
        expected_groups=np.arange(6), fill_value=-99.0,
        finalize_kwargs={"q": 0.75},
    )
    # sort=False: groups in first-appearance order (factorize.py:96)
    ub = np.array([30, 5, 30, 17, 5, 2, 17, 30, 2, 9])
    uv = rng.standard_normal(10)
    for func in ["sum", "mean", "count"]:
        yield f"{func}_nosort", dict(array=uv, by=ub, func=func, sort=False)
    # sort=False with an axis subset (first-appearance order on the
    # original by layout) and with discovered-group quantiles
    ub3 = rng.choice([7, 3, 9, 1], (4, 5, 30))
    yield "mean_nosort_subset", dict(
        array=rng.standard_normal((4, 5, 30)), by=ub3, func="mean",
        axis=(1,), sort=False, fill_value=-7.0,
    )
    yield "quantile_nosort", dict(
        array=rng.standard_normal(80), by=rng.choice([7, 3, 9, 1], 80),
        func="quantile", sort=False, finalize_kwargs={"q": 0.5},
    )
    # empty groups at the tail of the range
    yield "mean_sparse_groups", dict(
        array=rng.standard_normal(50),
        by=rng.integers(0, 3, 50),
        func="mean",
        expected_groups=np.arange(10),
    )
    yield "sum_sparse_groups_fill", dict(
        array=rng.standard_normal(50),
        by=rng.integers(0, 3, 50),
        func="sum",
        expected_groups=np.arange(10),
        fill_value=-1.0,
    )
    # order-dependent reductions with leading array dims (the lead-fold path)
    al = rng.standard_normal((3, 4, 60))
    al[rng.random(al.shape) < 0.2] = np.nan
    bl = rng.integers(0, 5, 60)
    yield "median_lead", dict(array=al, by=bl, func="median", expected_groups=np.arange(5))
    yield "nanquantile_lead", dict(
        array=al, by=bl, func="nanquantile", expected_groups=np.arange(5),
        finalize_kwargs={"q": 0.7},
    )
    yield "quantile_lead_vec", dict(
        array=al, by=bl, func="quantile", expected_groups=np.arange(5),
        finalize_kwargs={"q": [0.25, 0.75]},
    )
    # size-1 by dims broadcast against the array's trailing dims (the
    # dim=... case, reference core.py:300-309)
    ab = rng.standard_normal((30, 40))
    bb = rng.integers(0, 5, (1, 40))
    for func in ["sum", "mean", "nanmax"]:
        yield f"{func}_by_size1_bcast", dict(
            array=ab, by=bb, func=func, expected_groups=np.arange(5)
        )
    yield "mean_by_size1_bcast3d", dict(
        array=rng.standard_normal((6, 10, 20)),
        by=rng.integers(0, 4, (10, 1)),
        func="mean",
        expected_groups=np.arange(4),
    )


def gen_scan_cases():
    rng = np.random.default_rng(77)
    # reference scan.py:286-291 identity shortcut: length-1 trailing axis /
    # all-distinct 1-D by return the input unchanged — nancumsum of a NaN
    # row stays NaN instead of the identity 0 (fuzz seed 606162)
    yield "scan_nancumsum_quirklen1", dict(
        array=np.array([np.nan]), by=np.array([0.0]), func="nancumsum")
    yield "scan_nancumsum_quirkdistinct", dict(
        array=np.array([np.nan, 2.0, np.nan, 7.5]),
        by=np.array([4, 1, 9, 2]), func="nancumsum")
    yield "scan_nancumsum_quirku8", dict(
        array=np.arange(5).astype(np.uint8),
        by=np.array([3, 1, 4, 0, 2]), func="nancumsum")
    vals = rng.standard_normal(200)
    vals[rng.random(200) < 0.15] = np.nan
    by = rng.integers(0, 7, 200)
    for func in ["cumsum", "nancumsum", "ffill", "bfill"]:
        yield f"scan_{func}_f64", dict(array=vals, by=by, func=func)
        yield f"scan_{func}_f32", dict(array=vals.astype(np.float32), by=by, func=func)
    ints = rng.integers(-50, 50, 200).astype(np.int32)
    yield "scan_cumsum_i32", dict(array=ints, by=by, func="cumsum")
    yield "scan_cumsum_u8", dict(
        array=rng.integers(0, 250, 200).astype(np.uint8), by=by, func="cumsum")
    yield "scan_cumsum_i16", dict(
        array=rng.integers(-120, 120, 200).astype(np.int16), by=by, func="cumsum")
    f16 = rng.standard_normal(200).astype(np.float16)
    f16[rng.random(200) < 0.2] = np.nan
    yield "scan_ffill_f16", dict(array=f16, by=by, func="ffill")
    yield "scan_cumsum_f16", dict(array=f16, by=by, func="cumsum")
    cvs = (rng.standard_normal(200) + 1j * rng.standard_normal(200)).astype(np.complex128)
    cvs[rng.random(200) < 0.2] = np.nan
    for func in ["cumsum", "nancumsum", "ffill", "bfill"]:
        yield f"scan_{func}_c128", dict(array=cvs.copy(), by=by, func=func)
    # NaN labels -> sentinel group scans together
    nby = by.astype(float)
    nby[rng.random(200) < 0.1] = np.nan
    yield "scan_nancumsum_nanby", dict(array=vals, by=nby, func="nancumsum")
    # datetime values: int64 view, NaT passes through as a plain value
    tv = (np.datetime64("2021-06-01") + rng.integers(0, 5000, 120).astype("timedelta64[m]"))
    tv[rng.random(120) < 0.2] = np.datetime64("NaT")
    tb = rng.integers(0, 5, 120)
    for func in ["ffill", "bfill", "cumsum"]:
        yield f"scan_{func}_datetime", dict(array=tv, by=tb, func=func)
    # leading array dims: each column scans independently along the last axis
    a2 = rng.standard_normal((4, 120))
    a2[rng.random(a2.shape) < 0.25] = np.nan
    b2 = rng.integers(0, 6, 120)
    for func in ["cumsum", "ffill", "bfill"]:
        yield f"scan_{func}_lead", dict(array=a2, by=b2, func=func)


def main():
    core = load_reference()
    import importlib

    refscan = importlib.import_module("floxref.scan")
    out = {}
    n_done, n_skip = 0, 0
    # custom Aggregation (docs 'Custom Aggregations'): pin the reference's
    # chunk/combine/finalize execution; the loader reconstructs the same
    # CustomAggregation on the flox_amd side (tests/golden_util.py)
    refaggs = importlib.import_module("floxref.aggregations")
    c_rng = np.random.default_rng(99)
    c_vals = c_rng.standard_normal(250)
    c_by = c_rng.integers(0, 7, 250)
    c_agg = refaggs.Aggregation(
        name="custommean", numpy="mean", chunk=("sum", "nanlen"),
        combine=("sum", "sum"), finalize=lambda s, c: s / c,
        fill_value=0, final_fill_value=np.nan)
    try:
        c_res, c_grp = core.groupby_reduce(
            c_vals, c_by, func=c_agg, engine="flox", expected_groups=np.arange(9))
        out["customagg_mean::result"] = np.asarray(c_res)
        out["customagg_mean::groups0"] = np.asarray(c_grp)
        out["customagg_mean::array"] = c_vals
        out["customagg_mean::by0"] = c_by
        out["customagg_mean::expected0"] = np.arange(9)
        out["customagg_mean::customagg"] = np.asarray(True)
        n_done += 1
    except Exception as e:  # pragma: no cover
        print(f"SKIP customagg_mean: {type(e).__name__}: {e}")
        n_skip += 1
    for name, kw in gen_cases():
        by = kw.pop("by")
        bys = by if isinstance(by, tuple) else (by,)
        try:
            result, *groups = core.groupby_reduce(
                kw.pop("array1") if "array1" in kw else kw.pop("array"),
                *bys,
                engine="flox",
                **kw,
            )
        except Exception as e:  # pragma: no cover
            print(f"SKIP {name}: {type(e).__name__}: {e}")
            n_skip += 1
            continue
        # store inputs + outputs
        case = dict(kw)
        arr = None
        out[f"{name}::result"] = np.asarray(result)
        for i, g in enumerate(groups):
            ga = np.asarray(g)
            if ga.dtype != object:  # interval groups are pinned via the edges
                out[f"{name}::groups{i}"] = ga
        n_done += 1
    # re-run to also store the inputs (gen_cases is deterministic)
    for name, kw in gen_cases():
        if f"{name}::result" not in out:
            continue
        by = kw.pop("by")
        bys = by if isinstance(by, tuple) else (by,)
        out[f"{name}::array"] = np.asarray(kw.pop("array"))
        for i, b in enumerate(bys):
            out[f"{name}::by{i}"] = np.asarray(b)
        eg = kw.pop("expected_groups", None)
        if eg is not None:
            egs = eg if isinstance(eg, tuple) else (eg,)
            for i, e in enumerate(egs):
                import pandas as pd
                if isinstance(e, pd.IntervalIndex):
                    out[f"{name}::expected{i}"] = np.append(e.left.to_numpy(), e.right.to_numpy()[-1])
                    out[f"{name}::interval{i}"] = np.asarray(True)
                else:
                    out[f"{name}::expected{i}"] = np.asarray(e)
        if kw.get("isbin"):
            out[f"{name}::isbin"] = np.asarray(True)
        if kw.get("sort") is False:
            out[f"{name}::nosort"] = np.asarray(True)
        if kw.get("axis") is not None:
            out[f"{name}::axis"] = np.asarray(kw["axis"])
        if kw.get("fill_value") is not None:
            out[f"{name}::fill_value"] = np.asarray(kw["fill_value"])
        if kw.get("min_count") is not None:
            out[f"{name}::min_count"] = np.asarray(kw["min_count"])
        if kw.get("dtype") is not None:
            out[f"{name}::dtype_s"] = np.asarray(np.dtype(kw["dtype"]).str)
        if kw.get("finalize_kwargs"):
            fk = kw["finalize_kwargs"]
            if "ddof" in fk:
                out[f"{name}::ddof"] = np.asarray(fk["ddof"])
            if "q" in fk:
                out[f"{name}::q"] = np.asarray(fk["q"])
                out[f"{name}::q_scalar"] = np.asarray(np.isscalar(fk["q"]))
    for name, kw in gen_scan_cases():
        by = kw.pop("by")
        bys = by if isinstance(by, tuple) else (by,)
        arr = kw.pop("array")
        try:
            result = refscan.groupby_scan(arr, *bys, **kw)
        except Exception as e:  # pragma: no cover
            print(f"SKIP {name}: {type(e).__name__}: {e}")
            n_skip += 1
            continue
        out[f"{name}::result"] = np.asarray(result)
        out[f"{name}::array"] = np.asarray(arr)
        for i, b in enumerate(bys):
            out[f"{name}::by{i}"] = np.asarray(b)
        out[f"{name}::scan"] = np.asarray(True)
        n_done += 1
    np.savez_compressed(OUT, **out)
    print(f"wrote {OUT}: {n_done} cases ({n_skip} skipped)")


if __name__ == "__main__":
    main()

"""Property-based parity tests (the reference's test_properties.py strategy):
randomized arrays/labels through the product API vs the pinned oracle, plus
structural identities that need no oracle."""

import os

import numpy as np
import pytest
from hypothesis import HealthCheck, given, settings
from hypothesis import strategies as st

import flox_amd
from oracle import groupby_reduce as oracle_reduce
from oracle import groupby_scan as oracle_scan

pytestmark = pytest.mark.gpu

SETTINGS = dict(
    max_examples=int(os.environ.get("HYP_EXAMPLES", "25")),
    deadline=None,
    suppress_health_check=[HealthCheck.too_slow, HealthCheck.data_too_large],
)

array_strat = st.builds(
    lambda seed, n, scale, nanfrac, dt: _make(seed, n, scale, nanfrac, dt),
    seed=st.integers(0, 2**31 - 1),
    n=st.integers(1, 5000),
    scale=st.sampled_from([1.0, 100.0, 1e6]),
    nanfrac=st.sampled_from([0.0, 0.05, 0.5]),
    dt=st.sampled_from(["float32", "float64", "int64"]),
)


def _make(seed, n, scale, nanfrac, dt):
    rng = np.random.default_rng(seed)
    if np.dtype(dt).kind == "f":
        v = (rng.standard_normal(n) * scale).astype(dt)
        v[rng.random(n) < nanfrac] = np.nan
    else:
        v = rng.integers(-1000, 1000, n).astype(dt)
    ng = int(rng.integers(1, 50))
    labels = rng.integers(0, ng, n)
    return v, labels, ng


def _tol(dtype):
    if np.dtype(dtype).kind in "iub":
        return dict(rtol=0, atol=0)
    if np.dtype(dtype).itemsize == 4:
        return dict(rtol=1e-4, atol=1e-4)
    return dict(rtol=1e-9, atol=1e-9)


@settings(**SETTINGS)
@given(data=array_strat, func=st.sampled_from(
    ["sum", "nansum", "mean", "nanmean", "count", "min", "nanmin", "max",
     "nanmax", "var", "nanvar", "first", "nanlast", "argmax", "nanargmin"]))
def test_reduce_matches_oracle(data, func):
    vals, labels, ng = data
    want, *_ = oracle_reduce(vals, labels, func=func, expected_groups=np.arange(ng))
    got, *_ = flox_amd.groupby_reduce(vals, labels, func=func, expected_groups=np.arange(ng))
    assert got.dtype == want.dtype, (func, got.dtype, want.dtype)
    # relative tolerance scaled by the group magnitude for fp cancellation
    atol_extra = 1e-9 * float(np.nansum(np.abs(vals.astype("f8")))) if vals.dtype.kind == "f" else 0
    tol = _tol(want.dtype)
    np.testing.assert_allclose(got, want, equal_nan=True,
                               rtol=tol["rtol"], atol=tol["atol"] + atol_extra)


@settings(**SETTINGS)
@given(data=array_strat, func=st.sampled_from(["cumsum", "nancumsum", "ffill", "bfill"]))
def test_scan_matches_oracle(data, func):
    vals, labels, ng = data
    want = oracle_scan(vals, labels, func=func, expected_groups=np.arange(ng))
    got = flox_amd.groupby_scan(vals, labels, func=func, expected_groups=np.arange(ng))
    assert got.dtype == want.dtype
    atol_extra = 1e-4 * float(np.nanmax(np.abs(vals.astype("f8"))) + 1) if vals.dtype.kind == "f" else 0
    tol = _tol(want.dtype)
    np.testing.assert_allclose(got, want, equal_nan=True,
                               rtol=tol["rtol"], atol=tol["atol"] + atol_extra)


@settings(**SETTINGS)
@given(data=array_strat)
def test_bfill_is_reversed_ffill(data):
    """reference test_properties.py ffill/bfill reversal identity."""
    vals, labels, ng = data
    b = flox_amd.groupby_scan(vals, labels, func="bfill", expected_groups=np.arange(ng))
    f = flox_amd.groupby_scan(vals[::-1].copy(), labels[::-1].copy(), func="ffill",
                              expected_groups=np.arange(ng))
    np.testing.assert_array_equal(np.asarray(b), np.asarray(f)[::-1])


@settings(**SETTINGS)
@given(
    seed=st.integers(0, 2**31 - 1),
    n_t=st.integers(1, 400),
    m1=st.integers(1, 12),
    m2=st.integers(1, 40),
    func=st.sampled_from(["sum", "mean", "nanmean", "var", "nanvar", "count", "min", "nanmax"]),
)
def test_lead_dims_match_oracle(seed, n_t, m1, m2, func):
    """Column path (leading array dims) vs oracle on random shapes."""
    rng = np.random.default_rng(seed)
    arr = rng.standard_normal((m1, m2, n_t))
    arr[rng.random(arr.shape) < 0.05] = np.nan
    ng = int(rng.integers(1, 30))
    labels = rng.integers(0, ng, n_t)
    want, *_ = oracle_reduce(arr, labels, func=func, expected_groups=np.arange(ng))
    got, *_ = flox_amd.groupby_reduce(arr, labels, func=func, expected_groups=np.arange(ng))
    assert got.shape == want.shape and got.dtype == want.dtype
    np.testing.assert_allclose(got, want, equal_nan=True, rtol=1e-9, atol=1e-9)


@settings(**SETTINGS)
@given(data=array_strat)
def test_single_group_equals_numpy(data):
    """reference test_properties.py:93-176: one group -> plain numpy."""
    vals, _, _ = data
    labels = np.zeros(len(vals), dtype=np.int64)
    for func, npf in [("sum", np.sum), ("nanmax", np.nanmax), ("mean", np.mean)]:
        if vals.dtype.kind == "f" and np.isnan(vals).all() and func == "nanmax":
            continue
        got, *_ = flox_amd.groupby_reduce(vals, labels, func=func, expected_groups=np.arange(1))
        want = npf(vals.astype(np.float64 if vals.dtype.kind == "f" else vals.dtype))
        atol = 1e-9 * (float(np.nansum(np.abs(vals.astype("f8")))) + 1)
        np.testing.assert_allclose(float(got[0]), float(want), rtol=1e-6, atol=atol, equal_nan=True)


@settings(**SETTINGS)
@given(data=array_strat, func=st.sampled_from(
    ["argmin", "argmax", "nanargmin", "nanargmax", "first", "last",
     "nanfirst", "nanlast"]))
def test_packed_equals_two_pass(data, func):
    """The packed-key form and the two-pass (extremum + index-match) form of
    the order reductions must agree bit-for-bit on any input."""
    import flox_amd.core as core

    vals, labels, ng = data
    if vals.dtype == np.int64 and func.startswith(("arg", "nanarg")):
        vals = vals.astype(np.int32)  # packed args cover f32/i32
    old = core.PACKED_ARG_THRESHOLD
    try:
        core.PACKED_ARG_THRESHOLD = 1
        a, *_ = flox_amd.groupby_reduce(vals, labels, func=func, expected_groups=np.arange(ng))
        core.PACKED_ARG_THRESHOLD = 1 << 62
        b, *_ = flox_amd.groupby_reduce(vals, labels, func=func, expected_groups=np.arange(ng))
    finally:
        core.PACKED_ARG_THRESHOLD = old
    np.testing.assert_array_equal(np.asarray(a), np.asarray(b))


@settings(**SETTINGS)
@given(
    seed=st.integers(0, 2**31 - 1),
    n=st.integers(1, 3000),
    natfrac=st.sampled_from([0.0, 0.2, 0.9]),
    func=st.sampled_from(["min", "max", "nanmin", "nanmax", "count",
                          "first", "nanlast", "median"]),
)
def test_datetime_matches_oracle(seed, n, natfrac, func):
    rng = np.random.default_rng(seed)
    vals = np.datetime64("2020-01-01") + rng.integers(0, 10**6, n).astype("timedelta64[s]")
    vals[rng.random(n) < natfrac] = np.datetime64("NaT")
    ng = int(rng.integers(1, 20))
    labels = rng.integers(0, ng, n)
    want, *_ = oracle_reduce(vals, labels, func=func, expected_groups=np.arange(ng))
    got, *_ = flox_amd.groupby_reduce(vals, labels, func=func, expected_groups=np.arange(ng))
    assert got.dtype == want.dtype
    np.testing.assert_array_equal(got, want)


@settings(**SETTINGS)
@given(
    seed=st.integers(0, 2**31 - 1),
    m=st.integers(1, 8),
    n=st.integers(1, 800),
    func=st.sampled_from(["argmax", "nanargmin", "first", "nanlast", "median", "cumsum", "ffill"]),
)
def test_lead_dims_order_funcs_match_oracle(seed, m, n, func):
    rng = np.random.default_rng(seed)
    arr = rng.standard_normal((m, n))
    arr[rng.random(arr.shape) < 0.2] = np.nan
    ng = int(rng.integers(1, 15))
    labels = rng.integers(0, ng, n)
    eg = np.arange(ng)
    if func in ("cumsum", "ffill"):
        want = oracle_scan(arr, labels, func=func, expected_groups=eg)
        got = flox_amd.groupby_scan(arr, labels, func=func, expected_groups=eg)
    else:
        want, *_ = oracle_reduce(arr, labels, func=func, expected_groups=eg)
        got, *_ = flox_amd.groupby_reduce(arr, labels, func=func, expected_groups=eg)
    atol = 1e-9 * (float(np.nansum(np.abs(arr))) + 1)
    np.testing.assert_allclose(np.asarray(got), want, equal_nan=True, rtol=1e-9, atol=atol)

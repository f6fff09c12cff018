"""Double-entry bookkeeping for the oracle's numpy-semantics funcs.

arg/first/last/any/all/mode cannot be pinned against the reference in this
container (its engine="flox" lacks them and numpy_groupies is absent —
SURVEY.md §8c), so their oracle semantics anchor on plain numpy per-group
behaviour, exactly as the reference's own tests do
(test_core.py:222-385: expected = getattr(np, func)(array[..., ~nanmask],
axis=-1) on single-group data). These hypothesis tests state that anchor
executably, on single-group and multi-group data."""

import numpy as np
import pytest
from hypothesis import HealthCheck, given, settings
from hypothesis import strategies as st

from oracle import groupby_reduce as oracle_reduce

SETTINGS = dict(
    max_examples=60,
    deadline=None,
    suppress_health_check=[HealthCheck.too_slow],
)


def _arr(seed, n, nanfrac, dt):
    rng = np.random.default_rng(seed)
    if np.dtype(dt).kind == "f":
        v = (rng.standard_normal(n) * 100).astype(dt)
        v[rng.random(n) < nanfrac] = np.nan
        if np.isnan(v).all():
            v[0] = 1.0  # keep nan* funcs defined
    else:
        v = rng.integers(-50, 50, n).astype(dt)
    return v


arrays = st.builds(
    _arr,
    seed=st.integers(0, 2**31 - 1),
    n=st.integers(1, 400),
    nanfrac=st.sampled_from([0.0, 0.1, 0.6]),
    dt=st.sampled_from(["float64", "float32", "int64"]),
)


@settings(**SETTINGS)
@given(v=arrays)
def test_single_group_matches_numpy(v):
    by = np.zeros(len(v), dtype=np.int64)
    for func, npf in [
        ("argmax", np.argmax), ("argmin", np.argmin),
        ("first", lambda a: a[0]), ("last", lambda a: a[-1]),
        ("sum", np.sum), ("max", np.max), ("mean", np.mean),
    ]:
        got, _ = oracle_reduce(v, by, func=func, expected_groups=np.arange(1))
        want = npf(v)
        np.testing.assert_allclose(np.asarray(got).reshape(()), want,
                                   equal_nan=True, rtol=2e-6 if v.dtype.itemsize == 4 else 1e-12,
                                   atol=1e-5 if v.dtype.itemsize == 4 else 1e-12,
                                   err_msg=func)
    if v.dtype.kind == "f" and not np.isnan(v).all():
        for func, npf in [
            ("nanargmax", np.nanargmax), ("nanargmin", np.nanargmin),
            ("nanfirst", lambda a: a[~np.isnan(a)][0] if (~np.isnan(a)).any() else np.nan),
            ("nanlast", lambda a: a[~np.isnan(a)][-1] if (~np.isnan(a)).any() else np.nan),
        ]:
            got, _ = oracle_reduce(v, by, func=func, expected_groups=np.arange(1))
            np.testing.assert_allclose(np.asarray(got).reshape(()), npf(v),
                                       equal_nan=True, rtol=0, atol=0, err_msg=func)


@settings(**SETTINGS)
@given(v=arrays, ng=st.integers(1, 17), seed=st.integers(0, 2**31 - 1))
def test_multi_group_matches_per_group_numpy(v, ng, seed):
    rng = np.random.default_rng(seed)
    by = rng.integers(0, ng, len(v))
    bools = (np.nan_to_num(v) > 0)
    for func in ["argmax", "first", "nanlast", "any", "all"]:
        arr = bools if func in ("any", "all") else v
        if func == "nanlast" and v.dtype.kind != "f":
            continue
        got, _ = oracle_reduce(arr, by, func=func, expected_groups=np.arange(ng))
        got = np.asarray(got)
        for g in range(ng):
            rows = np.flatnonzero(by == g)
            if rows.size == 0:
                continue
            sub = arr[rows]
            if func == "argmax":
                want = rows[np.argmax(sub)]
            elif func == "first":
                want = sub[0]
            elif func == "nanlast":
                ok = ~np.isnan(sub)
                if not ok.any():
                    continue
                want = sub[ok][-1]
            elif func == "any":
                want = bool(np.any(sub))
            else:
                want = bool(np.all(sub))
            np.testing.assert_allclose(np.asarray(got[g], dtype=np.float64),
                                       np.float64(want), equal_nan=True,
                                       rtol=0, atol=0, err_msg=f"{func} g={g}")


@settings(**SETTINGS)
@given(v=arrays, ng=st.integers(1, 9), seed=st.integers(0, 2**31 - 1))
def test_permutation_invariance_commutative(v, ng, seed):
    """sum/min/max/count are row-order invariant (the combine-tree property
    the multi-GPU all-reduce depends on)."""
    rng = np.random.default_rng(seed)
    by = rng.integers(0, ng, len(v))
    perm = rng.permutation(len(v))
    for func in ["nansum", "min", "nanmax", "count"]:
        a, _ = oracle_reduce(v, by, func=func, expected_groups=np.arange(ng))
        b, _ = oracle_reduce(v[perm], by[perm], func=func, expected_groups=np.arange(ng))
        np.testing.assert_allclose(np.asarray(a, dtype=np.float64),
                                   np.asarray(b, dtype=np.float64),
                                   equal_nan=True, rtol=1e-12, atol=1e-9,
                                   err_msg=func)

"""GPU parity tests: the HIP engine vs the pinned oracle and the golden
fixtures, through the product API and through the engine seam."""

import zlib

import numpy as np
import pytest
import torch

import flox_amd
from oracle import groupby_reduce as oracle_reduce
from tests.golden_util import load_golden_cases, tolerance_for

pytestmark = pytest.mark.gpu

CASES = list(load_golden_cases())


@pytest.mark.parametrize("name,inputs,expected,groups", CASES, ids=[c[0] for c in CASES])
def test_golden_parity(name, inputs, expected, groups):
    kw = dict(inputs)
    arr = kw.pop("array")
    bys = kw.pop("by")
    if kw.pop("_scan", False):
        result = flox_amd.groupby_scan(arr, *bys, **kw)
        found = []
    else:
        result, *found = flox_amd.groupby_reduce(arr, *bys, **kw)
    assert result.shape == expected.shape
    assert result.dtype == expected.dtype, (result.dtype, expected.dtype)
    if expected.dtype.kind in "Mm":
        np.testing.assert_array_equal(result, expected)
    else:
        tol = tolerance_for(name, expected.dtype)
        np.testing.assert_allclose(result, expected, equal_nan=True, **tol)


FUNCS = [
    "count", "sum", "nansum", "mean", "nanmean", "var", "nanvar",
    "std", "nanstd", "min", "nanmin", "max", "nanmax",
]


def _tol(func, dtype):
    if np.dtype(dtype).kind in "iu":
        return dict(rtol=0, atol=0)
    if np.dtype(dtype).itemsize == 4:
        return dict(rtol=3e-6, atol=1e-6)
    # atol absorbs fp-cancellation under differing summation order (group
    # sums near zero from O(100)-magnitude addends)
    return dict(rtol=1e-12, atol=1e-9)


@pytest.mark.parametrize("func", FUNCS)
@pytest.mark.parametrize("dtype", ["float32", "float64", "int64"])
@pytest.mark.parametrize("ngroups", [7, 1000, 40_000])  # LDS path and global-atomic path
def test_random_sweep_vs_oracle(func, dtype, ngroups):
    rng = np.random.default_rng(zlib.crc32(f"{func}-{dtype}-{ngroups}".encode()))
    n = 200_000
    labels = rng.integers(0, ngroups, n)
    if np.dtype(dtype).kind == "f":
        vals = (rng.standard_normal(n) * 100).astype(dtype)
        vals[rng.random(n) < 0.03] = np.nan
    else:
        vals = rng.integers(-1000, 1000, n).astype(dtype)
    want, *_ = oracle_reduce(vals, labels, func=func, expected_groups=np.arange(ngroups))
    got, *_ = flox_amd.groupby_reduce(vals, labels, func=func, expected_groups=np.arange(ngroups))
    assert got.dtype == want.dtype
    np.testing.assert_allclose(got, want, equal_nan=True, **_tol(func, want.dtype))


@pytest.mark.parametrize("force_path", [1, 2])  # 1=LDS, 2=global-atomic
def test_both_kernel_paths_agree(force_path):
    from flox_amd.aggregate_hip import grouped_partials
    from flox_amd import _ffi

    rng = np.random.default_rng(7)
    n, ng = 500_000, 513
    vals = torch.tensor(rng.standard_normal(n), dtype=torch.float32, device="cuda")
    labels = torch.tensor(rng.integers(0, ng, n), device="cuda")
    p = grouped_partials(_ffi.SET_SUM_COUNT, vals, labels, ng, force_path=force_path)
    assert p["_path"] == force_path
    want_cnt = np.bincount(labels.cpu().numpy(), minlength=ng)
    np.testing.assert_array_equal(p["count"].cpu().numpy(), want_cnt)
    want_sum = np.bincount(labels.cpu().numpy(), weights=vals.cpu().numpy().astype(np.float64), minlength=ng)
    np.testing.assert_allclose(p["sum"].cpu().numpy(), want_sum, rtol=1e-12)


EXT_FUNCS = ["argmax", "argmin", "nanargmax", "nanargmin", "first", "last", "nanfirst", "nanlast"]


@pytest.mark.parametrize("func", EXT_FUNCS)
@pytest.mark.parametrize("dtype", ["float32", "float64", "int64"])
@pytest.mark.parametrize("ngroups", [7, 500, 20_000])
def test_arg_first_last_vs_oracle(func, dtype, ngroups):
    rng = np.random.default_rng(zlib.crc32(f"{func}-{dtype}-{ngroups}".encode()))
    n = 60_000
    labels = rng.integers(0, ngroups, n)
    if np.dtype(dtype).kind == "f":
        vals = (rng.standard_normal(n) * 100).astype(dtype)
        vals[rng.random(n) < 0.03] = np.nan
    else:
        vals = rng.integers(-1000, 1000, n).astype(dtype)
    want, *_ = oracle_reduce(vals, labels, func=func, expected_groups=np.arange(ngroups))
    got, *_ = flox_amd.groupby_reduce(vals, labels, func=func, expected_groups=np.arange(ngroups))
    assert got.dtype == want.dtype, (got.dtype, want.dtype)
    np.testing.assert_allclose(got, want, equal_nan=True, rtol=0, atol=0)


@pytest.mark.parametrize("func", ["median", "nanmedian", "quantile", "nanquantile"])
@pytest.mark.parametrize("dtype", ["float32", "float64", "int64"])
def test_quantile_family_vs_oracle(func, dtype):
    rng = np.random.default_rng(zlib.crc32(f"q-{func}-{dtype}".encode()))
    n, ng = 100_000, 257
    labels = rng.integers(0, ng, n)
    if np.dtype(dtype).kind == "f":
        vals = (rng.standard_normal(n) * 100).astype(dtype)
        vals[rng.random(n) < 0.02] = np.nan
    else:
        vals = rng.integers(-1000, 1000, n).astype(dtype)
    fk = {"q": [0.1, 0.5, 0.9]} if "quantile" in func else None
    want, *_ = oracle_reduce(vals, labels, func=func, expected_groups=np.arange(ng), finalize_kwargs=fk)
    got, *_ = flox_amd.groupby_reduce(vals, labels, func=func, expected_groups=np.arange(ng), finalize_kwargs=fk)
    assert got.dtype == want.dtype and got.shape == want.shape
    tol = dict(rtol=3e-6, atol=1e-5) if np.dtype(dtype).itemsize == 4 and np.dtype(dtype).kind == "f" else dict(rtol=1e-12, atol=1e-12)
    np.testing.assert_allclose(got, want, equal_nan=True, **tol)


@pytest.mark.parametrize("func", ["mode", "nanmode"])
@pytest.mark.parametrize("dtype", ["float32", "float64", "int64"])
def test_mode_vs_oracle(func, dtype):
    """Oracle calls scipy.stats.mode — the exact function the reference's
    mode wraps (aggregate_npg.py:185-215)."""
    rng = np.random.default_rng(zlib.crc32(f"m-{func}-{dtype}".encode()))
    n, ng = 50_000, 97
    labels = rng.integers(0, ng, n)
    if np.dtype(dtype).kind == "f":
        vals = rng.integers(-20, 20, n).astype(dtype)  # repeats so modes exist
        vals[rng.random(n) < 0.02] = np.nan
    else:
        vals = rng.integers(-20, 20, n).astype(dtype)
    want, *_ = oracle_reduce(vals, labels, func=func, expected_groups=np.arange(ng))
    got, *_ = flox_amd.groupby_reduce(vals, labels, func=func, expected_groups=np.arange(ng))
    assert got.dtype == want.dtype
    np.testing.assert_allclose(got, want, equal_nan=True, rtol=0, atol=0)


@pytest.mark.parametrize("func", ["cumsum", "nancumsum", "ffill", "bfill"])
@pytest.mark.parametrize("dtype", ["float32", "float64", "int64"])
def test_scan_vs_oracle(func, dtype):
    from oracle import groupby_scan as oracle_scan

    rng = np.random.default_rng(zlib.crc32(f"s-{func}-{dtype}".encode()))
    n, ng = 200_000, 512
    labels = rng.integers(0, ng, n)
    if np.dtype(dtype).kind == "f":
        vals = (rng.standard_normal(n)).astype(dtype)
        vals[rng.random(n) < 0.05] = np.nan
    else:
        vals = rng.integers(-100, 100, n).astype(dtype)
    want = oracle_scan(vals, labels, func=func, expected_groups=np.arange(ng))
    got = flox_amd.groupby_scan(vals, labels, func=func, expected_groups=np.arange(ng))
    assert got.dtype == want.dtype, (got.dtype, want.dtype)
    # fp32 cumsums: the device scan's tree association differs from the
    # sequential oracle by ~ulp x running-sum magnitude
    tol = dict(rtol=0, atol=0) if np.dtype(dtype).kind in "iu" else (
        dict(rtol=1e-4, atol=1e-4) if np.dtype(dtype).itemsize == 4 else dict(rtol=1e-12, atol=1e-10)
    )
    np.testing.assert_allclose(got, want, equal_nan=True, **tol)


def test_any_all_bool():
    rng = np.random.default_rng(17)
    n, ng = 100_000, 300
    b = rng.random(n) < 0.02
    labels = rng.integers(0, ng, n)
    for func in ("any", "all"):
        want, *_ = oracle_reduce(b, labels, func=func, expected_groups=np.arange(ng))
        got, *_ = flox_amd.groupby_reduce(b, labels, func=func, expected_groups=np.arange(ng))
        assert got.dtype == np.dtype(bool)
        np.testing.assert_array_equal(got, want)


def test_any_all_seam_float_nan_truthy():
    """Through the generic_aggregate seam any/all accept float input; NaN is
    truthy (np.any of a NaN-containing group is True) — a raw float->int
    cast would turn NaN into 0 on the GPU."""
    from flox_amd import generic_aggregate

    rng = np.random.default_rng(41)
    n, ng = 20_000, 64
    g = rng.integers(0, ng, n)
    a = (rng.random(n) < 0.05).astype(np.float64)  # mostly 0.0
    a[rng.random(n) < 0.03] = np.nan
    for func in ("any", "all"):
        got = np.asarray(generic_aggregate(g, a, engine="hip", func=func, size=ng))
        red = np.logical_or if func == "any" else np.logical_and
        want = np.array([
            red.reduce((a[g == i] != 0) | np.isnan(a[g == i])) if (g == i).any() else False
            for i in range(ng)
        ])
        np.testing.assert_array_equal(got, want, err_msg=func)


def test_partition_overflow_falls_back_exact():
    """Extreme label skew overflows the optimistic capacity regions; the
    exact counted path must kick in and produce correct results."""
    rng = np.random.default_rng(99)
    n, ng = 2_000_000, 2_000_000
    labels = np.zeros(n, dtype=np.int64)          # everything in group 0
    labels[: n // 100] = rng.integers(0, ng, n // 100)  # a sprinkle elsewhere
    vals = rng.standard_normal(n)
    want, *_ = oracle_reduce(vals, labels, func="sum", expected_groups=np.arange(ng))
    got, *_ = flox_amd.groupby_reduce(vals, labels, func="sum", expected_groups=np.arange(ng))
    np.testing.assert_allclose(got, want, equal_nan=True, rtol=1e-10, atol=1e-6)


def test_isbin_with_lead_dims():
    """Binned grouping through the column path (binned climatology)."""
    rng = np.random.default_rng(31)
    n_t, y = 500, 64
    arr = rng.standard_normal((y, n_t))
    by = rng.standard_normal(n_t) * 2
    edges = np.array([-3.0, -1.0, 0.0, 1.0, 3.0])
    want, *_ = oracle_reduce(arr, by, func="mean", expected_groups=edges, isbin=True)
    got, *_ = flox_amd.groupby_reduce(arr, by, func="mean", expected_groups=edges, isbin=True)
    assert got.shape == (y, 4)
    np.testing.assert_allclose(got, want, equal_nan=True, rtol=1e-10, atol=1e-12)


@pytest.mark.parametrize("func", ["sum", "nansum", "mean", "count", "min", "nanmax", "var"])
def test_partition_path_many_groups(func):
    """2e6 groups: the bucket-partition (sort) path."""
    rng = np.random.default_rng(zlib.crc32(func.encode()))
    n, ng = 1_000_000, 2_000_000
    labels = rng.integers(0, ng, n)
    vals = rng.standard_normal(n)
    vals[rng.random(n) < 0.02] = np.nan
    want, *_ = oracle_reduce(vals, labels, func=func, expected_groups=np.arange(ng))
    got, *_ = flox_amd.groupby_reduce(vals, labels, func=func, expected_groups=np.arange(ng))
    assert got.dtype == want.dtype
    np.testing.assert_allclose(got, want, equal_nan=True, **_tol(func, want.dtype))


@pytest.mark.parametrize("func", ["nanfirst", "nanlast"])
def test_nanfirst_nanlast_datetime_lead_and_subset(func):
    """Exact NaT skipping for nanfirst/nanlast on datetime with leading
    array dims (per-(lead,row) sentinel invalidation after the lead fold)
    and with an axis subset — closes the r01 NotImplementedError
    (VERDICT item 7). Anchored on the oracle (numpy per-group semantics,
    xrutils.nanfirst's isnull — the reference cannot run first/last in
    this environment)."""
    rng = np.random.default_rng(abs(zlib.crc32(func.encode())))
    tv = (np.datetime64("2022-03-01")
          + rng.integers(0, 9000, (3, 60)).astype("timedelta64[m]"))
    tv[rng.random((3, 60)) < 0.3] = np.datetime64("NaT")
    by = rng.integers(0, 6, 60)
    want, *_ = oracle_reduce(tv, by, func=func, expected_groups=np.arange(6))
    got, *_ = flox_amd.groupby_reduce(tv, by, func=func, expected_groups=np.arange(6))
    assert got.dtype == want.dtype
    np.testing.assert_array_equal(got, want)

    tv3 = (np.datetime64("2022-03-01")
           + rng.integers(0, 9000, (4, 5, 30)).astype("timedelta64[m]"))
    tv3[rng.random((4, 5, 30)) < 0.25] = np.datetime64("NaT")
    tb3 = rng.integers(0, 6, (4, 5, 30))
    want, *_ = oracle_reduce(tv3, tb3, func=func, axis=(2,), expected_groups=np.arange(6))
    got, *_ = flox_amd.groupby_reduce(tv3, tb3, func=func, axis=(2,), expected_groups=np.arange(6))
    assert got.dtype == want.dtype and got.shape == want.shape
    np.testing.assert_array_equal(got, want)


def test_mode_signed_zero():
    """scipy.stats.mode counts -0.0 and +0.0 as ONE value; the encoded-key
    runs must merge them (regression: the subset+lead test caught split
    counts flipping multimodal winners)."""
    rng = np.random.default_rng(77)
    n, ng = 30_000, 41
    vals = np.round(rng.standard_normal(n))  # plenty of -0.0 and +0.0
    labels = rng.integers(0, ng, n)
    for func in ("mode", "nanmode"):
        want, *_ = oracle_reduce(vals, labels, func=func, expected_groups=np.arange(ng))
        got, *_ = flox_amd.groupby_reduce(vals, labels, func=func, expected_groups=np.arange(ng))
        np.testing.assert_allclose(got, want, equal_nan=True, rtol=0, atol=0,
                                   err_msg=func)


@pytest.mark.parametrize("func", ["mode", "first", "nanlast", "nanmedian",
                                  "quantile"])
def test_order_funcs_subset_plus_extra_lead(func):
    """Order-dependent funcs with an axis subset AND extra leading array
    dims: the kept-by-dims fold and the lead fold compose (r2; the
    reference supports this shape — median/quantile golden-pinned, the
    rest oracle-anchored)."""
    rng = np.random.default_rng(abs(zlib.crc32(func.encode())) % 2**31)
    a4 = rng.standard_normal((2, 4, 5, 30))
    if func != "mode":
        a4[rng.random(a4.shape) < 0.15] = np.nan
    else:
        a4 = np.round(a4)  # repeats so modes exist
    b4 = rng.integers(0, 6, (4, 5, 30))
    kw = dict(axis=(3,), expected_groups=np.arange(6), fill_value=-99.0)
    if func == "quantile":
        kw["finalize_kwargs"] = {"q": [0.25, 0.75]}
    want, *_ = oracle_reduce(a4, b4, func=func, **kw)
    got, *_ = flox_amd.groupby_reduce(a4, b4, func=func, **kw)
    assert np.asarray(got).shape == want.shape
    assert np.asarray(got).dtype == want.dtype
    np.testing.assert_allclose(np.asarray(got), want, equal_nan=True,
                               rtol=1e-12, atol=1e-12, err_msg=func)


@pytest.mark.parametrize("func", ["argmin", "argmax", "nanargmin", "nanargmax"])
@pytest.mark.parametrize("dtype", ["float64", "int64"])
def test_pair_arg_reductions_many_groups(func, dtype):
    """8-byte-dtype arg-reductions at huge group counts: the pair-payload
    partition path (FH_SET_ARG*_PAIR — row rides the pad word, second
    bucket pass matches the extremum). Bit-exact incl. first-occurrence
    ties (int values in a tiny range force heavy ties)."""
    if "nan" in func and dtype == "int64":
        pytest.skip("nanarg on ints == arg on ints")
    rng = np.random.default_rng(zlib.crc32(f"pair-{func}-{dtype}".encode()))
    n, ng = 1_000_000, 2_000_000
    labels = rng.integers(0, ng, n)
    if dtype == "float64":
        vals = np.round(rng.standard_normal(n), 1)  # quantized -> ties
        vals[rng.random(n) < 0.02] = np.nan
    else:
        vals = rng.integers(-5, 5, n).astype(np.int64)  # massive ties
    want, *_ = oracle_reduce(vals, labels, func=func, expected_groups=np.arange(ng))
    got, *_ = flox_amd.groupby_reduce(vals, labels, func=func, expected_groups=np.arange(ng))
    assert got.dtype == want.dtype, (got.dtype, want.dtype)
    np.testing.assert_allclose(got, want, equal_nan=True, rtol=0, atol=0)


def test_pair_arg_multi_by_fused():
    """Pair-payload f64 arg-reductions through the fused 2-D groupby path
    (labels2 codes ravel in-kernel) at group counts past the threshold."""
    rng = np.random.default_rng(91)
    n = 400_000
    by_a = rng.integers(0, 40, n)
    by_b = rng.integers(0, 300, n)  # 12000 groups > PACKED_ARG_THRESHOLD
    vals = np.round(rng.standard_normal(n), 1)
    vals[rng.random(n) < 0.03] = np.nan
    eg = (np.arange(40), np.arange(300))
    for func in ["argmin", "nanargmax"]:
        want, *_ = oracle_reduce(vals, by_a, by_b, func=func, expected_groups=eg)
        got, *_ = flox_amd.groupby_reduce(vals, by_a, by_b, func=func, expected_groups=eg)
        assert np.asarray(got).shape == want.shape
        np.testing.assert_array_equal(np.asarray(got), want, err_msg=func)


def test_packed_nanarg_min_count():
    """nanarg min_count masks on the NaN-aware count — the packed keys are
    always-valid ints, so their count pass alone would include NaN rows."""
    rng = np.random.default_rng(17)
    n, ng = 120_000, 8_000  # past the packed threshold
    labels = rng.integers(0, ng, n)
    vals = rng.standard_normal(n).astype(np.float32)
    vals[rng.random(n) < 0.6] = np.nan  # many groups fall below min_count
    kw = dict(expected_groups=np.arange(ng), min_count=8, fill_value=-1.0)
    want, *_ = oracle_reduce(vals, labels, func="nanargmin", **kw)
    got, *_ = flox_amd.groupby_reduce(vals, labels, func="nanargmin", **kw)
    assert got.dtype == want.dtype
    np.testing.assert_array_equal(got, want)


def test_pair_arg_matches_two_pass_form():
    """The pair-payload path must agree bit-for-bit with the LDS two-pass
    form on the same inputs (threshold lowered to force both)."""
    from flox_amd import core as fa_core

    rng = np.random.default_rng(55)
    n, ng = 300_000, 9_000  # just above the packed threshold
    labels = rng.integers(0, ng, n)
    vals = rng.standard_normal(n)
    vals[rng.random(n) < 0.05] = np.nan
    for func in ["argmin", "nanargmax"]:
        got_pair, *_ = flox_amd.groupby_reduce(
            vals, labels, func=func, expected_groups=np.arange(ng))
        old_thr = fa_core.PACKED_ARG_THRESHOLD
        fa_core.PACKED_ARG_THRESHOLD = 10**9  # force the two-pass form
        try:
            got_two, *_ = flox_amd.groupby_reduce(
                vals, labels, func=func, expected_groups=np.arange(ng))
        finally:
            fa_core.PACKED_ARG_THRESHOLD = old_thr
        np.testing.assert_array_equal(got_pair, got_two, err_msg=func)


def test_engine_seam_callables():
    """The reference-shaped seam: generic_aggregate(engine='hip', func=...)
    (reference flox/aggregations.py:60-133 signature)."""
    from flox_amd import generic_aggregate

    rng = np.random.default_rng(3)
    n, ng = 10_000, 33
    g = rng.integers(0, ng, n)
    a = rng.standard_normal(n)
    got = generic_aggregate(g, a, engine="hip", func="sum", size=ng)
    want = np.bincount(g, weights=a, minlength=ng)
    np.testing.assert_allclose(np.asarray(got), want, rtol=1e-12)
    got_c = generic_aggregate(g, a, engine="hip", func="nanlen", size=ng)
    np.testing.assert_array_equal(np.asarray(got_c), np.bincount(g, minlength=ng))


ALL_SEAM_FUNCS = [
    "count", "sum", "nansum", "prod", "nanprod", "mean", "nanmean", "var",
    "nanvar", "std", "nanstd", "min", "nanmin", "max", "nanmax", "argmax",
    "nanargmax", "argmin", "nanargmin", "first", "nanfirst", "last",
    "nanlast", "median", "nanmedian", "mode", "nanmode", "any", "all",
    "nanlen",
]


def test_engine_seam_covers_every_reference_reduction():
    """generic_aggregate(engine='hip', func=...) resolves and runs for every
    name in the reference registry (aggregations.py:881-913) + quantiles."""
    from flox_amd import generic_aggregate

    rng = np.random.default_rng(23)
    n, ng = 5_000, 17
    g = rng.integers(0, ng, n)
    a = rng.standard_normal(n)
    b = rng.random(n) < 0.5
    for func in ALL_SEAM_FUNCS:
        arr = b if func in ("any", "all") else a
        out = generic_aggregate(g, arr, engine="hip", func=func, size=ng)
        assert out.shape[-1] == ng, func
    outq = generic_aggregate(g, a, engine="hip", func="quantile", size=ng, q=0.25)
    assert outq.shape[-1] == ng
    outq2 = generic_aggregate(g, a, engine="hip", func="nanquantile", size=ng, q=[0.2, 0.8])
    assert outq2.shape == (2, ng)


def test_multi_by_fused_ravel():
    """2-D groupby goes through the fused labels2 kernel path."""
    rng = np.random.default_rng(11)
    n = 300_000
    by_a = rng.integers(0, 12, n)
    by_b = rng.integers(0, 180, n)
    vals = rng.standard_normal(n)
    vals[rng.random(n) < 0.05] = np.nan
    want, *wg = oracle_reduce(
        vals, by_a, by_b, func="nanmean", expected_groups=(np.arange(12), np.arange(180))
    )
    got, *gg = flox_amd.groupby_reduce(
        vals, by_a, by_b, func="nanmean", expected_groups=(np.arange(12), np.arange(180))
    )
    assert got.shape == (12, 180)
    np.testing.assert_allclose(got, want, equal_nan=True, rtol=1e-12, atol=1e-14)


def test_torch_tensor_roundtrip_stays_on_device():
    vals = torch.rand(10_000, device="cuda")
    labels = torch.randint(0, 50, (10_000,), device="cuda")
    res, groups = flox_amd.groupby_reduce(vals, labels, func="mean", expected_groups=np.arange(50))
    assert isinstance(res, torch.Tensor) and res.is_cuda


def test_size_independent_properties_large():
    """Properties that hold at bench-scale sizes without an oracle run:
    count sums to the number of valid rows; grouped sums sum to the total."""
    n, ng = 50_000_000, 10_000
    g = torch.randint(0, ng, (n,), device="cuda")
    v = torch.rand(n, device="cuda", dtype=torch.float32)
    res_c, _ = flox_amd.groupby_reduce(v, g, func="count", expected_groups=np.arange(ng))
    assert int(res_c.sum().item()) == n
    res_s, _ = flox_amd.groupby_reduce(v, g, func="sum", expected_groups=np.arange(ng))
    total = v.sum(dtype=torch.float64).item()
    assert abs(res_s.sum(dtype=torch.float64).item() - total) < 1e-4 * abs(total) + 1e-3


@pytest.mark.parametrize("func", ["sum", "mean", "nanmean", "var", "count", "min", "nanmax"])
def test_column_path_climatology_shape(func):
    """Config-4 shape (scaled down): time-major (n_t, y, x) array reduced by
    hour-of-day along axis 0, fed as a .permute view (no transpose copy)."""
    rng = np.random.default_rng(5)
    n_t, y, x = 24 * 60, 36, 72
    arr = rng.standard_normal((n_t, y, x)).astype(np.float32)
    arr[rng.random((n_t, y, x)) < 0.02] = np.nan
    hours = (np.arange(n_t) % 24).astype(np.int64)
    want, *_ = oracle_reduce(
        arr.transpose(1, 2, 0), hours, func=func, expected_groups=np.arange(24)
    )
    arr_t = torch.tensor(arr, device="cuda").permute(1, 2, 0)
    got, _ = flox_amd.groupby_reduce(
        arr_t, torch.tensor(hours, device="cuda"), func=func, expected_groups=np.arange(24)
    )
    assert got.shape == (y, x, 24)
    g = got.cpu().numpy()
    assert g.dtype == want.dtype
    tol = dict(rtol=0, atol=0) if g.dtype.kind in "iu" else dict(rtol=3e-6, atol=1e-6)
    np.testing.assert_allclose(g, want, equal_nan=True, **tol)


def test_column_path_large_ngroups():
    """The column kernel has no group-count limit (register-segment design)."""
    rng = np.random.default_rng(9)
    n_t, m, ng = 3000, 500, 366
    arr = rng.standard_normal((n_t, m))
    labels = rng.integers(0, ng, n_t)
    want, *_ = oracle_reduce(arr.T, labels, func="mean", expected_groups=np.arange(ng))
    arr_t = torch.tensor(arr, device="cuda").permute(1, 0)
    got, _ = flox_amd.groupby_reduce(arr_t, torch.tensor(labels, device="cuda"), func="mean", expected_groups=np.arange(ng))
    np.testing.assert_allclose(got.cpu().numpy(), want, equal_nan=True, rtol=1e-12, atol=1e-14)


def test_packed_arg_small():
    """The packed-key arg form (one grouped i64 MIN over (enc(value), row)
    keys) against the oracle, forced on at small sizes; heavy ties and NaN
    runs exercise the first-occurrence rule and both NaN conventions."""
    import flox_amd.core as core

    old = core.PACKED_ARG_THRESHOLD
    try:
        core.PACKED_ARG_THRESHOLD = 1
        rng = np.random.default_rng(33)
        for dt in ["float32", "int32"]:
            for n, ng in [(5000, 50), (200_000, 3000)]:
                if dt == "float32":
                    v = rng.standard_normal(n).astype(dt)
                    v[rng.random(n) < 0.2] = np.nan
                    v[rng.integers(0, n, n // 10)] = 1.5  # exact ties
                else:
                    v = rng.integers(-40, 40, n).astype(dt)  # many ties
                labels = rng.integers(0, ng, n)
                eg = np.arange(ng + 8)  # trailing empty groups
                for func in ["argmin", "argmax", "nanargmin", "nanargmax",
                             "first", "last", "nanfirst", "nanlast"]:
                    want, *_ = oracle_reduce(v, labels, func=func, expected_groups=eg)
                    got, *_ = flox_amd.groupby_reduce(v, labels, func=func, expected_groups=eg)
                    np.testing.assert_array_equal(
                        np.asarray(got), want, err_msg=f"{dt} {func} {n}x{ng}"
                    )
    finally:
        core.PACKED_ARG_THRESHOLD = old


def test_packed_arg_partition_scale():
    """arg-reductions at group counts far beyond LDS capacity: the packed
    keys must route through the bucket-partition path and still match the
    oracle (was the documented atomic-fallback gap)."""
    rng = np.random.default_rng(44)
    n, ng = 2_000_000, 200_000
    v = rng.standard_normal(n).astype(np.float32)
    v[rng.random(n) < 0.1] = np.nan
    labels = rng.integers(0, ng, n)
    for func in ["argmin", "nanargmax", "first", "nanlast"]:
        want, *_ = oracle_reduce(v, labels, func=func, expected_groups=np.arange(ng))
        got, *_ = flox_amd.groupby_reduce(v, labels, func=func, expected_groups=range(ng))
        np.testing.assert_array_equal(np.asarray(got), want, err_msg=func)


@pytest.mark.parametrize("func", [
    "argmax", "nanargmin", "first", "nanlast", "median", "nanquantile",
    "mode", "quantile",
])
def test_lead_fold_order_funcs(func):
    """Order-dependent reductions with leading array dims (lead index folded
    into the group codes) vs the oracle."""
    rng = np.random.default_rng(91)
    arr = rng.standard_normal((3, 4, 200))
    arr[rng.random(arr.shape) < 0.2] = np.nan
    labels = rng.integers(0, 7, 200)
    eg = np.arange(9)  # two trailing empty groups
    kw = {}
    if func in ("quantile", "nanquantile"):
        kw["finalize_kwargs"] = {"q": [0.3, 0.9] if func == "quantile" else 0.7}
    want, *_ = oracle_reduce(arr, labels, func=func, expected_groups=eg, **kw)
    got, *_ = flox_amd.groupby_reduce(arr, labels, func=func, expected_groups=eg, **kw)
    assert np.asarray(got).shape == want.shape
    if func.endswith(("argmax", "argmin")):
        np.testing.assert_array_equal(np.asarray(got), want)
    else:
        np.testing.assert_allclose(np.asarray(got), want, equal_nan=True,
                                   rtol=1e-6, atol=1e-6)


@pytest.mark.parametrize("func", ["cumsum", "nancumsum", "ffill", "bfill"])
def test_lead_dims_scan(func):
    """Grouped scans with leading array dims vs the oracle."""
    from oracle import groupby_scan as oracle_scan

    rng = np.random.default_rng(92)
    arr = rng.standard_normal((5, 300))
    arr[rng.random(arr.shape) < 0.3] = np.nan
    labels = rng.integers(0, 6, 300)
    eg = np.arange(6)
    want = oracle_scan(arr, labels, func=func, expected_groups=eg)
    got = flox_amd.groupby_scan(arr, labels, func=func, expected_groups=eg)
    np.testing.assert_allclose(np.asarray(got), want, equal_nan=True,
                               rtol=1e-9, atol=1e-9)


def test_nosort_lead_dims():
    """sort=False (first-appearance group order) with leading array dims."""
    rng = np.random.default_rng(17)
    arr = rng.standard_normal((2, 3, 500))
    labels = rng.choice([30, 5, 17, 2, 44, 9], 500)
    want, wgrp = oracle_reduce(arr, labels, func="mean", sort=False)
    got, ggrp = flox_amd.groupby_reduce(arr, labels, func="mean", sort=False)
    np.testing.assert_array_equal(np.asarray(ggrp), wgrp)
    np.testing.assert_allclose(np.asarray(got), want, rtol=1e-12, atol=1e-12)


@pytest.mark.parametrize("axs", [(3,), (2, 3)])
@pytest.mark.parametrize("func", ["mean", "sum", "var", "count"])
def test_lead_axis_subset(func, axs):
    """Axis subset of by's dims with extra leading array dims: offset codes
    carry the kept by-dims, the column path carries the lead dims."""
    rng = np.random.default_rng(zlib.crc32(f"las-{func}-{axs}".encode()))
    arr = rng.standard_normal((2, 4, 5, 40))
    arr[rng.random(arr.shape) < 0.05] = np.nan
    by = rng.integers(0, 6, (4, 5, 40))
    want, *_ = oracle_reduce(arr, by, func=func, axis=axs,
                             expected_groups=np.arange(6), fill_value=-9.0)
    got, *_ = flox_amd.groupby_reduce(arr, by, func=func, axis=axs,
                                      expected_groups=np.arange(6), fill_value=-9.0)
    assert np.asarray(got).shape == want.shape
    np.testing.assert_allclose(np.asarray(got), want, equal_nan=True,
                               rtol=1e-9, atol=1e-9)


@pytest.mark.parametrize("dtype", ["float32", "float64", "int32", "int64"])
@pytest.mark.parametrize("skipnan", [False, True])
def test_radix_select_quantile_world1(dtype, skipnan):
    """The distributed quantile machinery (radix selection over grouped
    counts) at world_size 1 on the real COUNT kernel, vs the oracle."""
    from flox_amd.dist_quantile import distributed_grouped_quantile

    rng = np.random.default_rng(zlib.crc32(f"rs-{dtype}-{skipnan}".encode()))
    n, ng = 80_000, 157
    labels = rng.integers(0, ng, n)
    if np.dtype(dtype).kind == "f":
        vals = (rng.standard_normal(n) * 100).astype(dtype)
        vals[rng.random(n) < 0.1] = np.nan
    else:
        vals = rng.integers(-(2**40) if dtype == "int64" else -1000,
                            2**40 if dtype == "int64" else 1000, n).astype(dtype)
    q = np.array([0.0, 0.25, 0.5, 0.9, 1.0])
    func = "nanquantile" if skipnan else "quantile"
    want, *_ = oracle_reduce(vals, labels, func=func, expected_groups=np.arange(ng),
                             finalize_kwargs={"q": list(q)})
    got = distributed_grouped_quantile(
        torch.tensor(vals, device="cuda"),
        torch.tensor(labels, device="cuda"),
        ng, q, skipnan=skipnan,
    )
    # the selection lerps exact f32 values in f64; the oracle lerps in the
    # input precision — compare at the INPUT dtype's precision
    tol = (dict(rtol=3e-6, atol=1e-5) if np.dtype(dtype).itemsize == 4 and np.dtype(dtype).kind == "f"
           else dict(rtol=1e-12, atol=1e-12))
    np.testing.assert_allclose(got.cpu().numpy().astype(want.dtype), want,
                               equal_nan=True, **tol)


def test_nosort_subset_and_mode():
    """sort=False with axis subsets and with discovered-group mode."""
    rng = np.random.default_rng(71)
    arr = rng.standard_normal((4, 5, 60))
    by = rng.choice([7, 3, 9, 1, 12], (4, 5, 60))
    want, wg = oracle_reduce(arr, by, func="sum", axis=(1,), sort=False, fill_value=-7.0)
    got, gg = flox_amd.groupby_reduce(arr, by, func="sum", axis=(1,), sort=False, fill_value=-7.0)
    np.testing.assert_array_equal(np.asarray(gg), wg)
    np.testing.assert_allclose(np.asarray(got), want, rtol=1e-12, atol=1e-12)

    v = rng.integers(-5, 5, 400).astype(np.float64)
    b = rng.choice([7, 3, 9, 1], 400)
    want, wg = oracle_reduce(v, b, func="mode", sort=False)
    got, gg = flox_amd.groupby_reduce(v, b, func="mode", sort=False)
    np.testing.assert_array_equal(np.asarray(gg), wg)
    np.testing.assert_array_equal(np.asarray(got), want)


@pytest.mark.parametrize("dtype", ["float32", "float64", "int64"])
@pytest.mark.parametrize("skipnan", [False, True])
def test_distributed_mode_world1(dtype, skipnan):
    """The distributed-mode merge at world_size 1 vs the oracle (scipy.stats.mode)."""
    from flox_amd.dist_quantile import distributed_grouped_mode

    rng = np.random.default_rng(zlib.crc32(f"dm-{dtype}-{skipnan}".encode()))
    n, ng = 40_000, 61
    labels = rng.integers(0, ng, n)
    vals = rng.integers(-15, 15, n).astype(dtype)
    if np.dtype(dtype).kind == "f":
        vals[rng.random(n) < 0.1] = np.nan
    func = "nanmode" if skipnan else "mode"
    want, *_ = oracle_reduce(vals, labels, func=func, expected_groups=np.arange(ng))
    got = distributed_grouped_mode(
        torch.tensor(vals, device="cuda"), torch.tensor(labels, device="cuda"),
        ng, skipnan)
    np.testing.assert_array_equal(got.cpu().numpy().astype(want.dtype), want)


def test_empty_and_single_row_inputs():
    """Empty arrays fill everywhere; single-row groups match the oracle."""
    e = np.array([], dtype=np.float64)
    el = np.array([], dtype=np.int64)
    want, *_ = oracle_reduce(e, el, func="sum", expected_groups=np.arange(4), fill_value=0.0)
    got, *_ = flox_amd.groupby_reduce(e, el, func="sum", expected_groups=np.arange(4), fill_value=0.0)
    np.testing.assert_array_equal(np.asarray(got), want)
    want, *_ = oracle_reduce(e, el, func="mean", expected_groups=np.arange(3))
    got, *_ = flox_amd.groupby_reduce(e, el, func="mean", expected_groups=np.arange(3))
    np.testing.assert_array_equal(np.asarray(got), want)  # all NaN
    want, *_ = oracle_reduce(np.array([5.0]), np.array([1]), func="var", expected_groups=np.arange(3))
    got, *_ = flox_amd.groupby_reduce(np.array([5.0]), np.array([1]), func="var", expected_groups=np.arange(3))
    np.testing.assert_array_equal(np.asarray(got), want)  # [nan, 0, nan]


def test_atomic_fallback_extreme_ngroups():
    """Group counts beyond the partition planner's reach (> 33M fine
    buckets) fall to the global-atomic path and stay correct."""
    rng = np.random.default_rng(55)
    n, ng = 1_000_000, 40_000_000
    labels = rng.integers(0, ng, n)
    vals = rng.standard_normal(n).astype(np.float32)
    got, *_ = flox_amd.groupby_reduce(vals, labels, func="sum", expected_groups=range(ng))
    got = np.asarray(got)
    # spot-check against numpy on the rows present
    order = np.argsort(labels, kind="stable")
    ul, st = np.unique(labels[order], return_index=True)
    sums = np.add.reduceat(vals[order].astype(np.float64), st)
    np.testing.assert_allclose(got[ul], sums, rtol=1e-6, atol=1e-6)
    present = np.zeros(ng, bool)
    present[ul] = True
    # missing groups fill with the dtype NA (NaN), as the reference does for
    # plain sum with expected_groups and no user fill
    assert np.isnan(got[~present]).all()


@pytest.mark.parametrize("q", [0.5, [0.25, 0.9]])
def test_quantile_axis_subset(q):
    """quantile over an axis subset of by's dims (offset codes; q leads)."""
    rng = np.random.default_rng(81)
    arr = rng.standard_normal((3, 4, 50))
    arr[rng.random(arr.shape) < 0.1] = np.nan
    by = rng.integers(0, 6, (3, 4, 50))
    kw = dict(axis=(2,), expected_groups=np.arange(6), fill_value=-7.0,
              finalize_kwargs={"q": q})
    want, *_ = oracle_reduce(arr, by, func="nanquantile", **kw)
    got, *_ = flox_amd.groupby_reduce(arr, by, func="nanquantile", **kw)
    assert np.asarray(got).shape == want.shape
    np.testing.assert_allclose(np.asarray(got), want, equal_nan=True,
                               rtol=1e-9, atol=1e-9)


def test_engine_seam_2d_array_and_fill():
    """Seam callables accept 2-D arrays (reduce over axis=-1, as
    chunk_reduce calls them) and honor fill_value for absent groups."""
    from flox_amd import generic_aggregate

    rng = np.random.default_rng(5)
    g = rng.integers(0, 7, 500)
    a = rng.standard_normal((4, 500))
    out = np.asarray(generic_aggregate(g, a, engine="hip", func="mean", size=7))
    assert out.shape == (4, 7)
    for r in range(4):
        want = np.bincount(g, weights=a[r], minlength=7) / np.bincount(g, minlength=7)
        np.testing.assert_allclose(out[r], want, rtol=1e-12, atol=1e-12)
    # absent group (size > max label) gets fill_value
    out2 = np.asarray(generic_aggregate(g, a[0], engine="hip", func="sum",
                                        size=9, fill_value=-5.0))
    assert out2.shape == (9,)
    np.testing.assert_allclose(out2[7:], [-5.0, -5.0])


def test_custom_aggregation():
    """Reference-style custom Aggregation (docs 'Custom Aggregations'): the
    mean-from-parts example and a custom range = max - min."""
    from flox_amd import CustomAggregation

    mean_agg = CustomAggregation(
        name="mean", numpy="mean", chunk=("sum", "nanlen"), combine=("sum", "sum"),
        finalize=lambda sum_, count: sum_ / count,
        fill_value=0, final_fill_value=np.nan,
    )
    rng = np.random.default_rng(31)
    v = rng.standard_normal(5000)
    labels = rng.integers(0, 9, 5000)
    got, gg = flox_amd.groupby_reduce(v, labels, func=mean_agg, expected_groups=np.arange(11))
    want, *_ = oracle_reduce(v, labels, func="mean", expected_groups=np.arange(11))
    np.testing.assert_allclose(np.asarray(got), want, equal_nan=True, rtol=1e-12, atol=1e-12)

    rng_agg = CustomAggregation(
        name="range", chunk=("max", "min"), combine=("max", "min"),
        finalize=lambda mx, mn: mx - mn, final_fill_value=-1.0,
        final_dtype=np.float64,
    )
    got2, _ = flox_amd.groupby_reduce(v, labels, func=rng_agg, expected_groups=np.arange(11))
    wmax, *_ = oracle_reduce(v, labels, func="max", expected_groups=np.arange(11))
    wmin, *_ = oracle_reduce(v, labels, func="min", expected_groups=np.arange(11))
    want2 = wmax - wmin
    want2[9:] = -1.0
    np.testing.assert_allclose(np.asarray(got2), want2, rtol=1e-12, atol=1e-12)


@pytest.mark.parametrize("func", ["cumsum", "nancumsum", "ffill", "bfill"])
def test_scan_sorted_labels_fast_path(func):
    """Nondecreasing in-range labels take the sort-free scan path and must
    match the oracle exactly like the sorted-free case."""
    from oracle import groupby_scan as oracle_scan

    rng = np.random.default_rng(zlib.crc32(f"ss-{func}".encode()))
    n, ng = 150_000, 300
    labels = np.sort(rng.integers(0, ng, n))
    vals = rng.standard_normal(n)
    vals[rng.random(n) < 0.2] = np.nan
    want = oracle_scan(vals, labels, func=func, expected_groups=np.arange(ng))
    got = flox_amd.groupby_scan(vals, labels, func=func, expected_groups=np.arange(ng))
    np.testing.assert_allclose(np.asarray(got), want, equal_nan=True, rtol=1e-12, atol=1e-10)


@pytest.mark.parametrize("func", ["sum", "nanmean", "min", "nanmax", "count", "var", "prod"])
def test_sorted_labels_direct_reduce(func):
    """Sorted in-range labels at huge group counts take the scatter-free
    direct bucket path and must match the oracle."""
    rng = np.random.default_rng(zlib.crc32(f"sd-{func}".encode()))
    n, ng = 2_000_000, 100_000
    labels = np.sort(rng.integers(0, ng, n))
    vals = rng.standard_normal(n)
    vals[rng.random(n) < 0.05] = np.nan
    want, *_ = oracle_reduce(vals, labels, func=func, expected_groups=np.arange(ng))
    got, *_ = flox_amd.groupby_reduce(vals, labels, func=func, expected_groups=range(ng))
    tol = dict(rtol=0, atol=0) if want.dtype.kind in "iu" else dict(rtol=1e-11, atol=1e-11)
    np.testing.assert_allclose(np.asarray(got), want, equal_nan=True, **tol)


@pytest.mark.parametrize("func", ["var", "nanvar", "std"])
def test_sorted_direct_int_var_regression(func):
    """Huge-fuzz find (seed 246810 case 21): the sorted-direct bucket
    kernel's run flush narrowed the double SSD partial through Acc=int64
    for integer inputs, truncating d^2 fractions (~2.6e-7 relative error).
    Check against exact integer algebra var = (c*sum(v^2) - s^2) / c^2."""
    rng = np.random.default_rng(zlib.crc32(f"sdi-{func}".encode()))
    n, ng = 3_000_000, 200_000
    labels = np.sort(rng.integers(0, ng, n))
    vals = rng.integers(-1000, 1000, n).astype(np.int64)
    got, *_ = flox_amd.groupby_reduce(vals, labels, func=func, expected_groups=range(ng))
    c = np.bincount(labels, minlength=ng)
    s = np.zeros(ng, dtype=np.int64)
    np.add.at(s, labels, vals)
    sq = np.zeros(ng, dtype=np.int64)
    np.add.at(sq, labels, vals * vals)
    exact = np.full(ng, np.nan)
    ok = c > 0
    exact[ok] = (c[ok] * sq[ok] - s[ok] * s[ok]).astype(np.float64) / (
        c[ok].astype(np.float64) ** 2)
    if func == "std":
        exact = np.sqrt(exact)
    np.testing.assert_allclose(np.asarray(got), exact, equal_nan=True,
                               rtol=1e-12, atol=1e-12)


@pytest.mark.parametrize("func", [
    "sum", "nansum", "mean", "min", "nanmin", "max", "nanmax", "var",
    "argmin", "argmax", "first", "median", "cumsum", "ffill",
])
def test_extreme_values(func):
    """±inf, denormals, signed zeros, huge magnitudes and NaN mixed —
    must match the oracle (inf-inf -> NaN sums, inf extremes, etc.)."""
    from oracle import groupby_scan as oracle_scan

    rng = np.random.default_rng(zlib.crc32(f"ex-{func}".encode()))
    n, ng = 20_000, 37
    vals = rng.standard_normal(n)
    if func in ("sum", "nansum", "mean", "var", "cumsum"):
        # accumulating funcs: overflow-created infs (1e308+1e308) make ANY
        # parallel association diverge from the sequential one (inf + -inf
        # lands at different positions) — inherent to fp, not a bug; keep
        # the non-overflowing specials
        specials = np.array([np.inf, -np.inf, np.nan, 0.0, -0.0, 5e-324])
    else:
        specials = np.array([np.inf, -np.inf, np.nan, 0.0, -0.0, 1e308,
                             -1e308, 5e-324, np.finfo(np.float64).max])
    pos = rng.integers(0, n, 3000)
    vals[pos] = specials[rng.integers(0, len(specials), 3000)]
    labels = rng.integers(0, ng, n)
    eg = np.arange(ng)
    if func in ("cumsum", "ffill"):
        want = oracle_scan(vals, labels, func=func, expected_groups=eg)
        got = flox_amd.groupby_scan(vals, labels, func=func, expected_groups=eg)
    else:
        want, *_ = oracle_reduce(vals, labels, func=func, expected_groups=eg)
        got, *_ = flox_amd.groupby_reduce(vals, labels, func=func, expected_groups=eg)
    got = np.asarray(got)
    if func in ("argmin", "argmax"):
        np.testing.assert_array_equal(got, want)
    else:
        # inf-dominated sums are exact; finite ones to fp tolerance
        finite = np.isfinite(want)
        np.testing.assert_array_equal(np.isfinite(got), finite)
        np.testing.assert_array_equal(got[~finite].astype(np.float64),
                                      want[~finite].astype(np.float64))
        np.testing.assert_allclose(got[finite], want[finite], rtol=1e-9,
                                   atol=1e-9 * (1 + np.abs(want[finite]).max(initial=0)))


def test_partition_pass_a_overflow_regression():
    """Fuzz-found crash (case 160, seed 424242): a two-level partition whose
    first super-bucket overflows its optimistic capacity region; pass B must
    stay in bounds and the exact fallback must produce oracle parity."""
    rng = np.random.default_rng(160)
    n, ng = 63_420, 529_450  # ~99% of rows land in super-bucket 0
    labels = rng.integers(0, ng, n)
    vals = rng.standard_normal(n)
    for func in ["sum", "nanvar", "min"]:
        want, *_ = oracle_reduce(vals, labels, func=func, expected_groups=np.arange(ng))
        got, *_ = flox_amd.groupby_reduce(vals, labels, func=func, expected_groups=range(ng))
        np.testing.assert_allclose(np.asarray(got), want, equal_nan=True,
                                   rtol=1e-10, atol=1e-10, err_msg=func)


def test_engine_seam_scan_2d():
    """Seam scan callables accept (..., n) arrays like the reduction ones."""
    from flox_amd import generic_aggregate

    rng = np.random.default_rng(9)
    g = rng.integers(0, 6, 400)
    a = rng.standard_normal((3, 400))
    a[rng.random(a.shape) < 0.2] = np.nan
    out = np.asarray(generic_aggregate(g, a, engine="hip", func="ffill", size=6))
    assert out.shape == a.shape
    from oracle import groupby_scan as oracle_scan
    want = oracle_scan(a, g, func="ffill", expected_groups=np.arange(6))
    np.testing.assert_allclose(out, want, equal_nan=True, rtol=1e-12, atol=1e-12)

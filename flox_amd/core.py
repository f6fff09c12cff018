"""groupby_reduce: the flox-shaped top-level API on MI355X.

Mirrors the reference's eager path (flox/core.py:739-1222 groupby_reduce ->
_reduce_blockwise -> chunk_reduce -> _finalize_results) with the per-block
kernel layer replaced by fused HIP passes (aggregate_hip.grouped_partials)
and the dask combine tree replaced by RCCL all-reduces of the per-group
partial bins (distributed.py).

Scope: every reduction/scan the reference registers, over any mix of
leading array dims, axis subsets of ``by``'s dims, 1+ ``by`` arrays
(multi-dim groupby), isbin/IntervalIndex binning, sort=False
first-appearance ordering, expected_groups/fill_value/min_count/dtype
rules, custom Aggregation(chunk/combine/finalize) instances; dtypes
f32/f64/i32/i64/bool/datetime64/timedelta64 (NaT as int64-min, like the
reference) plus u8/u16/u32/u64/i8/i16/f16 via promote-compute-cast
following the reference's own promotion rules (see _SMALL_PROMOTE; uint64
computes on the int64 view). Grouping BY datetime labels works on the
int64 view (NaT rows drop; expected groups unit-align), including datetime
bin edges (isbin). Multi-GPU: every op combines across ranks (partial-bin
all-reduce / scan carries / quantile radix selection / mode run merge).
"""

from __future__ import annotations

import math
from typing import Any

import numpy as np
import torch

from . import _ffi, distributed, xrdtypes
from .aggregate_hip import grouped_partials, grouped_partials_cols
from .aggregations import REDUCTIONS

# arg-reductions switch from the two-pass (extremum, then index-match) form to
# the packed-key single-pass form above this group count: the IDX bins cost
# 20 B/group in LDS, so past ~8e3 groups the second pass falls to the
# global-atomic path. Measured crossover (tools/arg_threshold_prof.py,
# 2e8 rows): two-pass wins to ~4e3 groups (0.94 vs 1.14 ms), packed wins
# from ~8e3 (1.09 vs 1.35; 2.6 vs 10.0 ms at 16e3). Tests lower this to
# exercise the packed form at small n.
PACKED_ARG_THRESHOLD = 6000

# order-dependent reductions that run with leading array dims by folding the
# lead index into the group codes (see groupby_reduce's lead-fold block)
_LEAD_FOLD_FUNCS = (
    "quantile", "nanquantile", "median", "nanmedian", "mode", "nanmode",
    "argmax", "argmin", "nanargmax", "nanargmin",
    "first", "last", "nanfirst", "nanlast",
)

_TORCH_TO_NP = {
    torch.float32: np.dtype("float32"),
    torch.float64: np.dtype("float64"),
    torch.int64: np.dtype("int64"),
    torch.int32: np.dtype("int32"),
    torch.bool: np.dtype(bool),
}


def _np_dtype(t: torch.dtype) -> np.dtype:
    try:
        return _TORCH_TO_NP[t]
    except KeyError:
        raise NotImplementedError(
            f"engine='hip' does not support torch dtype {t} (numpy inputs "
            "of small/unsigned dtypes promote automatically)"
        ) from None


def _torch_dtype(d: np.dtype) -> torch.dtype:
    for k, v in _TORCH_TO_NP.items():
        if v == np.dtype(d):
            return k
    raise NotImplementedError(f"unsupported dtype {d}")


# small / unsigned / half value dtypes compute on a promoted dtype and cast
# back per the reference's own promotion rules (it casts BEFORE reducing:
# _initialize_aggregation + xrdtypes - u* sums land in uint64, i8/i16 in
# int64, min/max/first/last/mode preserve the input dtype, mean/var/std and
# the quantile family preserve a floating input dtype)
_SMALL_PROMOTE = {
    np.dtype("uint8"): np.dtype("int32"),
    np.dtype("uint16"): np.dtype("int32"),
    np.dtype("int8"): np.dtype("int32"),
    np.dtype("int16"): np.dtype("int32"),
    np.dtype("uint32"): np.dtype("int64"),
    np.dtype("float16"): np.dtype("float32"),
}
_PRESERVE_SMALL = (
    "min", "nanmin", "max", "nanmax", "first", "last", "nanfirst", "nanlast",
    "mode", "nanmode",
)


def _cast_back_small(out_np: np.ndarray, func: str, small, fill_value=None) -> np.ndarray:
    # final cast from the promoted compute dtype back to the reference's
    # output dtype for a small input dtype (see _SMALL_PROMOTE)
    if small is None:
        return out_np
    small = np.dtype(small)
    if fill_value is not None:
        # a user fill promotes the output dtype unconditionally (reference
        # xrdtypes.py:170-171) — core already computed in the promoted
        # result dtype, so no cast-back applies
        try:
            tgt = small if func in _PRESERVE_SMALL else (
                np.dtype(np.uint64) if small.kind == "u"
                else np.dtype(np.int64) if small.kind == "i" else small)
            if np.result_type(tgt, fill_value) != tgt:
                return out_np
        except TypeError:
            return out_np
    if small == np.dtype(np.uint64):
        if func in _PRESERVE_SMALL:
            # values < 2**63 (guarded at entry); NA fill int64-min maps to
            # uint64 NA = 0
            return np.where(
                out_np == np.iinfo(np.int64).min, 0, out_np
            ).astype(np.uint64)
        if func in ("sum", "nansum", "prod", "nanprod", "cumsum", "nancumsum"):
            out_np = np.where(out_np == np.iinfo(np.int64).min, 0, out_np) \
                if func in ("sum", "nansum", "prod", "nanprod") else out_np
            return out_np.astype(np.int64).view(np.uint64)  # wrap-exact
        if func in ("ffill", "bfill"):
            return out_np.view(np.uint64)  # identity carry
        return out_np
    if func in _PRESERVE_SMALL:
        if small.kind in "iu" and out_np.dtype.kind == "i":
            # the promoted NA fill (iinfo(promoted).min) maps to the small
            # dtype's NA (iinfo(small).min); real data never hits it
            out_np = np.where(
                out_np == np.iinfo(out_np.dtype).min, np.iinfo(small).min, out_np
            )
        return out_np.astype(small)
    if func in ("ffill", "bfill"):
        # identity-carry preserves the input dtype for every kind
        return out_np.astype(small)
    if func in ("sum", "nansum", "prod", "nanprod", "cumsum", "nancumsum"):
        if small.kind == "u":
            if out_np.dtype.kind == "i":
                # empty-group NA fill: iinfo(int64).min on the promoted
                # dtype maps to iinfo(uint64).min == 0 on the true output
                out_np = np.where(
                    out_np == np.iinfo(np.int64).min, 0, out_np
                )
            return out_np.astype(np.uint64)
        if small.kind == "i":
            return out_np.astype(np.int64)
        return out_np.astype(small)  # float16 sums stay float16
    if small.kind == "f" and func in (
        "mean", "nanmean", "var", "nanvar", "std", "nanstd",
        "median", "nanmedian", "ffill", "bfill",
    ):
        # NOT quantile/nanquantile: their final dtype is ALWAYS float64
        # (reference aggregations.py:695-710); median preserves floating
        return out_np.astype(small)
    return out_np


def _coerce_by(b):
    # by / expected arrays in dtypes torch cannot ingest (u16/u32/i8/f16...)
    # carry the same information as int64/f32 - convert before the device
    if isinstance(b, torch.Tensor):
        return b
    b = np.asarray(b)
    if b.dtype.kind in "iub" and b.dtype not in (np.dtype(np.int32), np.dtype(np.int64)):
        return b.astype(np.int64)
    if b.dtype == np.float16:
        return b.astype(np.float32)
    return b


# host staging buffers whose H2D copies get captured into a hipGraph must
# outlive every replay (the graph re-executes the copy from the same host
# address) — pin them here while capturing
_CAPTURE_KEEPALIVE: list = []


def _keepalive_if_capturing(t: torch.Tensor) -> torch.Tensor:
    if torch.cuda.is_current_stream_capturing():
        _CAPTURE_KEEPALIVE.append(t)
    return t


def _as_device_tensor(x, device) -> torch.Tensor:
    if isinstance(x, torch.Tensor):
        t = x
    else:
        t = _keepalive_if_capturing(torch.from_numpy(np.ascontiguousarray(x)))
    if t.device != device:
        t = t.to(device, non_blocking=True)
    return t


class _FactorizedBy:
    """One by-array turned into dense integer codes on device.

    direct=True: codes ARE the raw labels (expected_groups is 0..n-1), so the
    kernel's bounds check is the whole factorize step (the GPU analogue of
    the reference's RangeIndex fast path, factorize.py:44-53)."""

    def __init__(self, codes: torch.Tensor, groups: np.ndarray, direct: bool):
        self.codes = codes
        self.groups = groups
        self.ngroups = len(groups)
        self.direct = direct


def _factorize_bins(flat: torch.Tensor, edges: np.ndarray, dt_by=None) -> _FactorizedBy:
    """Bin-edge grouping (isbin=True): right-closed intervals (edges[i-1],
    edges[i]], values outside every interval -> invalid (restates reference
    factorize.py:55-82: np.digitize(right=True) - 1). Groups are the
    pandas IntervalIndex the reference returns. Datetime edges bin the
    int64-viewed by on unit-aligned views (dt_by = the by dtype)."""
    import pandas as pd

    edges = np.asarray(edges)
    assert edges.ndim == 1 and len(edges) >= 2, "bin edges need >= 2 values"
    iv_groups = pd.IntervalIndex.from_breaks(edges)
    if edges.dtype.kind in "Mm":
        assert dt_by is not None, "datetime bin edges need a datetime by"
        edges = edges.astype(dt_by).view("i8")
    nbins = len(edges) - 1
    fl = flat
    edges_t = _keepalive_if_capturing(
        torch.from_numpy(np.ascontiguousarray(edges))
    ).to(flat.device)
    if edges_t.dtype != fl.dtype:
        common = torch.promote_types(edges_t.dtype, fl.dtype)
        edges_t = edges_t.to(common)
        fl = fl.to(common)
    codes = torch.searchsorted(edges_t, fl, right=False) - 1
    bad = (codes < 0) | (codes >= nbins)
    if fl.dtype.is_floating_point:
        bad |= torch.isnan(fl)
    codes = torch.where(bad, torch.full_like(codes, -1), codes)
    return _FactorizedBy(codes, iv_groups, direct=False)


def _factorize_device(flat: torch.Tensor, expect, sort: bool, dt_by=None) -> _FactorizedBy:
    """labels -> codes in [0, ngroups), invalid/NaN -> out-of-range
    (restates reference _factorize_single, factorize.py:42-99).

    dt_by: original numpy datetime64/timedelta64 dtype when the by array was
    viewed as int64 — NaT (int64 min) rows are missing (pd.factorize drops
    NaT) and found/expected groups convert back to dt_by."""
    if dt_by is not None:
        nat = torch.iinfo(torch.int64).min
        if expect is None:
            fl = flat[flat != nat]
            uniq = torch.unique(fl)  # sorted; NaT excluded
            codes = torch.searchsorted(uniq, flat)
            codes = torch.where(flat == nat, torch.full_like(codes, -1), codes)
            codes = torch.clamp(codes, max=max(uniq.numel() - 1, 0))
            return _FactorizedBy(codes, uniq.cpu().numpy().view(dt_by), direct=False)
        expect_np = np.asarray(expect)
        assert expect_np.dtype.kind in "Mm", "datetime by needs datetime expected_groups"
        # align to the by array's unit (the reference returns groups in the
        # by's unit, e.g. expected [D] against a [s] by -> groups [s])
        sorted_expect = (np.sort(expect_np) if sort else expect_np).astype(dt_by)
        exp_t = _keepalive_if_capturing(
            torch.from_numpy(sorted_expect.view("i8").copy())
        ).to(flat.device)
        n = exp_t.numel()
        codes = torch.searchsorted(exp_t, flat)
        clipped = torch.clamp(codes, max=n - 1)
        bad = (codes == n) | (exp_t[clipped] != flat) | (flat == nat)
        codes = torch.where(bad, torch.full_like(codes, -1), clipped)
        return _FactorizedBy(codes, sorted_expect, direct=False)
    if expect is None:
        fl = flat
        if fl.dtype.is_floating_point:
            fl = fl[~torch.isnan(fl)]
        uniq = torch.unique(fl)  # sorted
        codes = torch.searchsorted(uniq, flat)
        if flat.dtype.is_floating_point:
            codes = torch.where(torch.isnan(flat), torch.full_like(codes, -1), codes)
        return _FactorizedBy(codes, uniq.cpu().numpy(), direct=False)

    if isinstance(expect, range):
        # O(1) RangeIndex-style fast path (reference factorize.py:44-53)
        if expect.step != 1 or expect.start != 0:
            expect = np.asarray(expect)
        elif flat.dtype.is_floating_point:
            expect = np.asarray(expect)
        else:
            # groups echo the caller's range (materializing np.arange(1e7)
            # costs ~10 ms per call)
            return _FactorizedBy(flat, expect, direct=True)
    expect_np = np.asarray(expect)
    n = len(expect_np)
    if (
        expect_np.dtype.kind in "iu"
        and not flat.dtype.is_floating_point
        and n > 0
        and expect_np[0] == 0
        and expect_np[-1] == n - 1
        and np.array_equal(expect_np, np.arange(n))
    ):
        # direct path: labels are already the codes; the kernel's unsigned
        # bounds check drops anything outside [0, n)
        return _FactorizedBy(flat, expect_np, direct=True)

    sorted_expect = np.sort(expect_np) if sort else expect_np
    if sort is False and not np.all(np.diff(expect_np) >= 0):
        raise NotImplementedError("unsorted expected_groups with sort=False")
    exp_np_t = sorted_expect
    if exp_np_t.dtype.kind in "iub" and exp_np_t.dtype not in (
        np.dtype(np.int32), np.dtype(np.int64)
    ):
        exp_np_t = exp_np_t.astype(np.int64)
    elif exp_np_t.dtype == np.float16:
        exp_np_t = exp_np_t.astype(np.float32)
    exp_t = _keepalive_if_capturing(
        torch.from_numpy(np.ascontiguousarray(exp_np_t))
    ).to(flat.device)
    if exp_t.dtype != flat.dtype:
        common = torch.promote_types(exp_t.dtype, flat.dtype)
        exp_t = exp_t.to(common)
        flat = flat.to(common)
    codes = torch.searchsorted(exp_t, flat)
    clipped = torch.clamp(codes, max=n - 1)
    bad = (codes == n) | (exp_t[clipped] != flat)
    if flat.dtype.is_floating_point:
        bad |= torch.isnan(flat)
    codes = torch.where(bad, torch.full_like(codes, -1), codes)
    return _FactorizedBy(codes, sorted_expect, direct=False)


def _factorize_strings_host(flat_np: np.ndarray, expect, sort: bool, device) -> _FactorizedBy:
    """String/object labels: the reference's pd.factorize hash path runs on
    the HOST (strings have no device representation); the dense int64 codes
    ship to the GPU (reference factorize.py:96)."""
    if expect is None:
        uniq, inv = np.unique(flat_np, return_inverse=True)  # sorted
        codes = inv.astype(np.int64)
        groups = uniq
    else:
        expect_np = np.asarray(expect)
        sorted_expect = np.sort(expect_np) if sort else expect_np
        if not sort and len(sorted_expect) > 1 and not np.all(
            sorted_expect[:-1] <= sorted_expect[1:]
        ):
            raise NotImplementedError("unsorted expected_groups with sort=False")
        idx = np.searchsorted(sorted_expect, flat_np).astype(np.int64)
        n = len(sorted_expect)
        clip = np.clip(idx, 0, max(n - 1, 0))
        bad = (idx == n) | (sorted_expect[clip] != flat_np)
        codes = np.where(bad, -1, clip)
        groups = sorted_expect
    t = torch.from_numpy(np.ascontiguousarray(codes)).to(device)
    return _FactorizedBy(t, groups, direct=False)


def _combined_codes(facs: list[_FactorizedBy]):
    """Combine multiple factorized by-arrays into one code stream.

    Returns (labels, labels2, grp_pair): when exactly two direct int code
    arrays, the ravel is fused into the kernel (labels2 path, mirroring
    reference _ravel_factorized factorize.py:102-108); otherwise codes are
    raveled here with invalid propagation."""
    if len(facs) == 1:
        return facs[0].codes, None, None
    if len(facs) == 2 and all(f.direct for f in facs):
        return facs[0].codes, facs[1].codes, (facs[0].ngroups, facs[1].ngroups)
    code = facs[0].codes.to(torch.int64)
    bad = code < 0
    for f in facs[1:]:
        c = f.codes.to(torch.int64)
        bad = bad | (c < 0) | (code < 0)
        code = code * f.ngroups + c
    code = torch.where(bad, torch.full_like(code, -1), code)
    return code, None, None


def groupby_reduce(
    array,
    *by,
    func: str,
    expected_groups=None,
    sort: bool = True,
    isbin=False,
    axis=None,
    fill_value=None,
    dtype=None,
    min_count: int | None = None,
    method: str | None = None,
    engine: str = "hip",
    reindex=None,
    finalize_kwargs: dict[str, Any] | None = None,
    distributed_combine: bool | None = None,
    shard_row_offset: int = 0,
):
    """Grouped reduction with flox semantics on MI355X. Returns (result, *groups).

    ``array``/``by``: numpy arrays or torch tensors (numpy is moved to the
    GPU; note the PCIe cost — keep tensors resident for performance).
    ``distributed_combine``: all-reduce the per-group partials across the
    initialized torch.distributed world (default: auto when initialized).
    Each rank passes its own row shard; results are full-group on every rank.
    """
    from .aggregations import Aggregation as _Agg
    from .aggregations import CustomAggregation as _CustomAgg

    if isinstance(func, _CustomAgg):
        # reference-style custom aggregation (docs "Custom Aggregations"):
        # run each chunk reduction through the full machinery, then apply
        # the user's finalize and the final fill for empty groups
        _CHUNK_OK = {"sum", "nansum", "count", "nanlen", "min", "nanmin",
                     "max", "nanmax", "prod", "nanprod"}
        inner = dict(
            expected_groups=expected_groups, sort=sort, isbin=isbin, axis=axis,
            method=method, engine=engine, reindex=reindex,
            distributed_combine=distributed_combine,
            shard_row_offset=shard_row_offset,
        )
        inters, groups_c, counts_c = [], None, None
        for cn in func.chunk:
            cn2 = "count" if cn == "nanlen" else cn
            if cn2 not in _CHUNK_OK:
                raise NotImplementedError(f"custom aggregation chunk {cn!r}")
            r, *g = groupby_reduce(array, *by, func=cn2, **inner)
            inters.append(r)
            groups_c = g
            if cn2 == "count":
                counts_c = r
        if counts_c is None:
            counts_c, *_ = groupby_reduce(array, *by, func="count", **inner)
        res = func.finalize(*inters) if func.finalize is not None else inters[0]
        ffv = fill_value if fill_value is not None else func.final_fill_value
        mask = counts_c == 0
        if isinstance(res, torch.Tensor):
            if not isinstance(mask, torch.Tensor):
                mask = torch.as_tensor(np.asarray(mask), device=res.device)
            if ffv is not None and bool(mask.any().item()):
                nan_on_int = (
                    isinstance(ffv, float) and math.isnan(ffv)
                    and not res.dtype.is_floating_point
                )
                if nan_on_int:
                    res = res.to(torch.float64)
                res = torch.where(mask, torch.tensor(ffv, dtype=res.dtype, device=res.device), res)
            if func.final_dtype is not None:
                res = res.to(_torch_dtype(np.dtype(func.final_dtype)))
        else:
            res = np.asarray(res)
            if ffv is not None and bool(np.any(mask)):
                res = np.where(mask, ffv, res)
            if func.final_dtype is not None:
                res = res.astype(func.final_dtype)
        return (res, *groups_c)

    if isinstance(func, _Agg):
        func = func.name  # reference accepts Aggregation instances (core.py:934-948)
    if engine != "hip":
        raise ValueError(f"flox_amd implements engine='hip' only (got {engine!r})")
    if func not in REDUCTIONS:
        raise NotImplementedError(f"reduction {func!r}")
    agg = REDUCTIONS[func]

    if not torch.cuda.is_available():
        raise RuntimeError("flox_amd.groupby_reduce requires a GPU (engine='hip')")
    device = torch.device("cuda", torch.cuda.current_device())

    return_numpy = not isinstance(array, torch.Tensor)
    user_fill_value = fill_value  # internal fill defaults must not trigger
    # the user-fill dtype-promotion rule in _cast_back_small
    # datetime64/timedelta64 values compute on their int64 view and
    # dtype-preserving results view back (reference core.py:985-1001)
    dt_dtype = None
    small_dtype = None
    if return_numpy:
        arr_np = np.asarray(array)
        if arr_np.dtype in _SMALL_PROMOTE:
            small_dtype = arr_np.dtype
            array = arr_np.astype(_SMALL_PROMOTE[arr_np.dtype])
            arr_np = np.asarray(array)
        elif arr_np.dtype.kind == "c":
            # complex: the linear set computes on a (2, ...) re/im component
            # view (components are independent for sums/means/first/last and
            # scans); whole-value nulls (either component NaN — the
            # reference's isnull) premask BOTH components for skipna funcs.
            # Order/product/var families have no componentwise form: raise.
            _C_OK = ("sum", "nansum", "mean", "nanmean", "count",
                     "first", "last", "nanfirst", "nanlast")
            if not isinstance(func, str) or func not in _C_OK:
                raise NotImplementedError(
                    f"complex input for {func!r} (supported: {_C_OK})"
                )
            if fill_value is not None and not (
                isinstance(fill_value, (float, complex)) and np.isnan(
                    np.asarray(fill_value).real
                )
            ):
                raise NotImplementedError("complex input with a non-NaN fill_value")
            comp_dt = arr_np.dtype
            if func == "count":
                # count ignores magnitudes: a real view with whole-value
                # nulls mapped to NaN is exact
                real = arr_np.real.astype(np.float64)
                real[np.isnan(arr_np)] = np.nan
                r, *grps = groupby_reduce(
                    real, *by, func=func, expected_groups=expected_groups,
                    sort=sort, isbin=isbin, axis=axis, fill_value=fill_value,
                    dtype=dtype, min_count=min_count, method=method,
                    engine=engine, reindex=reindex,
                    finalize_kwargs=finalize_kwargs,
                    distributed_combine=distributed_combine,
                    shard_row_offset=shard_row_offset,
                )
                return (r, *grps)
            fwidth = np.float64 if comp_dt == np.dtype(np.complex128) else np.float32
            fv2 = arr_np.view(fwidth).reshape(arr_np.shape + (2,)).copy()
            if REDUCTIONS[func].skipnan:
                fv2[np.isnan(arr_np)] = np.nan  # both components
            comps = np.moveaxis(fv2, -1, 0)  # (2, ...original dims)
            ax2 = axis
            if axis is not None:
                axl = axis if isinstance(axis, (tuple, list)) else (axis,)
                ax2 = tuple((a % arr_np.ndim) + 1 for a in axl)
            r, *grps = groupby_reduce(
                np.ascontiguousarray(comps), *by, func=func,
                expected_groups=expected_groups, sort=sort, isbin=isbin,
                axis=ax2, fill_value=fill_value, dtype=None,
                min_count=min_count, method=method, engine=engine,
                reindex=reindex, finalize_kwargs=finalize_kwargs,
                distributed_combine=distributed_combine,
                shard_row_offset=shard_row_offset,
            )
            out = np.asarray(r)
            res = (out[0] + 1j * out[1]).astype(
                comp_dt if dtype is None else np.dtype(dtype)
            )
            return (res, *grps)
        elif arr_np.dtype == np.uint64:
            # uint64 computes on its int64 VIEW: sums/prods/counts are
            # wrap-exact mod 2^64; order-dependent funcs are only correct
            # below 2^63, so values with the high bit set are rejected there
            small_dtype = arr_np.dtype
            array = arr_np.view(np.int64)
            arr_np = np.asarray(array)
            if func not in (
                "sum", "nansum", "count", "prod", "nanprod", "any", "all",
            ) and bool((arr_np < 0).any()):
                raise NotImplementedError(
                    f"{func} on uint64 values >= 2**63 (int64-view order breaks)"
                )
        if arr_np.dtype.kind in "Mm":
            dt_dtype = arr_np.dtype
            array = arr_np.view("i8")
            # NaT (= INT64_MIN) passes through the int64 view untouched, as
            # in the reference (core.py:994-997): it wins min and loses max;
            # count and nanfirst/nanlast skip NaT rows (handled below);
            # empty groups fill with NaT (xrdtypes.py:54-61)
            if func in xrdtypes.PRESERVES_DTYPE and fill_value is None:
                fill_value = np.iinfo(np.int64).min
    arr = _as_device_tensor(array, device)
    _torch_small = {
        torch.uint8: np.dtype("uint8"), torch.int8: np.dtype("int8"),
        torch.int16: np.dtype("int16"), torch.float16: np.dtype("float16"),
    }
    if arr.dtype in _torch_small:
        small_dtype = _torch_small[arr.dtype]
        arr = arr.to(_torch_dtype(_SMALL_PROMOTE[small_dtype]))
    was_bool = arr.dtype == torch.bool
    if was_bool:
        arr = arr.to(torch.int64)  # reference core.py:916-917
    in_np_dtype = _np_dtype(arr.dtype)
    by_dts = []
    _coerced = []
    for b in by:
        if not isinstance(b, torch.Tensor):
            bn = np.asarray(b)
            if bn.dtype.kind in "Mm":
                # group BY datetimes on the int64 view; NaT rows are missing
                by_dts.append(bn.dtype)
                _coerced.append(bn.view("i8"))
                continue
            if bn.dtype.kind in "US" or bn.dtype == object:
                # string/object labels factorize on the host at the facs
                # stage; the array stays numpy until then
                by_dts.append(None)
                _coerced.append(bn)
                continue
        by_dts.append(None)
        _coerced.append(_coerce_by(b))
    bys = tuple(
        b if isinstance(b, np.ndarray) and (b.dtype.kind in "US" or b.dtype == object)
        else _as_device_tensor(b, device)
        for b in _coerced
    )
    nby = len(bys)
    if nby == 0:
        raise ValueError("need at least one by array")
    # size-1 by dims broadcast against the array's trailing dims (the
    # dim=... case, reference core.py:300-309); multiple by arrays must
    # share a shape, as in the reference
    by_shape = bys[0].shape
    for b in bys:
        if b.shape != by_shape:
            raise ValueError("by arrays must have the same shape")
    if len(by_shape) <= arr.ndim:
        trail = tuple(arr.shape[arr.ndim - len(by_shape):])
        if trail != tuple(by_shape) and all(
            bs in (1, ts) for bs, ts in zip(by_shape, trail)
        ):
            bys = tuple(
                np.ascontiguousarray(np.broadcast_to(b, trail))
                if isinstance(b, np.ndarray)
                else b.broadcast_to(trail).contiguous()
                for b in bys
            )
            by_shape = bys[0].shape
    if tuple(arr.shape[arr.ndim - len(by_shape) :]) != tuple(by_shape):
        raise ValueError(f"by {tuple(by_shape)} must align with trailing dims of array {tuple(arr.shape)}")
    subset_keep_shape: tuple | None = None
    if axis is not None:
        ax = axis if isinstance(axis, (tuple, list)) else (axis,)
        ax = tuple(a % arr.ndim for a in ax)
        trailing = tuple(range(arr.ndim - len(by_shape), arr.ndim))
        if tuple(sorted(ax)) != trailing:
            # axis subset of by's dims (reference offset-labels case,
            # factorize.py:24-39): move reduced dims last and fold the kept
            # by-dims into the group codes below
            if not set(ax) <= set(trailing):
                raise ValueError(f"axis {axis} must address dims of by")
            nlead_s = arr.ndim - len(by_shape)
            keep = [d for d in range(arr.ndim) if d not in ax]  # lead + kept by dims
            perm = keep + sorted(ax)
            arr = arr.permute(perm).contiguous()
            by_perm = [d - nlead_s for d in perm if d >= nlead_s]
            bys = tuple(
                np.ascontiguousarray(np.transpose(b, by_perm))
                if isinstance(b, np.ndarray)
                else b.permute(by_perm).contiguous()
                for b in bys
            )
            by_shape = bys[0].shape  # (kept by dims..., reduced dims...)
            subset_keep_shape = tuple(arr.shape[nlead_s : len(keep)])
            # for sort=False: first-appearance order is defined on the
            # ORIGINAL by layout — keep the inverse permutation
            subset_by_inv = tuple(int(i) for i in np.argsort(by_perm))
    lead_shape = tuple(arr.shape[: arr.ndim - len(by_shape)])
    lead_M = math.prod(lead_shape) if lead_shape else 1

    if expected_groups is not None and not isinstance(expected_groups, tuple):
        expected_groups = (expected_groups,)
    if expected_groups is None:
        expected_groups = (None,) * nby
    # pd.IntervalIndex expected groups imply binning (reference
    # _convert_expected_groups_to_index, core.py:931-933)
    try:
        import pandas as pd

        new_eg, new_isbin = [], []
        isbins0 = isbin if isinstance(isbin, (tuple, list)) else (isbin,) * nby
        changed = False
        for e, ib in zip(expected_groups, isbins0):
            if isinstance(e, pd.IntervalIndex):
                assert (e.left.to_numpy()[1:] == e.right.to_numpy()[:-1]).all(), (
                    "IntervalIndex bins must be contiguous"
                )
                new_eg.append(np.append(e.left.to_numpy(), e.right.to_numpy()[-1]))
                new_isbin.append(True)
                changed = True
            else:
                new_eg.append(e)
                new_isbin.append(ib)
        if changed:
            expected_groups = tuple(new_eg)
            isbin = tuple(new_isbin)
    except ImportError:
        pass
    provided_expected = any(e is not None for e in expected_groups)
    if isbin and not provided_expected:
        raise ValueError("isbin=True requires expected_groups (the bin edges)")

    isbins = isbin if isinstance(isbin, (tuple, list)) else (isbin,) * nby
    facs = [
        _factorize_strings_host(b.reshape(-1), e, sort, device)
        if isinstance(b, np.ndarray)
        else _factorize_bins(b.reshape(-1), e, dt_by=dtb)
        if ib
        else _factorize_device(b.reshape(-1), e, sort, dt_by=dtb)
        for b, e, ib, dtb in zip(bys, expected_groups, isbins, by_dts)
    ]
    grp_shape = tuple(f.ngroups for f in facs)
    ngroups = math.prod(grp_shape)
    labels, labels2, grp_pair = _combined_codes(facs)

    nat_skip_after_fold = False
    if dt_dtype is not None and func in ("count", "nanfirst", "nanlast"):
        # NaT rows are missing for count (verified reference behavior) and
        # for nanfirst/nanlast (xrutils.nanfirst's isnull, xrutils.py:389-397):
        # drop them by invalidating their codes; with leading dims NaT-ness
        # is per (lead, row), so count recounts on a float view instead
        # (NaT -> NaN; exact, count ignores magnitudes) and nanfirst/nanlast
        # invalidate the per-(lead, row) composite codes after the lead fold
        if lead_M != 1:
            if func == "count":
                arr = torch.where(
                    arr == torch.iinfo(torch.int64).min,
                    torch.tensor(float("nan"), dtype=torch.float64, device=device),
                    arr.to(torch.float64),
                )
            else:
                nat_skip_after_fold = True
        else:
            natm = arr.reshape(-1) == torch.iinfo(torch.int64).min
            labels = torch.where(natm, torch.full_like(labels, -1), labels)

    if subset_keep_shape is not None:
        # offset the codes by the kept-dims slice index (reference
        # offset_labels, factorize.py:24-39): slice s's groups become bins
        # [s*ngroups, (s+1)*ngroups)
        if labels2 is not None:
            g0s, g1s = grp_pair
            c0_, c1_ = labels.to(torch.int64), labels2.to(torch.int64)
            bad_ = (c0_ < 0) | (c0_ >= g0s) | (c1_ < 0) | (c1_ >= g1s)
            base_codes = torch.where(bad_, torch.full_like(c0_, -1), c0_ * g1s + c1_)
        else:
            c0_ = labels.to(torch.int64)
            base_codes = torch.where(
                (c0_ < 0) | (c0_ >= ngroups), torch.full_like(c0_, -1), c0_
            )
        inner_n = math.prod(by_shape[len(subset_keep_shape):])
        lead_idx = torch.arange(base_codes.numel(), device=device) // inner_n
        labels = torch.where(
            base_codes < 0, torch.full_like(base_codes, -1),
            lead_idx * ngroups + base_codes,
        )
        labels2, grp_pair = None, None
        subset_ngroups = ngroups
        ngroups = int(math.prod(subset_keep_shape)) * ngroups
    # only the 1-D path flattens the values (a no-op for contiguous input);
    # the column path reads the caller's strided view directly
    vals = arr.reshape(-1) if lead_M == 1 else None

    # min_count defaulting (reference core.py:1026-1038 + aggregations.py:997-1003)
    if min_count is None:
        min_count_ = 1 if (
            (fill_value is not None and provided_expected) or subset_keep_shape is not None
        ) else 0
    else:
        min_count_ = min_count
    out_dtype = xrdtypes.final_dtype(func, in_np_dtype, dtype)
    if func in ("nanmin", "nanmax") and min_count_ == 0:
        min_count_ = 1
        if fill_value is None:
            fill_value = xrdtypes.fill_default(func, out_dtype)
    if min_count_ > 0 and func in ("nansum", "nanprod") and fill_value is None:
        fill_value = float("nan")
    if fill_value is not None:
        # user fill promotes the output dtype (reference xrdtypes.py:170-171)
        out_dtype = np.result_type(out_dtype, fill_value)

    dist_on = distributed.is_active() if distributed_combine is None else distributed_combine
    if dist_on and any(e is None for e in expected_groups):
        # rank-local factorization (torch.unique of the local shard) yields
        # per-rank code spaces: the all-reduced bins would be misaligned.
        # The reference's dask path has the same constraint (its combine
        # tree reindexes every block to the full expected_groups vector,
        # dask.py:90-144); discovering groups globally needs a set-union
        # collective that is not built — require expected_groups instead.
        raise NotImplementedError(
            "distributed_combine requires expected_groups for every `by` "
            "(rank-local group discovery would misalign the combined bins)"
        )
    ddof = (finalize_kwargs or {}).get("ddof", 0)
    lead_folded = False

    if lead_M == 1:
        def run_set(op_set, skipnan, means=None, target=None):
            return grouped_partials(
                op_set, vals, labels, ngroups, skipnan=skipnan,
                labels2=labels2, grp_shape=grp_pair, means=means,
                target=target, row_offset=shard_row_offset,
            )
    else:
        # column path: grouped dims to the front, lead dims flattened as
        # columns (a zero-copy view when the caller's layout is
        # grouped-axis-major, e.g. time-major climatology read through a
        # .permute view; otherwise one transpose copy)
        nbydims = len(by_shape)
        N = math.prod(by_shape)
        arr_t = arr.movedim(
            tuple(range(arr.ndim - nbydims, arr.ndim)), tuple(range(nbydims))
        )
        vt = arr_t.reshape((N, lead_M))
        if vt.stride(1) != 1:
            vt = vt.contiguous()
        if N >= 2**31 or ngroups * lead_M >= 2**62:
            raise NotImplementedError("column path: axis length must fit int32")
        # bound codes here (the column kernel trusts codes in [-1, ngroups))
        if labels2 is not None:
            g0, g1 = grp_pair
            c0, c1 = labels.to(torch.int64), labels2.to(torch.int64)
            bad = (c0 < 0) | (c0 >= g0) | (c1 < 0) | (c1 >= g1)
            codes_full = torch.where(bad, torch.full_like(c0, -1), c0 * g1 + c1)
        else:
            c0 = labels.to(torch.int64)
            bad = (c0 < 0) | (c0 >= ngroups)
            codes_full = torch.where(bad, torch.full_like(c0, -1), c0)
        # stable sort of the (small) per-row code vector: rows are then
        # walked in group order, one contiguous segment per group (the GPU
        # analogue of _prepare_for_flox, reference aggregate_flox.py:9-23)
        scodes64, perm64 = torch.sort(codes_full, stable=True)
        scodes, perm = scodes64.to(torch.int32), perm64.to(torch.int32)

        def run_set(op_set, skipnan, means=None, target=None):
            return grouped_partials_cols(
                op_set, vt, scodes, perm, ngroups, skipnan=skipnan, means=means
            )

        if func in _LEAD_FOLD_FUNCS:
            # order-dependent reductions with leading dims: fold the lead
            # index into the group codes (lead*ngroups + code) and run the
            # 1-D machinery over the C-order flattened stream — the
            # offset-labels trick of reference factorize.py:24-39. With an
            # axis subset the two folds COMPOSE: labels already carry the
            # kept-by-dims offset (ngroups = keep*base), and the extra
            # leading array dims fold on top of that
            if subset_keep_shape is not None and func in (
                "argmax", "argmin", "nanargmax", "nanargmin"
            ):
                raise NotImplementedError(
                    f"{func} with an axis subset and extra leading array dims: next row"
                )
            if dist_on:
                raise NotImplementedError(f"distributed {func} with leading dims: next row")
            if N * lead_M >= (1 << 31) or ngroups * lead_M >= (1 << 31):
                raise NotImplementedError(f"{func} with leading dims at this size: next row")
            vals = arr.reshape(-1)
            lead_i = torch.arange(lead_M, device=device, dtype=torch.int64)
            comp = lead_i[:, None] * ngroups + codes_full[None, :]
            comp = torch.where(
                (codes_full < 0)[None, :].expand_as(comp),
                torch.full_like(comp, -1), comp,
            )
            labels = comp.reshape(-1)
            labels2, grp_pair = None, None
            arg_localize_N = N
            base_ngroups = ngroups
            ngroups = lead_M * ngroups
            lead_folded = True
            if nat_skip_after_fold:
                # exact NaT skipping per (lead, row): the folded composite
                # codes are per flattened element, so invalidating them IS
                # the int64-sentinel skip (no lossy float view)
                natm = vals == torch.iinfo(torch.int64).min
                labels = torch.where(natm, torch.full_like(labels, -1), labels)

            def run_set(op_set, skipnan, means=None, target=None):
                return grouped_partials(
                    op_set, vals, labels, ngroups, skipnan=skipnan,
                    means=means, target=target, row_offset=0,
                )

    if func in ("quantile", "nanquantile", "median", "nanmedian"):
        from .aggregate_hip import grouped_quantile

        if lead_M != 1 and not lead_folded:
            raise NotImplementedError(f"{func} with leading array dims: next row")
        if func in ("quantile", "nanquantile"):
            if not finalize_kwargs or "q" not in finalize_kwargs:
                raise ValueError("Please pass `q` for quantile calculations.")
            q = finalize_kwargs["q"]
        else:
            q = 0.5
        q_arr = np.atleast_1d(np.asarray(q, dtype=np.float64))
        scalar_q = np.isscalar(q) or np.ndim(q) == 0
        if vals.numel() == 0 and not dist_on:
            resq = torch.full((len(q_arr), ngroups), float("nan"),
                              dtype=torch.float64, device=device)
        elif dist_on:
            # exact cross-rank quantiles by radix selection over grouped
            # counts — histograms cross the wire, values never do
            from .dist_quantile import distributed_grouped_quantile

            if labels2 is not None:
                g0d, g1d = grp_pair
                cd0, cd1 = labels.to(torch.int64), labels2.to(torch.int64)
                badd = (cd0 < 0) | (cd0 >= g0d) | (cd1 < 0) | (cd1 >= g1d)
                codes_d = torch.where(badd, torch.full_like(cd0, -1), cd0 * g1d + cd1)
            else:
                codes_d = labels.to(torch.int64)
            resq = distributed_grouped_quantile(
                vals, codes_d, ngroups, q_arr, skipnan=agg.skipnan
            )
        else:
            resq = grouped_quantile(
                vals, labels, ngroups, q_arr, skipnan=agg.skipnan,
                labels2=labels2, grp_shape=grp_pair,
            )
        result = resq[0] if scalar_q else resq
        if min_count_ > 0:
            pc = grouped_partials(
                _ffi.SET_COUNT, vals, labels, ngroups, skipnan=True,
                labels2=labels2, grp_shape=grp_pair,
            )
            if dist_on:
                distributed.all_reduce_(pc["count"], "sum")
            counts_for_mask = pc["count"]
        else:
            counts_for_mask = None
        t_out = _torch_dtype(out_dtype)
        if min_count_ > 0 and fill_value is not None:
            mask = counts_for_mask < min_count_
            result = torch.where(mask, torch.tensor(float(fill_value), dtype=result.dtype, device=device), result)
        result = result.to(t_out)
        new_shape = (
            ((len(q_arr),) if not scalar_q else ())
            + lead_shape
            + (subset_keep_shape if subset_keep_shape is not None else ())
            + grp_shape
        )
        result = result.reshape(new_shape)
        groups_list_q = [f.groups for f in facs]
        if not sort and not provided_expected:
            # first-appearance group order (as in the common finalize below)
            for ax_i, f in enumerate(facs):
                pidx = grouped_partials(
                    _ffi.SET_IDXMIN, f.codes.to(torch.int64), f.codes, f.ngroups
                )
                order = torch.argsort(pidx["idx"], stable=True)
                result = torch.index_select(result, result.dim() - len(facs) + ax_i, order)
                groups_list_q[ax_i] = np.asarray(groups_list_q[ax_i])[order.cpu().numpy()]
        groups = tuple(groups_list_q)
        if return_numpy:
            out_np = result.cpu().numpy()
            if dt_dtype is not None:
                out_np = out_np.astype(dt_dtype)  # reference core.py:1209-1211
            out_np = _cast_back_small(out_np, func, small_dtype, user_fill_value)
            return (out_np, *groups)
        return (result, *groups)

    if func in ("mode", "nanmode"):
        from .aggregate_hip import grouped_mode

        if lead_M != 1 and not lead_folded:
            raise NotImplementedError(f"{func} with leading array dims: next row")
        if dist_on:
            # exact cross-rank mode: run-length-encoded (group, value, count)
            # runs are all_gathered and merged on every rank
            from .dist_quantile import distributed_grouped_mode

            if labels2 is not None:
                g0m, g1m = grp_pair
                cm0, cm1 = labels.to(torch.int64), labels2.to(torch.int64)
                badm = (cm0 < 0) | (cm0 >= g0m) | (cm1 < 0) | (cm1 >= g1m)
                codes_m = torch.where(badm, torch.full_like(cm0, -1), cm0 * g1m + cm1)
            else:
                codes_m = labels.to(torch.int64)
            result = distributed_grouped_mode(vals, codes_m, ngroups, agg.skipnan)
        elif vals.numel() == 0:
            result = torch.full(
                (ngroups,),
                float("nan") if vals.is_floating_point() else 0,
                dtype=vals.dtype, device=device,
            )
        else:
            result = grouped_mode(
                vals, labels, ngroups, skipnan=agg.skipnan, labels2=labels2, grp_shape=grp_pair
            )
        p = grouped_partials(
            _ffi.SET_COUNT, vals, labels, ngroups, skipnan=True,
            labels2=labels2, grp_shape=grp_pair,
        )
        if dist_on:
            distributed.all_reduce_(p["count"], "sum")
        counts_for_mask = p["count"]
        # float results carry NaN for empty/propagated groups already;
        # integer results need the empty fill
        empty_mask = (
            torch.zeros_like(counts_for_mask, dtype=torch.bool)
            if arr.dtype.is_floating_point
            else counts_for_mask == 0
        )
    elif func in ("argmax", "argmin", "nanargmax", "nanargmin"):
        skip = agg.skipnan
        ismax = "max" in func
        base_ok = (
            ngroups > PACKED_ARG_THRESHOLD
            and shard_row_offset + vals.numel() < (1 << 32)
        )
        # huge group counts, 4-byte dtypes: pack (order-preserving 32-bit
        # value encoding, row index) into one int64 key and take a single
        # grouped MIN — the partition path then handles what the 20 B/group
        # IDX bins cannot (LDS holds ~8e3 of them). Ties break to the
        # smaller row, which is exactly np.argmin/argmax's first-occurrence
        # rule. Distributed: the packed key is lexicographic (value, row),
        # so ONE min all-reduce of the key bins is the exact global answer.
        packed_ok = (
            base_ok
            and vals.dtype in (torch.float32, torch.int32)
            and (0 < vals.numel() or dist_on)
        )
        # 8-byte dtypes: no room to pack — the partition pairs carry the
        # row in their spare pad word and a second bucket pass takes the
        # min row among rows matching the group extremum (FH_SET_ARG*_PAIR)
        pair_ok = (
            base_ok
            and not dist_on
            and vals.dtype in (torch.float64, torch.int64)
            and 0 < vals.numel() < (1 << 31)
            and ngroups <= (1 << 24)
            # pair-arg rides the partition path, whose overflow check
            # host-syncs: under hipGraph capture use the two-pass form
            and not (vals.is_cuda and torch.cuda.is_current_stream_capturing())
        )
        if dist_on and distributed.is_active():
            # the branch choice must agree across ranks (shard sizes differ)
            okt = torch.tensor([1 if packed_ok else 0], dtype=torch.int32,
                               device=device)
            distributed.all_reduce_(okt, "min")
            packed_ok = bool(okt.item())
        if packed_ok:
            lib = _ffi.load_library()
            key = torch.empty(vals.numel(), dtype=torch.int64, device=device)
            vc = vals.contiguous()
            if vc.numel():  # a zero-row dist rank still joins the collectives
                _ffi.check(lib.fh_pack_argkeys(
                    vc.data_ptr(),
                    _ffi.F32 if vals.dtype == torch.float32 else _ffi.I32,
                    vc.numel(), shard_row_offset, int(ismax), int(skip),
                    key.data_ptr(),
                    torch.cuda.current_stream(device).cuda_stream,
                ))
                vc.record_stream(torch.cuda.current_stream(device))
            p = grouped_partials(
                _ffi.SET_MIN_COUNT, key, labels, ngroups,
                skipnan=False, labels2=labels2, grp_shape=grp_pair,
            )
            if dist_on:
                distributed.all_reduce_(p["min"], "min")
                distributed.all_reduce_(p["count"], "sum")
            kmin = p["min"]
            empty_mask = kmin == ((1 << 63) - 1)
            result = (kmin ^ (-(1 << 63))) & 0xFFFFFFFF
            counts_for_mask = p["count"]
            if min_count_ > 0 and skip:
                # the packed keys are always-valid ints, so p["count"]
                # counts NaN rows too; nanarg min_count masks on the
                # NaN-aware count
                pc = grouped_partials(
                    _ffi.SET_COUNT, vals, labels, ngroups, skipnan=True,
                    labels2=labels2, grp_shape=grp_pair,
                )
                if dist_on:
                    distributed.all_reduce_(pc["count"], "sum")
                counts_for_mask = pc["count"]
        elif pair_ok:
            p = grouped_partials(
                _ffi.SET_ARGMAX_PAIR if ismax else _ffi.SET_ARGMIN_PAIR,
                vals, labels, ngroups, skipnan=skip,
                labels2=labels2, grp_shape=grp_pair,
                row_offset=shard_row_offset,
            )
            idx = p["idx"]
            empty_mask = idx == ((1 << 63) - 1)
            result = idx
            counts_for_mask = p["count"]
        else:
            # pass 1: the per-group extremum; pass 2: the smallest row index
            # whose value matches it (ties -> first occurrence, like
            # np.argmax; a NaN target matches NaN rows, so non-skip arg*
            # land on the first NaN)
            if skip:
                p1 = run_set(_ffi.SET_MAX_COUNT if ismax else _ffi.SET_MIN_COUNT, True)
            else:
                p1 = run_set(_ffi.SET_MAX_FULL if ismax else _ffi.SET_MIN_FULL, False)
            if dist_on:
                distributed.all_reduce_(p1["max" if ismax else "min"], "max" if ismax else "min")
                distributed.all_reduce_(p1["count"], "sum")
                if "nanflag" in p1:
                    distributed.all_reduce_(p1["nanflag"], "max")
            target = p1["max" if ismax else "min"]
            if "nanflag" in p1 and arr.dtype.is_floating_point:
                target = torch.where(p1["nanflag"] != 0, torch.full_like(target, float("nan")), target)
            p2 = run_set(_ffi.SET_IDXMIN, skip, target=target)
            if dist_on:
                # present/count are rank-local (the kernel marks presence
                # only for rows this rank holds): combine them too, or a
                # rank with no rows of a group would fill -1 while others
                # return the index
                distributed.all_reduce_(p2["idx"], "min")
                distributed.all_reduce_(p2["present"], "max")
                distributed.all_reduce_(p2["count"], "sum")
            idx = p2["idx"]
            sentinel = (1 << 63) - 1
            result = idx
            empty_mask = (p2["present"] == 0) | (idx == sentinel)
            counts_for_mask = p2["count"]
    elif func in ("first", "last", "nanfirst", "nanlast") and (
        ngroups > PACKED_ARG_THRESHOLD
        and not dist_on
        and 0 < vals.numel()
        and shard_row_offset + vals.numel() < (1 << 32)
    ):
        # huge group counts: first/last is a grouped MIN/MAX over the row
        # indices themselves (any value dtype), so the partition path applies
        # where the 20 B/group IDX bins cannot; NaN rows take the opposite
        # extreme under nan* so all-NaN groups land on the empty sentinel
        isfirst = func in ("first", "nanfirst")
        rows_key = torch.arange(vals.numel(), device=device, dtype=torch.int64) + shard_row_offset
        if agg.skipnan and vals.is_floating_point():
            bad_v = (1 << 63) - 1 if isfirst else -1
            rows_key = torch.where(torch.isnan(vals), torch.full_like(rows_key, bad_v), rows_key)
        p = grouped_partials(
            _ffi.SET_MIN_COUNT if isfirst else _ffi.SET_MAX_COUNT,
            rows_key, labels, ngroups,
            skipnan=False, labels2=labels2, grp_shape=grp_pair,
        )
        idx = p["min"] if isfirst else p["max"]
        valid = (idx != ((1 << 63) - 1)) if isfirst else (idx >= 0)
        local = idx - shard_row_offset
        safe = torch.clamp(local, 0, max(vals.numel() - 1, 0))
        result = vals[safe]
        empty_mask = ~valid
        if min_count_ > 0:
            pc = grouped_partials(
                _ffi.SET_COUNT, vals, labels, ngroups, skipnan=agg.skipnan,
                labels2=labels2, grp_shape=grp_pair,
            )
            counts_for_mask = pc["count"]
        else:
            counts_for_mask = torch.where(
                valid, torch.ones_like(idx), torch.zeros_like(idx)
            )
    elif func in ("first", "last", "nanfirst", "nanlast"):
        p = run_set(agg.op_set, agg.skipnan)
        if dist_on:
            distributed.combine_partials(p, agg.combine)
        idx = p["idx"]
        sentinel = ((1 << 63) - 1) if func in ("first", "nanfirst") else -1
        valid = idx != sentinel
        local = idx - shard_row_offset
        n_local = vals.numel()
        safe = torch.clamp(local, 0, max(n_local - 1, 0))
        if dist_on:
            # the owning rank contributes the value; others contribute exact 0
            owner = valid & (local >= 0) & (local < n_local)
            gathered = torch.where(owner, vals[safe], torch.zeros((), dtype=vals.dtype, device=device))
            distributed.all_reduce_(gathered, "sum")
        else:
            gathered = vals[safe]
        result = gathered
        empty_mask = ~valid
        counts_for_mask = p["count"]
    elif func in ("var", "nanvar", "std", "nanstd") and lead_M > 1:
        # single fused pass on the column path: the kernel accumulates
        # mean-shifted sums per (group, column) segment and emits the
        # var_chunk triple (ssd, sum, count); cross-rank combine closes the
        # reference's _var_combine adjustment in one formula
        skip = agg.skipnan
        p = run_set(_ffi.SET_WELFORD, skip)
        ssd, sums, counts = p["wssd"], p["wsum"], p["count"]
        if dist_on:
            t = torch.where(counts > 0, sums * sums / counts, torch.zeros_like(sums))
            a = ssd + t
            distributed.all_reduce_sum_many([a, sums, counts])
            ssd = a - torch.where(counts > 0, sums * sums / counts, torch.zeros_like(sums))
        den_w = counts.to(torch.float64) - ddof
        result = ssd / den_w
        nan_t = torch.full_like(result, float("nan"))
        result = torch.where((den_w < 0) | (counts == 0), nan_t, result)
        if func in ("std", "nanstd"):
            result = torch.sqrt(result)
        counts_for_mask = counts
        empty_mask = counts == 0
    elif func in ("var", "nanvar", "std", "nanstd"):
        skip = agg.skipnan
        p1 = run_set(_ffi.SET_SUM_COUNT, skip)
        if dist_on:
            distributed.all_reduce_sum_many([p1["sum"], p1["count"]])
        counts = p1["count"]
        # deviations about the (global) per-group mean: the cross-rank ssd
        # combine is then a plain sum (the reference's _var_combine,
        # aggregations.py:392-451, with zero adjustment terms)
        means = (p1["sum"].to(torch.float64) / counts).contiguous()
        p2 = run_set(_ffi.SET_SSD, skip, means=means)
        if dist_on:
            distributed.all_reduce_(p2["sum"], "sum")
        ssd = p2["sum"]
        den = counts.to(torch.float64) - ddof
        result = ssd / den
        nan_t = torch.full_like(result, float("nan"))
        result = torch.where((den < 0) | (counts == 0), nan_t, result)
        if func in ("std", "nanstd"):
            result = torch.sqrt(result)
        counts_for_mask = counts
        empty_mask = counts == 0
    else:
        p = run_set(agg.op_set, agg.skipnan)
        if dist_on:
            distributed.combine_partials(p, agg.combine)
        if func == "count":
            result = p["count"]
            empty_mask = p["count"] == 0
            counts_for_mask = p["count"]
        elif func in ("sum", "nansum", "prod", "nanprod"):
            result = p["sum"]
            empty_mask = p["present"] == 0
            counts_for_mask = p["count"]
        elif func in ("mean", "nanmean"):
            result = p["sum"].to(torch.float64) / p["count"]
            empty_mask = p["count"] == 0
            counts_for_mask = p["count"]
        elif func in ("min", "nanmin", "max", "nanmax"):
            result = p["min" if "min" in p else "max"]
            if "nanflag" in p and arr.dtype.is_floating_point:
                nan_t = torch.full_like(result, float("nan"))
                result = torch.where(p["nanflag"] != 0, nan_t, result)
            empty_mask = (p["present"] == 0) if "present" in p else (p["count"] == 0)
            counts_for_mask = p["count"]
        elif func in ("any", "all"):
            if not was_bool:
                raise NotImplementedError("any/all support bool input only")
            result = (p["max" if func == "any" else "min"] == 1)
            empty_mask = p["count"] == 0
            counts_for_mask = p["count"]
        else:  # pragma: no cover
            raise NotImplementedError(func)

    if lead_folded and func in ("argmax", "argmin", "nanargmax", "nanargmin"):
        # composite-group indices are flat (lead*N + t); the API returns the
        # index within the reduced trailing axes (t), like np.argmax(axis=-1)
        offs = (
            torch.arange(lead_M, device=device, dtype=torch.int64) * arg_localize_N
        ).repeat_interleave(base_ngroups)
        result = result - offs  # empty slots are overwritten by the fill below

    # --- finalize: masking + fills + final dtype (reference core.py:410-475) ---
    # fills are applied with an unconditional where (no host sync); the one
    # case that must inspect the mask on the host is a NaN fill on an
    # integer result, which promotes the output dtype (reference
    # core.py:446-449 maybe_promote)
    t_out_dtype = _torch_dtype(out_dtype)
    is_float_out = torch.empty(0, dtype=t_out_dtype).is_floating_point()
    if min_count_ > 0:
        mask = counts_for_mask < min_count_
        nan_fill = isinstance(fill_value, float) and math.isnan(fill_value)
        if fill_value is None or (nan_fill and not is_float_out):
            if bool(mask.any().item()):
                if fill_value is None:
                    raise ValueError("Filling is required but fill_value is None.")
                t_out_dtype = torch.float64
                result = result.to(t_out_dtype)
                result = torch.where(
                    mask, torch.tensor(fill_value, dtype=t_out_dtype, device=device), result
                )
        else:
            result = result.to(t_out_dtype)
            result = torch.where(
                mask, torch.tensor(fill_value, dtype=t_out_dtype, device=device), result
            )
    else:
        fv = fill_value if fill_value is not None else xrdtypes.fill_default(func, out_dtype)
        nan_fv = isinstance(fv, float) and math.isnan(fv)
        if nan_fv and func in ("mean", "nanmean", "var", "nanvar", "std", "nanstd"):
            # 0/0 division already produced NaN exactly where the fill goes
            pass
        elif nan_fv and not is_float_out:
            if bool(empty_mask.any().item()):
                result = result.to(torch.float64)
                t_out_dtype = torch.float64
                result = torch.where(
                    empty_mask, torch.tensor(fv, dtype=t_out_dtype, device=device), result
                )
        else:
            result = result.to(t_out_dtype)
            result = torch.where(
                empty_mask, torch.tensor(fv, dtype=t_out_dtype, device=device), result
            )

    result = result.to(t_out_dtype)
    if lead_M > 1 and not lead_folded:
        # column partials are (ngroups, M) group-major; the API result puts
        # the group dims last
        result = result.reshape(ngroups, lead_M).t().contiguous()
    if subset_keep_shape is not None:
        result = result.reshape(lead_shape + subset_keep_shape + grp_shape)
    else:
        result = result.reshape(lead_shape + grp_shape)

    groups_list = [f.groups for f in facs]
    if not sort and not provided_expected:
        # groups in first-appearance order (reference pd.factorize(sort=False),
        # factorize.py:96): find each group's first row with an index-min pass
        # and permute the result bins — rows need no relabeling; the group
        # dims are always the trailing len(facs) axes of the result
        for ax_i, f in enumerate(facs):
            fcodes = f.codes
            if subset_keep_shape is not None:
                # restore the original by layout before taking first rows
                fcodes = (
                    fcodes.reshape(by_shape).permute(subset_by_inv).reshape(-1)
                ).contiguous()
            pidx = grouped_partials(
                _ffi.SET_IDXMIN, fcodes.to(torch.int64), fcodes, f.ngroups
            )
            order = torch.argsort(pidx["idx"], stable=True)
            result = torch.index_select(result, result.dim() - len(facs) + ax_i, order)
            groups_list[ax_i] = np.asarray(groups_list[ax_i])[order.cpu().numpy()]
    groups = tuple(groups_list)
    if return_numpy:
        out_np = result.cpu().numpy()
        if dt_dtype is not None and func not in ("count", "any", "all"):
            # the reference casts every non-count result back to the datetime
            # dtype (core.py:1209-1211): int64 counts reinterpret, float
            # results truncate, NaN -> NaT
            out_np = out_np.astype(dt_dtype)
        out_np = _cast_back_small(out_np, func, small_dtype, user_fill_value)
        return (out_np, *groups)
    if small_dtype is not None and func in _PRESERVE_SMALL:
        # torch outputs for torch small-dtype inputs: preserve-funcs cast
        # back (torch has these dtypes); sums stay in the promoted
        # accumulator dtype (torch has no uint64) - documented
        _back = {np.dtype("uint8"): torch.uint8, np.dtype("int8"): torch.int8,
                 np.dtype("int16"): torch.int16, np.dtype("float16"): torch.float16}
        if np.dtype(small_dtype) in _back:
            result = result.to(_back[np.dtype(small_dtype)])
    return (result, *groups)

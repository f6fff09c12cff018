"""engine="hip": grouped aggregations on MI355X via the floxhip C ABI.

Two levels:

* ``grouped_partials(...)`` — the fused low-level call used by
  ``flox_amd.core.groupby_reduce``: one kernel pass produces every per-group
  partial an aggregation needs (e.g. mean -> {sum(f64), count(i64)}), which
  is also exactly the per-rank state the RCCL combine reduces.

* reference-shaped per-reduction callables (``sum``, ``nansum``, ``mean``,
  ``nanlen`` ...) with the signature of the reference engine seam
  (flox/aggregations.py:60-133 generic_aggregate ->
  f(group_idx, array, *, axis=-1, size=None, fill_value=None, dtype=None)),
  so ``engine="hip"`` registers beside the reference's "flox"/"numpy" engines.

All tensors are torch CUDA tensors; there is NO CPU fallback — if the HIP
library or a GPU is missing these raise.
"""

from __future__ import annotations

import builtins
import ctypes

import numpy as np
import torch

from . import _ffi
from ._ffi import (
    FLAG_SKIPNAN,
    SET_COUNT,
    SET_IDXMAX,
    SET_IDXMIN,
    SET_MAX_COUNT,
    SET_MAX_FULL,
    SET_MIN_COUNT,
    SET_MIN_FULL,
    SET_PROD,
    SET_SSD,
    SET_SUM_COUNT,
    SET_SUM_COUNT_PRESENT,
    SET_WELFORD,
    FhCall,
)

_TORCH_VDTYPE = {
    torch.float32: _ffi.F32,
    torch.float64: _ffi.F64,
    torch.int64: _ffi.I64,
    torch.int32: _ffi.I32,
}
_TORCH_LDTYPE = {torch.int64: _ffi.L_I64, torch.int32: _ffi.L_I32}

# op-set membership: which output tensors each set produces
_SET_MEMBERS = {
    SET_SUM_COUNT: ("sum", "count"),
    SET_SUM_COUNT_PRESENT: ("sum", "count", "present"),
    SET_COUNT: ("count",),
    SET_MIN_FULL: ("min", "count", "present", "nanflag"),
    SET_MIN_COUNT: ("min", "count"),
    SET_MAX_FULL: ("max", "count", "present", "nanflag"),
    SET_MAX_COUNT: ("max", "count"),
    SET_SSD: ("sum",),
    SET_PROD: ("sum", "count", "present"),
    SET_IDXMIN: ("idx", "count", "present"),
    SET_IDXMAX: ("idx", "count", "present"),
    SET_WELFORD: ("wssd", "wsum", "count"),  # cols path only
    # pair-payload arg-reductions (8-B dtypes, partition path only):
    # idx = int64 row (INT64_MAX empty), min/max = the decoded extremum
    _ffi.SET_ARGMIN_PAIR: ("idx", "count", "present", "min", "nanflag"),
    _ffi.SET_ARGMAX_PAIR: ("idx", "count", "present", "max", "nanflag"),
}

IDX_SENTINEL_MIN = (1 << 63) - 1  # untouched IDXMIN bin
IDX_SENTINEL_MAX = -1             # untouched IDXMAX bin



def _record(t, stream):
    """record_stream guard: illegal during hipGraph capture (graph-pool
    allocations have static lifetime there, so it is also unnecessary)."""
    if isinstance(t, torch.Tensor) and not torch.cuda.is_current_stream_capturing():
        t.record_stream(stream)


def _acc_dtype(value_dtype: torch.dtype) -> torch.dtype:
    """sum/ssd accumulator dtype: f64 for floats (npg contract), i64 for ints."""
    return torch.float64 if value_dtype.is_floating_point else torch.int64


def _require_gpu_tensor(t: torch.Tensor, name: str) -> None:
    if not t.is_cuda:
        raise RuntimeError(
            f"flox_amd engine='hip' requires {name} on the GPU (got device={t.device}); "
            "there is deliberately no CPU fallback."
        )


def grouped_partials(
    op_set: int,
    values: torch.Tensor,
    labels: torch.Tensor,
    ngroups: int,
    *,
    labels2: torch.Tensor | None = None,
    grp_shape: tuple[int, int] | None = None,
    means: torch.Tensor | None = None,
    target: torch.Tensor | None = None,
    row_offset: int = 0,
    skipnan: bool = False,
    force_path: int = 0,
) -> dict[str, torch.Tensor]:
    """One fused factorize+reduce pass. Returns per-group partial tensors.

    values, labels: contiguous 1-D CUDA tensors of equal length.
    labels2/grp_shape: fused 2-D groupby (reference factorize.py:102-108).
    means: f64[ngroups] for SET_SSD (var pass 2).
    target/row_offset: SET_IDXMIN/IDXMAX (arg-reductions, first/last).
    force_path: 0 auto, 1 LDS-binned, 2 global-atomic (testing).
    """
    lib = _ffi.load_library()
    _require_gpu_tensor(values, "values")
    _require_gpu_tensor(labels, "labels")
    if ngroups == 0:
        dev0 = values.device
        acc = _acc_dtype(values.dtype)
        empty = {
            "idx": torch.empty(0, dtype=torch.int64, device=dev0),
            "sum": torch.empty(0, dtype=acc, device=dev0),
            "count": torch.empty(0, dtype=torch.int64, device=dev0),
            "present": torch.empty(0, dtype=torch.int32, device=dev0),
            "min": torch.empty(0, dtype=values.dtype, device=dev0),
            "max": torch.empty(0, dtype=values.dtype, device=dev0),
            "nanflag": torch.empty(0, dtype=torch.int32, device=dev0),
        }
        return {k: empty[k] for k in _SET_MEMBERS[op_set]}
    if values.numel() == 0:
        # zero rows: the kernel rejects null device pointers, so feed one
        # dummy row with an out-of-range code (dropped by the bounds check);
        # the init paths then produce the correct empty bins
        values = torch.zeros(1, dtype=values.dtype, device=values.device)
        labels = torch.full((1,), -1, dtype=labels.dtype, device=labels.device)
        if labels2 is not None:
            labels2 = torch.full((1,), -1, dtype=labels2.dtype, device=labels2.device)
    assert values.ndim == 1 and labels.ndim == 1 and values.numel() == labels.numel()
    values = values.contiguous()
    labels = labels.contiguous()
    if labels.dtype not in _TORCH_LDTYPE:
        labels = labels.to(torch.int64)
    if values.dtype not in _TORCH_VDTYPE:
        raise NotImplementedError(f"engine='hip' does not support values dtype {values.dtype}")
    dev = values.device

    c = FhCall()
    c.op_set = op_set
    c.vdtype = _TORCH_VDTYPE[values.dtype]
    c.ldtype = _TORCH_LDTYPE[labels.dtype]
    c.flags = (FLAG_SKIPNAN if skipnan else 0) | (
        _ffi.FLAG_FORCE_LDS if force_path == 1 else _ffi.FLAG_FORCE_ATOMIC if force_path == 2 else 0
    )
    if torch.cuda.is_current_stream_capturing():
        # hipGraph capture: the partition paths read back to the host
        # (overflow check / counting pre-pass), which would invalidate the
        # capture — restrict the C dispatcher to sync-free paths
        c.flags |= _ffi.FLAG_NO_HOST_SYNC
    # sorted-labels direct path at huge group counts: skips every scatter
    # pass (~44 -> 12 B/row). A 4K-pair sample rejects random labels for
    # ~20 us; the full O(n) verification only runs when the sample passes
    # (the time-ordered layout this targets).
    if (
        labels2 is None
        and force_path == 0
        and ngroups >= 8192
        and values.numel() >= 1_000_000
        and labels.numel() > 1
        and not torch.cuda.is_current_stream_capturing()
    ):
        idx = torch.randint(0, labels.numel() - 1, (4096,), device=labels.device)
        if bool((labels[idx + 1] >= labels[idx]).all().item()):
            ok = bool((labels[1:] >= labels[:-1]).all().item())
            if ok and bool((labels[0] >= 0).item()) and bool((labels[-1] < ngroups).item()):
                c.flags |= _ffi.FLAG_SORTED_LABELS
    c.n = values.numel()
    c.ngroups = ngroups
    c.values = values.data_ptr()
    c.labels = labels.data_ptr()
    if labels2 is not None:
        _require_gpu_tensor(labels2, "labels2")
        labels2 = labels2.contiguous()
        if labels2.dtype != labels.dtype:
            labels2 = labels2.to(labels.dtype)
        assert grp_shape is not None and grp_shape[0] * grp_shape[1] == ngroups
        c.labels2 = labels2.data_ptr()
        c.g0, c.g1 = grp_shape
    if means is not None:
        assert means.dtype == torch.float64 and means.is_cuda
        means = means.contiguous()
        c.means = means.data_ptr()
    if target is not None:
        assert target.dtype == values.dtype and target.is_cuda
        target = target.contiguous()
        c.target = target.data_ptr()
    c.row_offset = row_offset

    members = _SET_MEMBERS[op_set]
    out: dict[str, torch.Tensor] = {}
    if "idx" in members:
        out["idx"] = torch.empty(ngroups, dtype=torch.int64, device=dev)
        c.out_sum = out["idx"].data_ptr()
    if "sum" in members:
        out["sum"] = torch.empty(ngroups, dtype=_acc_dtype(values.dtype), device=dev)
        if op_set == SET_SSD:
            out["sum"] = torch.empty(ngroups, dtype=torch.float64, device=dev)
        c.out_sum = out["sum"].data_ptr()
    if "count" in members:
        out["count"] = torch.empty(ngroups, dtype=torch.int64, device=dev)
        c.out_count = out["count"].data_ptr()
    if "present" in members:
        out["present"] = torch.empty(ngroups, dtype=torch.int32, device=dev)
        c.out_present = out["present"].data_ptr()
    if "min" in members:
        out["min"] = torch.empty(ngroups, dtype=values.dtype, device=dev)
        c.out_min = out["min"].data_ptr()
    if "max" in members:
        out["max"] = torch.empty(ngroups, dtype=values.dtype, device=dev)
        c.out_max = out["max"].data_ptr()
    if "nanflag" in members:
        out["nanflag"] = torch.empty(ngroups, dtype=torch.int32, device=dev)
        c.out_nanflag = out["nanflag"].data_ptr()

    nscratch = lib.fh_scratch_bytes(ctypes.byref(c))
    if nscratch < 0:
        raise RuntimeError("fh_scratch_bytes failed")
    scratch = None
    if nscratch:
        scratch = torch.empty(nscratch, dtype=torch.uint8, device=dev)
        c.scratch = scratch.data_ptr()
        c.scratch_bytes = nscratch
    c.stream = torch.cuda.current_stream(dev).cuda_stream

    _ffi.check(lib.fh_grouped_reduce(ctypes.byref(c)))
    out["_path"] = c.path_used  # type: ignore[assignment]
    # keep tensors alive until the stream consumes them (torch caching allocator
    # ties lifetime to the stream via recorded events only for torch ops; we
    # record explicitly)
    for t in (values, labels, labels2, means, target, scratch):
        _record(t, torch.cuda.current_stream(dev))
    return out


# ---------------------------------------------------------------------------
# reference-shaped engine callables (the generic_aggregate seam)
# ---------------------------------------------------------------------------


def _prep(group_idx, array):
    """Coerce engine-seam inputs (torch CUDA tensors; numpy moves to GPU).

    Small/unsigned/half dtypes promote (the reference itself casts arrays to
    the aggregation's intermediate dtype before calling the engine, so what
    arrives here for e.g. a uint8 sum is already uint64): uint64 computes on
    its int64 VIEW — sums are wrap-exact mod 2^64; min/max are correct for
    values < 2^63 (documented) — and u8..u32/i8/i16 promote exactly."""
    if not isinstance(array, torch.Tensor):
        a_np = np.ascontiguousarray(array)
        if a_np.dtype == np.uint64:
            a_np = a_np.view(np.int64)
        elif a_np.dtype.kind in "iu" and a_np.dtype.itemsize < 8 and a_np.dtype not in (
            np.dtype(np.int32), np.dtype(np.int64)
        ):
            a_np = a_np.astype(np.int64)
        elif a_np.dtype == np.float16:
            a_np = a_np.astype(np.float32)
        array = torch.as_tensor(a_np)
    if not isinstance(group_idx, torch.Tensor):
        g_np = np.ascontiguousarray(group_idx)
        if g_np.dtype.kind in "iub" and g_np.dtype not in (
            np.dtype(np.int32), np.dtype(np.int64)
        ):
            g_np = g_np.astype(np.int64)
        group_idx = torch.as_tensor(g_np)
    if not array.is_cuda:
        if not torch.cuda.is_available():
            raise RuntimeError("engine='hip' requires a GPU; none is available")
        array = array.cuda()
    group_idx = group_idx.to(array.device)
    return group_idx.reshape(-1), array.reshape(-1)


def _size_of(group_idx, size):
    if size is not None:
        return int(size)
    return int(group_idx.max().item()) + 1 if group_idx.numel() else 0


def _fill(result: torch.Tensor, mask: torch.Tensor, fill_value):
    if fill_value is None:
        return result
    if mask.any():
        fv = torch.as_tensor(fill_value, dtype=result.dtype, device=result.device)
        result = torch.where(mask, fv, result)
    return result


def _to_dtype(result: torch.Tensor, dtype):
    if dtype is None:
        return result
    import torch.utils.dlpack  # noqa: F401

    td = torch.from_numpy(np.empty(0, dtype=np.dtype(dtype))).dtype
    return result.to(td)


def _sum_like(group_idx, array, *, skipnan, axis=-1, size=None, fill_value=None, dtype=None):
    group_idx, array = _prep(group_idx, array)
    ng = _size_of(group_idx, size)
    p = grouped_partials(SET_SUM_COUNT_PRESENT, array, group_idx, ng, skipnan=skipnan)
    res = p["sum"]
    if array.dtype.is_floating_point:
        res = res.to(array.dtype if dtype is None else res.dtype)
    res = _fill(res, p["present"] == 0, fill_value)
    return _to_dtype(res, dtype)


def sum(group_idx, array, *, axis=-1, size=None, fill_value=None, dtype=None, **kw):  # noqa: A001
    return _sum_like(group_idx, array, skipnan=False, size=size, fill_value=fill_value, dtype=dtype)


def nansum(group_idx, array, *, axis=-1, size=None, fill_value=None, dtype=None, **kw):
    return _sum_like(group_idx, array, skipnan=True, size=size, fill_value=fill_value, dtype=dtype)


def nanlen(group_idx, array, *, axis=-1, size=None, fill_value=None, dtype=None, **kw):
    group_idx, array = _prep(group_idx, array)
    ng = _size_of(group_idx, size)
    p = grouped_partials(SET_COUNT, array, group_idx, ng, skipnan=True)
    return _to_dtype(p["count"], dtype if dtype is not None else np.intp)


count = nanlen


def _mean_like(group_idx, array, *, skipnan, size=None, fill_value=None, dtype=None):
    group_idx, array = _prep(group_idx, array)
    ng = _size_of(group_idx, size)
    p = grouped_partials(SET_SUM_COUNT, array, group_idx, ng, skipnan=skipnan)
    # f64 division explicitly: torch resolves int64/int64 to float32
    res = p["sum"].to(torch.float64) / p["count"]
    if array.dtype.is_floating_point and dtype is None:
        res = res.to(array.dtype)
    res = _fill(res, p["count"] == 0, fill_value)
    return _to_dtype(res, dtype)


def mean(group_idx, array, *, axis=-1, size=None, fill_value=None, dtype=None, **kw):
    return _mean_like(group_idx, array, skipnan=False, size=size, fill_value=fill_value, dtype=dtype)


def nanmean(group_idx, array, *, axis=-1, size=None, fill_value=None, dtype=None, **kw):
    return _mean_like(group_idx, array, skipnan=True, size=size, fill_value=fill_value, dtype=dtype)


def _minmax(group_idx, array, *, ismin, skipnan, size=None, fill_value=None, dtype=None):
    group_idx, array = _prep(group_idx, array)
    ng = _size_of(group_idx, size)
    op = (SET_MIN_COUNT if skipnan else SET_MIN_FULL) if ismin else (
        SET_MAX_COUNT if skipnan else SET_MAX_FULL
    )
    p = grouped_partials(op, array, group_idx, ng, skipnan=skipnan)
    res = p["min" if ismin else "max"]
    if not skipnan and array.dtype.is_floating_point:
        res = torch.where(p["nanflag"] != 0, torch.full_like(res, float("nan")), res)
    empty = (p["present"] == 0) if "present" in p else (p["count"] == 0)
    res = _fill(res, empty, fill_value)
    return _to_dtype(res, dtype)


def min(group_idx, array, *, axis=-1, size=None, fill_value=None, dtype=None, **kw):  # noqa: A001
    return _minmax(group_idx, array, ismin=True, skipnan=False, size=size, fill_value=fill_value, dtype=dtype)


def nanmin(group_idx, array, *, axis=-1, size=None, fill_value=None, dtype=None, **kw):
    return _minmax(group_idx, array, ismin=True, skipnan=True, size=size, fill_value=fill_value, dtype=dtype)


def max(group_idx, array, *, axis=-1, size=None, fill_value=None, dtype=None, **kw):  # noqa: A001
    return _minmax(group_idx, array, ismin=False, skipnan=False, size=size, fill_value=fill_value, dtype=dtype)


def nanmax(group_idx, array, *, axis=-1, size=None, fill_value=None, dtype=None, **kw):
    return _minmax(group_idx, array, ismin=False, skipnan=True, size=size, fill_value=fill_value, dtype=dtype)


def prod(group_idx, array, *, axis=-1, size=None, fill_value=None, dtype=None, **kw):
    group_idx, array = _prep(group_idx, array)
    ng = _size_of(group_idx, size)
    p = grouped_partials(SET_PROD, array, group_idx, ng, skipnan=False)
    res = p["sum"]
    if array.dtype.is_floating_point:
        res = res.to(array.dtype if dtype is None else res.dtype)
    res = _fill(res, p["present"] == 0, fill_value)
    return _to_dtype(res, dtype)


def nanprod(group_idx, array, *, axis=-1, size=None, fill_value=None, dtype=None, **kw):
    group_idx, array = _prep(group_idx, array)
    ng = _size_of(group_idx, size)
    p = grouped_partials(SET_PROD, array, group_idx, ng, skipnan=True)
    res = p["sum"]
    if array.dtype.is_floating_point:
        res = res.to(array.dtype if dtype is None else res.dtype)
    res = _fill(res, p["present"] == 0, fill_value)
    return _to_dtype(res, dtype)


def grouped_partials_cols(
    op_set: int,
    values2d: torch.Tensor,
    codes_sorted: torch.Tensor,
    perm: torch.Tensor,
    ngroups: int,
    *,
    means: torch.Tensor | None = None,
    skipnan: bool = False,
) -> dict[str, torch.Tensor]:
    """Grouped reduce over the LEADING axis of a 2-D array (n_t, m) with
    column stride 1 (row stride = values2d.stride(0)). codes_sorted/perm:
    int32 tensors from a stable sort of the per-row group codes. Returns
    per-group partials of shape (ngroups, m)."""
    lib = _ffi.load_library()
    _require_gpu_tensor(values2d, "values")
    assert values2d.ndim == 2 and values2d.stride(1) == 1
    assert codes_sorted.dtype == torch.int32 and perm.dtype == torch.int32
    n_t, m = values2d.shape
    dev = values2d.device
    if values2d.dtype not in _TORCH_VDTYPE:
        raise NotImplementedError(f"engine=hip does not support values dtype {values2d.dtype}")

    c = FhCall()
    c.op_set = op_set
    c.vdtype = _TORCH_VDTYPE[values2d.dtype]
    c.ldtype = _ffi.L_I32
    c.flags = FLAG_SKIPNAN if skipnan else 0
    c.n = n_t
    c.m = m
    c.ldm = values2d.stride(0)
    c.ngroups = ngroups
    c.values = values2d.data_ptr()
    codes_sorted = codes_sorted.contiguous()
    perm = perm.contiguous()
    c.labels = codes_sorted.data_ptr()
    c.perm = perm.data_ptr()
    # group-aligned row chunks: enough workgroups to fill the chip with NO
    # partials slab (each chunk owns disjoint groups since codes are sorted)
    chunk_t = None
    col_blocks = builtins.max(1, m // 1024)
    desired = int(builtins.min(64, builtins.max(1, -(-2048 // col_blocks))))
    if desired > 1 and n_t > 1 and not torch.cuda.is_current_stream_capturing():
        # host-side group-aligned chunk planning needs a D2H sync — illegal
        # during hipGraph capture; captured column ops use the slab path
        sc = codes_sorted.cpu().numpy()
        seg = np.flatnonzero(np.diff(sc)) + 1
        target = builtins.max(1, -(-n_t // desired))
        bounds = [0]
        for b in seg:
            if b - bounds[-1] >= target:
                bounds.append(int(b))
        bounds.append(int(n_t))
        if len(bounds) > 2:
            chunk_t = torch.tensor(bounds, dtype=torch.int64, device=dev)
            c.chunk_offsets = chunk_t.data_ptr()
            c.nchunks = len(bounds) - 1
    if means is not None:
        assert means.dtype == torch.float64 and means.is_cuda
        means = means.contiguous()
        c.means = means.data_ptr()

    members = _SET_MEMBERS[op_set]
    out: dict[str, torch.Tensor] = {}
    shape = (ngroups, m)
    if "wssd" in members:
        out["wssd"] = torch.empty(shape, dtype=torch.float64, device=dev)
        c.out_sum = out["wssd"].data_ptr()
        out["wsum"] = torch.empty(shape, dtype=torch.float64, device=dev)
        c.out_min = out["wsum"].data_ptr()  # sum-of-x rides the out_min slot
    if "sum" in members:
        dt = torch.float64 if op_set == SET_SSD else _acc_dtype(values2d.dtype)
        out["sum"] = torch.empty(shape, dtype=dt, device=dev)
        c.out_sum = out["sum"].data_ptr()
    if "count" in members:
        out["count"] = torch.empty(shape, dtype=torch.int64, device=dev)
        c.out_count = out["count"].data_ptr()
    if "present" in members:
        out["present"] = torch.empty(shape, dtype=torch.int32, device=dev)
        c.out_present = out["present"].data_ptr()
    if "min" in members:
        out["min"] = torch.empty(shape, dtype=values2d.dtype, device=dev)
        c.out_min = out["min"].data_ptr()
    if "max" in members:
        out["max"] = torch.empty(shape, dtype=values2d.dtype, device=dev)
        c.out_max = out["max"].data_ptr()
    if "nanflag" in members:
        out["nanflag"] = torch.empty(shape, dtype=torch.int32, device=dev)
        c.out_nanflag = out["nanflag"].data_ptr()

    nscratch = lib.fh_scratch_bytes(ctypes.byref(c))
    if nscratch < 0:
        raise RuntimeError("fh_scratch_bytes failed")
    scratch = None
    if nscratch:
        scratch = torch.empty(nscratch, dtype=torch.uint8, device=dev)
        c.scratch = scratch.data_ptr()
        c.scratch_bytes = nscratch
    c.stream = torch.cuda.current_stream(dev).cuda_stream
    _ffi.check(lib.fh_grouped_reduce_cols(ctypes.byref(c)))
    out["_path"] = c.path_used  # type: ignore[assignment]
    for t in (values2d, codes_sorted, perm, means, scratch, chunk_t):
        _record(t, torch.cuda.current_stream(dev))
    return out


def grouped_quantile(
    values: torch.Tensor,
    labels: torch.Tensor,
    ngroups: int,
    q,
    *,
    skipnan: bool = False,
    labels2: torch.Tensor | None = None,
    grp_shape: tuple[int, int] | None = None,
) -> torch.Tensor:
    """Grouped linear-interpolation quantiles via the sorted path
    (fh_grouped_quantile). Returns f64 (nq, ngroups); NaN for empty groups,
    all-NaN groups (skipnan) and NaN-containing groups (not skipnan)."""
    lib = _ffi.load_library()
    _require_gpu_tensor(values, "values")
    _require_gpu_tensor(labels, "labels")
    values = values.contiguous()
    labels = labels.contiguous()
    if labels.dtype not in _TORCH_LDTYPE:
        labels = labels.to(torch.int64)
    dev = values.device
    q_t = torch.as_tensor(np.atleast_1d(np.asarray(q, dtype=np.float64)), device=dev)
    nq = q_t.numel()
    out = torch.empty((nq, ngroups), dtype=torch.float64, device=dev)

    c = FhCall()
    c.vdtype = _TORCH_VDTYPE[values.dtype]
    c.ldtype = _TORCH_LDTYPE[labels.dtype]
    c.flags = FLAG_SKIPNAN if skipnan else 0
    c.n = values.numel()
    c.ngroups = ngroups
    c.values = values.data_ptr()
    c.labels = labels.data_ptr()
    if labels2 is not None:
        labels2 = labels2.contiguous()
        if labels2.dtype != labels.dtype:
            labels2 = labels2.to(labels.dtype)
        c.labels2 = labels2.data_ptr()
        c.g0, c.g1 = grp_shape
    c.means = q_t.contiguous().data_ptr()
    c.out_sum = out.data_ptr()
    nscratch = lib.fh_quantile_scratch_bytes(ctypes.byref(c))
    scratch = torch.empty(builtins.max(int(nscratch), 1), dtype=torch.uint8, device=dev)
    c.scratch = scratch.data_ptr()
    c.scratch_bytes = nscratch
    c.stream = torch.cuda.current_stream(dev).cuda_stream
    _ffi.check(lib.fh_grouped_quantile(ctypes.byref(c), nq))
    for t in (values, labels, labels2, q_t, scratch, out):
        _record(t, torch.cuda.current_stream(dev))
    return out


def grouped_mode(
    values: torch.Tensor,
    labels: torch.Tensor,
    ngroups: int,
    *,
    skipnan: bool = False,
    labels2: torch.Tensor | None = None,
    grp_shape: tuple[int, int] | None = None,
) -> torch.Tensor:
    """Grouped mode via the sorted path (scipy.stats.mode semantics: most
    frequent value, ties -> smallest; NaN propagates unless skipnan)."""
    lib = _ffi.load_library()
    _require_gpu_tensor(values, "values")
    if values.is_floating_point():
        # scipy counts by NUMERIC equality: -0.0 and +0.0 are one value, but
        # their order-preserving encodings differ — canonicalize to +0.0 so
        # the sorted runs merge
        values = torch.where(values == 0, torch.zeros_like(values), values)
    values = values.contiguous()
    labels = labels.contiguous()
    if labels.dtype not in _TORCH_LDTYPE:
        labels = labels.to(torch.int64)
    dev = values.device
    out = torch.empty(ngroups, dtype=values.dtype, device=dev)
    c = FhCall()
    c.vdtype = _TORCH_VDTYPE[values.dtype]
    c.ldtype = _TORCH_LDTYPE[labels.dtype]
    c.flags = FLAG_SKIPNAN if skipnan else 0
    c.n = values.numel()
    c.ngroups = ngroups
    c.values = values.data_ptr()
    c.labels = labels.data_ptr()
    if labels2 is not None:
        labels2 = labels2.contiguous()
        if labels2.dtype != labels.dtype:
            labels2 = labels2.to(labels.dtype)
        c.labels2 = labels2.data_ptr()
        c.g0, c.g1 = grp_shape
    c.out_sum = out.data_ptr()
    nscratch = lib.fh_quantile_scratch_bytes(ctypes.byref(c))
    scratch = torch.empty(builtins.max(int(nscratch), 1), dtype=torch.uint8, device=dev)
    c.scratch = scratch.data_ptr()
    c.scratch_bytes = nscratch
    c.stream = torch.cuda.current_stream(dev).cuda_stream
    _ffi.check(lib.fh_grouped_quantile(ctypes.byref(c), 0))
    for t in (values, labels, labels2, scratch, out):
        _record(t, torch.cuda.current_stream(dev))
    return out


def var_partials(
    group_idx, array, *, skipnan, size, labels2=None, grp_shape=None,
    global_counts=None, global_sums=None,
):
    """The (ssd, sum, len) triple of the reference's var_chunk
    (flox/aggregations.py:348-389): pass 1 sum+count -> means, pass 2
    sum of squared deviations about those means.

    When global_counts/sums are given (multi-GPU), deviations are taken about
    the GLOBAL mean so the cross-rank combine is a plain sum (algebraically
    the reference's _var_combine with zero adjustment terms)."""
    group_idx, array = _prep(group_idx, array)
    ng = _size_of(group_idx, size)
    kw = dict(labels2=labels2, grp_shape=grp_shape)
    if global_counts is None:
        p1 = grouped_partials(SET_SUM_COUNT, array, group_idx, ng, skipnan=skipnan, **kw)
        sums, counts = p1["sum"], p1["count"]
    else:
        sums, counts = global_sums, global_counts
    # empty groups give mean = 0/0 = NaN, but no row carries their code, so the
    # SSD kernel never reads those entries
    means = sums.to(torch.float64) / counts
    p2 = grouped_partials(SET_SSD, array, group_idx, ng, skipnan=skipnan, means=means, **kw)
    return p2["sum"], sums, counts


# ---------------------------------------------------------------------------
# remaining seam callables: var/std, arg-reductions, first/last, quantiles,
# mode, any/all — so every reference reduction name resolves through
# generic_aggregate(engine="hip") (reference aggregations.py:60-133)
# ---------------------------------------------------------------------------


def _var_like(group_idx, array, *, skipnan, std, size=None, fill_value=None, dtype=None, ddof=0):
    group_idx, array = _prep(group_idx, array)
    ng = _size_of(group_idx, size)
    ssd, sums, counts = var_partials(group_idx, array, skipnan=skipnan, size=ng)
    den = counts.to(torch.float64) - ddof
    res = ssd / den
    nan_t = torch.full_like(res, float("nan"))
    res = torch.where((den < 0) | (counts == 0), nan_t, res)
    if std:
        res = torch.sqrt(res)
    if array.dtype.is_floating_point and dtype is None:
        res = res.to(array.dtype)
    res = _fill(res, counts == 0, fill_value)
    return _to_dtype(res, dtype)


def var(group_idx, array, *, axis=-1, size=None, fill_value=None, dtype=None, ddof=0, **kw):
    return _var_like(group_idx, array, skipnan=False, std=False, size=size, fill_value=fill_value, dtype=dtype, ddof=ddof)


def nanvar(group_idx, array, *, axis=-1, size=None, fill_value=None, dtype=None, ddof=0, **kw):
    return _var_like(group_idx, array, skipnan=True, std=False, size=size, fill_value=fill_value, dtype=dtype, ddof=ddof)


def std(group_idx, array, *, axis=-1, size=None, fill_value=None, dtype=None, ddof=0, **kw):
    return _var_like(group_idx, array, skipnan=False, std=True, size=size, fill_value=fill_value, dtype=dtype, ddof=ddof)


def nanstd(group_idx, array, *, axis=-1, size=None, fill_value=None, dtype=None, ddof=0, **kw):
    return _var_like(group_idx, array, skipnan=True, std=True, size=size, fill_value=fill_value, dtype=dtype, ddof=ddof)


def _arg_like(group_idx, array, *, ismax, skipnan, size=None, fill_value=None, dtype=None):
    group_idx, array = _prep(group_idx, array)
    ng = _size_of(group_idx, size)
    if skipnan:
        p1 = grouped_partials(SET_MAX_COUNT if ismax else SET_MIN_COUNT, array, group_idx, ng, skipnan=True)
    else:
        p1 = grouped_partials(SET_MAX_FULL if ismax else SET_MIN_FULL, array, group_idx, ng, skipnan=False)
    target = p1["max" if ismax else "min"]
    if "nanflag" in p1 and array.dtype.is_floating_point:
        target = torch.where(p1["nanflag"] != 0, torch.full_like(target, float("nan")), target)
    p2 = grouped_partials(SET_IDXMIN, array, group_idx, ng, skipnan=skipnan, target=target)
    idx = p2["idx"]
    sentinel = (1 << 63) - 1
    missing = (p2["present"] == 0) | (idx == sentinel)
    res = torch.where(missing, torch.full_like(idx, -1 if fill_value is None else int(fill_value)), idx)
    return _to_dtype(res, dtype if dtype is not None else np.intp)


def argmax(group_idx, array, *, axis=-1, size=None, fill_value=None, dtype=None, **kw):
    return _arg_like(group_idx, array, ismax=True, skipnan=False, size=size, fill_value=fill_value, dtype=dtype)


def argmin(group_idx, array, *, axis=-1, size=None, fill_value=None, dtype=None, **kw):
    return _arg_like(group_idx, array, ismax=False, skipnan=False, size=size, fill_value=fill_value, dtype=dtype)


def nanargmax(group_idx, array, *, axis=-1, size=None, fill_value=None, dtype=None, **kw):
    return _arg_like(group_idx, array, ismax=True, skipnan=True, size=size, fill_value=fill_value, dtype=dtype)


def nanargmin(group_idx, array, *, axis=-1, size=None, fill_value=None, dtype=None, **kw):
    return _arg_like(group_idx, array, ismax=False, skipnan=True, size=size, fill_value=fill_value, dtype=dtype)


def _pos_like(group_idx, array, *, last, skipnan, size=None, fill_value=None, dtype=None):
    group_idx, array = _prep(group_idx, array)
    ng = _size_of(group_idx, size)
    p = grouped_partials(SET_IDXMAX if last else SET_IDXMIN, array, group_idx, ng, skipnan=skipnan)
    idx = p["idx"]
    sentinel = -1 if last else (1 << 63) - 1
    valid = idx != sentinel
    safe = torch.clamp(idx, 0, builtins.max(array.numel() - 1, 0))
    res = array[safe]
    if fill_value is not None:
        res = _fill(res, ~valid, fill_value)
    elif array.dtype.is_floating_point:
        res = torch.where(~valid, torch.full_like(res, float("nan")), res)
    return _to_dtype(res, dtype)


def first(group_idx, array, *, axis=-1, size=None, fill_value=None, dtype=None, **kw):
    return _pos_like(group_idx, array, last=False, skipnan=False, size=size, fill_value=fill_value, dtype=dtype)


def last(group_idx, array, *, axis=-1, size=None, fill_value=None, dtype=None, **kw):
    return _pos_like(group_idx, array, last=True, skipnan=False, size=size, fill_value=fill_value, dtype=dtype)


def nanfirst(group_idx, array, *, axis=-1, size=None, fill_value=None, dtype=None, **kw):
    return _pos_like(group_idx, array, last=False, skipnan=True, size=size, fill_value=fill_value, dtype=dtype)


def nanlast(group_idx, array, *, axis=-1, size=None, fill_value=None, dtype=None, **kw):
    return _pos_like(group_idx, array, last=True, skipnan=True, size=size, fill_value=fill_value, dtype=dtype)


def _quantile_like(group_idx, array, *, q, skipnan, size=None, fill_value=None, dtype=None):
    group_idx, array = _prep(group_idx, array)
    ng = _size_of(group_idx, size)
    q_arr = np.atleast_1d(np.asarray(q, dtype=np.float64))
    res = grouped_quantile(array, group_idx, ng, q_arr, skipnan=skipnan)
    if np.isscalar(q) or np.ndim(q) == 0:
        res = res[0]
    if array.dtype.is_floating_point and dtype is None:
        res = res.to(array.dtype)
    return _to_dtype(res, dtype)


def quantile(group_idx, array, *, q, axis=-1, size=None, fill_value=None, dtype=None, **kw):
    return _quantile_like(group_idx, array, q=q, skipnan=False, size=size, fill_value=fill_value, dtype=dtype)


def nanquantile(group_idx, array, *, q, axis=-1, size=None, fill_value=None, dtype=None, **kw):
    return _quantile_like(group_idx, array, q=q, skipnan=True, size=size, fill_value=fill_value, dtype=dtype)


def median(group_idx, array, *, axis=-1, size=None, fill_value=None, dtype=None, **kw):
    return _quantile_like(group_idx, array, q=0.5, skipnan=False, size=size, fill_value=fill_value, dtype=dtype)


def nanmedian(group_idx, array, *, axis=-1, size=None, fill_value=None, dtype=None, **kw):
    return _quantile_like(group_idx, array, q=0.5, skipnan=True, size=size, fill_value=fill_value, dtype=dtype)


def mode(group_idx, array, *, axis=-1, size=None, fill_value=None, dtype=None, **kw):
    group_idx, array = _prep(group_idx, array)
    ng = _size_of(group_idx, size)
    return _to_dtype(grouped_mode(array, group_idx, ng, skipnan=False), dtype)


def nanmode(group_idx, array, *, axis=-1, size=None, fill_value=None, dtype=None, **kw):
    group_idx, array = _prep(group_idx, array)
    ng = _size_of(group_idx, size)
    return _to_dtype(grouped_mode(array, group_idx, ng, skipnan=True), dtype)


def _scan_like(group_idx, array, func, size=None, dtype=None):
    from .scan import groupby_scan

    # keep the array's leading dims — groupby_scan folds them natively
    if not isinstance(array, torch.Tensor):
        array = torch.as_tensor(np.ascontiguousarray(array))
    if not array.is_cuda:
        if not torch.cuda.is_available():
            raise RuntimeError("engine='hip' requires a GPU; none is available")
        array = array.cuda()
    gi = group_idx if isinstance(group_idx, torch.Tensor) else torch.as_tensor(
        np.ascontiguousarray(group_idx))
    gi = gi.to(array.device).reshape(-1)
    ng = _size_of(gi, size)
    return groupby_scan(array, gi, func=func, expected_groups=range(ng))


def cumsum(group_idx, array, *, axis=-1, size=None, fill_value=None, dtype=None, **kw):
    return _scan_like(group_idx, array, "cumsum", size=size, dtype=dtype)


def nancumsum(group_idx, array, *, axis=-1, size=None, fill_value=None, dtype=None, **kw):
    return _scan_like(group_idx, array, "nancumsum", size=size, dtype=dtype)


def ffill(group_idx, array, *, axis=-1, size=None, fill_value=None, dtype=None, **kw):
    return _scan_like(group_idx, array, "ffill", size=size, dtype=dtype)


def bfill(group_idx, array, *, axis=-1, size=None, fill_value=None, dtype=None, **kw):
    return _scan_like(group_idx, array, "bfill", size=size, dtype=dtype)


def _truthy_i64(array: torch.Tensor) -> torch.Tensor:
    """numpy truthiness as int64: for floats, NaN is truthy (np.any of a
    NaN-containing group is True), so (v != 0) | isnan(v) — a raw
    float->int cast would turn NaN into 0 on the GPU."""
    if array.dtype.is_floating_point:
        return ((array != 0) | torch.isnan(array)).to(torch.int64)
    return array.to(torch.int64)


def any_(group_idx, array, *, axis=-1, size=None, fill_value=None, dtype=None, **kw):
    group_idx, array = _prep(group_idx, array)
    ng = _size_of(group_idx, size)
    p = grouped_partials(SET_MAX_COUNT, _truthy_i64(array), group_idx, ng)
    return (p["max"] != 0) & (p["count"] > 0)


def all_(group_idx, array, *, axis=-1, size=None, fill_value=None, dtype=None, **kw):
    group_idx, array = _prep(group_idx, array)
    ng = _size_of(group_idx, size)
    p = grouped_partials(SET_MIN_COUNT, _truthy_i64(array), group_idx, ng)
    return (p["min"] != 0) & (p["count"] > 0)


# ---------------------------------------------------------------------------
# leading array dims through the engine seam: the reference's engine
# callables accept (lead..., n) arrays with (n,) group_idx, reducing over
# axis=-1 (aggregate_flox.py:133-192 operates on the trailing axis). Fold
# the lead index into composite codes (lead*ng + code) and reshape back.
# ---------------------------------------------------------------------------


def _fold_lead(group_idx, array, size):
    if not isinstance(array, torch.Tensor):
        array = torch.as_tensor(np.ascontiguousarray(array))
    if not isinstance(group_idx, torch.Tensor):
        group_idx = torch.as_tensor(np.ascontiguousarray(group_idx))
    if not array.is_cuda:
        if not torch.cuda.is_available():
            raise RuntimeError("engine='hip' requires a GPU; none is available")
        array = array.cuda()
    gi = group_idx.to(array.device).reshape(-1).to(torch.int64)
    n = gi.numel()
    if array.shape[-1] != n:
        raise ValueError(f"group_idx ({n},) must match array's trailing axis {tuple(array.shape)}")
    lead_shape = tuple(array.shape[:-1])
    M = 1
    for d in lead_shape:
        M *= d
    ng = _size_of(gi, size)
    lead = torch.arange(M, device=array.device, dtype=torch.int64)
    comp = lead[:, None] * ng + gi[None, :]
    comp = torch.where((gi < 0)[None, :].expand_as(comp), torch.full_like(comp, -1), comp)
    return comp.reshape(-1), array.reshape(-1), M, ng, lead_shape, n


def _lead_wrap(kind):
    import functools

    def deco(f):
        @functools.wraps(f)
        def wrapper(group_idx, array, *, axis=-1, size=None, fill_value=None, dtype=None, **kw):
            a_nd = getattr(array, "ndim", None)
            if a_nd is None:
                a_nd = np.ndim(array)
            g_nd = getattr(group_idx, "ndim", None)
            if g_nd is None:
                g_nd = np.ndim(group_idx)
            if a_nd <= builtins.max(g_nd, 1):
                return f(group_idx, array, axis=axis, size=size,
                         fill_value=fill_value, dtype=dtype, **kw)
            comp, flat, M, ng, lead_shape, n = _fold_lead(group_idx, array, size)
            out = f(comp, flat, axis=-1, size=M * ng,
                    fill_value=fill_value, dtype=dtype, **kw)
            if kind == "q" and out.ndim == 2:  # (nq, M*ng)
                return out.reshape((out.shape[0],) + lead_shape + (ng,))
            out = out.reshape(lead_shape + (ng,))
            if kind == "arg":
                # composite indices are flat (lead*n + t); the API returns t
                offs = (torch.arange(M, device=out.device, dtype=out.dtype) * n).reshape(
                    lead_shape + (1,))
                out = torch.where(out >= 0, out - offs, out)
            return out
        return wrapper
    return deco


_LEAD_KINDS = {
    "sum": "reduce", "nansum": "reduce", "nanlen": "reduce", "count": "reduce",
    "prod": "reduce", "nanprod": "reduce", "mean": "reduce", "nanmean": "reduce",
    "min": "reduce", "nanmin": "reduce", "max": "reduce", "nanmax": "reduce",
    "var": "reduce", "nanvar": "reduce", "std": "reduce", "nanstd": "reduce",
    "argmax": "arg", "argmin": "arg", "nanargmax": "arg", "nanargmin": "arg",
    "first": "reduce", "last": "reduce", "nanfirst": "reduce", "nanlast": "reduce",
    "median": "reduce", "nanmedian": "reduce", "quantile": "q", "nanquantile": "q",
    "mode": "reduce", "nanmode": "reduce", "any_": "reduce", "all_": "reduce",
}
for _n, _k in _LEAD_KINDS.items():
    globals()[_n] = _lead_wrap(_k)(globals()[_n])
del _n, _k


# ---------------------------------------------------------------------------
# numpy-in -> numpy-out at the seam: the reference's orchestrator
# (chunk_reduce / _finalize_results) operates on numpy arrays, so when the
# caller hands numpy in, hand numpy back (callee-allocates convention,
# reference aggregate_flox.py:174-180). torch callers keep torch tensors
# on-device. Outermost wrapper (after the lead-dim fold).
# ---------------------------------------------------------------------------


def _numpy_io_wrap(f):
    import functools

    @functools.wraps(f)
    def wrapper(group_idx, array, **kw):
        np_in = not isinstance(array, torch.Tensor)
        out = f(group_idx, array, **kw)
        if np_in and isinstance(out, torch.Tensor):
            return out.cpu().numpy()
        return out

    return wrapper


for _n in list(_LEAD_KINDS) + ["cumsum", "nancumsum", "ffill", "bfill"]:
    globals()[_n] = _numpy_io_wrap(globals()[_n])
del _n


# ---------------------------------------------------------------------------
# complex input at the seam: the linear funcs compute componentwise on the
# re/im planes (the reference casts arrays to the aggregation's intermediate
# dtype before the engine call, so complex sums arrive as complex128).
# Whole-value nulls (either component NaN — the reference's isnull) premask
# BOTH components for the skipna funcs. Outermost wrapper.
# ---------------------------------------------------------------------------

_COMPLEX_SKIPNA = {"nansum", "nanmean", "nanfirst", "nanlast", "nancumsum",
                   "nanlen", "count"}


def _complex_io_wrap(f, name):
    import functools

    @functools.wraps(f)
    def wrapper(group_idx, array, **kw):
        if isinstance(array, torch.Tensor) or np.asarray(array).dtype.kind != "c":
            return f(group_idx, array, **kw)
        a = np.asarray(array)
        comp_dt = a.dtype
        fw = np.float64 if comp_dt == np.dtype(np.complex128) else np.float32
        if name in ("nanlen", "count"):
            real = a.real.astype(np.float64)
            real[np.isnan(a)] = np.nan  # whole-value null
            return f(group_idx, real, **kw)
        fv = a.view(fw).reshape(a.shape + (2,)).copy()
        if name in _COMPLEX_SKIPNA:
            fv[np.isnan(a)] = np.nan
        kw.pop("dtype", None)
        rr = f(group_idx, np.ascontiguousarray(fv[..., 0]), **kw)
        ri = f(group_idx, np.ascontiguousarray(fv[..., 1]), **kw)
        return (np.asarray(rr) + 1j * np.asarray(ri)).astype(comp_dt)

    return wrapper


for _n in ["sum", "nansum", "mean", "nanmean", "nanlen", "count",
           "first", "last", "nanfirst", "nanlast", "cumsum", "nancumsum"]:
    globals()[_n] = _complex_io_wrap(globals()[_n], _n)
del _n

"""Aggregation blueprints and the engine-dispatch seam.

Keeps the reference's operator API intact (flox/aggregations.py:161-301
``Aggregation``; :60-133 ``generic_aggregate``): every reduction is described
by the per-block partials it needs (``chunk``), how partials merge across
blocks/GPUs (``combine``), and how partials become the result (``finalize``).
Here ``chunk`` maps onto ONE fused HIP op-set per pass (include/floxhip.h),
and ``combine`` is the RCCL reduction applied to each partial.
"""

from __future__ import annotations

from dataclasses import dataclass, field

from . import _ffi


@dataclass
class Aggregation:
    """Blueprint for a grouped reduction (mirrors reference
    flox/aggregations.py:161-301, re-expressed for the fused HIP engine).

    op_set: the fused kernel pass (fh_opset) producing this aggregation's
        partials; None for multi-pass aggregations handled specially (var).
    skipnan: NaN values contribute nothing (the reference's nan* family).
    combine: per-partial cross-block/cross-GPU reduction: name -> "sum" |
        "min" | "max" | "or" (reference combine recipes,
        flox/aggregations.py:304-546).
    """

    name: str
    op_set: int | None
    skipnan: bool = False
    combine: dict = field(default_factory=dict)
    preserves_dtype: bool = False


# present rides the same coalesced SUM all-reduce as sum/count: the
# per-rank flag is {0,1} and every consumer tests present != 0
_SUMLIKE_COMBINE = {"sum": "sum", "count": "sum", "present": "sum"}

REDUCTIONS: dict[str, Aggregation] = {
    "count": Aggregation("count", _ffi.SET_COUNT, skipnan=True, combine={"count": "sum"}),
    "sum": Aggregation("sum", _ffi.SET_SUM_COUNT_PRESENT, combine=dict(_SUMLIKE_COMBINE)),
    "nansum": Aggregation(
        "nansum", _ffi.SET_SUM_COUNT_PRESENT, skipnan=True, combine=dict(_SUMLIKE_COMBINE)
    ),
    "prod": Aggregation("prod", _ffi.SET_PROD, combine={"sum": "prod", "count": "sum", "present": "sum"}),
    "nanprod": Aggregation(
        "nanprod", _ffi.SET_PROD, skipnan=True, combine={"sum": "prod", "count": "sum", "present": "sum"}
    ),
    "mean": Aggregation("mean", _ffi.SET_SUM_COUNT, combine={"sum": "sum", "count": "sum"}),
    "nanmean": Aggregation(
        "nanmean", _ffi.SET_SUM_COUNT, skipnan=True, combine={"sum": "sum", "count": "sum"}
    ),
    "min": Aggregation(
        "min",
        _ffi.SET_MIN_FULL,
        combine={"min": "min", "count": "sum", "present": "sum", "nanflag": "max"},
        preserves_dtype=True,
    ),
    "nanmin": Aggregation(
        "nanmin", _ffi.SET_MIN_COUNT, skipnan=True, combine={"min": "min", "count": "sum"},
        preserves_dtype=True,
    ),
    "max": Aggregation(
        "max",
        _ffi.SET_MAX_FULL,
        combine={"max": "max", "count": "sum", "present": "sum", "nanflag": "max"},
        preserves_dtype=True,
    ),
    "nanmax": Aggregation(
        "nanmax", _ffi.SET_MAX_COUNT, skipnan=True, combine={"max": "max", "count": "sum"},
        preserves_dtype=True,
    ),
    # var family: two fused passes (SUM_COUNT then SSD); see core._reduce_var
    "var": Aggregation("var", None),
    "nanvar": Aggregation("nanvar", None, skipnan=True),
    # arg-reductions: extremum pass then min-index-among-matches pass
    # (reference argmax/argmin/nanarg* recipes, aggregations.py:582-649)
    "argmax": Aggregation("argmax", None),
    "argmin": Aggregation("argmin", None),
    "nanargmax": Aggregation("nanargmax", None, skipnan=True),
    "nanargmin": Aggregation("nanargmin", None, skipnan=True),
    # first/last: index extremum + gather (reference aggregations.py:635-649)
    "first": Aggregation(
        "first", _ffi.SET_IDXMIN,
        combine={"idx": "min", "count": "sum", "present": "sum"},
        preserves_dtype=True,
    ),
    "last": Aggregation(
        "last", _ffi.SET_IDXMAX,
        combine={"idx": "max", "count": "sum", "present": "sum"},
        preserves_dtype=True,
    ),
    "nanfirst": Aggregation(
        "nanfirst", _ffi.SET_IDXMIN, skipnan=True,
        combine={"idx": "min", "count": "sum", "present": "sum"},
        preserves_dtype=True,
    ),
    "nanlast": Aggregation(
        "nanlast", _ffi.SET_IDXMAX, skipnan=True,
        combine={"idx": "max", "count": "sum", "present": "sum"},
        preserves_dtype=True,
    ),
    # quantile family: sorted path (blockwise-only in the reference too,
    # aggregations.py:672-712); median = quantile(q=0.5)
    "quantile": Aggregation("quantile", None),
    "nanquantile": Aggregation("nanquantile", None, skipnan=True),
    "median": Aggregation("median", None),
    "nanmedian": Aggregation("nanmedian", None, skipnan=True),
    "mode": Aggregation("mode", None, preserves_dtype=True),
    "nanmode": Aggregation("nanmode", None, skipnan=True, preserves_dtype=True),
    # bool reductions via min/max of the 0/1-cast input
    # (reference aggregations.py:651-676 all_/any_)
    "any": Aggregation("any", _ffi.SET_MAX_COUNT, combine={"max": "max", "count": "sum"}),
    "all": Aggregation("all", _ffi.SET_MIN_COUNT, combine={"min": "min", "count": "sum"}),
    "std": Aggregation("std", None),
    "nanstd": Aggregation("nanstd", None, skipnan=True),
}


def generic_aggregate(
    group_idx,
    array,
    *,
    engine: str,
    func: str,
    axis=-1,
    size=None,
    fill_value=None,
    dtype=None,
    **kwargs,
):
    """Engine dispatch, the reference's plugin seam
    (flox/aggregations.py:60-133). flox_amd registers exactly one engine,
    "hip"; anything else is an explicit error — there is no CPU fallback."""
    if engine != "hip":
        raise ValueError(
            f"flox_amd implements engine='hip' only (got {engine!r}). "
            "For CPU semantics use the reference implementation."
        )
    from . import aggregate_hip

    seam_name = {"any": "any_", "all": "all_"}.get(func, func)
    method = getattr(aggregate_hip, seam_name, None)
    if method is None:
        raise NotImplementedError(f"engine='hip' does not implement {func!r} yet")
    return method(group_idx, array, axis=axis, size=size, fill_value=fill_value, dtype=dtype, **kwargs)


class CustomAggregation:
    """Reference-compatible custom aggregation (flox.Aggregation,
    reference aggregations.py:161-301 / docs "Custom Aggregations"):
    compose existing chunk reductions with a user ``finalize``.

    Supported ``chunk`` names: sum nansum count nanlen min nanmin max nanmax
    prod nanprod. ``finalize`` receives one torch tensor per chunk
    intermediate (use array operators, not numpy functions). ``combine``
    names drive the cross-GPU all-reduce of each intermediate.
    """

    def __init__(
        self,
        name,
        *,
        numpy=None,
        chunk,
        combine,
        preprocess=None,
        aggregate=None,
        finalize=None,
        fill_value=None,
        final_fill_value=float("nan"),
        dtypes=None,
        final_dtype=None,
        reduction_type="reduce",
    ):
        if preprocess is not None:
            raise NotImplementedError("CustomAggregation.preprocess")
        self.name = name
        self.numpy = numpy
        self.chunk = (chunk,) if isinstance(chunk, str) else tuple(chunk)
        self.combine = (combine,) if isinstance(combine, str) else tuple(combine)
        self.finalize = finalize
        self.fill_value = fill_value
        self.final_fill_value = final_fill_value
        self.final_dtype = final_dtype

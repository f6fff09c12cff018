"""Cross-GPU combine of per-group partial bins over RCCL/xGMI.

Replaces the reference's dask combine tree (flox/dask.py:90-144
_simple_combine: concatenate block intermediates, reduce over the block
axis) with a single collective per partial: the partial bins computed by
each rank's fused kernel ARE the intermediates, and the combine recipe per
partial (sum/min/max — reference flox/aggregations.py:304-546) maps 1:1 to
an all-reduce op. One process per GPU; backend "nccl" (RCCL on ROCm) on
GPUs, "gloo" for the CPU-only correctness tests.
"""

from __future__ import annotations

import torch
import torch.distributed as dist

_OPS = None


def _ops():
    global _OPS
    if _OPS is None:
        _OPS = {
            "sum": dist.ReduceOp.SUM,
            "min": dist.ReduceOp.MIN,
            "max": dist.ReduceOp.MAX,
            "prod": dist.ReduceOp.PRODUCT,
        }
    return _OPS


def is_active() -> bool:
    return dist.is_available() and dist.is_initialized() and dist.get_world_size() > 1


def combine_partials(partials: dict[str, torch.Tensor], combine: dict[str, str]) -> dict:
    """All-reduce each partial bin across ranks with its combine op, in place.

    min/max bins hold +inf/-inf for groups a rank never saw (the reference's
    intermediate fill values, flox/aggregations.py:529-546), so the
    collective needs no validity mask.
    """
    if not is_active():
        return partials
    for name, op in combine.items():
        t = partials.get(name)
        if t is None:
            continue
        dist.all_reduce(t, op=_ops()[op])
    return partials


def all_reduce_(t: torch.Tensor, op: str = "sum") -> torch.Tensor:
    if is_active():
        dist.all_reduce(t, op=_ops()[op])
    return t

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


# group-count threshold for the shard-aware sparse combine (the cohorts
# label-locality idea of reference cohorts.py:109-301 restated for ranks:
# when each rank's shard touches only a subset of the groups, exchanging
# compressed (index, value) bins beats all-reducing the full 1e7-group
# buffers). Below it the dense all-reduce always wins (bins are a few MB).
SPARSE_NGROUPS = 1 << 20
# engage only when the gathered compressed traffic is under this fraction
# of the dense all-reduce's
SPARSE_FRACTION = 0.25


def _sparse_combine(partials: dict[str, torch.Tensor], combine: dict[str, str]) -> bool:
    """Shard-aware combine: each rank gathers only its TOUCHED bins.

    A bin is touched when any partial differs from its combine identity
    (count/present/nanflag nonzero, or a NaN-poisoned fp sum — the all-NaN
    non-skip group whose count is 0). One scalar all-reduce decides whether
    the locality is real; uniform shards fall back to the dense path
    (returns False). Exact for every combine op (sum/prod/min/max are
    applied only at touched indices; untouched bins already hold the
    identity on every rank)."""
    names = [n for n in combine if n in partials]
    if not names:
        return True
    t0 = partials[names[0]]
    mask = torch.zeros(t0.shape, dtype=torch.bool, device=t0.device)
    for name in names:
        t = partials[name]
        if name in ("count", "present", "nanflag"):
            mask |= t != 0
        elif t.is_floating_point():
            mask |= torch.isnan(t)
    if "count" not in partials and "present" not in partials:
        return False  # no touch signal: stay dense
    world = dist.get_world_size()
    rank = dist.get_rank()
    idx = mask.nonzero(as_tuple=False).flatten()
    nnz = torch.tensor([idx.numel()], dtype=torch.int64, device=t0.device)
    tot = nnz.clone()
    dist.all_reduce(tot, op=dist.ReduceOp.SUM)
    mx = nnz.clone()
    dist.all_reduce(mx, op=dist.ReduceOp.MAX)
    ngroups = t0.numel()
    if float(tot.item()) > SPARSE_FRACTION * world * ngroups:
        return False
    m = int(mx.item())
    if m == 0:
        return True
    pad_idx = torch.full((m,), -1, dtype=torch.int64, device=t0.device)
    pad_idx[: idx.numel()] = idx
    g_idx = [torch.empty_like(pad_idx) for _ in range(world)]
    dist.all_gather(g_idx, pad_idx)
    for name in names:
        t = partials[name]
        vals = t.flatten()[idx] if idx.numel() else t.new_empty(0)
        pad_v = t.new_zeros(m)
        pad_v[: idx.numel()] = vals
        g_v = [torch.empty_like(pad_v) for _ in range(world)]
        dist.all_gather(g_v, pad_v)
        op = combine[name]
        red = {"sum": "sum", "prod": "prod", "min": "amin", "max": "amax"}[op]
        flat = t.flatten()
        for r in range(world):
            if r == rank:
                continue
            gi = g_idx[r]
            ok = gi >= 0
            if not bool(ok.any().item()):
                continue
            flat.scatter_reduce_(0, gi[ok], g_v[r][ok], reduce=red)
        partials[name] = flat.view(t.shape)
    return True


def combine_partials(partials: dict[str, torch.Tensor], combine: dict[str, str]) -> dict:
    """All-reduce each partial bin across ranks with its combine op, in place.

    min/max bins hold +inf/-inf for groups a rank never saw (the reference's
    intermediate fill values, flox/aggregations.py:529-546), so the
    collective needs no validity mask. Partials sharing a combine op ride
    ONE coalesced all-reduce (mean's sum/count/present = a single
    collective per step instead of three — collective launch latency is
    the scaling tax at small bin sizes). At huge group counts the
    shard-aware sparse combine is tried first (see _sparse_combine).
    """
    if not is_active():
        return partials
    first = next((partials[n] for n in combine if n in partials), None)
    if first is not None and first.numel() >= SPARSE_NGROUPS:
        if _sparse_combine(partials, combine):
            return partials
    by_op: dict[str, list[torch.Tensor]] = {}
    for name, op in combine.items():
        t = partials.get(name)
        if t is None:
            continue
        by_op.setdefault(op, []).append(t)
    for op, tensors in by_op.items():
        if len(tensors) == 1 or not hasattr(dist, "all_reduce_coalesced"):
            for t in tensors:
                dist.all_reduce(t, op=_ops()[op])
            continue
        if len({t.dtype for t in tensors}) == 1:
            dist.all_reduce_coalesced(tensors, op=_ops()[op])
        elif op == "sum" and all(
            t.dtype in (torch.float64, torch.int64, torch.int32) for t in tensors
        ):
            # counts/presence ride the f64 sum collective exactly (values
            # are far below 2^53); converting ngroups-sized bins costs
            # less than a second collective launch
            up = [t if t.dtype == torch.float64 else t.to(torch.float64) for t in tensors]
            dist.all_reduce_coalesced(up, op=dist.ReduceOp.SUM)
            for t, u in zip(tensors, up):
                if u is not t:
                    t.copy_(u.to(t.dtype))
        else:
            for t in tensors:
                dist.all_reduce(t, op=_ops()[op])
    return partials


def all_reduce_(t: torch.Tensor, op: str = "sum") -> torch.Tensor:
    if is_active():
        dist.all_reduce(t, op=_ops()[op])
    return t


def all_reduce_sum_many(tensors: list[torch.Tensor]) -> None:
    """Sum-all-reduce several bins in one coalesced collective (mixed
    f64/i64/i32 ride as f64 — exact below 2^53), in place."""
    if not is_active() or not tensors:
        return
    combine_partials({str(i): t for i, t in enumerate(tensors)},
                     {str(i): "sum" for i in range(len(tensors))})


def scan_carry_exchange(
    out: torch.Tensor,
    vals: torch.Tensor,
    codes: torch.Tensor,
    ngroups: int,
    func: str,
) -> torch.Tensor:
    """Cross-rank carry for grouped scans: rank r holds rows [r·shard, …) of
    the global row order, and its locally-scanned ``out`` is corrected with
    the per-group state of earlier (later, for bfill) ranks.

    This is the flat-rank form of the reference's Blelloch scan combine
    (flox/aggregations.py:792-845 scan_binary_op, flox/dask.py:628-652):
    the per-rank state is the per-group total (cumsum — apply_binary_op with
    add) or the last/first valid value (ffill/bfill — concat_then_scan), the
    tree is replaced by one all_gather of the (ngroups,) state vector and an
    exclusive walk over ranks.

    ``codes`` are the factorized group codes per row; rows with codes outside
    [0, ngroups) (the reference's NaN-sentinel group, factorize.py:201-210)
    carry locally only. Works on CPU (gloo) and GPU (RCCL) tensors alike.
    """
    if not is_active() or ngroups <= 0 or out.numel() == 0:
        return out
    world, rank = dist.get_world_size(), dist.get_rank()
    codes = codes.to(torch.int64)
    valid = (codes >= 0) & (codes < ngroups)
    cidx = codes.clamp(0, ngroups - 1)

    if func in ("cumsum", "nancumsum"):
        # per-group total; for cumsum a NaN poisons the total (and so every
        # later rank's rows of that group), matching np.cumsum propagation
        v = vals
        if func == "nancumsum" and v.is_floating_point():
            v = torch.nan_to_num(v, nan=0.0)
        totals = torch.zeros(ngroups, dtype=out.dtype, device=out.device)
        totals.index_add_(0, cidx[valid], v[valid].to(out.dtype))
        gathered = [torch.empty_like(totals) for _ in range(world)]
        dist.all_gather(gathered, totals)
        prefix = torch.zeros_like(totals)
        for k in range(rank):
            prefix += gathered[k]
        add = prefix[cidx]
        add = torch.where(valid, add, torch.zeros_like(add))
        return out + add

    if func in ("ffill", "bfill"):
        if not out.is_floating_point():
            return out  # integer fills have no missing rows to carry into
        nan = torch.tensor(float("nan"), dtype=out.dtype, device=out.device)
        idx = torch.arange(out.numel(), device=out.device)
        pos = torch.full((ngroups,), -1, dtype=torch.int64, device=out.device)
        # the edge row of each group in the local scan already holds that
        # rank's state: last row -> last-valid (ffill), first row -> the
        # first-valid seen scanning backwards (bfill)
        red = "amax" if func == "ffill" else "amin"
        if func == "bfill":
            pos = torch.full_like(pos, out.numel())
        pos.index_reduce_(0, cidx[valid], idx[valid], red, include_self=False)
        has = (pos >= 0) & (pos < out.numel())
        state = torch.where(has, out[pos.clamp(0, max(out.numel() - 1, 0))], nan)
        gathered = [torch.empty_like(state) for _ in range(world)]
        dist.all_gather(gathered, state)
        carry = torch.full_like(state, float("nan"))
        ranks = range(rank) if func == "ffill" else range(world - 1, rank, -1)
        for k in ranks:  # nearest valid wins (forward for ffill, backward for bfill)
            carry = torch.where(torch.isnan(gathered[k]), carry, gathered[k])
        fill = torch.isnan(out) & valid
        return torch.where(fill, carry[cidx], out)

    raise NotImplementedError(f"distributed scan {func!r}")

"""Distributed exact grouped quantiles: radix selection over grouped counts.

The reference's quantile (flox/aggregate_flox.py:189-265 quantile_ /
nanquantile_) sorts each group and lerps between the two neighbouring order
statistics. Across ranks a sort would shuffle the whole 12 GB/GPU shard over
xGMI; instead we select the k-th and (k+1)-th order statistics exactly with
8-bit radix selection over the order-preserving integer encoding of the
values: each round counts, per (selection, group, byte), how many values
match the prefix fixed so far — one grouped COUNT kernel pass over the
resident shard plus one all_reduce of the (nsel*ngroups*256) histogram.
32-bit values finish in 4 rounds; 64-bit values run two 32-bit phases
(high half, then low half among rows matching the selected high half).

Only the histogram crosses the wire; values never move. The final lerp
v_lo + (h - k_lo) * (v_hi - v_lo) reproduces np.quantile(method="linear"),
the same formula the single-GPU kernel and the oracle use.
"""

from __future__ import annotations

import numpy as np
import torch

from . import distributed

_M32 = 0xFFFFFFFF


def _default_count_fn(values_i64, labels, nbins):
    """Grouped COUNT through the HIP kernel (partition path at large nbins)."""
    from . import _ffi
    from .aggregate_hip import grouped_partials

    p = grouped_partials(_ffi.SET_COUNT, values_i64, labels, nbins, skipnan=False)
    return p["count"]


def _enc_halves(vals):
    """Order-preserving encoding split into non-negative 32-bit halves.

    Returns (halves, decode) where halves is [enc32] for 4-byte dtypes or
    [hi32, lo32] for 8-byte dtypes; each half is an int64 tensor in
    [0, 2^32). decode maps selected encodings back to float64 values.
    """
    if vals.dtype == torch.float32:
        u = vals.view(torch.int32).to(torch.int64) & _M32
        enc = torch.where(u & 0x80000000 != 0, (~u) & _M32, u | 0x80000000)

        def dec(e):
            u = torch.where(e & 0x80000000 != 0, e ^ 0x80000000, (~e) & _M32)
            return u.to(torch.int32).view(torch.float32).to(torch.float64)

        return [enc], dec
    if vals.dtype == torch.int32:
        enc = (vals.to(torch.int64) & _M32) ^ 0x80000000

        def dec(e):
            return (e ^ 0x80000000).to(torch.int32).to(torch.float64)

        return [enc], dec
    if vals.dtype == torch.float64:
        u = vals.view(torch.int64)
        # i64-ordered key (signed compare == value order, as in the
        # packed-arg encoding): u64-order map, then flip the sign bit
        key = torch.where(u < 0, ~u, u ^ (-(1 << 63))) ^ (-(1 << 63))

        def dec(e):
            hi, lo = e
            key = ((hi ^ 0x80000000) << 32) | lo  # i64 bit pattern
            k2 = key ^ (-(1 << 63))  # u64-ordered pattern
            u = torch.where(k2 < 0, k2 ^ (-(1 << 63)), ~k2)
            return u.view(torch.float64)

        hi = ((key >> 32) & _M32) ^ 0x80000000  # signed-high -> unsigned order
        lo = key & _M32
        return [hi, lo], dec
    if vals.dtype == torch.int64:
        key = vals  # i64 order is the value order

        def dec(e):
            hi, lo = e
            key = ((hi ^ 0x80000000) << 32) | lo
            return key.to(torch.float64)

        hi = ((key >> 32) & _M32) ^ 0x80000000
        lo = key & _M32
        return [hi, lo], dec
    raise NotImplementedError(f"distributed quantile for {vals.dtype}")


def _radix_select32(enc, codes, ngroups, valid, k, count_fn):
    """Per-(selection, group) k-th smallest 32-bit encoding among valid rows.

    enc: (n,) int64 in [0,2^32); codes: (n,) int64 in [-1, ngroups);
    valid: (nsel, n) bool (which rows participate in each selection);
    k: (nsel, ngroups) int64 target ranks. One grouped COUNT + one
    all_reduce per (round, selection). Returns (nsel, ngroups) encodings.
    """
    nsel = k.shape[0]
    prefix = torch.zeros((nsel, ngroups), dtype=torch.int64, device=enc.device)
    k = k.clone()
    cidx = codes.clamp(0, max(ngroups - 1, 0))
    for r in range(4):
        shift = 24 - 8 * r
        byte = (enc >> shift) & 0xFF
        hi = enc >> (shift + 8)
        sel_bins = ngroups * 256
        cums = []
        for s in range(nsel):
            m = valid[s] & (hi == prefix[s][cidx]) & (codes >= 0)
            lab = torch.where(m, codes * 256 + byte, torch.full_like(codes, -1))
            cnt = count_fn(enc, lab, sel_bins).to(torch.int64)
            distributed.all_reduce_(cnt, "sum")
            cums.append(cnt.reshape(ngroups, 256).cumsum(-1))
        c = torch.stack(cums)  # (nsel, ngroups, 256)
        bin_idx = (c <= k.unsqueeze(-1)).sum(-1).clamp(max=255)
        prev = torch.gather(c, 2, (bin_idx - 1).clamp(min=0).unsqueeze(-1)).squeeze(-1)
        prev = torch.where(bin_idx > 0, prev, torch.zeros_like(prev))
        k = k - prev
        prefix = prefix * 256 + bin_idx
    return prefix


def distributed_grouped_quantile(
    vals, codes, ngroups, q_arr, skipnan, count_fn=None
):
    """Exact grouped quantiles across ranks. vals/codes are this rank's
    shard; q_arr: float64 array of quantiles. Returns a float64 tensor
    (nq, ngroups) (NaN for empty groups and, without skipnan, for groups
    containing NaN — the reference's propagation, aggregate_flox.py:225).
    Ranks must call in lockstep (collective counts inside).
    """
    if count_fn is None:
        count_fn = _default_count_fn
    device = vals.device
    codes = codes.to(torch.int64)
    halves, dec = _enc_halves(vals)
    isnan = torch.isnan(vals) if vals.is_floating_point() else torch.zeros_like(codes, dtype=torch.bool)
    okc = (codes >= 0) & (codes < ngroups)
    valid_row = okc & ~isnan

    # global per-group counts: valid rows, and NaN rows (for propagation)
    lab_valid = torch.where(valid_row, codes, torch.full_like(codes, -1))
    n_g = count_fn(halves[0], lab_valid, ngroups).to(torch.int64)
    distributed.all_reduce_(n_g, "sum")
    lab_nan = torch.where(okc & isnan, codes, torch.full_like(codes, -1))
    nan_g = count_fn(halves[0], lab_nan, ngroups).to(torch.int64)
    distributed.all_reduce_(nan_g, "sum")

    nq = len(q_arr)
    q_t = torch.tensor(np.asarray(q_arr, dtype=np.float64), device=device)
    # virtual index h = (n-1)q per (q, group); bracketing ranks k_lo/k_hi
    nn = (n_g.to(torch.float64) - 1).clamp(min=0)
    h = q_t[:, None] * nn[None, :]
    k_lo = h.floor().to(torch.int64)
    k_hi = h.ceil().to(torch.int64)
    k = torch.cat([k_lo, k_hi])  # (2nq, ngroups)

    nsel = 2 * nq
    valid = valid_row.unsqueeze(0).expand(nsel, -1)
    sel_hi = _radix_select32(halves[0], codes, ngroups, valid, k, count_fn)
    if len(halves) == 1:
        vv = dec(sel_hi)
    else:
        # phase 2: rows matching each selection's high half; rank within them
        # = k minus the count of rows with a smaller high half
        cidx = codes.clamp(0, max(ngroups - 1, 0))
        below = []
        for s in range(nsel):
            m = valid[s] & (halves[0] < sel_hi[s][cidx]) & (codes >= 0)
            lab = torch.where(m, codes, torch.full_like(codes, -1))
            cnt = count_fn(halves[0], lab, ngroups).to(torch.int64)
            distributed.all_reduce_(cnt, "sum")
            below.append(cnt)
        k2 = k - torch.stack(below)
        valid2 = torch.stack([
            valid[s] & (halves[0] == sel_hi[s][cidx]) for s in range(nsel)
        ])
        sel_lo = _radix_select32(halves[1], codes, ngroups, valid2, k2, count_fn)
        vv = dec((sel_hi, sel_lo))
    v_lo, v_hi = vv[:nq], vv[nq:]
    t = h - k_lo.to(torch.float64)
    res = v_lo + t * (v_hi - v_lo)
    nan = torch.tensor(float("nan"), dtype=torch.float64, device=device)
    bad = (n_g == 0) if skipnan else ((n_g == 0) | (nan_g > 0))
    return torch.where(bad[None, :], nan, res)


def _enc_full(vals):
    """Full-width order-preserving integer key per value (int64), with ALL
    NaNs mapped to one canonical largest key (sorts last, loses count ties
    — scipy.stats.mode's ordering). Returns (keys, decode-to-vals.dtype)."""
    if vals.dtype == torch.float32:
        u = vals.view(torch.int32).to(torch.int64) & _M32
        enc = torch.where(u & 0x80000000 != 0, (~u) & _M32, u | 0x80000000)
        enc = torch.where(torch.isnan(vals), torch.full_like(enc, _M32), enc)

        def dec(e):
            u = torch.where(e & 0x80000000 != 0, e ^ 0x80000000, (~e) & _M32)
            return u.to(torch.int32).view(torch.float32)

        return enc, dec
    if vals.dtype == torch.float64:
        u = vals.view(torch.int64)
        key = torch.where(u < 0, ~u, u ^ (-(1 << 63))) ^ (-(1 << 63))
        key = torch.where(torch.isnan(vals), torch.full_like(key, (1 << 63) - 1), key)

        def dec(e):
            k2 = e ^ (-(1 << 63))
            u = torch.where(k2 < 0, k2 ^ (-(1 << 63)), ~k2)
            return u.view(torch.float64)

        return key, dec
    if vals.dtype in (torch.int32, torch.int64):
        t = vals.dtype

        def dec(e):
            return e.to(t)

        return vals.to(torch.int64), dec
    raise NotImplementedError(f"distributed mode for {vals.dtype}")


def _rle(keys, codes):
    """Run-length encode sorted-(code, key) pairs -> (code, key, count)."""
    if keys.numel() == 0:
        z = torch.zeros(0, dtype=torch.int64, device=keys.device)
        return z, z.clone(), z.clone()
    sk, si = torch.sort(keys)
    sc = codes[si]
    sc, sj = torch.sort(sc, stable=True)
    sk = sk[sj]
    new = torch.ones_like(sc, dtype=torch.bool)
    new[1:] = (sc[1:] != sc[:-1]) | (sk[1:] != sk[:-1])
    starts = torch.nonzero(new).squeeze(1)
    ends = torch.cat([starts[1:], torch.tensor([sc.numel()], device=sc.device)])
    return sc[starts], sk[starts], ends - starts


def distributed_grouped_mode(vals, codes, ngroups, skipnan):
    """Exact grouped mode across ranks: each rank run-length encodes its
    (group, value) pairs, the compressed runs are all_gathered, and every
    rank merges counts and picks scipy.stats.mode's answer (most frequent;
    ties -> smallest value; NaN competes as one value sorting last, so it
    wins only when strictly most frequent — aggregate_npg.py:185-215).
    Returns a tensor (ngroups,) in vals.dtype (NaN/0 for empty groups;
    core's finalize fills integer empties)."""
    import torch.distributed as dist

    device = vals.device
    codes = codes.to(torch.int64)
    ok = (codes >= 0) & (codes < ngroups)
    if vals.is_floating_point():
        # mode counts by NUMERIC equality: canonicalize -0.0 -> +0.0 so the
        # encoded runs merge (scipy.stats.mode treats them as one value)
        vals = torch.where(vals == 0, torch.zeros_like(vals), vals)
    if skipnan and vals.is_floating_point():
        ok &= ~torch.isnan(vals)
    keys, dec = _enc_full(vals[ok])
    rc, rk, rn = _rle(keys, codes[ok])

    if distributed.is_active():
        world = dist.get_world_size()
        n_local = torch.tensor([rc.numel()], dtype=torch.int64, device=device)
        sizes = [torch.zeros_like(n_local) for _ in range(world)]
        dist.all_gather(sizes, n_local)
        mx = int(max(int(s.item()) for s in sizes))
        packed = torch.zeros((3, max(mx, 1)), dtype=torch.int64, device=device)
        packed[0, : rc.numel()] = rc
        packed[1, : rc.numel()] = rk
        packed[2, : rc.numel()] = rn
        gathered = [torch.empty_like(packed) for _ in range(world)]
        dist.all_gather(gathered, packed)
        parts = [g[:, : int(s.item())] for g, s in zip(gathered, sizes)]
        allp = torch.cat(parts, dim=1)
        # merge duplicate (code, key) runs from different ranks
        sk, si = torch.sort(allp[1])
        sc = allp[0][si]
        sn = allp[2][si]
        sc, sj = torch.sort(sc, stable=True)
        sk, sn = sk[sj], sn[sj]
        new = torch.ones_like(sc, dtype=torch.bool)
        new[1:] = (sc[1:] != sc[:-1]) | (sk[1:] != sk[:-1])
        seg = torch.cumsum(new.to(torch.int64), 0) - 1
        nseg = int(seg[-1].item()) + 1 if sc.numel() else 0
        cnt = torch.zeros(max(nseg, 1), dtype=torch.int64, device=device)
        cnt.index_add_(0, seg, sn)
        starts = torch.nonzero(new).squeeze(1)
        rc, rk, rn = sc[starts], sk[starts], cnt[:nseg]

    # per group: max count, ties -> first run (runs are value-ascending)
    nan = float("nan") if vals.is_floating_point() else 0
    out = torch.full((ngroups,), nan, dtype=vals.dtype, device=device)
    if rc.numel():
        maxc = torch.zeros(ngroups, dtype=torch.int64, device=device)
        maxc.index_reduce_(0, rc, rn, "amax", include_self=False)
        best = rn == maxc[rc]
        big = rc.numel() + 1
        firstpos = torch.full((ngroups,), big, dtype=torch.int64, device=device)
        pos = torch.arange(rc.numel(), device=device)
        firstpos.index_reduce_(0, rc[best], pos[best], "amin", include_self=False)
        has = firstpos < big
        sel = firstpos.clamp(max=rc.numel() - 1)
        chosen = dec(rk[sel])
        out = torch.where(has, chosen.to(vals.dtype), out)
    return out

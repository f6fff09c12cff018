"""Multi-process (gloo, world_size=2) tests of the cross-GPU combine logic.

Runs on CPU: the combine recipes (distributed.combine_partials and the
two-phase var combine) are device-agnostic torch collectives; the kernels
that produce per-rank partials are exercised by the gpu-marked tests. Here
each rank builds its shard's partial bins in numpy/torch and the combined
result must equal the whole-data answer (bit-exact for count/min/max).
"""

import multiprocessing as mp
import os

import numpy as np
import pytest
import torch
import torch.distributed as dist

N, NG = 40_000, 97
SEED = 123


def _shard_data(rank, world):
    rng = np.random.default_rng(SEED)
    vals = rng.standard_normal(N)
    labels = rng.integers(0, NG, N)
    sl = slice(rank * N // world, (rank + 1) * N // world)
    return vals, labels, vals[sl], labels[sl]


def _partials_np(vals, labels):
    sums = np.bincount(labels, weights=vals, minlength=NG)
    counts = np.bincount(labels, minlength=NG)
    mins = np.full(NG, np.inf)
    np.minimum.at(mins, labels, vals)
    maxs = np.full(NG, -np.inf)
    np.maximum.at(maxs, labels, vals)
    return {
        "sum": torch.tensor(sums),
        "count": torch.tensor(counts),
        "min": torch.tensor(mins),
        "max": torch.tensor(maxs),
    }


def _worker(rank, world, port, fail_q):
    try:
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        os.environ["MASTER_PORT"] = str(port)
        dist.init_process_group("gloo", rank=rank, world_size=world)
        import flox_amd.distributed as fdist

        vals, labels, sv, sl = _shard_data(rank, world)

        # 1. plain partial combine == whole-data partials
        p = _partials_np(sv, sl)
        fdist.combine_partials(p, {"sum": "sum", "count": "sum", "min": "min", "max": "max"})
        whole = _partials_np(vals, labels)
        np.testing.assert_array_equal(p["count"].numpy(), whole["count"].numpy())
        np.testing.assert_array_equal(p["min"].numpy(), whole["min"].numpy())
        np.testing.assert_array_equal(p["max"].numpy(), whole["max"].numpy())
        np.testing.assert_allclose(p["sum"].numpy(), whole["sum"].numpy(), rtol=1e-13)

        # 2. two-phase var combine (deviations about the GLOBAL mean; the
        #    reference's _var_combine reduces to a plain sum of ssd then)
        sums = torch.tensor(np.bincount(sl, weights=sv, minlength=NG))
        counts = torch.tensor(np.bincount(sl, minlength=NG))
        fdist.all_reduce_(sums, "sum")
        fdist.all_reduce_(counts, "sum")
        means = (sums / counts).numpy()
        ssd_local = np.bincount(sl, weights=(sv - means[sl]) ** 2, minlength=NG)
        ssd = torch.tensor(ssd_local)
        fdist.all_reduce_(ssd, "sum")
        var_combined = ssd.numpy() / counts.numpy()

        from oracle import groupby_reduce as oracle_reduce

        want, _ = oracle_reduce(vals, labels, func="var", expected_groups=np.arange(NG))
        np.testing.assert_allclose(var_combined, want, rtol=1e-10, atol=1e-12)

        dist.destroy_process_group()
    except Exception as e:  # pragma: no cover
        fail_q.put(f"rank {rank}: {type(e).__name__}: {e}")
        raise


def test_gloo_world2_combine():
    ctx = mp.get_context("spawn")
    fail_q = ctx.Queue()
    port = 29517
    procs = [ctx.Process(target=_worker, args=(r, 2, port, fail_q)) for r in range(2)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=120)
    errs = []
    while not fail_q.empty():
        errs.append(fail_q.get())
    assert not errs, errs
    assert all(p.exitcode == 0 for p in procs), [p.exitcode for p in procs]


# ---------------------------------------------------------------------------
# full groupby_reduce distributed branch on CPU: stub the kernel layer with a
# numpy partials producer so core's all-reduce paths execute under gloo
# ---------------------------------------------------------------------------


def _fake_partials_factory(np_mod, torch_mod):
    import flox_amd._ffi as F

    def fake(op_set, values, labels, ngroups, *, labels2=None, grp_shape=None,
             means=None, target=None, row_offset=0, skipnan=False, force_path=0):
        v = values.numpy() if hasattr(values, "numpy") else np_mod.asarray(values)
        l = labels.numpy() if hasattr(labels, "numpy") else np_mod.asarray(labels)
        valid = (l >= 0) & (l < ngroups)
        nan = np_mod.isnan(v) if v.dtype.kind == "f" else np_mod.zeros(len(v), bool)
        m = valid & ~nan if skipnan else valid
        out = {}
        if op_set in (F.SET_SUM_COUNT, F.SET_SUM_COUNT_PRESENT):
            out["sum"] = torch_mod.tensor(
                np_mod.bincount(l[m], weights=v[m].astype("f8"), minlength=ngroups))
            out["count"] = torch_mod.tensor(
                np_mod.bincount(l[valid & ~nan], minlength=ngroups))
            if op_set == F.SET_SUM_COUNT_PRESENT:
                out["present"] = torch_mod.tensor(
                    (np_mod.bincount(l[valid], minlength=ngroups) > 0).astype("i4"))
        elif op_set == F.SET_SSD:
            mu = means.numpy()
            d = v[m].astype("f8") - mu[l[m]]
            out["sum"] = torch_mod.tensor(np_mod.bincount(l[m], weights=d * d, minlength=ngroups))
        elif op_set == F.SET_COUNT:
            out["count"] = torch_mod.tensor(np_mod.bincount(l[valid & ~nan], minlength=ngroups))
        elif op_set in (F.SET_MIN_FULL, F.SET_MAX_FULL, F.SET_MIN_COUNT, F.SET_MAX_COUNT):
            # kernel semantics (floxhip.hip process()): min/max over non-NaN
            # valid rows (+/-inf when none), count = non-NaN valid rows,
            # present = any valid row, nanflag = any NaN row (FULL sets)
            ismin = op_set in (F.SET_MIN_FULL, F.SET_MIN_COUNT)
            ext = np_mod.full(ngroups, np_mod.inf if ismin else -np_mod.inf)
            mm = valid & ~nan
            if ismin:
                np_mod.minimum.at(ext, l[mm], v[mm].astype("f8"))
            else:
                np_mod.maximum.at(ext, l[mm], v[mm].astype("f8"))
            out["min" if ismin else "max"] = torch_mod.tensor(ext)
            out["count"] = torch_mod.tensor(np_mod.bincount(l[mm], minlength=ngroups))
            if op_set in (F.SET_MIN_FULL, F.SET_MAX_FULL):
                out["present"] = torch_mod.tensor(
                    (np_mod.bincount(l[valid], minlength=ngroups) > 0).astype("i4"))
                out["nanflag"] = torch_mod.tensor(
                    (np_mod.bincount(l[valid & nan], minlength=ngroups) > 0).astype("i4"))
        elif op_set in (F.SET_IDXMIN, F.SET_IDXMAX):
            # idx = extreme global row index among candidate rows: valid,
            # (skipnan -> non-NaN), and matching the per-group target when
            # given (a NaN target matches NaN rows). Sentinels INT64_MAX / -1.
            ismin = op_set == F.SET_IDXMIN
            cand = valid & ~nan if skipnan else valid
            if target is not None:
                t = target.numpy()
                tl = t[np_mod.clip(l, 0, ngroups - 1)]
                if v.dtype.kind == "f":
                    match = np_mod.where(
                        np_mod.isnan(v), np_mod.isnan(tl),
                        ~np_mod.isnan(tl) & (v == tl))
                else:
                    match = v == tl
                cand = cand & match
            rows = np_mod.arange(len(v), dtype="i8") + row_offset
            sent = np_mod.iinfo("i8").max if ismin else -1
            idx = np_mod.full(ngroups, sent, dtype="i8")
            if ismin:
                np_mod.minimum.at(idx, l[cand], rows[cand])
            else:
                np_mod.maximum.at(idx, l[cand], rows[cand])
            out["idx"] = torch_mod.tensor(idx)
            out["count"] = torch_mod.tensor(
                np_mod.bincount(l[valid & ~nan], minlength=ngroups))
            out["present"] = torch_mod.tensor(
                (np_mod.bincount(l[valid], minlength=ngroups) > 0).astype("i4"))
        else:
            raise NotImplementedError(op_set)
        return out

    return fake


def _worker_full_branch(rank, world, port, fail_q):
    try:
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        os.environ["MASTER_PORT"] = str(port)
        dist.init_process_group("gloo", rank=rank, world_size=world)
        import numpy as np
        import torch
        import flox_amd.core as core
        from oracle import groupby_reduce as oracle_reduce

        rng = np.random.default_rng(7)
        n, ng = 20_000, 37
        vals = rng.standard_normal(n)
        vals[rng.random(n) < 0.05] = np.nan
        labels = rng.integers(0, ng - 2, n)
        # groups 35/36 live on exactly one rank each: a rank holding no rows
        # of a group must still return the group's global result (the
        # p2-present/count combine in the arg-reduction path)
        labels[:50] = ng - 2
        labels[-50:] = ng - 1
        sl = slice(rank * n // world, (rank + 1) * n // world)

        fake = _fake_partials_factory(np, torch)
        orig_cuda = torch.cuda.is_available
        torch.cuda.is_available = lambda: True  # let core proceed on CPU
        torch.cuda.current_device = lambda: 0
        core.grouped_partials = fake
        core._as_device_tensor = lambda x, d: (
            x if isinstance(x, torch.Tensor) else torch.from_numpy(np.ascontiguousarray(x)))
        _orig_device = torch.device
        core.torch.device = lambda *a, **k: _orig_device("cpu")
        try:
            for func in ["nanmean", "sum", "count", "var"]:
                got, *_ = core.groupby_reduce(
                    vals[sl], labels[sl], func=func, expected_groups=np.arange(ng))
                want, *_ = oracle_reduce(vals, labels, func=func, expected_groups=np.arange(ng))
                np.testing.assert_allclose(
                    got.numpy() if hasattr(got, "numpy") else got, want,
                    equal_nan=True, rtol=1e-10, atol=1e-10)
            # order-dependent funcs need the global row offset of this shard;
            # the arg-reductions cover the p2 present/count combine (a rank
            # holding no rows of a group must still return the global index)
            for func in ["min", "max", "nanmin", "nanmax",
                         "argmax", "argmin", "nanargmax", "nanargmin",
                         "first", "last", "nanfirst", "nanlast"]:
                got, *_ = core.groupby_reduce(
                    vals[sl], labels[sl], func=func,
                    expected_groups=np.arange(ng), shard_row_offset=sl.start)
                want, *_ = oracle_reduce(vals, labels, func=func, expected_groups=np.arange(ng))
                np.testing.assert_allclose(
                    got.numpy() if hasattr(got, "numpy") else got, want,
                    equal_nan=True, rtol=1e-12, atol=0, err_msg=func)
            # rank-local group discovery is rejected under the distributed
            # combine (per-rank code spaces would misalign the bins)
            try:
                core.groupby_reduce(vals[sl], labels[sl], func="sum")
            except NotImplementedError:
                pass
            else:
                raise AssertionError("expected NotImplementedError for dist + expected_groups=None")
        finally:
            torch.cuda.is_available = orig_cuda
        dist.destroy_process_group()
    except Exception as e:  # pragma: no cover
        import traceback
        fail_q.put(f"rank {rank}: {type(e).__name__}: {e}\n{traceback.format_exc()}")
        raise


def _worker_scan(rank, world, port, fail_q):
    try:
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        os.environ["MASTER_PORT"] = str(port)
        dist.init_process_group("gloo", rank=rank, world_size=world)
        import numpy as np
        import torch
        import flox_amd.distributed as fdist
        from oracle import groupby_scan as oracle_scan

        rng = np.random.default_rng(11)
        n, ng = 9_000, 23
        fvals = rng.standard_normal(n)
        fvals[rng.random(n) < 0.3] = np.nan  # big NaN runs so carries matter
        ivals = rng.integers(-50, 50, n).astype(np.int64)
        labels = rng.integers(0, ng, n)
        # ~4% of rows fall outside expected_groups: the reference's
        # NaN-sentinel group (factorize_:201-210) must carry across ranks
        # too (via its dedicated slot ngroups, mirroring scan.py's call site)
        labels[rng.random(n) < 0.04] = ng + 76
        eg = np.arange(ng)
        sl = slice(rank * n // world, (rank + 1) * n // world)

        for vals, funcs in [
            (fvals, ["cumsum", "nancumsum", "ffill", "bfill"]),
            (ivals, ["cumsum"]),
        ]:
            for func in funcs:
                local = oracle_scan(vals[sl], labels[sl], func=func, expected_groups=eg)
                codes = labels[sl].copy()
                codes[(codes < 0) | (codes >= ng)] = ng
                got = fdist.scan_carry_exchange(
                    torch.from_numpy(np.ascontiguousarray(local)),
                    torch.from_numpy(np.ascontiguousarray(vals[sl])),
                    torch.from_numpy(np.ascontiguousarray(codes)),
                    ng + 1,
                    func,
                )
                want = oracle_scan(vals, labels, func=func, expected_groups=eg)[sl]
                np.testing.assert_allclose(
                    got.numpy(), want, equal_nan=True, rtol=1e-10, atol=1e-10,
                    err_msg=f"{func} {vals.dtype}",
                )
        dist.destroy_process_group()
    except Exception as e:  # pragma: no cover
        import traceback
        fail_q.put(f"rank {rank}: {type(e).__name__}: {e}\n{traceback.format_exc()}")
        raise


def test_gloo_scan_carry_exchange():
    """Distributed grouped scans at world_size 2: each rank scans its shard
    locally (via the oracle, standing in for the HIP scan kernel) and the
    carry exchange must reproduce the whole-array scan — the flat-rank form
    of the reference's scan_binary_op combine (flox/aggregations.py:792-845)."""
    ctx = mp.get_context("spawn")
    fail_q = ctx.Queue()
    procs = [ctx.Process(target=_worker_scan, args=(r, 2, 29521, fail_q)) for r in range(2)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=180)
    errs = []
    while not fail_q.empty():
        errs.append(fail_q.get())
    assert not errs, errs[0]
    assert all(p.exitcode == 0 for p in procs), [p.exitcode for p in procs]


def _worker_quantile(rank, world, port, fail_q):
    try:
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        os.environ["MASTER_PORT"] = str(port)
        dist.init_process_group("gloo", rank=rank, world_size=world)
        import numpy as np
        import torch
        from flox_amd.dist_quantile import distributed_grouped_quantile
        from oracle import groupby_reduce as oracle_reduce

        def np_count(values_i64, labels, nbins):
            l = labels.numpy()
            return torch.tensor(np.bincount(l[l >= 0], minlength=nbins), dtype=torch.int64)

        rng = np.random.default_rng(13)
        n, ng = 30_000, 41
        for dtype, skipnan in [("float32", True), ("float64", True), ("float64", False), ("int64", True)]:
            if np.dtype(dtype).kind == "f":
                vals = (rng.standard_normal(n) * 50).astype(dtype)
                vals[rng.random(n) < 0.08] = np.nan
            else:
                vals = rng.integers(-(2**45), 2**45, n).astype(dtype)
            labels = rng.integers(0, ng, n)
            q = np.array([0.1, 0.5, 0.95])
            sl = slice(rank * n // world, (rank + 1) * n // world)
            got = distributed_grouped_quantile(
                torch.from_numpy(np.ascontiguousarray(vals[sl])),
                torch.from_numpy(np.ascontiguousarray(labels[sl])),
                ng, q, skipnan=skipnan, count_fn=np_count,
            )
            func = "nanquantile" if skipnan else "quantile"
            want, *_ = oracle_reduce(vals, labels, func=func,
                                     expected_groups=np.arange(ng),
                                     finalize_kwargs={"q": list(q)})
            # the selection is exact in f64; the oracle lerps f32 inputs in
            # f32 (np.quantile preserves float dtype) — compare at the
            # output dtype's precision
            tol = dict(rtol=3e-6, atol=1e-5) if want.dtype.itemsize == 4 else dict(rtol=1e-12, atol=1e-12)
            np.testing.assert_allclose(got.numpy().astype(want.dtype), want,
                                       equal_nan=True, err_msg=f"{dtype} skipnan={skipnan}", **tol)
        dist.destroy_process_group()
    except Exception as e:  # pragma: no cover
        import traceback
        fail_q.put(f"rank {rank}: {type(e).__name__}: {e}\n{traceback.format_exc()}")
        raise


def test_gloo_distributed_quantile():
    """Exact cross-rank quantiles at world_size 2: sharded values, radix
    selection with a numpy count stub (the HIP COUNT kernel is exercised by
    the world-1 GPU test) — must equal the whole-data oracle bit-for-bit."""
    ctx = mp.get_context("spawn")
    fail_q = ctx.Queue()
    procs = [ctx.Process(target=_worker_quantile, args=(r, 2, 29523, fail_q)) for r in range(2)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=240)
    errs = []
    while not fail_q.empty():
        errs.append(fail_q.get())
    assert not errs, errs[0]
    assert all(p.exitcode == 0 for p in procs), [p.exitcode for p in procs]


def test_gloo_full_groupby_reduce_branch():
    """The exact distributed code in core.groupby_reduce (partial all-reduce,
    global-mean var, finalize) at world_size 2, with the kernel layer stubbed
    by numpy — shard results must equal the whole-data oracle."""
    ctx = mp.get_context("spawn")
    fail_q = ctx.Queue()
    procs = [ctx.Process(target=_worker_full_branch, args=(r, 2, 29519, fail_q)) for r in range(2)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=180)
    errs = []
    while not fail_q.empty():
        errs.append(fail_q.get())
    assert not errs, errs[0]
    assert all(p.exitcode == 0 for p in procs), [p.exitcode for p in procs]


def _worker_mode(rank, world, port, fail_q):
    try:
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        os.environ["MASTER_PORT"] = str(port)
        dist.init_process_group("gloo", rank=rank, world_size=world)
        import numpy as np
        import torch
        from flox_amd.dist_quantile import distributed_grouped_mode
        from oracle import groupby_reduce as oracle_reduce

        rng = np.random.default_rng(29)
        n, ng = 20_000, 33
        for dtype, skipnan in [("float64", False), ("float32", True), ("int64", False)]:
            vals = rng.integers(-10, 10, n).astype(dtype)
            if np.dtype(dtype).kind == "f":
                vals[rng.random(n) < 0.15] = np.nan
            labels = rng.integers(0, ng, n)
            sl = slice(rank * n // world, (rank + 1) * n // world)
            got = distributed_grouped_mode(
                torch.from_numpy(np.ascontiguousarray(vals[sl])),
                torch.from_numpy(np.ascontiguousarray(labels[sl])),
                ng, skipnan)
            func = "nanmode" if skipnan else "mode"
            want, *_ = oracle_reduce(vals, labels, func=func, expected_groups=np.arange(ng))
            np.testing.assert_array_equal(got.numpy().astype(want.dtype), want,
                                          err_msg=f"{dtype} skipnan={skipnan}")
        dist.destroy_process_group()
    except Exception as e:  # pragma: no cover
        import traceback
        fail_q.put(f"rank {rank}: {type(e).__name__}: {e}\n{traceback.format_exc()}")
        raise


def test_gloo_distributed_mode():
    """Exact cross-rank mode at world_size 2: compressed (group, value,
    count) runs are exchanged and merged — must equal the whole-data oracle."""
    ctx = mp.get_context("spawn")
    fail_q = ctx.Queue()
    procs = [ctx.Process(target=_worker_mode, args=(r, 2, 29525, fail_q)) for r in range(2)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=240)
    errs = []
    while not fail_q.empty():
        errs.append(fail_q.get())
    assert not errs, errs[0]
    assert all(p.exitcode == 0 for p in procs), [p.exitcode for p in procs]


def _worker_sparse_combine(rank, world, port, fail_q):
    try:
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        os.environ["MASTER_PORT"] = str(port)
        dist.init_process_group("gloo", rank=rank, world_size=world)
        import numpy as np
        import torch
        import flox_amd.distributed as fdist

        ng = 50_000
        old_thr = fdist.SPARSE_NGROUPS
        fdist.SPARSE_NGROUPS = 1024  # engage the sparse path at test sizes
        try:
            rng = np.random.default_rng(101)
            n = 30_000
            vals = rng.standard_normal(world * n)
            vals[rng.random(world * n) < 0.02] = np.nan
            # strong label locality: rank r's labels live in its own slice of
            # the group space (the cohorts case), plus a few shared groups
            labels = np.concatenate([
                np.where(
                    rng.random(n) < 0.05,
                    rng.integers(0, 64, n),  # shared groups
                    rng.integers(r * ng // world, (r + 1) * ng // world, n),
                )
                for r in range(world)
            ])
            sv, sl = vals[rank * n:(rank + 1) * n], labels[rank * n:(rank + 1) * n]

            def local_partials(v, l, skipnan, nbins=ng):
                sums = np.zeros(nbins)
                counts = np.zeros(nbins, dtype=np.int64)
                present = np.zeros(nbins, dtype=np.int32)
                mins = np.full(nbins, np.inf)
                np.add.at(present, l, 1)
                m = ~np.isnan(v) if skipnan else np.ones(len(v), bool)
                np.add.at(sums, l[m], np.where(np.isnan(v[m]), np.nan, v[m]))
                np.add.at(counts, l[~np.isnan(v)], 1)
                np.minimum.at(mins, l[~np.isnan(v)], v[~np.isnan(v)])
                return {
                    "sum": torch.tensor(sums), "count": torch.tensor(counts),
                    "present": torch.tensor((present > 0).astype(np.int32)),
                    "min": torch.tensor(mins),
                }

            combine = {"sum": "sum", "count": "sum", "present": "sum", "min": "min"}
            # sparse path (auto-engaged: locality makes tot_nnz << world*ng)
            p_sparse = local_partials(sv, sl, skipnan=False)
            fdist.combine_partials(p_sparse, combine)
            # dense path for the same inputs
            fdist.SPARSE_NGROUPS = 1 << 60
            p_dense = local_partials(sv, sl, skipnan=False)
            fdist.combine_partials(p_dense, combine)
            fdist.SPARSE_NGROUPS = 1024
            for k in combine:
                np.testing.assert_allclose(
                    p_sparse[k].numpy(), p_dense[k].numpy(),
                    equal_nan=True, rtol=1e-13, atol=0, err_msg=k)
            # whole-data ground truth
            whole = local_partials(vals, labels, skipnan=False)
            np.testing.assert_array_equal(p_sparse["count"].numpy(), whole["count"].numpy())
            np.testing.assert_allclose(p_sparse["sum"].numpy(), whole["sum"].numpy(),
                                       equal_nan=True, rtol=1e-12)
            np.testing.assert_array_equal(p_sparse["min"].numpy(), whole["min"].numpy())
            # uniform labels must REJECT the sparse path (decision collective
            # agrees on every rank) and still combine exactly
            labels_u = np.concatenate([
                rng2.integers(0, 2000, n) for rng2 in
                [np.random.default_rng(7 + r) for r in range(world)]
            ])
            slu = labels_u[rank * n:(rank + 1) * n]
            fdist.SPARSE_NGROUPS = 1024
            p_u = local_partials(sv, slu, skipnan=False, nbins=2000)
            fdist.combine_partials(p_u, combine)
            whole_u = local_partials(vals, labels_u, skipnan=False, nbins=2000)
            np.testing.assert_array_equal(p_u["count"].numpy(), whole_u["count"].numpy())
            np.testing.assert_allclose(p_u["sum"].numpy(), whole_u["sum"].numpy(),
                                       equal_nan=True, rtol=1e-12)
        finally:
            fdist.SPARSE_NGROUPS = old_thr
        dist.destroy_process_group()
    except Exception as e:  # pragma: no cover
        import traceback
        fail_q.put(f"rank {rank}: {type(e).__name__}: {e}\n{traceback.format_exc()}")
        raise


def test_gloo_sparse_combine():
    """Shard-aware sparse combine (cohorts label-locality restated for
    ranks): compressed touched-bin exchange must equal both the dense
    all-reduce and the whole-data partials; uniform labels auto-reject."""
    ctx = mp.get_context("spawn")
    fail_q = ctx.Queue()
    procs = [ctx.Process(target=_worker_sparse_combine, args=(r, 2, 29527, fail_q))
             for r in range(2)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=240)
    errs = []
    while not fail_q.empty():
        errs.append(fail_q.get())
    assert not errs, errs[0]
    assert all(p.exitcode == 0 for p in procs), [p.exitcode for p in procs]


def test_gloo_world3_scan_quantile_mode():
    """world_size=3: middle ranks both receive and forward carries/counts."""
    ctx = mp.get_context("spawn")
    for worker, port in [(_worker_scan, 29531), (_worker_quantile, 29533), (_worker_mode, 29535)]:
        fail_q = ctx.Queue()
        procs = [ctx.Process(target=worker, args=(r, 3, port, fail_q)) for r in range(3)]
        for p in procs:
            p.start()
        for p in procs:
            p.join(timeout=240)
        errs = []
        while not fail_q.empty():
            errs.append(fail_q.get())
        assert not errs, errs[0]
        assert all(p.exitcode == 0 for p in procs), (worker.__name__, [p.exitcode for p in procs])

"""Headline benchmark: grouped-reduce GB/s vs the 8 TB/s HBM roofline.

Workload (BASELINE.json configs[1], the largest single-GPU config quoted for
the metric): float32 (1e9,) values, int64 labels uniform over 1e4 groups,
func="mean", engine="hip" — one "step" = one full grouped mean over the
resident shard (fused factorize+sum+count kernel -> combine -> finalize),
inputs already in HBM when the timed region starts.

Run:  python bench.py [--gpus N] [--steps K] [--warmup W]
N>1 is launched by the driver via torch.distributed.run (one rank per GPU,
RCCL); each rank holds its own 1e9-row shard (weak scaling) and the per-group
partial bins are combined with one all-reduce per partial.
"""

from __future__ import annotations

import argparse
import json
import os
import sys
import time

import numpy as np
import torch

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

ROWS = 1_000_000_000
NGROUPS = 10_000
HBM_PEAK_GBPS = 8000.0  # spec peak, /opt/skills/guides/MI355X_MICROARCH.md


def log(msg):
    if int(os.environ.get("RANK", "0")) == 0:
        print(msg, file=sys.stderr, flush=True)


def cpu_baseline_leg(rows_sample: int, ngroups: int) -> dict:
    """Time flox's own CPU ``engine="flox"`` path on this box's host cores —
    the REAL reference (vendored shimmed tree, oracle/_ref, shipped with the
    snapshot), not a restatement — on a bounded sample of the same workload
    (north_star: "flox's own CPU engine='flox' path timed on the host cores
    of the same box in the same run"). kind="reference". Reported baseline,
    not the target. The oracle (numpy restatement, engine="numpy"/npg
    semantics) is timed beside it as a secondary figure. Falls back to
    kind="port" (oracle only) if the vendored tree is absent."""
    from oracle import groupby_reduce as oracle_reduce

    try:
        from oracle.ref_loader import available as ref_available
        from oracle.ref_loader import load_reference
        have_ref = ref_available()
    except Exception:  # pragma: no cover
        have_ref = False

    rng = np.random.default_rng(1)
    vals = rng.random(rows_sample, dtype=np.float32)
    labels = rng.integers(0, ngroups, rows_sample)
    t0 = time.perf_counter()
    oracle_reduce(vals, labels, func="mean", expected_groups=np.arange(ngroups))
    dt = time.perf_counter() - t0
    oracle_leg = {
        "value": round(rows_sample * 12 / dt / 1e9, 4),
        "unit": "GB/s",
        "cores": 1,  # the oracle's numpy ops are single-threaded
        "kind": "port",
        "sample": f"{rows_sample:.0e} of {ROWS:.0e} rows, {dt:.1f}s (oracle, npg semantics)",
    }
    if not have_ref:
        return oracle_leg

    ref_core = load_reference()
    rs = min(rows_sample, 20_000_000)  # reference runs ~0.025 GB/s: ~10 s
    t1 = time.perf_counter()
    ref_core.groupby_reduce(
        vals[:rs], labels[:rs], func="mean",
        expected_groups=np.arange(ngroups), engine="flox",
    )
    dt_ref = time.perf_counter() - t1
    return {
        "value": round(rs * 12 / dt_ref / 1e9, 4),
        "unit": "GB/s",
        "cores": 1,  # factorize/argsort/reduceat are single-threaded numpy
        "kind": "reference",
        "sample": (
            f"{rs:.0e} of {ROWS:.0e} rows, {dt_ref:.1f}s — the reference's own "
            "groupby_reduce(engine='flox') (vendored oracle/_ref tree) on this host"
        ),
        "oracle_port": oracle_leg,
    }


def read_traffic(config: str):
    """Per-launch HBM bytes of the dominant kernel from the committed rocprof
    PMC summary (profiles/traffic.json, written by profiles/collect.sh and
    collect_pmc.sh on the GPU box)."""
    p = os.path.join(os.path.dirname(os.path.abspath(__file__)), "profiles", "traffic.json")
    if os.path.exists(p):
        with open(p) as f:
            d = json.load(f)
        return d.get("configs", {}).get(config, {}).get("bytes_per_launch")
    return None


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=10)
    ap.add_argument("--warmup", type=int, default=3)
    ap.add_argument("--rows", type=int, default=ROWS)
    ap.add_argument("--ngroups", type=int, default=NGROUPS)
    ap.add_argument("--func", default="mean")
    ap.add_argument("--no-cpu-baseline", action="store_true")
    ap.add_argument(
        "--no-graph", action="store_true",
        help="disable hipGraph step replay (auto-enabled at world 1 when the "
             "step captures cleanly; the captured graph re-executes every "
             "kernel each replay — it only removes per-step launch/orchestration overhead)",
    )
    ap.add_argument(
        "--config",
        default="2",
        choices=["2", "3", "4", "5"],
        help="BASELINE.json config to measure (default 2 = the headline 1e9-row/1e4-group mean)",
    )
    args = ap.parse_args()

    import flox_amd
    from flox_amd import aggregate_hip, core as fa_core

    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))
    if world > 1:
        import torch.distributed as dist

        torch.cuda.set_device(local_rank)
        dist.init_process_group("nccl")
    assert torch.cuda.is_available(), "bench.py needs an MI355X"
    device = torch.device("cuda", local_rank if world > 1 else 0)
    torch.cuda.set_device(device)

    gen = torch.Generator(device=device).manual_seed(1234 + rank)
    if args.config == "2":
        n, ng = args.rows, args.ngroups
        func = args.func
        log(f"config 2: {n:.0e} rows, {ng} groups, {func} (rank {rank}/{world})")
        vals = torch.rand(n, generator=gen, dtype=torch.float32, device=device)
        labels = torch.randint(0, ng, (n,), generator=gen, dtype=torch.int64, device=device)
        expected = range(ng)
        bytes_per_step_per_gpu = n * (4 + 8)
        workload = "configs[1]: fp32 (1e9,) values, int64 labels, 1e4 uniform groups, func=mean, engine=hip"
        cfg_extra = {"rows": n, "ngroups": ng, "func": func, "labels": "int64"}

        def step():
            res, _ = flox_amd.groupby_reduce(vals, labels, func=func, expected_groups=expected)
            return res
    elif args.config == "3":
        n, ng = args.rows, 10_000_000
        func = "sum"
        log(f"config 3: {n:.0e} rows, 1e7 groups, sum (rank {rank}/{world})")
        vals = torch.rand(n, generator=gen, dtype=torch.float32, device=device)
        labels = torch.randint(0, ng, (n,), generator=gen, dtype=torch.int64, device=device)
        expected = range(ng)  # O(1) range detection; np.arange(1e7) costs ~10 ms/call to verify
        bytes_per_step_per_gpu = n * (4 + 8)
        workload = "configs[2]: fp32 (1e9,) values, 1e7 groups, func=sum (bucket-partition path)"
        cfg_extra = {"rows": n, "ngroups": ng, "func": func, "labels": "int64"}

        def step():
            res, _ = flox_amd.groupby_reduce(vals, labels, func=func, expected_groups=expected)
            return res
    elif args.config == "4":
        n_t, y, x = 8760, 720, 1440
        func = args.func if args.func in ("sum", "mean", "var") else "mean"
        log(f"config 4: ({n_t},{y},{x}) by hour-of-day, {func} (rank {rank}/{world})")
        # weak scaling: each rank holds its own year of hourly data
        arr = torch.rand((n_t, y, x), generator=gen, dtype=torch.float32, device=device)
        hours = (torch.arange(n_t, device=device) % 24).to(torch.int64)
        view = arr.permute(1, 2, 0)
        expected = np.arange(24)
        bytes_per_step_per_gpu = arr.numel() * 4
        workload = "configs[3]: fp32 (8760,720,1440) by hour-of-day (24 groups, axis 0), RCCL combine of (24,720,1440) partials"
        cfg_extra = {"shape": [n_t, y, x], "ngroups": 24, "func": func}

        def step():
            res, _ = flox_amd.groupby_reduce(view, hours, func=func, expected_groups=expected)
            return res
    else:  # config 5
        n = 100_000_000
        func = "nanmean"
        log(f"config 5: fp64 (1e8,) 2-D groupby 12x180, nanmean (rank {rank}/{world})")
        vals = torch.rand(n, generator=gen, dtype=torch.float64, device=device)
        vals[torch.rand(n, generator=gen, device=device) < 0.05] = float("nan")
        months = torch.randint(0, 12, (n,), generator=gen, dtype=torch.int64, device=device)
        latbin = torch.randint(0, 180, (n,), generator=gen, dtype=torch.int64, device=device)
        expected = (np.arange(12), np.arange(180))
        bytes_per_step_per_gpu = n * (8 + 16)
        workload = "configs[4]: fp64 (1e8,) ~5% NaN by (month, lat-bin) 12x180, func=nanmean, reindexed to expected"
        cfg_extra = {"rows": n, "grp_shape": [12, 180], "func": func}

        def step():
            res, *_g = flox_amd.groupby_reduce(
                vals, months, latbin, func=func, expected_groups=expected
            )
            return res

    def barrier():
        if world > 1:
            torch.distributed.barrier()
        torch.cuda.synchronize()

    # warmup
    for _ in range(args.warmup):
        step()
    barrier()

    # hipGraph step replay (auto): capture the whole step once, replay K
    # times — every kernel re-executes each replay; only the per-step
    # launch/orchestration overhead goes away. Eager fallback on any capture
    # failure (e.g. the partition path's in-step overflow sync) and on a
    # mismatch between replayed and eager results.
    graph = None
    graph_res = None
    if not args.no_graph and world == 1:
        # eager probe BEFORE any capture attempt: graphs only pay where
        # launch overhead is a visible fraction of the step, and a capture
        # attempt that dies mid-step (partition path's host sync, allocator
        # growth — both capture-forbidden on HIP) leaves torch allocator /
        # RNG state subtly damaged even when caught. Long-step configs never
        # attempt capture at all.
        step()
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(3):
            step()
        torch.cuda.synchronize()
        t_probe = (time.perf_counter() - t0) / 3
        if t_probe > 3e-3:
            log(f"eager step {t_probe * 1e3:.2f} ms: above the hipGraph "
                "payoff range; skipping capture")
            args.no_graph = True
    if not args.no_graph and world == 1:
        try:
            g = torch.cuda.CUDAGraph()
            # own outer stream context: if step() raises mid-capture, the
            # graph ctx's __exit__ raises from capture_end BEFORE restoring
            # the current stream, which would strand later eager steps on
            # the invalidated capture stream (hipError 901). A python
            # `with` unwinds this one even when the inner ctx throws.
            cap_stream = torch.cuda.Stream()
            with torch.cuda.stream(cap_stream):
                with torch.cuda.graph(g):
                    graph_res = step()
            g.replay()
            torch.cuda.synchronize()
            eager_res = step()
            torch.cuda.synchronize()
            gr = graph_res.to(torch.float64)
            er = eager_res.to(torch.float64)
            ok = bool(
                torch.isclose(gr, er, rtol=1e-10, atol=1e-10, equal_nan=True)
                .all().item()
            )
            if ok:
                # keep the graph only if replay is actually faster (a
                # captured column op falls to the slab path, which can lose
                # to the eager group-aligned chunks)
                t0 = time.perf_counter()
                for _ in range(3):
                    g.replay()
                torch.cuda.synchronize()
                t_replay = time.perf_counter() - t0
                t0 = time.perf_counter()
                for _ in range(3):
                    step()
                torch.cuda.synchronize()
                t_eager = time.perf_counter() - t0
                if t_replay < t_eager:
                    graph = g
                    log(f"hipGraph step replay enabled (verified; "
                        f"{t_replay / 3 * 1e3:.2f} vs eager {t_eager / 3 * 1e3:.2f} ms)")
                else:
                    log(f"hipGraph replay slower than eager "
                        f"({t_replay / 3 * 1e3:.2f} vs {t_eager / 3 * 1e3:.2f} ms); eager steps")
            else:
                log("hipGraph replay mismatch vs eager; falling back to eager")
        except Exception as e:  # pragma: no cover - path depends on config
            log(f"hipGraph capture unavailable ({type(e).__name__}: {e}); eager steps")
            graph = None
            try:  # drain any half-ended capture state before eager steps
                torch.cuda.synchronize()
            except Exception:
                pass

    # roofline instrumentation: HIP events around every fused-kernel launch
    # on the launch stream (kernel + its O(ngroups) slab-combine tail)
    events = []
    orig = aggregate_hip.grouped_partials
    orig_cols = aggregate_hip.grouped_partials_cols

    def _timed(fn):
        def wrapper(*a, **kw):
            e0 = torch.cuda.Event(enable_timing=True)
            e1 = torch.cuda.Event(enable_timing=True)
            e0.record()
            out = fn(*a, **kw)
            e1.record()
            events.append((e0, e1))
            return out
        return wrapper

    aggregate_hip.grouped_partials = _timed(orig)
    fa_core.grouped_partials = aggregate_hip.grouped_partials  # core binds the name
    aggregate_hip.grouped_partials_cols = _timed(orig_cols)
    fa_core.grouped_partials_cols = aggregate_hip.grouped_partials_cols

    if graph is not None:
        # graph replays bypass the python wrapper: collect the roofline
        # events from instrumented eager steps OUTSIDE the timed region
        for _ in range(min(3, args.steps)):
            step()
        barrier()
        graph_events = events[:]
        events = []

    barrier()
    t0 = time.perf_counter()
    if graph is not None:
        for _ in range(args.steps):
            graph.replay()
    else:
        for _ in range(args.steps):
            step()
    barrier()
    elapsed = time.perf_counter() - t0
    if graph is not None:
        events = graph_events
    aggregate_hip.grouped_partials = orig
    fa_core.grouped_partials = orig
    aggregate_hip.grouped_partials_cols = orig_cols
    fa_core.grouped_partials_cols = orig_cols

    if world > 1:
        t = torch.tensor([elapsed], device=device)
        torch.distributed.all_reduce(t, op=torch.distributed.ReduceOp.MAX)
        elapsed = t.item()

    kernel_ms = [e0.elapsed_time(e1) for e0, e1 in events]
    avg_kernel_ms = float(np.mean(kernel_ms)) if kernel_ms else None
    achieved_gbps = bytes_per_step_per_gpu / (avg_kernel_ms / 1e3) / 1e9 if avg_kernel_ms else None

    if rank == 0:
        total_bytes = bytes_per_step_per_gpu * world * args.steps
        value = total_bytes / elapsed / 1e9
        cpu = None
        if args.config == "2" and not args.no_cpu_baseline:
            cpu = cpu_baseline_leg(min(args.rows, 30_000_000), args.ngroups)
        out = {
            "metric": "grouped-reduce GB/s (input bytes/s) + fraction of HBM peak, 1e9 rows",
            "value": round(value, 2),
            "unit": "GB/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(elapsed / args.steps * 1e3, 3),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,  # reference publishes no number
            "dtype": "f64" if args.config == "5" else "f32",  # accumulation is f64 (npg contract)
            "data": "synthetic",
            "config": {
                "workload": workload,
                **cfg_extra,
                "parallelism": f"dp{world}" if world > 1 else "single",
                "hipgraph": graph is not None,
            },
            "roofline": {
                "bound": "hbm",
                "achieved": round(achieved_gbps, 2) if achieved_gbps else None,
                "peak": HBM_PEAK_GBPS,
                "unit": "GB/s",
                "frac": round(achieved_gbps / HBM_PEAK_GBPS, 4) if achieved_gbps else None,
                "traffic": read_traffic(args.config),
            },
            "cpu_baseline": cpu,
        }
        print(json.dumps(out), flush=True)

    if world > 1:
        torch.distributed.destroy_process_group()


if __name__ == "__main__":
    main()

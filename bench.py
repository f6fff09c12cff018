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
    """Time the oracle (the CPU restatement of the reference algorithm,
    engine="numpy"/numpy_groupies semantics) on a bounded sample of the same
    workload, on this box's host cores. Reported baseline, not the target."""
    from oracle import groupby_reduce as oracle_reduce

    rng = np.random.default_rng(1)
    vals = rng.random(rows_sample, dtype=np.float32)
    labels = rng.integers(0, ngroups, rows_sample)
    t0 = time.perf_counter()
    oracle_reduce(vals, labels, func="mean", expected_groups=np.arange(ngroups))
    dt = time.perf_counter() - t0
    gbps = rows_sample * 12 / dt / 1e9
    return {
        "value": round(gbps, 4),
        "unit": "GB/s",
        "cores": 1,  # the oracle's numpy ops are single-threaded
        "kind": "port",
        "sample": f"{rows_sample:.0e} of {ROWS:.0e} rows, {dt:.1f}s",
    }


def read_traffic():
    """Per-launch HBM bytes from the committed rocprof PMC summary, if any
    (profiles/traffic.json, written by profiles/collect.sh on the GPU box)."""
    p = os.path.join(os.path.dirname(os.path.abspath(__file__)), "profiles", "traffic.json")
    if os.path.exists(p):
        with open(p) as f:
            d = json.load(f)
        return d.get("bytes_per_launch")
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

    n, ng = args.rows, args.ngroups
    log(f"generating {n:.0e} rows, {ng} groups on {device} (rank {rank}/{world})")
    gen = torch.Generator(device=device).manual_seed(1234 + rank)
    vals = torch.rand(n, generator=gen, dtype=torch.float32, device=device)
    labels = torch.randint(0, ng, (n,), generator=gen, dtype=torch.int64, device=device)
    expected = np.arange(ng)
    bytes_per_step_per_gpu = n * (4 + 8)

    def step():
        res, _ = flox_amd.groupby_reduce(vals, labels, func=args.func, expected_groups=expected)
        return res

    def barrier():
        if world > 1:
            torch.distributed.barrier()
        torch.cuda.synchronize()

    # warmup
    for _ in range(args.warmup):
        step()
    barrier()

    # roofline instrumentation: HIP events around every fused-kernel launch
    # on the launch stream (kernel + its O(ngroups) slab-combine tail)
    events = []
    orig = aggregate_hip.grouped_partials

    def timed_partials(*a, **kw):
        e0 = torch.cuda.Event(enable_timing=True)
        e1 = torch.cuda.Event(enable_timing=True)
        e0.record()
        out = orig(*a, **kw)
        e1.record()
        events.append((e0, e1))
        return out

    aggregate_hip.grouped_partials = timed_partials
    fa_core.grouped_partials = timed_partials  # core binds the name directly

    barrier()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        step()
    barrier()
    elapsed = time.perf_counter() - t0
    aggregate_hip.grouped_partials = orig
    fa_core.grouped_partials = orig

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
        cpu = None if args.no_cpu_baseline else cpu_baseline_leg(min(n, 30_000_000), ng)
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
            "dtype": "f32",  # values f32, accumulated in f64 (npg contract)
            "data": "synthetic",
            "config": {
                "workload": "configs[1]: fp32 (1e9,) values, int64 labels, 1e4 uniform groups, func=mean, engine=hip",
                "rows": n,
                "ngroups": ng,
                "func": args.func,
                "labels": "int64",
                "parallelism": f"dp{world}" if world > 1 else "single",
            },
            "roofline": {
                "bound": "hbm",
                "achieved": round(achieved_gbps, 2) if achieved_gbps else None,
                "peak": HBM_PEAK_GBPS,
                "unit": "GB/s",
                "frac": round(achieved_gbps / HBM_PEAK_GBPS, 4) if achieved_gbps else None,
                "traffic": read_traffic(),
            },
            "cpu_baseline": cpu,
        }
        print(json.dumps(out), flush=True)

    if world > 1:
        torch.distributed.destroy_process_group()


if __name__ == "__main__":
    main()

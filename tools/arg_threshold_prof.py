"""Crossover probe: packed-key vs two-pass arg at mid group counts."""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch, flox_amd
import flox_amd.core as core

n = 200_000_000
g = torch.Generator(device="cuda").manual_seed(0)
v = torch.randn(n, generator=g, dtype=torch.float32, device="cuda")
for ng in [1000, 4000, 8000, 16000, 64000]:
    labels = torch.randint(0, ng, (n,), generator=g, dtype=torch.int64, device="cuda")
    times = {}
    for name, thr in [("packed", 1), ("twopass", 1 << 62)]:
        core.PACKED_ARG_THRESHOLD = thr
        for _ in range(2):
            r, _ = flox_amd.groupby_reduce(v, labels, func="argmin", expected_groups=range(ng))
        torch.cuda.synchronize(); t0 = time.perf_counter()
        for _ in range(3):
            r, _ = flox_amd.groupby_reduce(v, labels, func="argmin", expected_groups=range(ng))
        torch.cuda.synchronize()
        times[name] = (time.perf_counter() - t0) / 3 * 1e3
    print(f"ng={ng:6d}: packed {times['packed']:7.2f} ms   twopass {times['twopass']:7.2f} ms")

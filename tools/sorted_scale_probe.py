"""Timing probe: sorted-path families (quantile/median/mode/scan) at scale."""
import os, sys, time
sys.path.insert(0, os.environ.get("GRAFT_REPO_ROOT", "/root/repo"))
import numpy as np
import torch
import flox_amd

def t(fn, iters=3, warm=1):
    for _ in range(warm):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e3

g = torch.Generator(device="cuda").manual_seed(3)
n = 1_000_000_000
vals = torch.rand(n, generator=g, dtype=torch.float32, device="cuda")
lab4 = torch.randint(0, 10_000, (n,), generator=g, dtype=torch.int64, device="cuda")
ms = t(lambda: flox_amd.groupby_reduce(vals, lab4, func="median", expected_groups=range(10_000)))
print(f"median 1e9 rows / 1e4 groups: {ms:.1f} ms ({n*12/ms/1e6:.0f} GB/s algorithmic)")
ms = t(lambda: flox_amd.groupby_reduce(vals, lab4, func="quantile", expected_groups=range(10_000), finalize_kwargs={"q": [0.1, 0.5, 0.9]}))
print(f"vector-q(3) 1e9/1e4: {ms:.1f} ms")
ms = t(lambda: flox_amd.groupby_scan(vals, lab4, func="cumsum", expected_groups=range(10_000)))
print(f"cumsum 1e9/1e4 (random labels): {ms:.1f} ms")
del vals, lab4; torch.cuda.empty_cache()
# mode at moderate cardinality (values with repeats)
vi = torch.randint(-50, 50, (200_000_000,), generator=g, dtype=torch.int64, device="cuda")
li = torch.randint(0, 10_000, (200_000_000,), generator=g, dtype=torch.int64, device="cuda")
ms = t(lambda: flox_amd.groupby_reduce(vi, li, func="mode", expected_groups=range(10_000)))
print(f"mode 2e8 i64 / 1e4 groups: {ms:.1f} ms")

"""Timing probe: packed-key arg vs two-pass atomic at 1e9 rows / 1e7 groups."""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np
import torch
import flox_amd
import flox_amd.core as core

n, ng = 1_000_000_000, 10_000_000
g = torch.Generator(device="cuda").manual_seed(0)
v = torch.randn(n, generator=g, dtype=torch.float32, device="cuda")
labels = torch.randint(0, ng, (n,), generator=g, dtype=torch.int64, device="cuda")

def bench(tag):
    for _ in range(2):
        r, _ = flox_amd.groupby_reduce(v, labels, func="argmin", expected_groups=range(ng))
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(3):
        r, _ = flox_amd.groupby_reduce(v, labels, func="argmin", expected_groups=range(ng))
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / 3
    print(f"{tag}: {dt*1e3:8.2f} ms")
    return r

r_packed = bench("packed  (threshold 7e3)")
core.PACKED_ARG_THRESHOLD = 1 << 62  # force the old two-pass atomic form
r_atomic = bench("two-pass atomic")
assert torch.equal(r_packed, r_atomic), "paths disagree"
print("results identical")

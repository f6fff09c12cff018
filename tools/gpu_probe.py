"""First-contact GPU diagnostics (run manually via gpurun, not pytest)."""

import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np
import torch

print("device:", torch.cuda.get_device_name(0), flush=True)

import flox_amd
from flox_amd import _ffi
from flox_amd.aggregate_hip import grouped_partials

# 1. tiny LDS-path call
v = torch.tensor([1.0, 2.0, 3.0, 4.0, 5.0], device="cuda")
l = torch.tensor([0, 1, 0, 2, 1], device="cuda")
p = grouped_partials(_ffi.SET_SUM_COUNT, v, l, 3)
torch.cuda.synchronize()
print("tiny sums:", p["sum"].cpu().numpy(), "counts:", p["count"].cpu().numpy(), "path:", p["_path"])
assert np.allclose(p["sum"].cpu().numpy(), [4.0, 7.0, 4.0])

# 2. 1e4-group mean -> needs ~120KB dynamic LDS (the big question)
n, ng = 10_000_000, 10_000
g = torch.Generator(device="cuda").manual_seed(0)
vals = torch.rand(n, generator=g, device="cuda")
labels = torch.randint(0, ng, (n,), generator=g, device="cuda")
p = grouped_partials(_ffi.SET_SUM_COUNT, vals, labels, ng)
torch.cuda.synchronize()
print("120KB-LDS path used:", p["_path"], "count total:", int(p["count"].sum().item()))
assert int(p["count"].sum().item()) == n

# 3. quick timing at 1e9 rows (the bench shape)
n = 1_000_000_000
vals = torch.rand(n, generator=g, device="cuda")
labels = torch.randint(0, ng, (n,), generator=g, device="cuda")
for _ in range(2):
    p = grouped_partials(_ffi.SET_SUM_COUNT, vals, labels, ng)
torch.cuda.synchronize()
t0 = time.perf_counter()
iters = 5
for _ in range(iters):
    p = grouped_partials(_ffi.SET_SUM_COUNT, vals, labels, ng)
torch.cuda.synchronize()
dt = (time.perf_counter() - t0) / iters
print(f"1e9-row SUM_COUNT: {dt*1e3:.3f} ms = {n*12/dt/1e9:.0f} GB/s ({n*12/dt/8e12*100:.1f}% of 8TB/s peak)")

# 4. global-atomic path at 1e7 groups (config 3 shape), sum
ng7 = 10_000_000
labels7 = torch.randint(0, ng7, (n,), generator=g, device="cuda")
for _ in range(2):
    p = grouped_partials(_ffi.SET_SUM_COUNT_PRESENT, vals, labels7, ng7, skipnan=False)
torch.cuda.synchronize()
t0 = time.perf_counter()
for _ in range(3):
    p = grouped_partials(_ffi.SET_SUM_COUNT_PRESENT, vals, labels7, ng7)
torch.cuda.synchronize()
dt = (time.perf_counter() - t0) / 3
print(f"1e9-row 1e7-group atomic path: {dt*1e3:.3f} ms = {n*12/dt/1e9:.0f} GB/s, path={p['_path']}")
print("probe OK")

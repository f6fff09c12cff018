"""Trace workload: quantile, scan, and packed-argmin at representative sizes."""
import sys, os, time
sys.path.insert(0, os.environ.get("GRAFT_REPO_ROOT", "/root/repo"))
import numpy as np, torch, flox_amd
g = torch.Generator(device="cuda").manual_seed(3)
n, ng = 200_000_000, 10_000
v = torch.randn(n, generator=g, dtype=torch.float32, device="cuda")
lab = torch.randint(0, ng, (n,), generator=g, dtype=torch.int64, device="cuda")
for _ in range(2):
    q, _ = flox_amd.groupby_reduce(v, lab, func="median", expected_groups=range(ng))
torch.cuda.synchronize(); t0=time.perf_counter()
q, _ = flox_amd.groupby_reduce(v, lab, func="median", expected_groups=range(ng))
torch.cuda.synchronize(); print(f"median 2e8x1e4: {(time.perf_counter()-t0)*1e3:.2f} ms")
for _ in range(2):
    s = flox_amd.groupby_scan(v, lab, func="cumsum", expected_groups=range(ng))
torch.cuda.synchronize(); t0=time.perf_counter()
s = flox_amd.groupby_scan(v, lab, func="cumsum", expected_groups=range(ng))
torch.cuda.synchronize(); print(f"cumsum 2e8x1e4: {(time.perf_counter()-t0)*1e3:.2f} ms")
lab2 = torch.randint(0, 10_000_000, (n,), generator=g, dtype=torch.int64, device="cuda")
for _ in range(2):
    a, _ = flox_amd.groupby_reduce(v, lab2, func="argmin", expected_groups=range(10_000_000))
torch.cuda.synchronize(); t0=time.perf_counter()
a, _ = flox_amd.groupby_reduce(v, lab2, func="argmin", expected_groups=range(10_000_000))
torch.cuda.synchronize(); print(f"argmin 2e8x1e7 (packed): {(time.perf_counter()-t0)*1e3:.2f} ms")

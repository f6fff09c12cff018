import os, sys, time
sys.path.insert(0, os.environ.get("GRAFT_REPO_ROOT", "/root/repo"))
import torch
from flox_amd.aggregate_hip import grouped_partials
from flox_amd import _ffi
g = torch.Generator(device="cuda").manual_seed(0)
n, ng = 1_000_000_000, 10_000_000
vals = torch.rand(n, generator=g, dtype=torch.float32, device="cuda")
labels = torch.randint(0, ng, (n,), generator=g, dtype=torch.int64, device="cuda")
for _ in range(2):
    p = grouped_partials(_ffi.SET_SUM_COUNT_PRESENT, vals, labels, ng)
torch.cuda.synchronize()
t0 = time.perf_counter()
for _ in range(3):
    p = grouped_partials(_ffi.SET_SUM_COUNT_PRESENT, vals, labels, ng)
torch.cuda.synchronize()
print(f"{(time.perf_counter()-t0)/3*1e3:.2f} ms, path={p['_path']}")
# correctness invariants at full scale
assert int(p["count"].sum().item()) == n, int(p["count"].sum().item())
tot = vals.sum(dtype=torch.float64).item()
got = p["sum"].sum().item()
assert abs(got - tot) < 1e-4 * abs(tot) + 1e-3, (got, tot)
assert int(p["present"].min().item()) >= 0
print("invariants OK")

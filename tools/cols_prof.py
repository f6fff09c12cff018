import os, sys, time
sys.path.insert(0, os.environ.get("GRAFT_REPO_ROOT", "/root/repo"))
import numpy as np, torch, flox_amd
n_t, y, x = 8760, 240, 480
g = torch.Generator(device="cuda").manual_seed(0)
arr = torch.rand((n_t, y, x), generator=g, dtype=torch.float32, device="cuda")
hours = (torch.arange(n_t, device="cuda") % 24).to(torch.int64)
view = arr.permute(1, 2, 0)
expected = np.arange(24)
for _ in range(3):
    res, _ = flox_amd.groupby_reduce(view, hours, func="sum", expected_groups=expected)
torch.cuda.synchronize()
t0=time.perf_counter()
for _ in range(3):
    res, _ = flox_amd.groupby_reduce(view, hours, func="sum", expected_groups=expected)
torch.cuda.synchronize()
dt=(time.perf_counter()-t0)/3
print(f"4GB cols sum: {dt*1e3:.2f} ms = {arr.numel()*4/dt/1e9:.0f} GB/s")

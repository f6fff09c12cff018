import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch, flox_amd
n, ng = 200_000_000, 10_000
g = torch.Generator(device="cuda").manual_seed(0)
v = torch.randn(n, generator=g, dtype=torch.float32, device="cuda")
lab_r = torch.randint(0, ng, (n,), generator=g, dtype=torch.int64, device="cuda")
lab_s, _ = torch.sort(lab_r)
for name, lab in [("random", lab_r), ("sorted", lab_s)]:
    for _ in range(2):
        s = flox_amd.groupby_scan(v, lab, func="cumsum", expected_groups=range(ng))
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(3):
        s = flox_amd.groupby_scan(v, lab, func="cumsum", expected_groups=range(ng))
    torch.cuda.synchronize()
    print(f"cumsum 2e8 {name}: {(time.perf_counter()-t0)/3*1e3:7.2f} ms")

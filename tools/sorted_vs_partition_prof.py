import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch, flox_amd
n, ng = 1_000_000_000, 10_000_000
g = torch.Generator(device="cuda").manual_seed(0)
v = torch.rand(n, generator=g, dtype=torch.float32, device="cuda")
lab = torch.sort(torch.randint(0, ng, (n,), generator=g, dtype=torch.int64, device="cuda"))[0]
lab_broken = lab.clone(); lab_broken[0] = ng - 1  # breaks the sortedness sample -> partition path
for name, l in [("direct (sorted flag)", lab), ("partition (sorted data)", lab_broken)]:
    for _ in range(2):
        r, _ = flox_amd.groupby_reduce(v, l, func="sum", expected_groups=range(ng))
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(3):
        r, _ = flox_amd.groupby_reduce(v, l, func="sum", expected_groups=range(ng))
    torch.cuda.synchronize()
    print(f"{name}: {(time.perf_counter()-t0)/3*1e3:7.2f} ms")

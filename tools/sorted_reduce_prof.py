import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch, flox_amd
n, ng = 1_000_000_000, 10_000_000
g = torch.Generator(device="cuda").manual_seed(0)
v = torch.rand(n, generator=g, dtype=torch.float32, device="cuda")
lab_r = torch.randint(0, ng, (n,), generator=g, dtype=torch.int64, device="cuda")
lab_s, _ = torch.sort(lab_r)
for name, lab in [("random", lab_r), ("sorted", lab_s)]:
    for _ in range(2):
        r, _ = flox_amd.groupby_reduce(v, lab, func="sum", expected_groups=range(ng))
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(3):
        r, _ = flox_amd.groupby_reduce(v, lab, func="sum", expected_groups=range(ng))
    torch.cuda.synchronize()
    print(f"sum 1e9x1e7 {name}: {(time.perf_counter()-t0)/3*1e3:7.2f} ms")
import numpy as np
# invariant: total sum over all groups equals the plain sum either way
s_r, _ = flox_amd.groupby_reduce(v, lab_r, func="sum", expected_groups=range(ng))
s_s, _ = flox_amd.groupby_reduce(v, lab_s, func="sum", expected_groups=range(ng))
tot = float(v.sum(dtype=torch.float64).item())
for name, s in [("random", s_r), ("sorted", s_s)]:
    got = float(s.to(torch.float64).sum().item())
    assert abs(got - tot) < 1e-2 * abs(tot) + 1.0, (name, got, tot)
# and per-group counts match exactly (same label multiset)
c_r, _ = flox_amd.groupby_reduce(v, lab_r, func="count", expected_groups=range(ng))
c_s, _ = flox_amd.groupby_reduce(v, lab_s, func="count", expected_groups=range(ng))
np.testing.assert_array_equal(
    np.sort(c_r.cpu().numpy()), np.sort(c_s.cpu().numpy()))
print("invariants OK")

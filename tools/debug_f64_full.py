import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np, torch, flox_amd
from oracle import groupby_reduce as oracle_reduce

for seed in range(30):
    rng = np.random.default_rng(seed)
    n, ng = 200_000, 40_000
    labels = rng.integers(0, ng, n)
    vals = rng.standard_normal(n) * 100
    vals[rng.random(n) < 0.03] = np.nan
    want, *_ = oracle_reduce(vals, labels, func="sum", expected_groups=np.arange(ng))
    got, *_ = flox_amd.groupby_reduce(vals, labels, func="sum", expected_groups=np.arange(ng))
    ok = np.isclose(got, want, rtol=1e-12, atol=1e-14) | (np.isnan(got) & np.isnan(want))
    bad = np.where(~ok)[0]
    print(f"seed {seed}: bad={len(bad)}")
    if len(bad):
        for g in bad[:4]:
            rows = np.where(labels == g)[0]
            print(f"  g={g} got={got[g]!r} want={want[g]!r} nrows={len(rows)} vals={vals[rows][:8]}")
        break
print("done")

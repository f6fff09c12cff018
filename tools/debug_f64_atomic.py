"""Debug harness: f64 sum on the global-atomic path vs a torch-side check."""
import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np, torch
from flox_amd.aggregate_hip import grouped_partials
from flox_amd import _ffi

for seed in range(20):
    rng = np.random.default_rng(seed)
    n, ng = 200_000, 40_000
    labels = rng.integers(0, ng, n)
    vals = (rng.standard_normal(n) * 100)
    vals[rng.random(n) < 0.03] = np.nan
    vt = torch.tensor(vals, device="cuda")
    lt = torch.tensor(labels, device="cuda")
    p = grouped_partials(_ffi.SET_SUM_COUNT_PRESENT, vt, lt, ng, skipnan=False)
    assert p["_path"] == 2, p["_path"]
    got = p["sum"].cpu().numpy()
    want = np.zeros(ng)
    np.add.at(want, labels, vals)
    bad = np.where(~(np.isclose(got, want, rtol=1e-12, atol=1e-12) | (np.isnan(got) & np.isnan(want))))[0]
    cnt_got = p["count"].cpu().numpy()
    cnt_want = np.bincount(labels[~np.isnan(vals)], minlength=ng)
    cbad = np.where(cnt_got != cnt_want)[0]
    pres = p["present"].cpu().numpy()
    pres_want = (np.bincount(labels, minlength=ng) > 0).astype(np.int32)
    pbad = np.where(pres != pres_want)[0]
    print(f"seed {seed}: sum bad={len(bad)} cnt bad={len(cbad)} present bad={len(pbad)}")
    if len(bad):
        for g in bad[:5]:
            rows = np.where(labels == g)[0]
            print(f"  g={g} got={got[g]!r} want={want[g]!r} rows={rows} vals={vals[rows]}")
        break
print("done")

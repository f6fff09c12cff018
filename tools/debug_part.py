import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np, torch, zlib
from flox_amd.aggregate_hip import grouped_partials
from flox_amd import _ffi

rng = np.random.default_rng(zlib.crc32(b"sum"))
n, ng = 1_000_000, 2_000_000
labels = rng.integers(0, ng, n)
vals = rng.standard_normal(n)
vals[rng.random(n) < 0.02] = np.nan
vt = torch.tensor(vals, device="cuda")
lt = torch.tensor(labels, device="cuda")
p = grouped_partials(_ffi.SET_SUM_COUNT_PRESENT, vt, lt, ng, skipnan=False)
print("path", p["_path"])
got_sum = p["sum"].cpu().numpy()
got_cnt = p["count"].cpu().numpy()
got_pres = p["present"].cpu().numpy()
want_sum = np.zeros(ng); np.add.at(want_sum, labels, vals)
want_cnt = np.bincount(labels[~np.isnan(vals)], minlength=ng)
want_pres = (np.bincount(labels, minlength=ng) > 0).astype(np.int32)
ok = np.isclose(got_sum, want_sum, rtol=1e-10, atol=1e-9) | (np.isnan(got_sum) & np.isnan(want_sum))
bad = np.where(~ok)[0]
cbad = np.where(got_cnt != want_cnt)[0]
pbad = np.where(got_pres != want_pres)[0]
print(f"sum bad={len(bad)} cnt bad={len(cbad)} present bad={len(pbad)}")
for g in bad[:6]:
    rows = np.where(labels == g)[0]
    print(f"  g={g} bucketA={g>>18} fine={g>>12} got={got_sum[g]!r} want={want_sum[g]!r} rows={rows[:4]} vals={vals[rows][:4]}")
for g in cbad[:6]:
    print(f"  CNT g={g} fine={g>>12} got={got_cnt[g]} want={want_cnt[g]}")
if len(bad):
    bb = np.unique(bad >> 12)
    print("bad fine buckets:", bb[:20], "n:", len(bb))
    bs = np.unique(bad >> 18)
    print("bad super buckets:", bs[:20])

import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np, torch, flox_amd
from flox_amd import _ffi
from flox_amd.aggregate_hip import grouped_partials

rng = np.random.default_rng(7)
keep, ng = 10, 52945
n = 63420
ngt = keep * ng
codes = torch.randint(0, ngt, (n,), dtype=torch.int64, device="cuda")
vals = torch.randint(-1000, 1000, (n,), dtype=torch.int64, device="cuda")
print("pass1 SUM_COUNT...", flush=True)
p1 = grouped_partials(_ffi.SET_SUM_COUNT, vals, codes, ngt, skipnan=True)
torch.cuda.synchronize(); print("  ok path", p1.get("_path"), flush=True)
means = (p1["sum"].to(torch.float64) / p1["count"]).contiguous()
means = torch.nan_to_num(means, nan=0.0)
print("pass2 SSD...", flush=True)
p2 = grouped_partials(_ffi.SET_SSD, vals, codes, ngt, skipnan=True, means=means)
torch.cuda.synchronize(); print("  ok path", p2.get("_path"), flush=True)
# full API repro of the exact fuzz case
print("full API case...", flush=True)
arr = rng.integers(-1000, 1000, (2, 5, 6342)).astype(np.int64)
by = rng.integers(0, 52943, (2, 5, 6342))
r, *_ = flox_amd.groupby_reduce(arr, by, func="nanvar", axis=(2,),
                                expected_groups=np.arange(52945), fill_value=-7.0)
torch.cuda.synchronize(); print("  full ok", np.asarray(r).shape, flush=True)

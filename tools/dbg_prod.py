import sys, os
sys.path.insert(0, os.environ.get("GRAFT_REPO_ROOT", "/root/repo"))
sys.path.insert(0, os.path.join(os.environ.get("GRAFT_REPO_ROOT", "/root/repo"), "tools"))
import numpy as np
from fuzz_oracle_vs_reference import FUNCS, SCANS, make_case
rng = np.random.default_rng(424242)
for i in range(62):
    arr, by, kw = make_case(rng, big=bool(rng.random() < 0.2))
    is_scan = rng.random() < 0.2 and "axis" not in kw and "min_count" not in kw
    if is_scan:
        func = str(rng.choice(SCANS)); kw.pop("fill_value", None); kw.pop("sort", None)
    else:
        func = str(rng.choice(FUNCS))
        if func in ("quantile","nanquantile"):
            q = [0.25, 0.9] if rng.random() < 0.5 else float(rng.random()); kw["finalize_kwargs"]={"q":q}
        if func in ("var","nanvar","std","nanstd") and rng.random() < 0.3:
            kw["finalize_kwargs"]={"ddof":1}
a = np.asarray(arr)
ac = np.clip(a, 0, 1)
print("case:", func, a.dtype, a.shape, sorted(kw), "eg len:", len(kw["expected_groups"]))
import flox_amd
from oracle import groupby_reduce as oracle_reduce
want, *_ = oracle_reduce(ac, by, func=func, **kw)
got, *_ = flox_amd.groupby_reduce(ac, by, func=func, **kw)
got = np.asarray(got)
bad = np.flatnonzero(got.astype(np.int64) != want.astype(np.int64))
print("mismatches:", bad.size)
if bad.size:
    print("first bad groups:", bad[:10])
    print("got:", got[bad[:10]].astype(np.int64))
    print("want:", want[bad[:10]].astype(np.int64))
    g0 = int(bad[0])
    rows = np.flatnonzero(np.asarray(by) == g0)
    print("group", g0, "rows:", rows.size, "vals:", ac[rows][:20])
    # also direct grouped_partials probe
    import torch
    from flox_amd.aggregate_hip import grouped_partials
    from flox_amd import _ffi
    v = torch.tensor(ac.astype(np.int64), device="cuda")
    l = torch.tensor(np.asarray(by), device="cuda")
    ng = len(kw["expected_groups"])
    p = grouped_partials(_ffi.SET_PROD, v, l, ng, skipnan=True)
    print("path:", p["_path"], "sum bin:", p["sum"][bad[:10]].cpu().numpy(),
          "count:", p["count"][bad[:10]].cpu().numpy(),
          "present:", p["present"][bad[:10]].cpu().numpy())

import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
sys.path.insert(0, os.path.join(os.path.dirname(os.path.dirname(os.path.abspath(__file__))), "tools"))
import numpy as np
from fuzz_oracle_vs_reference import make_case, FUNCS, SCANS
rng = np.random.default_rng(424242)
case = None
for i in range(161):
    big = bool(rng.random() < 0.2)
    arr, by, kw = make_case(rng, big=big)
    is_scan = rng.random() < 0.2 and "axis" not in kw and "min_count" not in kw
    if is_scan:
        func = str(rng.choice(SCANS)); kw.pop("fill_value", None); kw.pop("sort", None)
    else:
        func = str(rng.choice(FUNCS))
        if func in ("quantile", "nanquantile"):
            q = [0.25, 0.9] if rng.random() < 0.5 else float(rng.random())
            kw["finalize_kwargs"] = {"q": q}
        if func in ("var", "nanvar", "std", "nanstd") and rng.random() < 0.3:
            kw["finalize_kwargs"] = {"ddof": 1}
    if i == 160:
        case = (arr, by, kw, func, is_scan)
arr, by, kw, func, is_scan = case
print("case160:", func, np.asarray(arr).dtype, np.shape(arr), sorted(kw), flush=True)
import torch, flox_amd
bys = by if isinstance(by, tuple) else (by,)
for rep in range(5):
    r, *_ = flox_amd.groupby_reduce(arr, *bys, func=func, **kw)
    torch.cuda.synchronize()
    print("rep", rep, "ok", flush=True)
print("DONE", flush=True)

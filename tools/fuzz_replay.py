"""Replay one fuzz case index with full detail. Usage: fuzz_replay.py IDX [seed]"""
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(os.path.dirname(os.path.abspath(__file__))), "tests", "golden"))
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np
from generate import load_reference
from fuzz_oracle_vs_reference import make_case, FUNCS, SCANS

idx = int(sys.argv[1])
seed = int(sys.argv[2]) if len(sys.argv) > 2 else 12345
core = load_reference()
import importlib
refscan = importlib.import_module("floxref.scan")
from oracle import groupby_reduce as oreduce
from oracle import groupby_scan as oscan

rng = np.random.default_rng(seed)
for i in range(idx + 1):
    arr, by, kw = make_case(rng)
    is_scan = rng.random() < 0.2 and "axis" not in kw and "min_count" not in kw
    if is_scan:
        func = str(rng.choice(SCANS))
        kw.pop("fill_value", None)
        kw.pop("sort", None)
    else:
        func = str(rng.choice(FUNCS))
        if func in ("quantile", "nanquantile"):
            q = [0.25, 0.9] if rng.random() < 0.5 else float(rng.random())
            kw["finalize_kwargs"] = {"q": q}
        if func in ("var", "nanvar", "std", "nanstd") and rng.random() < 0.3:
            kw["finalize_kwargs"] = {"ddof": 1}

print("func:", func, "scan:", is_scan)
print("arr dtype/shape:", np.asarray(arr).dtype, np.shape(arr))
print("kw:", kw)
bys = by if isinstance(by, tuple) else (by,)
np.set_printoptions(threshold=60, precision=10)
if is_scan:
    want = refscan.groupby_scan(arr, *bys, func=func, **kw)
    got = oscan(arr, *bys, func=func, **kw)
else:
    want, *wg = core.groupby_reduce(arr, *bys, func=func, engine="flox", **kw)
    got, *gg = oreduce(arr, *bys, func=func, **kw)
    print("ref groups:", wg)
    print("orc groups:", gg)
print("ref:", np.asarray(want), np.asarray(want).dtype)
print("orc:", np.asarray(got), np.asarray(got).dtype)
if np.shape(arr) and np.prod(np.shape(arr)) < 80:
    print("arr:", arr)
    print("by:", by)

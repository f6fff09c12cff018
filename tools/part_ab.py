"""A/B the partition-path variants (FH_PART_MODE env selects per process):
timings + result digests for cross-process comparison. Digests must match
bit-for-bit for count/present and to f64-roundoff for sums across modes."""
import os
import sys
import time

sys.path.insert(0, os.environ.get("GRAFT_REPO_ROOT", "/root/repo"))
import torch  # noqa: E402

from flox_amd import _ffi  # noqa: E402
from flox_amd.aggregate_hip import grouped_partials  # noqa: E402

MODE = os.environ.get("FH_PART_MODE", "0")
N = int(os.environ.get("AB_N", 1_000_000_000))
ITERS = int(os.environ.get("AB_ITERS", 3))


def digest(p):
    out = {}
    for k, t in p.items():
        if k == "_path":
            out[k] = t
            continue
        tt = t.to(torch.float64) if t.dtype.is_floating_point else t.to(torch.int64)
        out[k] = (
            f"{tt.sum().item():.6e}",
            f"{tt[::97].sum().item():.6e}",
        )
    return out


def run(tag, op_set, ng, dtype=torch.float32, skipnan=False, nan_frac=0.0):
    g = torch.Generator(device="cuda").manual_seed(12345)
    if dtype.is_floating_point:
        vals = torch.rand(N, generator=g, dtype=dtype, device="cuda")
        if nan_frac:
            m = torch.rand(N, generator=g, device="cuda") < nan_frac
            vals[m] = float("nan")
    else:
        vals = torch.randint(-1000, 1000, (N,), generator=g, dtype=dtype, device="cuda")
    labels = torch.randint(0, ng, (N,), generator=g, dtype=torch.int64, device="cuda")
    for _ in range(2):
        p = grouped_partials(op_set, vals, labels, ng, skipnan=skipnan)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(ITERS):
        p = grouped_partials(op_set, vals, labels, ng, skipnan=skipnan)
    torch.cuda.synchronize()
    ms = (time.perf_counter() - t0) / ITERS * 1e3
    alg_bytes = N * (dtype.itemsize + 8)  # values + int64 labels, read once
    print(f"[mode={MODE}] {tag}: {ms:.2f} ms path={p['_path']} "
          f"algGBs={alg_bytes / ms / 1e6:.0f}", flush=True)
    print(f"[mode={MODE}] {tag} digest: {digest(p)}", flush=True)
    del vals, labels, p
    torch.cuda.empty_cache()


CASES = {
    "sum": lambda: run("sum f32 1e7g", _ffi.SET_SUM_COUNT_PRESENT, 10_000_000),
    "mean": lambda: run("mean f32 1e7g", _ffi.SET_SUM_COUNT, 10_000_000),
    "nansum": lambda: run("nansum f32 1e7g 2%nan", _ffi.SET_SUM_COUNT_PRESENT,
                          10_000_000, skipnan=True, nan_frac=0.02),
    "min": lambda: run("min f32 1e7g", _ffi.SET_MIN_FULL, 10_000_000),
    "sum2e6": lambda: run("sum f32 2e6g", _ffi.SET_SUM_COUNT_PRESENT, 2_000_000),
    "sum1e5": lambda: run("sum f32 1e5g", _ffi.SET_SUM_COUNT_PRESENT, 100_000),
    "sumf64": lambda: run("sum f64 1e7g", _ffi.SET_SUM_COUNT_PRESENT,
                          10_000_000, dtype=torch.float64),
    "argf64": lambda: run("argmin-pair f64 1e7g", _ffi.SET_ARGMIN_PAIR,
                          10_000_000, dtype=torch.float64),
    "argi64": lambda: run("argmin-pair i64 1e7g", _ffi.SET_ARGMIN_PAIR,
                          10_000_000, dtype=torch.int64),
}

if __name__ == "__main__":
    only = os.environ.get("AB_ONLY")
    for name, fn in CASES.items():
        if only and name not in only.split(","):
            continue
        fn()

"""Standing reduced version of tools/sweep_degenerate.py: exhaustively
compare the oracle with the REFERENCE on degenerate tiny inputs (n <= 2,
float64) — the corner where the reference's scan identity shortcut hid
(fuzz seed 606162). Runs only where /root/reference is importable (the
build container); the GPU box pins the same surface via golden fixtures.
"""
import os
import sys

import numpy as np
import pytest

sys.path.insert(0, os.path.join(os.path.dirname(__file__), "..", "tools"))
sys.path.insert(0, os.path.join(os.path.dirname(__file__), "..", "oracle"))

from ref_loader import available  # noqa: E402

pytestmark = pytest.mark.skipif(
    not available(), reason="reference tree not present (GPU box)"
)


@pytest.mark.filterwarnings("ignore::RuntimeWarning")
def test_degenerate_sweep_n2():
    import itertools

    from ref_loader import load_reference
    from fuzz_oracle_vs_reference import FUNCS, SCANS
    from sweep_degenerate import label_patterns

    core = load_reference()
    import importlib

    refscan = importlib.import_module("floxref.scan")
    from oracle import groupby_reduce as oreduce
    from oracle import groupby_scan as oscan

    INF_ARTIFACT_FUNCS = {
        "nanmin", "nanmax", "nanvar", "nanstd", "median", "nanmedian",
        "cumsum", "nancumsum",
    }
    fpool = [float("nan"), 1.5, -0.0, float("inf"), float("-inf")]
    n_ok = 0
    for n in (1, 2):
        for vt in itertools.product(fpool, repeat=n):
            vals = np.array(vt)
            for lp in label_patterns(n):
                labels = np.array(lp)
                for func, is_scan in [(f, False) for f in FUNCS] + [
                    (s, True) for s in SCANS
                ]:
                    if func in INF_ARTIFACT_FUNCS and np.isinf(vals).any():
                        continue  # documented reference ±inf artifacts
                    kw = {"finalize_kwargs": {"q": 0.4}} if "quantile" in func else {}
                    try:
                        if is_scan:
                            want = refscan.groupby_scan(vals, labels, func=func, **kw)
                        else:
                            want, *_ = core.groupby_reduce(
                                vals, labels, func=func, engine="flox", **kw)
                    except Exception:
                        continue  # reference can't run it here (npg fallback etc.)
                    try:
                        if is_scan:
                            got = oscan(vals, labels, func=func, **kw)
                        else:
                            got, *_ = oreduce(vals, labels, func=func, **kw)
                    except NotImplementedError:
                        continue
                    want, got = np.asarray(want), np.asarray(got)
                    ctx = f"{func} scan={is_scan} v={vals} by={labels}"
                    assert got.shape == want.shape, ctx
                    assert got.dtype == want.dtype, ctx
                    if want.dtype.kind in "iub":
                        np.testing.assert_array_equal(got, want, err_msg=ctx)
                    else:
                        np.testing.assert_allclose(
                            got, want, equal_nan=True, rtol=1e-11, atol=1e-11,
                            err_msg=ctx)
                    n_ok += 1
    assert n_ok > 3000, n_ok

"""Exhaustive oracle-vs-reference sweep of DEGENERATE inputs.

Randomized fuzz reaches tiny shapes rarely (the scan identity-shortcut
quirk at n=1 hid in 1-of-30000 draws — seed 606162), so this sweeps them
exhaustively: every combination of values from a small adversarial pool
(NaN, +/-0.0, +/-inf), every label pattern (incl. NaN labels and the
all-distinct pattern that triggers the reference's scan.py:286-291
identity shortcut), n in {1,2,3}, across every reduction and scan the
reference can run in this container.

Usage: python tools/sweep_degenerate.py            # float64 sweep
       SWEEP_INT=1 python tools/sweep_degenerate.py  # + int64 sweep
"""
import itertools
import os
import sys

sys.path.insert(0, os.path.dirname(__file__))
sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import numpy as np  # noqa: E402

from fuzz_oracle_vs_reference import FUNCS, SCANS  # noqa: E402


def label_patterns(n):
    """All label tuples from {0, 1, NaN}^n plus the all-distinct pattern."""
    pats = set(itertools.product([0.0, 1.0, float("nan")], repeat=n))
    pats.add(tuple(float(i) for i in range(n)))  # all-distinct (identity shortcut)
    return sorted(pats, key=repr)


def main():
    sys.path.insert(0, os.path.join(os.path.dirname(__file__), "..", "oracle"))
    from ref_loader import load_reference

    core = load_reference()
    import importlib

    refscan = importlib.import_module("floxref.scan")
    from oracle import groupby_reduce as oreduce
    from oracle import groupby_scan as oscan

    fpool = [float("nan"), 1.5, -0.0, float("inf"), float("-inf")]
    ipool = [0, -3, 7]
    n_ok = n_skip = n_bad = n_artifact = 0

    # Reference implementation artifacts under +-inf values (excluded from
    # parity, documented in DESIGN.md "reference artifact regions"):
    # - nanmin/nanmax mask NaN by filling +-inf (aggregate_flox.py:215-217),
    #   so a GENUINE +-inf extremum is indistinguishable from an all-NaN
    #   group and comes back NaN;
    # - nanvar/nanstd accumulate nansum of squared deviations
    #   (aggregations.py var family): inf - inf = NaN deviations are
    #   silently dropped, e.g. nanvar([inf]) -> 0.0;
    # - median/nanmedian lerp between inf endpoints -> inf*0 = NaN terms;
    # - cumsum/nancumsum run ONE global accumulate then subtract each prior
    #   group's total (aggregate_flox.py:296-326): +-inf in any earlier
    #   group poisons every later group with inf - inf = NaN.
    # Our kernels/oracle keep numpy's own semantics for these inputs.
    INF_ARTIFACT_FUNCS = {
        "nanmin", "nanmax", "nanvar", "nanstd", "median", "nanmedian",
        "cumsum", "nancumsum",
    }

    def run_case(vals, labels, func, is_scan, kw):
        nonlocal n_ok, n_skip, n_bad, n_artifact
        if func in INF_ARTIFACT_FUNCS and np.isinf(np.asarray(vals, dtype="f8")).any():
            n_artifact += 1
            return
        bys = (labels,)
        try:
            if is_scan:
                want = refscan.groupby_scan(vals, *bys, func=func, **kw)
            else:
                want, *_ = core.groupby_reduce(vals, *bys, func=func, engine="flox", **kw)
        except Exception:
            n_skip += 1
            return
        try:
            if is_scan:
                got = oscan(vals, *bys, func=func, **kw)
            else:
                got, *_ = oreduce(vals, *bys, func=func, **kw)
        except NotImplementedError:
            n_skip += 1
            return
        except Exception as e:
            n_bad += 1
            print(f"ORACLE RAISED {type(e).__name__}: {e} | {func} v={vals} by={labels} kw={kw}")
            return
        want, got = np.asarray(want), np.asarray(got)
        try:
            assert got.shape == want.shape, (got.shape, want.shape)
            assert got.dtype == want.dtype, (got.dtype, want.dtype)
            if want.dtype.kind in "iubMm":
                np.testing.assert_array_equal(got, want)
            else:
                np.testing.assert_allclose(got, want, equal_nan=True,
                                           rtol=1e-11, atol=1e-11)
            n_ok += 1
        except AssertionError as e:
            n_bad += 1
            print(f"MISMATCH {func} scan={is_scan} v={vals} by={labels} kw={kw}:\n{e}")

    do_int = os.environ.get("SWEEP_INT") == "1"
    pools = [(fpool, np.float64)] + ([(ipool, np.int64)] if do_int else [])
    for pool, dt in pools:
        for n in (1, 2, 3):
            vcombos = list(itertools.product(pool, repeat=n))
            lpats = label_patterns(n)
            for vt in vcombos:
                vals = np.array(vt, dtype=dt)
                for lp in lpats:
                    labels = np.array(lp)
                    for func in FUNCS:
                        kw = {}
                        if func in ("quantile", "nanquantile"):
                            kw["finalize_kwargs"] = {"q": 0.4}
                        run_case(vals, labels, func, False, kw)
                    for func in SCANS:
                        run_case(vals, labels, func, True, {})
            print(f"[{dt.__name__} n={n}] cumulative: {n_ok} ok, {n_skip} skipped, "
                  f"{n_artifact} inf-artifact, {n_bad} mismatches", flush=True)

    # kwarg corners at n <= 2: expected supersets + fill_value, min_count
    # bigger than any group, ddof on 1-2 element groups
    for n in (1, 2):
        for vt in itertools.product(fpool, repeat=n):
            vals = np.array(vt)
            for lp in label_patterns(n):
                labels = np.array(lp)
                for exp in (np.array([0.0, 1.0]), np.array([0.0, 1.0, 5.0])):
                    for func in FUNCS:
                        kw = {"expected_groups": exp}
                        if func in ("quantile", "nanquantile"):
                            kw["finalize_kwargs"] = {"q": 0.4}
                        run_case(vals, labels, func, False, kw)
                    for func, fv in [("sum", -7.5), ("min", -7.5), ("last", -7.5)]:
                        run_case(vals, labels, func, False,
                                 {"expected_groups": exp, "fill_value": fv})
                    for mc in (1, 2, 3):
                        run_case(vals, labels, "nansum", False,
                                 {"expected_groups": exp, "min_count": mc,
                                  "fill_value": np.nan})
                for func in ("var", "nanvar", "std", "nanstd"):
                    run_case(vals, labels, func, False,
                             {"finalize_kwargs": {"ddof": 1}})
    print(f"[kwargs n<=2] cumulative: {n_ok} ok, {n_skip} skipped, "
          f"{n_artifact} inf-artifact, {n_bad} mismatches", flush=True)

    # datetime values with NaT (min/max/count/first/last/median + the
    # non-float ffill/bfill early identity, scan.py:199-201)
    dpool = [np.datetime64("NaT"), np.datetime64("2020-01-01"),
             np.datetime64("1970-01-01")]
    for n in (1, 2):
        for vt in itertools.product(dpool, repeat=n):
            vals = np.array(vt, dtype="datetime64[s]")
            for lp in label_patterns(n):
                labels = np.array(lp)
                for func in ("min", "nanmin", "max", "nanmax", "count",
                             "first", "last", "nanfirst", "nanlast",
                             "median", "nanmedian"):
                    run_case(vals, labels, func, False, {})
                for func in ("ffill", "bfill"):
                    run_case(vals, labels, func, True, {})
    print(f"[datetime n<=2] cumulative: {n_ok} ok, {n_skip} skipped, "
          f"{n_artifact} inf-artifact, {n_bad} mismatches", flush=True)

    # 2-D lead dims at degenerate sizes: exhaustive value grids over
    # (m, n) in {1,2}^2 (trimmed pool at 2x2), default axis (the random
    # fuzz covers axis subsets at larger shapes)
    for m, ncol in ((1, 1), (1, 2), (2, 1), (2, 2)):
        pool2 = fpool if m * ncol <= 2 else [float("nan"), 1.5, float("inf"), -0.0]
        for vt in itertools.product(pool2, repeat=m * ncol):
            vals = np.array(vt).reshape(m, ncol)
            for lp in label_patterns(ncol):
                labels = np.array(lp)
                for func in FUNCS:
                    kw = {}
                    if func in ("quantile", "nanquantile"):
                        kw["finalize_kwargs"] = {"q": 0.4}
                    run_case(vals, labels, func, False, kw)
    print(f"[2d lead m,n<=2] cumulative: {n_ok} ok, {n_skip} skipped, "
          f"{n_artifact} inf-artifact, {n_bad} mismatches", flush=True)

    # complex values, linear set (inf excluded: same artifact regions)
    cpool = [complex("nan"), 1.5 + 2.5j, complex(0, float("nan")), -0.0 + 0j]
    for n in (1, 2):
        for vt in itertools.product(cpool, repeat=n):
            vals = np.array(vt, dtype=np.complex128)
            for lp in label_patterns(n):
                labels = np.array(lp)
                for func in ("sum", "nansum", "mean", "nanmean", "count",
                             "first", "last", "nanfirst", "nanlast"):
                    run_case(vals, labels, func, False, {})
                for func in SCANS:
                    run_case(vals, labels, func, True, {})
    print(f"[complex n<=2] cumulative: {n_ok} ok, {n_skip} skipped, "
          f"{n_artifact} inf-artifact, {n_bad} mismatches", flush=True)
    print(f"degenerate sweep done: {n_ok} ok, {n_skip} skipped, "
          f"{n_artifact} inf-artifact, {n_bad} mismatches")
    sys.exit(1 if n_bad else 0)


if __name__ == "__main__":
    main()

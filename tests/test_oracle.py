"""Pin the oracle (oracle/flox_oracle.py) against golden vectors generated from
the reference's own implementation (tests/golden/generate.py)."""

import numpy as np
import pytest

from oracle import groupby_reduce as oracle_reduce
from oracle import groupby_scan as oracle_scan
from tests.golden_util import load_golden_cases, tolerance_for

CASES = list(load_golden_cases())


@pytest.mark.parametrize("name,inputs,expected,groups", CASES, ids=[c[0] for c in CASES])
def test_oracle_matches_reference(name, inputs, expected, groups):
    kw = dict(inputs)
    arr = kw.pop("array")
    bys = kw.pop("by")
    if kw.pop("_scan", False):
        result = oracle_scan(arr, *bys, **kw)
        found = []
    else:
        result, *found = oracle_reduce(arr, *bys, **kw)
    assert result.shape == expected.shape, (result.shape, expected.shape)
    assert result.dtype == expected.dtype, (result.dtype, expected.dtype)
    if expected.dtype.kind in "Mm":
        np.testing.assert_array_equal(result, expected)
    else:
        tol = tolerance_for(name, expected.dtype)
        np.testing.assert_allclose(result, expected, equal_nan=True, **tol)
    for f, g in zip(found, groups):
        np.testing.assert_array_equal(np.asarray(f, dtype=g.dtype), g)

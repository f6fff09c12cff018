"""Standing huge-size fuzz coverage (default 6 cases; tools/fuzz_huge_gpu.py
runs the full sweep). 2e6..2e7-row cases through partition / pair-arg /
packed / sorted-direct / overflow paths, vectorized-oracle and cross-path
checked."""

import os
import sys

import numpy as np
import pytest

sys.path.insert(0, os.path.join(
    os.path.dirname(os.path.dirname(os.path.abspath(__file__))), "tools"))

pytestmark = pytest.mark.gpu

N_CASES = int(os.environ.get("FUZZ_HUGE_CASES_TEST", "6"))


def test_fuzz_huge():
    import torch

    from fuzz_huge_gpu import one_case

    rng = np.random.default_rng(909)
    for i in range(N_CASES):
        one_case(i, rng)
        torch.cuda.empty_cache()

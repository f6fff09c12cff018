"""TEST INFRASTRUCTURE ONLY — CPU oracle for flox-semantics grouped reductions.

This package is the parity *checker* for the flox_amd HIP engine. It restates
the algorithm of the reference implementation (xarray-contrib/flox, snapshot at
/root/reference) in plain numpy, with float64 accumulation matching the
``engine="numpy"`` (numpy_groupies) numerics contract documented at
reference tests/test_properties.py:146-151.

Only ``tests/``, ``__graft_entry__.smoke()`` and ``bench.py``'s ``cpu_baseline``
leg may import this package. The product path (``flox_amd``) never imports it
and fails loudly when its HIP extension is missing.

Pinning: the oracle is validated against golden vectors generated from the
reference's own implementation (see tests/golden/generate.py, runnable only in
the build container where /root/reference is mounted) — tests/test_oracle.py.
"""

from .flox_oracle import ALL_FUNCS, groupby_reduce, groupby_scan  # noqa: F401

"""TEST INFRASTRUCTURE ONLY — loader for the vendored shimmed reference.

The vendored tree (oracle/_ref/floxref/, written by
tools/vendor_reference.py in the build container; gitignored, ships with
the gpurun snapshot) is the reference implementation of xarray-contrib/flox
with the SURVEY.md §8c syntax shims and the INTEGRATION.md §2 engine="hip"
maintainer patch applied. This module stubs the reference's absent hard
dependencies (numpy_groupies, toolz — untouched by the eager
engine="flox"/"hip" paths) and imports it.

Consumers: tests/test_reference_integration_gpu.py (the reference driving
the HIP engine end-to-end) and bench.py's cpu_baseline leg (the true
engine="flox" CPU timing, kind="reference"). The product path never
imports this.
"""

from __future__ import annotations

import os
import sys
import types

_REF_DIR = os.path.join(os.path.dirname(os.path.abspath(__file__)), "_ref")


def available() -> bool:
    return os.path.isfile(os.path.join(_REF_DIR, "floxref", "core.py"))


def _install_stubs() -> None:
    def _unavailable(*a, **k):  # pragma: no cover
        raise RuntimeError("numpy_groupies stub: not available in this container")

    npg_stub = types.ModuleType("numpy_groupies")
    npg_stub.aggregate_numpy = types.SimpleNamespace(aggregate=_unavailable)
    npg_stub.aggregate_numba = types.SimpleNamespace(aggregate=_unavailable)
    npg_stub.aggregate = _unavailable
    sys.modules.setdefault("numpy_groupies", npg_stub)

    toolz_stub = types.ModuleType("toolz")
    toolz_stub.partition_all = _unavailable
    toolz_stub.unique = _unavailable
    toolz_stub.memoize = lambda f=None, **k: (f if f is not None else (lambda g: g))
    sys.modules.setdefault("toolz", toolz_stub)


def load_reference():
    """Import the vendored reference; returns its core module (floxref.core)."""
    if not available():
        raise RuntimeError(
            "vendored reference missing (oracle/_ref/floxref). Run "
            "tools/vendor_reference.py in the build container first."
        )
    _install_stubs()
    if _REF_DIR not in sys.path:
        sys.path.insert(0, _REF_DIR)
    import floxref.core as core  # noqa: PLC0415

    return core

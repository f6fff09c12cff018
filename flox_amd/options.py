"""Runtime options (the reference's flox.set_options surface, options.py:9-64,
re-keyed to this engine's knobs).

Supported options:
  packed_arg_threshold: group count above which arg-reductions leave the
      LDS two-pass form for the packed/pair partition paths (core.py).
  sparse_combine_ngroups: group count from which the distributed combine
      tries the shard-aware sparse (touched-bin) exchange (distributed.py).
  sparse_combine_fraction: gathered-traffic fraction under which the sparse
      exchange engages (vs the dense all-reduce).
"""

from __future__ import annotations

from contextlib import contextmanager

OPTIONS = {
    "packed_arg_threshold": None,     # None = the measured default
    "sparse_combine_ngroups": None,
    "sparse_combine_fraction": None,
}


def _apply(name, value):
    from . import core, distributed

    if name == "packed_arg_threshold":
        old = core.PACKED_ARG_THRESHOLD
        if value is not None:
            core.PACKED_ARG_THRESHOLD = int(value)
        return ("packed_arg_threshold", old)
    if name == "sparse_combine_ngroups":
        old = distributed.SPARSE_NGROUPS
        if value is not None:
            distributed.SPARSE_NGROUPS = int(value)
        return ("sparse_combine_ngroups", old)
    if name == "sparse_combine_fraction":
        old = distributed.SPARSE_FRACTION
        if value is not None:
            distributed.SPARSE_FRACTION = float(value)
        return ("sparse_combine_fraction", old)
    raise ValueError(
        f"unknown option {name!r}; supported: {sorted(OPTIONS)}"
    )


def _restore(name, old):
    from . import core, distributed

    if name == "packed_arg_threshold":
        core.PACKED_ARG_THRESHOLD = old
    elif name == "sparse_combine_ngroups":
        distributed.SPARSE_NGROUPS = old
    elif name == "sparse_combine_fraction":
        distributed.SPARSE_FRACTION = old


@contextmanager
def set_options(**kwargs):
    """Context manager over the engine's runtime knobs (the reference's
    flox.set_options shape: `with flox_amd.set_options(x=...): ...`)."""
    saved = [_apply(k, v) for k, v in kwargs.items()]
    try:
        yield
    finally:
        for name, old in reversed(saved):
            _restore(name, old)

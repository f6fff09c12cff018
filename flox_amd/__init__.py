"""flox_amd — MI355X-native grouped-reduction engine with flox's API.

A from-scratch re-implementation of the hot path of xarray-contrib/flox
(``groupby_reduce``: factorize -> per-group sum/count/min/max/var) as
hand-written HIP/CDNA4 kernels behind flox's own engine-plugin boundary,
with an RCCL-over-xGMI combine across GPUs. See DESIGN.md.
"""

from .aggregations import (  # noqa: F401
    REDUCTIONS,
    Aggregation,
    CustomAggregation,
    generic_aggregate,
)
from .core import groupby_reduce  # noqa: F401
from .options import OPTIONS, set_options  # noqa: F401
from .scan import groupby_scan  # noqa: F401

__version__ = "0.1.0"

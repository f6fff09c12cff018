"""Vendor the shimmed reference into oracle/_ref/ (gitignored, ships to GPU).

TEST INFRASTRUCTURE ONLY. Runs in the build container, where the reference
(xarray-contrib/flox) is mounted read-only at /root/reference. It writes a
Python-3.10-compatible copy of the reference package to oracle/_ref/floxref/
with:

  * the three mechanical syntax shims verified in SURVEY.md §8c (PEP-646
    star-subscript, starred return annotation, typing.Self), identical to
    tests/golden/generate.py's in-memory loader;
  * the INTEGRATION.md §2 maintainer patch applied: an `elif engine ==
    "hip"` arm in generic_aggregate (reference aggregations.py:60-133) and
    "hip" added to the T_Engine literal (reference core.py:87) — so the
    REFERENCE's own groupby_reduce can drive the flox_amd engine end-to-end
    (tests/test_reference_integration_gpu.py).

oracle/_ref/ is listed in .gitignore (reference sources never enter the
repository history) but ships with the gpurun snapshot, so the GPU box can
run the reference in the driver's seat and time the true `engine="flox"`
CPU baseline (bench.py). Loader: oracle/ref_loader.py.
"""

from __future__ import annotations

import os
import re
import shutil
import sys

REF = "/root/reference/flox"
HERE = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
DEST = os.path.join(HERE, "oracle", "_ref", "floxref")

HIP_ARM = '''\
    elif engine == "hip":
        from flox_amd import aggregate_hip

        method = getattr(aggregate_hip, {"any": "any_", "all": "all_"}.get(func, func))

'''


def vendor() -> str:
    if not os.path.isdir(REF):
        raise RuntimeError(f"reference not present at {REF}")
    if os.path.isdir(DEST):
        shutil.rmtree(DEST)
    os.makedirs(DEST)
    patched_agg = patched_core = False
    for fname in sorted(os.listdir(REF)):
        if not fname.endswith(".py"):
            continue
        with open(os.path.join(REF, fname)) as f:
            src = f.read()
        # shim 1: PEP-646 star-subscript (aggregations.py:400,408)
        src = src.replace("array[*not_last]", "array[tuple(not_last)]")
        src = src.replace("array[*not_first]", "array[tuple(not_first)]")
        # shim 2: starred return annotation (core.py:754)
        src = re.sub(
            r"-> tuple\[DaskArray, \*tuple\[np\.ndarray \| DaskArray, \.\.\.\]\]:", ":", src
        )
        # shim 3: typing.Self (multiarray.py:2)
        src = src.replace("from typing import Self", "from typing_extensions import Self")
        # intra-package imports: flox.X -> floxref.X
        src = re.sub(r"\bfrom flox(\.|\b)", r"from floxref\1", src)
        src = re.sub(r"\bimport flox\b", "import floxref", src)
        # the three-line maintainer patch
        if fname == "aggregations.py":
            anchor = '    elif engine in ["numpy", "numba"]:'
            assert anchor in src, "generic_aggregate anchor moved"
            src = src.replace(anchor, HIP_ARM + anchor, 1)
            patched_agg = True
        if fname == "core.py":
            t_engine = 'T_Engine: TypeAlias = Literal["flox", "numpy", "numba", "numbagg"]'
            if t_engine in src:
                src = src.replace(
                    t_engine,
                    'T_Engine: TypeAlias = Literal["flox", "numpy", "numba", "numbagg", "hip"]',
                    1,
                )
                patched_core = True
        with open(os.path.join(DEST, fname), "w") as f:
            f.write(src)
    assert patched_agg, "failed to apply the generic_aggregate hip arm"
    assert patched_core, "failed to widen T_Engine"
    return DEST


if __name__ == "__main__":
    print(f"vendored shimmed+patched reference -> {vendor()}")
    # prove it imports and the seam dispatches (CPU-only check: the hip arm
    # resolves the callable; running it needs a GPU)
    sys.path.insert(0, HERE)
    from oracle.ref_loader import load_reference  # noqa: E402

    core = load_reference()
    print("imported:", core.__name__)

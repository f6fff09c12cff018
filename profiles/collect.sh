#!/bin/bash
# rocprofv3 evidence for the headline kernel — run ON the MI355X box.
# Counter passes are kept separate from trace passes (gpurun requirement).
cd /tmp && export TMPDIR=/tmp
REPO=${GRAFT_REPO_ROOT:-/root/repo}
OUT=$REPO/gpurun_out/prof
mkdir -p "$OUT"
B="python $REPO/bench.py --steps 3 --warmup 1 --no-cpu-baseline"
rocprofv3 --kernel-trace --stats -d "$OUT/trace" -- $B > "$OUT/bench_traced.json" 2> "$OUT/trace.log"
rocprofv3 --pmc FETCH_SIZE -d "$OUT/fetch" -- $B > /dev/null 2> "$OUT/fetch.log"
rocprofv3 --pmc WRITE_SIZE -d "$OUT/write" -- $B > /dev/null 2> "$OUT/write.log"
find "$OUT" -type f | head -40

#!/bin/bash
# rocprofv3 kernel-trace evidence for every kernel path (run on the MI355X box)
cd /tmp && export TMPDIR=/tmp
REPO=${GRAFT_REPO_ROOT:-/root/repo}
OUT=$REPO/gpurun_out/prof_all
mkdir -p "$OUT"
rocprofv3 --kernel-trace --stats -d "$OUT/c2" -- python $REPO/bench.py --steps 3 --warmup 1 --no-cpu-baseline > "$OUT/c2.json" 2> "$OUT/c2.log"
rocprofv3 --kernel-trace --stats -d "$OUT/c3" -- python $REPO/bench.py --config 3 --steps 3 --warmup 1 --no-cpu-baseline > "$OUT/c3.json" 2> "$OUT/c3.log"
rocprofv3 --kernel-trace --stats -d "$OUT/c4" -- python $REPO/bench.py --config 4 --steps 3 --warmup 1 --no-cpu-baseline > "$OUT/c4.json" 2> "$OUT/c4.log"
rocprofv3 --kernel-trace --stats -d "$OUT/c5" -- python $REPO/bench.py --config 5 --steps 3 --warmup 1 --no-cpu-baseline > "$OUT/c5.json" 2> "$OUT/c5.log"
echo collected

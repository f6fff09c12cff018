#!/bin/bash
# rocprofv3 PMC (FETCH_SIZE / WRITE_SIZE) evidence for configs 3,4,5 — run ON the
# MI355X box. Counter passes stay separate from trace passes (gpurun requirement).
# Summaries are printed to gpurun_out/pmc_summary.txt; DBs are deleted on-box
# (they would blow the 64 MiB copy-back limit).
cd /tmp && export TMPDIR=/tmp
REPO=${GRAFT_REPO_ROOT:-/root/repo}
OUT=$REPO/gpurun_out/pmc
mkdir -p "$OUT"
for CFG in 3 4 5; do
  B="python $REPO/bench.py --config $CFG --steps 3 --warmup 1 --no-cpu-baseline"
  rocprofv3 --pmc FETCH_SIZE -d "$OUT/c${CFG}_fetch" -- $B > /dev/null 2> "$OUT/c${CFG}_fetch.log"
  rocprofv3 --pmc WRITE_SIZE -d "$OUT/c${CFG}_write" -- $B > /dev/null 2> "$OUT/c${CFG}_write.log"
done
python $REPO/tools/pmc_summarize.py "$OUT"/c*_fetch "$OUT"/c*_write > $REPO/gpurun_out/pmc_summary.txt 2>&1
rm -rf "$OUT"/c*_fetch "$OUT"/c*_write
cat $REPO/gpurun_out/pmc_summary.txt

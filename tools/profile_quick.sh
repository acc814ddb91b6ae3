#!/bin/bash
set -x
cd /tmp && export TMPDIR=/tmp
R="$GRAFT_REPO_ROOT"
mkdir -p "$R/gpurun_out/prof3"
rocprofv3 --kernel-trace --stats -d /tmp/prof -o kt -- \
  python "$R/bench.py" --steps 3 --warmup 1 --scale full --windows 1 \
  > "$R/gpurun_out/prof3/bench.log" 2>&1
python "$R/tools/kstats.py" /tmp/prof/kt_results.db > "$R/gpurun_out/prof3/kstats.txt" 2>&1

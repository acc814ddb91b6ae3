#!/bin/bash
# Profiling recipe: per-kernel stats for one bench step (run on the GPU box).
set -e
cd /tmp && export TMPDIR=/tmp
cd "$GRAFT_REPO_ROOT"
mkdir -p gpurun_out/prof
python tools/timeparts.py 2>&1 | tee gpurun_out/timeparts.txt
cd /tmp && rocprofv3 --kernel-trace --stats -d "$GRAFT_REPO_ROOT/gpurun_out/prof" -o bench_prof -- \
  python "$GRAFT_REPO_ROOT/bench.py" --steps 3 --warmup 1 --scale full --windows 1 \
  > "$GRAFT_REPO_ROOT/gpurun_out/bench_prof.log" 2>&1
ls -la "$GRAFT_REPO_ROOT/gpurun_out/prof" >> "$GRAFT_REPO_ROOT/gpurun_out/bench_prof.log"

#!/bin/bash
# Round profiling evidence: kernel-trace stats + PMC counters for the custom
# kernels (separate runs — never combine --pmc with trace domains).
set -x
cd /tmp && export TMPDIR=/tmp
R="$GRAFT_REPO_ROOT"
mkdir -p "$R/gpurun_out/prof2"
# 1) kernel trace + stats over the graph-captured bench
rocprofv3 --kernel-trace --stats -d "$R/gpurun_out/prof2" -o ktrace -- \
  python "$R/bench.py" --steps 3 --warmup 1 --scale full --windows 1 --graphs on \
  > "$R/gpurun_out/prof2/bench.log" 2>&1
# 2) PMC run: occupancy/issue counters for the custom kernels
rocprofv3 --pmc SQ_WAVES,SQ_WAVE_CYCLES,SQ_BUSY_CYCLES,SQ_WAIT_ANY -d "$R/gpurun_out/prof2" -o pmc1 -- \
  python "$R/bench.py" --steps 1 --warmup 1 --scale full --windows 1 --graphs off \
  > "$R/gpurun_out/prof2/pmc1.log" 2>&1
ls -la "$R/gpurun_out/prof2" | tail -8

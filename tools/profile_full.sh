#!/bin/bash
# Round profiling evidence: kernel-trace stats + PMC counters, summarized
# in-run (rocpd DBs exceed the gpurun merge cap and boxes are ephemeral).
set -x
cd /tmp && export TMPDIR=/tmp
R="$GRAFT_REPO_ROOT"
mkdir -p "$R/gpurun_out/prof2"
rocprofv3 --kernel-trace --stats -d /tmp/prof -o ktrace -- \
  python "$R/bench.py" --steps 3 --warmup 1 --scale full --windows 1 --graphs on \
  > "$R/gpurun_out/prof2/bench.log" 2>&1
rocprofv3 --pmc SQ_WAVES,SQ_WAVE_CYCLES,SQ_BUSY_CYCLES,SQ_WAIT_ANY -d /tmp/prof -o pmc1 -- \
  python "$R/bench.py" --steps 1 --warmup 1 --scale full --windows 1 --graphs off \
  > "$R/gpurun_out/prof2/pmc1.log" 2>&1
python "$R/tools/kstats.py" /tmp/prof/ktrace_results.db > "$R/gpurun_out/prof2/kstats.txt" 2>&1
python "$R/tools/pmcstats.py" /tmp/prof/pmc1_results.db > "$R/gpurun_out/prof2/pmc1.txt" 2>&1
tail -3 "$R/gpurun_out/prof2/bench.log"

#!/bin/bash
# rocprofv3 capture on the GPU box (SURVEY.md §5 tracing row).
# Usage: tools/profile.sh [outdir]  -- writes kernel stats CSVs under outdir.
# NOTE: PMC counters must be collected in a separate run from trace domains
# (pool rule); this script does the two runs back-to-back.
set -x
OUT=${1:-/root/repo/gpurun_out/prof}
mkdir -p "$OUT"
cd /tmp && export TMPDIR=/tmp

# 1) kernel trace + stats (csv)
timeout 300 rocprofv3 --kernel-trace --stats --output-format csv -d "$OUT/trace" -o bench \
  -- python /root/repo/bench.py --steps 3 --warmup 2 > "$OUT/trace.log" 2>&1

# 2) PMC counters for the hot kernel (own run, counters only)
timeout 300 rocprofv3 --pmc SQ_BUSY_CYCLES SQ_WAVE_CYCLES SQ_VALU_MFMA_BUSY_CYCLES \
  --output-format csv -d "$OUT/pmc" -o bench \
  -- python /root/repo/tools/kernel_bench.py --iters 20 > "$OUT/pmc.log" 2>&1

ls -laR "$OUT" >> "$OUT/trace.log"

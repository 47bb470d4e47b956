#!/bin/bash
# rocprofv3 capture on the GPU box (SURVEY.md §5 tracing row).
# Usage: tools/profile.sh [outdir]
# Two separate runs by pool rule: (1) kernel trace+stats, (2) PMC counters.
set -x
OUT=${1:-/root/repo/gpurun_out/prof}
mkdir -p "$OUT"
cd /tmp && export TMPDIR=/tmp

# 1) kernel trace + stats (csv) over the flagship bench
timeout 300 rocprofv3 --kernel-trace --stats --output-format csv -d "$OUT/trace" -o bench \
  -- python /root/repo/bench.py --steps 3 --warmup 2 > "$OUT/trace.log" 2>&1

# 2) PMC: MFMA utilisation + occupancy + LDS conflicts for the hot kernels
#    (counters-only run; no trace domains -- pool rule)
timeout 300 rocprofv3 --pmc MfmaUtil OccupancyPercent VALUBusy LDSBankConflict \
  --output-format csv -d "$OUT/pmc" -o bench \
  -- python /root/repo/bench.py --steps 2 --warmup 1 > "$OUT/pmc.log" 2>&1

ls -laR "$OUT" >> "$OUT/trace.log" 2>&1

#!/bin/bash
# Final-state PMC: HBM traffic of the shipped hot loop (with the
# round-2 last-stage k-store elision), FETCH and WRITE in separate
# runs (both at once exceed gfx950 counter capacity).
set -x
export PYTHONPATH=/root/repo
export TMPDIR=/tmp
mkdir -p /root/repo/gpurun_out
cd /tmp
LOG=/root/repo/gpurun_out/final_pmc.log
: > "$LOG"
for C in FETCH_SIZE WRITE_SIZE; do
  echo "=== pmc $C : bench --steps 2 --warmup 1 --grid 256 ===" >> "$LOG"
  timeout 420 rocprofv3 --pmc $C -d /tmp/pmc_$C -- \
    python /root/repo/bench.py --steps 2 --warmup 1 --grid 256 >> "$LOG" 2>&1
  echo "exit=$?" >> "$LOG"
  timeout 120 python /root/repo/tools/rocpd_stats.py /tmp/pmc_$C >> "$LOG" 2>&1
done
tail -80 "$LOG"

#!/bin/bash
# Single-GPU proxy for the N=8 strong-scaling regime: per-rank grids
# 256^3 and 128^3 (what each GPU holds when 512^3 is split 8/64 ways),
# sweeping the in-kernel periodic-read default.  Writes JSON lines to
# gpurun_out/r02_strong_proxy.log
set -u
export HSA_ENABLE_IPC_MODE_LEGACY=0
out=gpurun_out/r02_strong_proxy.log
: > "$out"
for grid in 256 128; do
  for peri in 0 1; do
    echo "== grid=$grid PERIODIC=$peri" >> "$out"
    PYSTELLA_PERIODIC=$peri python bench.py --steps 40 --warmup 10 \
        --grid $grid 2>/dev/null | tail -1 >> "$out"
    echo "== grid=$grid PERIODIC=$peri gws" >> "$out"
    PYSTELLA_PERIODIC=$peri python bench.py --steps 20 --warmup 5 \
        --grid $grid --gws 2>/dev/null | tail -1 >> "$out"
  done
done
grep -E "^==|value" "$out" | cut -c1-160

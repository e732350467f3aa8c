#!/bin/bash
# One-box A/Bs: (1) MG Newton smoother LDS-stencil vs elementwise,
# (2) GW ring-kernel tile sweep, (3) rocprof kernel trace of the
# observables path (fused projectors: one kernel per op).
set -u
export HSA_ENABLE_IPC_MODE_LEGACY=0
out=gpurun_out/r02_ab2.log
: > "$out"

echo "=== MG newton 1024^3 fp32, LDS stencil smoothers" >> "$out"
python tools/bench_mg.py --n 1024 --cycles 3 --smoother newton 2>/dev/null | tail -1 >> "$out"
echo "=== MG newton 1024^3 fp32, elementwise smoothers" >> "$out"
PYSTELLA_STENCIL_LDS=0 python tools/bench_mg.py --n 1024 --cycles 3 --smoother newton 2>/dev/null | tail -1 >> "$out"

for t in "" "64,4,32" "64,4,64" "32,8,64"; do
  echo "=== gws tile=${t:-default}" >> "$out"
  PYSTELLA_TILE=$t python bench.py --steps 10 --warmup 3 --gws 2>/dev/null | tail -1 | head -c 200 >> "$out"
  echo >> "$out"
done

cd /tmp && export TMPDIR=/tmp
rocprofv3 --kernel-trace --stats -- python /root/repo/tools/bench_observables.py 256 \
    > /root/repo/gpurun_out/r02_obs_trace.log 2>&1
cd /root/repo
grep -E "proj_|tt_project|spectra_bin|NAME|KERNEL" gpurun_out/r02_obs_trace.log | head -30 >> "$out" || true
cat "$out"

#!/bin/bash
# GW stage-kernel A/B matrix on one box: sectioned emission x sched
# mask x in-kernel periodic reads.  Writes one JSON line per config to
# gpurun_out/r02_gw_matrix.log
set -u
export HSA_ENABLE_IPC_MODE_LEGACY=0
out=gpurun_out/r02_gw_matrix.log
: > "$out"
run() {
    local tag="$1" sec="$2" mask="$3" peri="$4"
    echo "== $tag (SECTIONS=$sec MASK=$mask PERIODIC=$peri)" >> "$out"
    PYSTELLA_SECTIONS=$sec PYSTELLA_SECTION_MASK=$mask \
    PYSTELLA_PERIODIC=$peri \
        python bench.py --steps 10 --warmup 3 --gws 2>/dev/null \
        | tail -1 >> "$out"
}
run "flat"          0 0 0
run "sec1-closed"   1 0 0
run "sec1-open"     1 0xffffffff 0
run "sec1-vmemrd"   1 0x20 0
run "flat-periodic" 0 0 1
run "sec1-open-peri" 1 0xffffffff 1
run "sec2-open"     2 0xffffffff 0
grep -E "^==|value" "$out" | sed 's/\(.\{200\}\).*/\1/'

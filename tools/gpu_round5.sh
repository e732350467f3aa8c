set -x
cd "$GRAFT_REPO_ROOT"
mkdir -p gpurun_out
timeout 300 python tools/mfma_probe.py > gpurun_out/mfma_probe.log 2>&1
echo "probe exit=$?"
timeout 600 python bench.py --steps 10 --warmup 3 --gws > gpurun_out/bench_gws.log 2>&1
echo "gws exit=$?"
export TMPDIR=/tmp; cd /tmp
timeout 600 rocprofv3 --kernel-trace --stats -d "$GRAFT_REPO_ROOT/gpurun_out/prof5" -- python "$GRAFT_REPO_ROOT/bench.py" --steps 3 --warmup 1 > "$GRAFT_REPO_ROOT/gpurun_out/prof_bench5.log" 2>&1
echo "rocprof exit=$?"

set -x
cd "$GRAFT_REPO_ROOT"
mkdir -p gpurun_out
timeout 600 python -m pytest tests -m gpu -x -q > gpurun_out/pytest_gpu4.log 2>&1
echo "pytest exit=$?"
timeout 600 python bench.py --steps 10 --warmup 3 > gpurun_out/bench4.log 2>&1
echo "bench exit=$?"
timeout 600 python bench.py --steps 10 --warmup 3 --no-device-friedmann > gpurun_out/bench4b.log 2>&1
echo "bench-nodev exit=$?"
export TMPDIR=/tmp; cd /tmp
timeout 600 rocprofv3 --kernel-trace --stats -d "$GRAFT_REPO_ROOT/gpurun_out/prof4" -- python "$GRAFT_REPO_ROOT/bench.py" --steps 3 --warmup 1 > "$GRAFT_REPO_ROOT/gpurun_out/prof_bench4.log" 2>&1
echo "rocprof exit=$?"

set -x
cd "$GRAFT_REPO_ROOT"
mkdir -p gpurun_out
python __graft_entry__.py smoke > gpurun_out/smoke.log 2>&1
echo "smoke exit=$?"
timeout 900 python -m pytest tests -m gpu -x -q > gpurun_out/pytest_gpu.log 2>&1
echo "pytest exit=$?"
timeout 600 python bench.py --steps 10 --warmup 3 > gpurun_out/bench1.log 2>&1
echo "bench exit=$?"
export TMPDIR=/tmp; cd /tmp
timeout 600 rocprofv3 --kernel-trace --stats -d "$GRAFT_REPO_ROOT/gpurun_out/prof" -- python "$GRAFT_REPO_ROOT/bench.py" --steps 3 --warmup 1 > "$GRAFT_REPO_ROOT/gpurun_out/prof_bench.log" 2>&1
echo "rocprof exit=$?"
tail -3 "$GRAFT_REPO_ROOT"/gpurun_out/pytest_gpu.log "$GRAFT_REPO_ROOT"/gpurun_out/bench1.log

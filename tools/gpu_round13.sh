set -x
cd "$GRAFT_REPO_ROOT"
mkdir -p gpurun_out
timeout 600 python -m pytest tests -m gpu -x -q > gpurun_out/pytest_gpu13.log 2>&1
echo "pytest exit=$?"
timeout 300 python __graft_entry__.py smoke > gpurun_out/smoke13.log 2>&1
echo "smoke exit=$?"
timeout 300 python bench.py --steps 10 --warmup 3 > gpurun_out/bench13.log 2>&1
echo "b512 exit=$?"
export TMPDIR=/tmp; cd /tmp
timeout 420 rocprofv3 --kernel-trace --stats -d "$GRAFT_REPO_ROOT/gpurun_out/prof13" -- python "$GRAFT_REPO_ROOT/bench.py" --steps 3 --warmup 1 > "$GRAFT_REPO_ROOT/gpurun_out/prof13.log" 2>&1
echo "rocprof exit=$?"

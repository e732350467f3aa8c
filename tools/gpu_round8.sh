set -x
cd "$GRAFT_REPO_ROOT"
mkdir -p gpurun_out
timeout 300 python __graft_entry__.py smoke > gpurun_out/smoke8.log 2>&1
echo "smoke exit=$?"
timeout 600 python -m pytest tests -m gpu -x -q > gpurun_out/pytest_gpu8.log 2>&1
echo "pytest exit=$?"
timeout 900 python tools/tune.py 512 > gpurun_out/tune8.log 2>&1
echo "tune exit=$?"
timeout 600 python bench.py --steps 10 --warmup 3 --gws > gpurun_out/bench8_gws.log 2>&1
echo "gws exit=$?"
export TMPDIR=/tmp; cd /tmp
timeout 600 rocprofv3 --kernel-trace --stats -d "$GRAFT_REPO_ROOT/gpurun_out/prof8" -- python "$GRAFT_REPO_ROOT/bench.py" --steps 2 --warmup 1 --gws > "$GRAFT_REPO_ROOT/gpurun_out/prof_gws.log" 2>&1
echo "rocprof exit=$?"

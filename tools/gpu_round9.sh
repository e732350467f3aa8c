set -x
cd "$GRAFT_REPO_ROOT"
mkdir -p gpurun_out
timeout 600 python -m pytest tests -m gpu -x -q > gpurun_out/pytest_gpu9.log 2>&1
echo "pytest exit=$?"
timeout 300 python bench.py --steps 10 --warmup 3 > gpurun_out/bench9_512.log 2>&1
echo "b512 exit=$?"
timeout 300 python bench.py --steps 10 --warmup 3 --grid 256 > gpurun_out/bench9_256.log 2>&1
echo "b256 exit=$?"

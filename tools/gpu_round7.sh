set -x
cd "$GRAFT_REPO_ROOT"
mkdir -p gpurun_out
timeout 600 python -m pytest tests -m gpu -x -q > gpurun_out/pytest_gpu7.log 2>&1
echo "pytest exit=$?"
timeout 300 python bench.py --steps 10 --warmup 3 > gpurun_out/bench7.log 2>&1
echo "bench exit=$?"
timeout 600 python bench.py --steps 5 --warmup 2 --gws > gpurun_out/bench7_gws.log 2>&1
echo "gws exit=$?"

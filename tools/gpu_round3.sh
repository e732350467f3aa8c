set -x
cd "$GRAFT_REPO_ROOT"
mkdir -p gpurun_out
timeout 600 python -m pytest tests -m gpu -x -q > gpurun_out/pytest_gpu3.log 2>&1
echo "pytest exit=$?"
timeout 900 python tools/tune.py 512 > gpurun_out/tune3.log 2>&1
echo "tune exit=$?"
timeout 600 python bench.py --steps 10 --warmup 3 > gpurun_out/bench3.log 2>&1
echo "bench exit=$?"

set -x
cd "$GRAFT_REPO_ROOT"
mkdir -p gpurun_out
timeout 600 python -m pytest tests -m gpu -x -q > gpurun_out/pytest_gpu10.log 2>&1
echo "pytest exit=$?"
timeout 300 python bench.py --steps 10 --warmup 3 > gpurun_out/bench10.log 2>&1
echo "b512 exit=$?"
timeout 600 python bench.py --steps 10 --warmup 3 --gws > gpurun_out/bench10_gws.log 2>&1
echo "gws exit=$?"
timeout 300 python bench.py --steps 10 --warmup 3 --no-fuse > gpurun_out/bench10_nofuse.log 2>&1
echo "nofuse exit=$?"

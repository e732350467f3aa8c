set -x
cd "$GRAFT_REPO_ROOT"
mkdir -p gpurun_out
timeout 600 python -m pytest tests -m gpu -x -q > gpurun_out/pytest_gpu12.log 2>&1
echo "pytest exit=$?"
timeout 420 python tools/bench_observables.py > gpurun_out/bench_obs3.log 2>&1
echo "obs exit=$?"
timeout 900 python tools/bench_mg.py --n 1024 --depth 5 --cycles 3 > gpurun_out/bench_mg_rbgs.log 2>&1
echo "mgrbgs exit=$?"
timeout 300 python bench.py --steps 10 --warmup 3 > gpurun_out/bench12.log 2>&1
echo "b512 exit=$?"

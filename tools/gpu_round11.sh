set -x
cd "$GRAFT_REPO_ROOT"
mkdir -p gpurun_out
timeout 600 python -m pytest tests -m gpu -x -q > gpurun_out/pytest_gpu11.log 2>&1
echo "pytest exit=$?"
timeout 600 python tools/bench_observables.py > gpurun_out/bench_obs.log 2>&1
echo "obs exit=$?"
timeout 900 python tools/bench_mg.py --n 1024 --depth 5 --cycles 3 > gpurun_out/bench_mg_jac.log 2>&1
echo "mgjac exit=$?"

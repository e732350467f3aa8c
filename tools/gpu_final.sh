set -x
cd "$GRAFT_REPO_ROOT"
mkdir -p gpurun_out
timeout 900 python -m pytest tests -m gpu -q > gpurun_out/final_pytest.log 2>&1
echo "pytest exit=$?"
timeout 300 python __graft_entry__.py smoke > gpurun_out/final_smoke.log 2>&1
echo "smoke exit=$?"
timeout 300 python bench.py --steps 15 --warmup 4 > gpurun_out/final_bench_a.log 2>&1
echo "ba exit=$?"
timeout 300 python bench.py --steps 15 --warmup 4 > gpurun_out/final_bench_b.log 2>&1
echo "bb exit=$?"
timeout 600 python bench.py --steps 8 --warmup 2 --gws > gpurun_out/final_gws.log 2>&1
echo "gws exit=$?"
timeout 420 python tools/bench_observables.py > gpurun_out/final_obs.log 2>&1
echo "obs exit=$?"
export TMPDIR=/tmp; cd /tmp
timeout 420 rocprofv3 --kernel-trace --stats -d "$GRAFT_REPO_ROOT/gpurun_out/final_prof" -- python "$GRAFT_REPO_ROOT/bench.py" --steps 3 --warmup 1 > "$GRAFT_REPO_ROOT/gpurun_out/final_prof.log" 2>&1
echo "prof exit=$?"

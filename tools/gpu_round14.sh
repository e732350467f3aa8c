set -x
cd "$GRAFT_REPO_ROOT"
mkdir -p gpurun_out
timeout 900 python tools/tune.py 512 > gpurun_out/tune_final.log 2>&1
echo "tune exit=$?"
timeout 420 python bench.py --steps 6 --warmup 2 --grid 768 > gpurun_out/bench_768.log 2>&1
echo "b768 exit=$?"
timeout 900 python bench.py --steps 30 --warmup 3 --gws > gpurun_out/soak_gws.log 2>&1
echo "gwsoak exit=$?"

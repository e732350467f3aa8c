set -x
cd "$GRAFT_REPO_ROOT"
mkdir -p gpurun_out
timeout 300 python bench.py --steps 10 --warmup 3 > gpurun_out/ab_512_a.log 2>&1
timeout 300 python bench.py --steps 10 --warmup 3 > gpurun_out/ab_512_b.log 2>&1
timeout 600 python bench.py --steps 8 --warmup 2 --gws > gpurun_out/ab_gws_split.log 2>&1
timeout 600 python bench.py --steps 8 --warmup 2 --gws --gws-no-split > gpurun_out/ab_gws_nosplit.log 2>&1
echo done

set -x
cd "$GRAFT_REPO_ROOT"
mkdir -p gpurun_out
timeout 900 python -m pytest tests -m gpu -x -q > gpurun_out/pytest_gpu.log 2>&1
echo "pytest exit=$?"
timeout 600 python tools/tune.py 512 > gpurun_out/tune2.log 2>&1
echo "tune exit=$?"
timeout 600 python bench.py --steps 10 --warmup 3 > gpurun_out/bench2.log 2>&1
echo "bench exit=$?"
export TMPDIR=/tmp; cd /tmp
timeout 600 rocprofv3 --kernel-trace --stats -d "$GRAFT_REPO_ROOT/gpurun_out/prof2" -- python "$GRAFT_REPO_ROOT/bench.py" --steps 3 --warmup 1 > "$GRAFT_REPO_ROOT/gpurun_out/prof_bench2.log" 2>&1
echo "rocprof exit=$?"
timeout 300 rocprofv3 --list-avail > "$GRAFT_REPO_ROOT/gpurun_out/avail.txt" 2>&1
echo "avail exit=$?"
timeout 600 rocprofv3 --pmc FETCH_SIZE WRITE_SIZE -d "$GRAFT_REPO_ROOT/gpurun_out/pmc1" -- python "$GRAFT_REPO_ROOT/bench.py" --steps 2 --warmup 1 --grid 256 > "$GRAFT_REPO_ROOT/gpurun_out/pmc_bench.log" 2>&1
echo "pmc exit=$?"

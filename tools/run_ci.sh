#!/bin/bash
# CI-style test matrix, mirroring the reference's .github/workflows/ci.yml
# (single-rank pytest + distributed re-runs; reference runs the same
# suite under mpirun -np 4 --proc_shape 2,2,1 / -np 3 --proc_shape 3,1,1;
# here the multi-process coverage is in-suite via gloo spawn tests).
set -e
cd "$(dirname "$0")/.."

echo "== build (hipcc, gfx950) =="
python -m pystella_amd.backend.build

echo "== CPU suite (includes gloo world-2 distributed tests) =="
python -m pytest tests/ -q -m "not gpu"

echo "== torchrun 2-rank bench smoke (gloo) =="
python -m torch.distributed.run --standalone --local-addr 127.0.0.1 \
    --nnodes=1 --nproc-per-node 2 bench.py \
    --gpus 2 --steps 2 --warmup 1 --grid 16 --device cpu

if python -c "import torch; exit(0 if torch.cuda.is_available() else 1)"; then
    echo "== GPU suite (MI355X) =="
    python -m pytest tests/ -q -m gpu
    echo "== bench =="
    python bench.py --steps 10 --warmup 3
fi

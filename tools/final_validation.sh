#!/bin/bash
# Final round-2 validation: the driver's exact round-end sequence.
set -x
export PYTHONPATH=/root/repo
mkdir -p gpurun_out
cd /root/repo
echo "=== pytest -m gpu -x -q ===" > gpurun_out/final_validation.log
timeout 600 python -m pytest tests/ -m gpu -x -q >> gpurun_out/final_validation.log 2>&1
echo "pytest exit: $?" >> gpurun_out/final_validation.log
echo "=== smoke() ===" >> gpurun_out/final_validation.log
timeout 300 python -c "import __graft_entry__ as g; g.build(); g.smoke(); print('SMOKE OK')" >> gpurun_out/final_validation.log 2>&1
echo "smoke exit: $?" >> gpurun_out/final_validation.log
echo "=== bench.py default x3 ===" >> gpurun_out/final_validation.log
for i in 1 2 3; do
  timeout 300 python bench.py >> gpurun_out/final_validation.log 2>&1
  echo "bench run $i exit: $?" >> gpurun_out/final_validation.log
done
tail -30 gpurun_out/final_validation.log

#!/bin/bash
# batch 32: final confirm at final HEAD — full GPU suite + smoke +
# driver-shaped headline.
set -x
cd /root/repo
export PYTHONUNBUFFERED=1
timeout 900 python -m pytest tests -m gpu -q > gpurun_out/r032_pytest.log 2>&1
echo "pytest rc=$?"; tail -1 gpurun_out/r032_pytest.log
timeout 300 python -c 'import __graft_entry__ as g; g.smoke(); print("smoke OK")' \
  > gpurun_out/r032_smoke.log 2>&1
echo "smoke rc=$?"; tail -1 gpurun_out/r032_smoke.log
timeout 900 python bench.py --gpus 1 --steps 20 --warmup 5 --no-extras \
  > gpurun_out/r032_bench.json 2> gpurun_out/r032_bench.log
echo "bench rc=$?"
grep -o '"value": [0-9.e+]*\|"ms_per_step": [0-9.]*\|"frac": [0-9.]*' \
  gpurun_out/r032_bench.json | head -4

#!/bin/bash
# Final round-2 rehearsal at final HEAD: full GPU suite + smoke + the exact
# driver bench command (all default legs: cpu baselines + extras).
set -x
cd /root/repo
export PYTHONUNBUFFERED=1
timeout 900 python -m pytest tests -m gpu -q > gpurun_out/r02final_pytest.log 2>&1
echo "pytest rc=$?"; tail -2 gpurun_out/r02final_pytest.log
timeout 300 python -c 'import __graft_entry__ as g; g.smoke(); print("smoke OK")' \
  > gpurun_out/r02final_smoke.log 2>&1
echo "smoke rc=$?"; tail -1 gpurun_out/r02final_smoke.log
timeout 1500 python bench.py --gpus 1 --steps 20 --warmup 5 \
  > gpurun_out/r02final_bench.json 2> gpurun_out/r02final_bench.log
echo "bench rc=$?"
grep -o '"value": [0-9.e+]*\|"ms_per_step": [0-9.]*\|"frac": [0-9.]*' \
  gpurun_out/r02final_bench.json | head -4
grep '^\[bench-extra\]' gpurun_out/r02final_bench.log | head -6

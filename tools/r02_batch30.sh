#!/bin/bash
# batch 30: validate the bucket-mode two-phase compaction (pytest gpu,
# soak with bucket cases, dist nccl config) at the extended build.
set -x
cd /root/repo
export PYTHONUNBUFFERED=1
timeout 900 python -m pytest tests -m gpu -q > gpurun_out/r030_pytest.log 2>&1
echo "pytest rc=$?"; tail -1 gpurun_out/r030_pytest.log
timeout 400 python tools/soak_parity.py 300 47 > gpurun_out/r030_soak.log 2>&1
echo "soak rc=$?"; tail -1 gpurun_out/r030_soak.log
# bucketed headline-shaped timing (direct-indexed bucket path)
timeout 900 python bench.py --steps 2 --warmup 1 --no-cpu-baseline --no-extras \
  > gpurun_out/r030_gen.log 2>&1
timeout 600 python bench.py --steps 10 --warmup 3 --bucket-ms 60000 \
  --no-cpu-baseline --no-extras > gpurun_out/r030_bucket.json \
  2> gpurun_out/r030_bucket.log
echo "bucket bench rc=$?"
grep -o '"ms_per_step": [0-9.]*' gpurun_out/r030_bucket.json | head -1

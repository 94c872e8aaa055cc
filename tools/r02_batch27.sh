#!/bin/bash
# batch 27: two-phase compaction — parity + headline A/B vs the legacy
# single-pass kernel (HX_COMPACT_LEGACY=1), plus a short soak.
set -x
cd /root/repo
export PYTHONUNBUFFERED=1
timeout 900 python -m pytest tests -m gpu -q > gpurun_out/r02y_pytest.log 2>&1
echo "pytest rc=$?"; tail -1 gpurun_out/r02y_pytest.log
timeout 900 python bench.py --steps 2 --warmup 1 --no-cpu-baseline --no-extras \
  > gpurun_out/r02y_gen.log 2>&1
echo "== gen rc=$?"
B="--steps 20 --warmup 5 --no-cpu-baseline --no-extras"
for v in new legacy new; do
  tag=$v; [ -e gpurun_out/r02y_${v}.json ] && tag=${v}b
  env=""; [ $v = legacy ] && env="HX_COMPACT_LEGACY=1"
  timeout 600 env $env HX_DEBUG=1 python bench.py $B \
    > gpurun_out/r02y_${tag}.json 2> gpurun_out/r02y_${tag}.log
  echo "== compact=$v ($tag)"
  grep -o '"ms_per_step": [0-9.]*' gpurun_out/r02y_${tag}.json | head -1
done
timeout 400 python tools/soak_parity.py 300 31 > gpurun_out/r02y_soak.log 2>&1
echo "soak rc=$?"; tail -1 gpurun_out/r02y_soak.log

#!/bin/bash
# batch 26: decoder occupancy ladder extension — MINW 6/7/8 headline A/B
set -x
cd /root/repo
export PYTHONUNBUFFERED=1
B="--steps 20 --warmup 5 --no-cpu-baseline --no-extras"
timeout 900 python bench.py --steps 2 --warmup 1 --no-cpu-baseline --no-extras \
  > gpurun_out/r02x_gen.log 2>&1
echo "== gen rc=$?"
for mw in 6 7 8 6; do
  tag=mw${mw}; [ -e gpurun_out/r02x_${tag}.json ] && tag=mw${mw}b
  timeout 600 env HX_SNAPPY_MINW=$mw python bench.py $B \
    > gpurun_out/r02x_${tag}.json 2> gpurun_out/r02x_${tag}.log
  echo "== MINW=$mw ($tag)"
  grep -o '"ms_per_step": [0-9.]*' gpurun_out/r02x_${tag}.json | head -1
done
# parity at the extended instantiations
timeout 600 env HX_SNAPPY_MINW=8 python -m pytest tests/test_gpu_parity.py -q \
  -k 'snappy or codec' > gpurun_out/r02x_pytest8.log 2>&1
echo "pytest mw8 rc=$?"; tail -1 gpurun_out/r02x_pytest8.log

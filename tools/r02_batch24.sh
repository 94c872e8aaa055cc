#!/bin/bash
# batch 24: headline-shaped same-box A/B of the snappy decoder occupancy
# (HX_SNAPPY_MINW 1/5/6) at the default pipeline depth; dataset generated
# once and reused from /tmp/hx_bench_data.
set -x
cd /root/repo
export PYTHONUNBUFFERED=1
B="--steps 20 --warmup 5 --no-cpu-baseline --no-extras"
# first run generates the dataset (counts against none of the variants)
timeout 900 python bench.py --steps 2 --warmup 1 --no-cpu-baseline --no-extras \
  > gpurun_out/r02w_gen.log 2>&1
echo "== gen rc=$?"
for mw in 5 6 1 5; do
  tag=mw${mw}; [ -e gpurun_out/r02w_${tag}.json ] && tag=mw${mw}b
  timeout 600 env HX_SNAPPY_MINW=$mw python bench.py $B \
    > gpurun_out/r02w_${tag}.json 2> gpurun_out/r02w_${tag}.log
  echo "== MINW=$mw ($tag)"
  grep -o '"ms_per_step": [0-9.]*' gpurun_out/r02w_${tag}.json | head -1
done

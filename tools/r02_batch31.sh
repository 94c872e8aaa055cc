#!/bin/bash
# batch 31: uncompressed at pipeline 3 (overlap the result path), bucket
# compact new-vs-legacy A/B, long parity soak.
set -x
cd /root/repo
export PYTHONUNBUFFERED=1
timeout 900 python bench.py --compression none --steps 2 --warmup 1 \
  --no-cpu-baseline --no-extras > gpurun_out/r031_gen.log 2>&1
echo "== gen rc=$?"
for p in 1 3; do
  timeout 600 python bench.py --compression none --pipeline $p --steps 20 \
    --warmup 5 --no-cpu-baseline --no-extras \
    > gpurun_out/r031_unc_p${p}.json 2> gpurun_out/r031_unc_p${p}.log
  echo "== uncompressed pipeline=$p"
  grep -o '"ms_per_step": [0-9.]*\|"value": [0-9.e+]*' \
    gpurun_out/r031_unc_p${p}.json | head -2
done
timeout 900 python bench.py --steps 2 --warmup 1 --no-cpu-baseline \
  --no-extras > gpurun_out/r031_gen2.log 2>&1
for v in new legacy; do
  env=""; [ $v = legacy ] && env="HX_COMPACT_LEGACY=1"
  timeout 600 env $env python bench.py --steps 10 --warmup 3 \
    --bucket-ms 60000 --no-cpu-baseline --no-extras \
    > gpurun_out/r031_bucket_${v}.json 2> gpurun_out/r031_bucket_${v}.log
  echo "== bucket compact=$v"
  grep -o '"ms_per_step": [0-9.]*' gpurun_out/r031_bucket_${v}.json | head -1
done
timeout 500 python tools/soak_parity.py 400 59 > gpurun_out/r031_soak.log 2>&1
echo "soak rc=$?"; tail -1 gpurun_out/r031_soak.log

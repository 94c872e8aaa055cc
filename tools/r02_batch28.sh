#!/bin/bash
# batch 28: result-D2H engine A/B (blit vs SDMA) + post-compact-fix ktrace
set -x
cd /root/repo
export PYTHONUNBUFFERED=1
timeout 900 python bench.py --steps 2 --warmup 1 --no-cpu-baseline --no-extras \
  > gpurun_out/r02z_gen.log 2>&1
echo "== gen rc=$?"
B="--steps 20 --warmup 5 --no-cpu-baseline --no-extras"
for v in def sdma1 sdma0 def; do
  tag=$v; [ -e gpurun_out/r02z_${v}.json ] && tag=${v}b
  env=""
  [ $v = sdma1 ] && env="HSA_ENABLE_SDMA=1"
  [ $v = sdma0 ] && env="HSA_ENABLE_SDMA=0"
  timeout 600 env $env python bench.py $B \
    > gpurun_out/r02z_${tag}.json 2> gpurun_out/r02z_${tag}.log
  echo "== $v ($tag)"
  grep -o '"ms_per_step": [0-9.]*' gpurun_out/r02z_${tag}.json | head -1
done
export TMPDIR=/tmp; cd /tmp
timeout 700 rocprofv3 --kernel-trace --stats -d /root/repo/gpurun_out/ktrace28 \
  -o ktrace28 -- python /root/repo/bench.py --steps 2 --warmup 1 \
  --no-cpu-baseline --no-extras > /root/repo/gpurun_out/r02z_kt.log 2>&1
echo "ktrace rc=$?"
find /root/repo/gpurun_out -size +20M -delete

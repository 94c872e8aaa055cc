#!/bin/bash
# r02 GPU batch 4: occupancy sweep (the kernel is latency-bound: ne sets
# LDS/block and thus waves/CU) + full parity suite for the pair kernel.
set -x
cd "${GRAFT_REPO_ROOT:-/root/repo}"
export PYTHONUNBUFFERED=1
mkdir -p gpurun_out

timeout 900 python -m pytest tests -m gpu -q > gpurun_out/r02d_pytest.log 2>&1
echo "pytest rc=$?" | tee -a gpurun_out/r02d_pytest.log
tail -3 gpurun_out/r02d_pytest.log

B="--steps 4 --warmup 1 --pipeline 1 --no-cpu-baseline --no-extras --compression none"
run() {
    local name=$1 envs=$2
    timeout 900 env HX_DEBUG=1 $envs python bench.py $B \
        > "gpurun_out/r02d_${name}.json" 2> "gpurun_out/r02d_${name}.log"
    echo "rc=$? name=${name}"
    grep -m1 "\[hx\] exec" "gpurun_out/r02d_${name}.log"
}
run base_ne2048_nb16384   "HX_RANGE_NE=2048 HX_RANGE_TARGET=800"
run ne1024_nb16384        "HX_RANGE_NE=1024 HX_RANGE_TARGET=800"
run ne1024_nb32768        "HX_RANGE_NE=1024 HX_RANGE_TARGET=400"
run ne512_nb32768         "HX_RANGE_NE=512  HX_RANGE_TARGET=400"
run ne1024_nb8192         "HX_RANGE_NE=1024 HX_RANGE_TARGET=1600"
run ne1024_nb16384_noint  "HX_RANGE_NE=1024 HX_RANGE_TARGET=800 HX_INTERP=0"
run ne1024_nb16384_skip1  "HX_RANGE_NE=1024 HX_RANGE_TARGET=800 HX_SKIP=1"
ls gpurun_out/r02d_* | tail

#!/bin/bash
set -x
cd "${GRAFT_REPO_ROOT:-/root/repo}"
export PYTHONUNBUFFERED=1
mkdir -p gpurun_out
timeout 900 python -m pytest tests -m gpu -q > gpurun_out/r02h_pytest.log 2>&1
echo "pytest rc=$?" | tee -a gpurun_out/r02h_pytest.log
tail -3 gpurun_out/r02h_pytest.log
B="--steps 4 --warmup 1 --pipeline 1 --no-cpu-baseline --no-extras"
run() {
    local name=$1 envs=$2 bargs=$3
    timeout 900 env HX_DEBUG=1 $envs python bench.py $B $bargs \
        > "gpurun_out/r02h_${name}.json" 2> "gpurun_out/r02h_${name}.log"
    echo "rc=$? name=${name}"
    grep -m1 "\[hx\] exec" "gpurun_out/r02h_${name}.log" | grep -o "kernel=[a-z]*\|kernel_ms=[0-9.]*"
    grep -o '"ms_per_step": [0-9.]*\|"value": [0-9.e+]*' "gpurun_out/r02h_${name}.json" | head -2
}
run sn        ""   ""
run sn_pipe3  ""   "--pipeline 3"
export TMPDIR=/tmp
cd /tmp
OUT="${GRAFT_REPO_ROOT:-/root/repo}/gpurun_out"
RB="python ${GRAFT_REPO_ROOT:-/root/repo}/bench.py --steps 2 --warmup 1 --pipeline 1 --no-cpu-baseline --no-extras"
timeout 900 rocprofv3 --kernel-trace --stats -d "$OUT/ktrace8" -o ktrace8 \
    -- $RB > "$OUT/r02h_ktrace.log" 2>&1 || true
find "$OUT" -size +20M -delete 2>/dev/null

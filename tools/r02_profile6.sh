#!/bin/bash
# r02 GPU batch 6: snappy-mirror kernel validation + kernel default A/B +
# traffic PMC for the headline record.
set -x
cd "${GRAFT_REPO_ROOT:-/root/repo}"
export PYTHONUNBUFFERED=1
mkdir -p gpurun_out

timeout 900 python -m pytest tests -m gpu -q > gpurun_out/r02f_pytest.log 2>&1
echo "pytest rc=$?" | tee -a gpurun_out/r02f_pytest.log
tail -3 gpurun_out/r02f_pytest.log

B="--steps 4 --warmup 1 --pipeline 1 --no-cpu-baseline --no-extras"
run() {
    local name=$1 envs=$2 bargs=$3
    timeout 900 env HX_DEBUG=1 $envs python bench.py $B $bargs \
        > "gpurun_out/r02f_${name}.json" 2> "gpurun_out/r02f_${name}.log"
    echo "rc=$? name=${name}"
    grep -m1 "\[hx\] exec" "gpurun_out/r02f_${name}.log"
    grep -o '"ms_per_step": [0-9.]*' "gpurun_out/r02f_${name}.json" | head -1
}
run sn_r2   ""            ""
run sn_r1   "HX_RANGE2=0" ""
run un_r2   ""            "--compression none"
run un_r1   "HX_RANGE2=0" "--compression none"
run un_r1_pipe3 "HX_RANGE2=0" "--compression none --pipeline 3"
run sn_r1_pipe3 "HX_RANGE2=0" "--pipeline 3"

# stats + traffic on the snappy headline (decode + agg kernels)
export TMPDIR=/tmp
cd /tmp
OUT="${GRAFT_REPO_ROOT:-/root/repo}/gpurun_out"
RB="python ${GRAFT_REPO_ROOT:-/root/repo}/bench.py --steps 2 --warmup 1 --pipeline 1 --no-cpu-baseline --no-extras"
timeout 900 rocprofv3 --kernel-trace --stats -d "$OUT/ktrace6" -o ktrace6 \
    -- env HX_RANGE2=0 $RB > "$OUT/r02f_ktrace.log" 2>&1 || true
timeout 900 rocprofv3 --pmc FETCH_SIZE -d "$OUT/pmc_fetch6" -o pmc_fetch6 \
    -- env HX_RANGE2=0 $RB > "$OUT/r02f_pmc_fetch.log" 2>&1 || true
timeout 900 rocprofv3 --pmc WRITE_SIZE -d "$OUT/pmc_write6" -o pmc_write6 \
    -- env HX_RANGE2=0 $RB > "$OUT/r02f_pmc_write.log" 2>&1 || true
find "$OUT" -size +20M -delete 2>/dev/null

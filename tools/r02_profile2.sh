#!/bin/bash
# r02 GPU batch 2: validate the partition fix at the 1B shape, A/B the LDS
# variants on the healthy kernel, PMC traffic for the roofline, and one
# full driver-like run (snappy headline + extras + native cpu baseline).
set -x
cd "${GRAFT_REPO_ROOT:-/root/repo}"
export PYTHONUNBUFFERED=1
mkdir -p gpurun_out

timeout 600 python -m pytest tests -m gpu -x -q \
    > gpurun_out/r02b_pytest.log 2>&1
echo "pytest rc=$?" | tee -a gpurun_out/r02b_pytest.log

B="--steps 4 --warmup 1 --pipeline 1 --no-cpu-baseline --no-extras"
run() {
    local name=$1 envs=$2 bargs=$3
    timeout 900 env HX_DEBUG=1 $envs python bench.py $B $bargs \
        > "gpurun_out/r02b_${name}.json" 2> "gpurun_out/r02b_${name}.log"
    echo "rc=$? name=${name}"
    grep -m1 "\[hx\] exec" "gpurun_out/r02b_${name}.log"
}

run fixed    ""                   "--compression none"
run interp0  "HX_INTERP=0"        "--compression none"
run ne4096   "HX_RANGE_NE=4096"   "--compression none"
run tgt300   "HX_RANGE_TARGET=300" "--compression none"
run nopoll   "HX_NO_POLL=1"       "--compression none"
run skip1    "HX_SKIP=1"          "--compression none"
run snappy   ""                   ""

# PMC traffic for the fixed kernel (uncompressed), separate passes
export TMPDIR=/tmp
cd /tmp
OUT="${GRAFT_REPO_ROOT:-/root/repo}/gpurun_out"
RB="python ${GRAFT_REPO_ROOT:-/root/repo}/bench.py --steps 2 --warmup 1 --pipeline 1 --no-cpu-baseline --no-extras"
timeout 900 rocprofv3 --pmc FETCH_SIZE -d "$OUT/pmc_fetch" -o pmc_fetch \
    -- $RB --compression none > "$OUT/r02b_pmc_fetch.log" 2>&1 || true
timeout 900 rocprofv3 --pmc WRITE_SIZE -d "$OUT/pmc_write" -o pmc_write \
    -- $RB --compression none > "$OUT/r02b_pmc_write.log" 2>&1 || true
timeout 900 rocprofv3 --pmc SQ_LDS_IDX_ACTIVE SQ_LDS_BANK_CONFLICT SQ_WAIT_ANY SQ_WAVE_CYCLES \
    -d "$OUT/pmc_lds2" -o pmc_lds2 -- $RB --compression none \
    > "$OUT/r02b_pmc_lds2.log" 2>&1 || true
timeout 900 rocprofv3 --kernel-trace --stats -d "$OUT/ktrace2" -o ktrace2 \
    -- $RB > "$OUT/r02b_ktrace2.log" 2>&1 || true

# full driver-like run last (snappy headline, pipeline 3, extras, baselines)
cd "${GRAFT_REPO_ROOT:-/root/repo}"
timeout 1200 python bench.py --steps 10 --warmup 3 \
    > gpurun_out/r02b_full.json 2> gpurun_out/r02b_full.log
echo "full rc=$?"
tail -3 gpurun_out/r02b_full.log
find gpurun_out -size +20M -delete 2>/dev/null
ls -la gpurun_out | head -40

#!/bin/bash
# r02 GPU batch 3: pair-load kernel (HX_RANGE2 default) — parity + A/B +
# bisects + PMC on the new kernel. Also runs the new dist GPU tests.
set -x
cd "${GRAFT_REPO_ROOT:-/root/repo}"
export PYTHONUNBUFFERED=1
mkdir -p gpurun_out

timeout 900 python -m pytest tests -m gpu -x -q \
    > gpurun_out/r02c_pytest.log 2>&1
echo "pytest rc=$?" | tee -a gpurun_out/r02c_pytest.log
tail -3 gpurun_out/r02c_pytest.log

B="--steps 4 --warmup 1 --pipeline 1 --no-cpu-baseline --no-extras"
run() {
    local name=$1 envs=$2 bargs=$3
    timeout 900 env HX_DEBUG=1 $envs python bench.py $B $bargs \
        > "gpurun_out/r02c_${name}.json" 2> "gpurun_out/r02c_${name}.log"
    echo "rc=$? name=${name}"
    grep -m1 "\[hx\] exec" "gpurun_out/r02c_${name}.log"
}

run r2          ""                        "--compression none"
run r2_nopoll   "HX_NO_POLL=1"            "--compression none"
run r1          "HX_RANGE2=0"             "--compression none"
run r2_skip1    "HX_SKIP=1"               "--compression none"
run r1_skip1    "HX_RANGE2=0 HX_SKIP=1"   "--compression none"
run r2_interp0  "HX_INTERP=0"             "--compression none"
run r2_nb8192   "HX_RANGE_TARGET=2800"    "--compression none"
run r2_nb16384  "HX_RANGE_TARGET=1400"    "--compression none"
run r2_ne4096   "HX_RANGE_NE=4096"        "--compression none"
run r2_snappy   ""                        ""

# PMC on the new kernel
export TMPDIR=/tmp
cd /tmp
OUT="${GRAFT_REPO_ROOT:-/root/repo}/gpurun_out"
RB="python ${GRAFT_REPO_ROOT:-/root/repo}/bench.py --steps 2 --warmup 1 --pipeline 1 --no-cpu-baseline --no-extras"
timeout 900 rocprofv3 --pmc SQ_LDS_IDX_ACTIVE SQ_LDS_BANK_CONFLICT SQ_WAIT_ANY SQ_WAVE_CYCLES \
    -d "$OUT/pmc_lds3" -o pmc_lds3 -- $RB --compression none \
    > "$OUT/r02c_pmc_lds3.log" 2>&1 || true
timeout 900 rocprofv3 --pmc FETCH_SIZE -d "$OUT/pmc_fetch3" -o pmc_fetch3 \
    -- $RB --compression none > "$OUT/r02c_pmc_fetch3.log" 2>&1 || true
timeout 900 rocprofv3 --pmc WRITE_SIZE -d "$OUT/pmc_write3" -o pmc_write3 \
    -- $RB --compression none > "$OUT/r02c_pmc_write3.log" 2>&1 || true

find "$OUT" -size +20M -delete 2>/dev/null
ls gpurun_out | tail -5

#!/bin/bash
# r02 GPU batch 1: pin the range-kernel floor (poll vs LDS-fallback vs
# LDS-array) with A/B solo-kernel timings + SQ PMC counters.
# Run via gpurun from the repo root; outputs land in gpurun_out/.
set -x
cd "${GRAFT_REPO_ROOT:-/root/repo}"
export PYTHONUNBUFFERED=1
mkdir -p gpurun_out

# 1. parity sanity (new interp default must stay parity-green)
timeout 600 python -m pytest tests -m gpu -x -q \
    > gpurun_out/r02a_pytest.log 2>&1
echo "pytest rc=$?" >> gpurun_out/r02a_pytest.log

B="--steps 3 --warmup 1 --pipeline 1 --no-cpu-baseline"
run() {
    local name=$1; shift
    timeout 900 env HX_DEBUG=1 "$@" python bench.py $B \
        > "gpurun_out/r02a_${name}.json" 2> "gpurun_out/r02a_${name}.log"
    echo "rc=$? name=${name}"
    tail -2 "gpurun_out/r02a_${name}.log"
}

run base            env
run interp0         HX_INTERP=0
run nopoll          HX_NO_POLL=1
run nopoll_interp0  HX_NO_POLL=1 HX_INTERP=0
run skip1           HX_SKIP=1
run skip4           HX_SKIP=4
run ne4096          HX_RANGE_NE=4096
run tgt2800         HX_RANGE_TARGET=2800
run wave            HX_RANGE=0

# 2. PMC counters on the default configuration (separate passes; never with
#    trace domains). Run from /tmp per the rocprofv3 recipe.
export TMPDIR=/tmp
cd /tmp
RB="python ${GRAFT_REPO_ROOT:-/root/repo}/bench.py --steps 2 --warmup 1 --pipeline 1 --no-cpu-baseline"
OUT="${GRAFT_REPO_ROOT:-/root/repo}/gpurun_out"
rocprofv3 -L > "$OUT/r02a_counters_avail.txt" 2>&1 || true
timeout 900 rocprofv3 --pmc SQ_LDS_IDX_ACTIVE SQ_LDS_BANK_CONFLICT SQ_WAIT_ANY SQ_WAVE_CYCLES \
    -d "$OUT/pmc_lds" -o pmc_lds -- $RB > "$OUT/r02a_pmc_lds.log" 2>&1 || true
timeout 900 rocprofv3 --pmc SQ_ACTIVE_INST_ANY SQ_WAIT_INST_ANY SQ_INSTS_LDS SQ_BUSY_CYCLES \
    -d "$OUT/pmc_inst" -o pmc_inst -- $RB > "$OUT/r02a_pmc_inst.log" 2>&1 || true
# kernel-trace stats pass for solo dispatch times under the profiler
timeout 900 rocprofv3 --kernel-trace --stats -d "$OUT/ktrace" -o ktrace \
    -- $RB > "$OUT/r02a_ktrace.log" 2>&1 || true
# keep outputs small: drop anything huge
find "$OUT" -size +20M -name '*.db' -delete 2>/dev/null
ls -la "$OUT"

#!/bin/bash
# r02 GPU batch 5: full suite (index + dist + parity) with new defaults,
# A/B prefetch kernel, end-to-end driver-like run, PMC for the record.
set -x
cd "${GRAFT_REPO_ROOT:-/root/repo}"
export PYTHONUNBUFFERED=1
mkdir -p gpurun_out

timeout 900 python -m pytest tests -m gpu -q > gpurun_out/r02e_pytest.log 2>&1
echo "pytest rc=$?" | tee -a gpurun_out/r02e_pytest.log
tail -4 gpurun_out/r02e_pytest.log

B="--steps 4 --warmup 1 --pipeline 1 --no-cpu-baseline --no-extras"
run() {
    local name=$1 envs=$2 bargs=$3
    timeout 900 env HX_DEBUG=1 $envs python bench.py $B $bargs \
        > "gpurun_out/r02e_${name}.json" 2> "gpurun_out/r02e_${name}.log"
    echo "rc=$? name=${name}"
    grep -m1 "\[hx\] exec" "gpurun_out/r02e_${name}.log"
}
run def          ""            "--compression none"
run def_oldk     "HX_RANGE2=0" "--compression none"
run def_skip1    "HX_SKIP=1"   "--compression none"
run def_pipe3    ""            "--compression none --pipeline 3"
run snappy_def   ""            ""

# full driver-like run with extras + cpu baselines (native leg fix check)
timeout 1200 python bench.py --steps 8 --warmup 2 \
    > gpurun_out/r02e_full.json 2> gpurun_out/r02e_full.log
echo "full rc=$?"
grep -E "bench-extra|native|index query" gpurun_out/r02e_full.log | tail -8

# PMC for the roofline record (new default kernel)
export TMPDIR=/tmp
cd /tmp
OUT="${GRAFT_REPO_ROOT:-/root/repo}/gpurun_out"
RB="python ${GRAFT_REPO_ROOT:-/root/repo}/bench.py --steps 2 --warmup 1 --pipeline 1 --no-cpu-baseline --no-extras"
timeout 900 rocprofv3 --pmc FETCH_SIZE -d "$OUT/pmc_fetch5" -o pmc_fetch5 \
    -- $RB --compression none > "$OUT/r02e_pmc_fetch.log" 2>&1 || true
timeout 900 rocprofv3 --pmc WRITE_SIZE -d "$OUT/pmc_write5" -o pmc_write5 \
    -- $RB --compression none > "$OUT/r02e_pmc_write.log" 2>&1 || true
timeout 900 rocprofv3 --kernel-trace --stats -d "$OUT/ktrace5" -o ktrace5 \
    -- $RB > "$OUT/r02e_ktrace.log" 2>&1 || true
find "$OUT" -size +20M -delete 2>/dev/null

#!/bin/bash
# Final driver rehearsal: exact driver-shaped commands end to end.
set -x
cd "${GRAFT_REPO_ROOT:-/root/repo}"
export PYTHONUNBUFFERED=1
mkdir -p gpurun_out

timeout 700 python -m pytest tests -m gpu -x -q > gpurun_out/r02m_pytest.log 2>&1
echo "pytest rc=$?"; tail -2 gpurun_out/r02m_pytest.log

timeout 300 python -c "import __graft_entry__ as g; g.smoke()" > gpurun_out/r02m_smoke.log 2>&1
echo "smoke rc=$?"; tail -1 gpurun_out/r02m_smoke.log

timeout 1200 python bench.py --gpus 1 --steps 20 --warmup 5 \
    > gpurun_out/r02m_bench.json 2> gpurun_out/r02m_bench.log
echo "bench rc=$?"
tail -1 gpurun_out/r02m_bench.json | head -c 600; echo
grep -E "bench-extra|native|index query|cpu_base" gpurun_out/r02m_bench.log | tail -6

# RCCL (nccl backend) exercise of the config-5 exchange on one GPU
timeout 600 env HX_BENCH_BACKEND=nccl python -m torch.distributed.run \
    --nnodes=1 --nproc-per-node 1 --master-addr 127.0.0.1 --master-port 29871 \
    bench.py --gpus 1 --config5 --rows 100000000 --series 1000000 --ssts 16 \
    --steps 3 --warmup 1 --no-cpu-baseline --no-extras \
    > gpurun_out/r02m_nccl5.json 2> gpurun_out/r02m_nccl5.log
echo "nccl config5 rc=$?"
grep -o '"value": [0-9.e+]*\|"workload": "[a-z0-9_]*"' gpurun_out/r02m_nccl5.json | head -2
tail -2 gpurun_out/r02m_nccl5.log

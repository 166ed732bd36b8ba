#!/bin/bash
set -x
cd "$GRAFT_REPO_ROOT" || cd /root/repo
mkdir -p gpurun_out
export HSA_ENABLE_IPC_MODE_LEGACY=0

# driver rehearsal: smoke + the exact round-end bench invocation
timeout 300 python -c "import __graft_entry__ as g; g.smoke()" > gpurun_out/smoke11.log 2>&1
echo "smoke rc=$?"; tail -1 gpurun_out/smoke11.log
timeout 900 python bench.py --gpus 1 --steps 20 --warmup 5 > gpurun_out/bench_driver_rehearsal.log 2>&1
echo "bench20 rc=$?"; grep -o '"value": [0-9.]*\|"ms_per_step": [0-9.]*' gpurun_out/bench_driver_rehearsal.log
# full-default 60-step run (same box, warm db)
timeout 600 python bench.py > gpurun_out/bench_60.log 2>&1
echo "bench60 rc=$?"; grep -o '"value": [0-9.]*\|"ms_per_step": [0-9.]*' gpurun_out/bench_60.log

# PMC for the production-structure MFMA GEMM, explicit CSV output
timeout 300 rocprofv3 --pmc MfmaUtil SQ_LDS_BANK_CONFLICT VALUBusy \
    --output-format csv -d gpurun_out/pmc11 -o pmc_r02 \
    -- ./tools/gemm_probe 4096 3 1 > gpurun_out/pmc_bench11.log 2>&1
echo "pmc rc=$?"
find gpurun_out/pmc11 -type f
CSV=$(find gpurun_out/pmc11 -name '*counter_collection.csv' | head -1)
python tools/pmc_summary.py "$CSV" > gpurun_out/pmc_gemm_r02.txt 2>&1 || true
rm -rf gpurun_out/pmc11
head -12 gpurun_out/pmc_gemm_r02.txt

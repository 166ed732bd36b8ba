#!/bin/bash
set -x
cd "$GRAFT_REPO_ROOT" || cd /root/repo
mkdir -p gpurun_out
export HSA_ENABLE_IPC_MODE_LEGACY=0
timeout 700 python -m pytest tests -m gpu -x -q > gpurun_out/pytest_gpu10.log 2>&1
echo "pytest rc=$?" | tee -a gpurun_out/pytest_gpu10.log
timeout 200 python tools/seg_bench.py > gpurun_out/seg_bench10.txt 2>&1
cat gpurun_out/seg_bench10.txt
# PMC retry with robust csv discovery
timeout 300 rocprofv3 --pmc MfmaUtil SQ_LDS_BANK_CONFLICT VALUBusy \
    -d gpurun_out/pmc10 -o pmc_r02 \
    -- ./tools/gemm_probe 4096 3 1 > gpurun_out/pmc_bench10.log 2>&1
echo "pmc rc=$?"
find gpurun_out/pmc10 -type f > gpurun_out/pmc_files.txt 2>&1
CSV=$(find gpurun_out/pmc10 -name '*.csv' | grep -i counter | head -1)
[ -z "$CSV" ] && CSV=$(find gpurun_out/pmc10 -name '*.csv' | head -1)
python tools/pmc_summary.py "$CSV" > gpurun_out/pmc_gemm_r02.txt 2>&1 || true
rm -rf gpurun_out/pmc10
head -12 gpurun_out/pmc_gemm_r02.txt

#!/bin/bash
# Round-2 GPU call 8: clean steady-state profile (warm db + immediate mode)
# + PMC counters for the production gemm_f32.
set -x
cd "$GRAFT_REPO_ROOT" || cd /root/repo
mkdir -p gpurun_out
export HSA_ENABLE_IPC_MODE_LEGACY=0

# 1. warm the MIOpen find-db (SEARCH runs in this process's warmup)
timeout 600 python bench.py --steps 5 --warmup 8 > gpurun_out/warm.log 2>&1
echo "warm rc=$?"

# 2. clean profile: immediate mode reads the tuned db entries, no find work
cd /tmp && export TMPDIR=/tmp && cd "$GRAFT_REPO_ROOT"
MIOPEN_FIND_MODE=FAST MIOPEN_FIND_ENFORCE=NONE CPD_BENCHMARK_FIND=0 \
timeout 420 rocprofv3 --kernel-trace --output-format rocpd \
    -d gpurun_out/prof8 -o r02c \
    -- python bench.py --steps 25 --warmup 8 --no-hip-graph \
    > gpurun_out/prof_bench8.log 2>&1
echo "prof rc=$?"; grep -o '"ms_per_step": [0-9.]*' gpurun_out/prof_bench8.log
DB=$(find gpurun_out/prof8 -name '*.db' | head -1)
python tools/prof_summary.py "$DB" 0.3 > gpurun_out/prof_steady_r02_clean.txt 2>&1 || true
rm -rf gpurun_out/prof8
head -16 gpurun_out/prof_steady_r02_clean.txt; tail -3 gpurun_out/prof_steady_r02_clean.txt

# 3. PMC counters on the production MFMA GEMM (counters-only run)
timeout 300 rocprofv3 --pmc MfmaUtil SQ_LDS_BANK_CONFLICT VALUBusy \
    -d gpurun_out/pmc8 -o pmc_r02 \
    -- ./tools/gemm_probe 4096 3 1 > gpurun_out/pmc_bench8.log 2>&1
echo "pmc rc=$?"
CSV=$(find gpurun_out/pmc8 -name '*counter_collection.csv' | head -1)
python tools/pmc_summary.py "$CSV" > gpurun_out/pmc_gemm_r02.txt 2>&1 || true
rm -rf gpurun_out/pmc8
cat gpurun_out/pmc_gemm_r02.txt | head -12

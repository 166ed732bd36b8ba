#!/bin/bash
# Round-2 GPU call 2: graph-test fix check, BN-reduce-fix measurement,
# conv layout/find-mode matrix, gemm v5 probe, clean steady-state profile.
set -x
cd "$GRAFT_REPO_ROOT" || cd /root/repo
mkdir -p gpurun_out
export HSA_ENABLE_IPC_MODE_LEGACY=0

timeout 600 python -m pytest tests -m gpu -x -q > gpurun_out/pytest_gpu2.log 2>&1
echo "pytest rc=$?" | tee -a gpurun_out/pytest_gpu2.log

# bench matrix: layout x BN x find-mode (short runs, one box)
timeout 420 python bench.py --steps 20 --warmup 8 \
    > gpurun_out/b_nchw_fused.log 2>&1
timeout 420 python bench.py --steps 15 --warmup 8 --no-fused-bn \
    > gpurun_out/b_nchw_eagerbn.log 2>&1
MIOPEN_FIND_MODE=NORMAL timeout 600 python bench.py --steps 15 --warmup 8 \
    --channels-last --no-fused-bn > gpurun_out/b_cl_normal.log 2>&1
MIOPEN_FIND_MODE=NORMAL MIOPEN_FIND_ENFORCE=SEARCH timeout 900 python \
    bench.py --steps 15 --warmup 8 --channels-last --no-fused-bn \
    > gpurun_out/b_cl_search.log 2>&1

# probes
timeout 300 ./tools/gemm_probe 4096 > gpurun_out/gemm_probe2_4096.log 2>&1
timeout 300 ./tools/gemm_probe 2048 > gpurun_out/gemm_probe2_2048.log 2>&1
timeout 300 ./tools/quant_gemm_probe 2048 3 3 4 > gpurun_out/qgp2_2048.log 2>&1

# clean kernel-trace profile of the eager NCHW bench steady state
cd /tmp && export TMPDIR=/tmp && cd "$GRAFT_REPO_ROOT"
timeout 600 rocprofv3 --kernel-trace --output-format rocpd \
    -d gpurun_out/prof_r02 -o r02 \
    -- python bench.py --steps 12 --warmup 6 --no-hip-graph \
    > gpurun_out/prof_bench.log 2>&1
echo "prof rc=$?"
DB=$(find gpurun_out/prof_r02 -name '*.db' | head -1)
python tools/prof_summary.py "$DB" > gpurun_out/prof_steady_r02.txt 2>&1 || true
rm -rf gpurun_out/prof_r02   # keep only the summary (merge-size budget)
tail -1 gpurun_out/b_nchw_fused.log gpurun_out/b_nchw_eagerbn.log gpurun_out/b_cl_normal.log gpurun_out/b_cl_search.log

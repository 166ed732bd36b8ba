#!/bin/bash
# Round-2 GPU call 6: finalize-fix validation (BN microbench + CL bench).
set -x
cd "$GRAFT_REPO_ROOT" || cd /root/repo
mkdir -p gpurun_out
export HSA_ENABLE_IPC_MODE_LEGACY=0

timeout 300 python -m pytest tests/test_fused_bn.py tests/test_gpu_numerics.py -m gpu -x -q > gpurun_out/pytest_gpu6.log 2>&1
echo "pytest rc=$?" | tee -a gpurun_out/pytest_gpu6.log
timeout 420 python tools/kernel_bench.py > gpurun_out/kernel_bench_r02c.txt 2>&1
grep -E "BN|gemm" gpurun_out/kernel_bench_r02c.txt
timeout 600 python bench.py --steps 30 --warmup 10 --channels-last \
    > gpurun_out/b4_cl_fused.log 2>&1
echo "cl rc=$?"; grep -o '"ms_per_step": [0-9.]*\|"value": [0-9.]*' gpurun_out/b4_cl_fused.log
timeout 420 python bench.py --steps 30 --warmup 10 \
    > gpurun_out/b4_nchw.log 2>&1
echo "nchw rc=$?"; grep -o '"ms_per_step": [0-9.]*\|"value": [0-9.]*' gpurun_out/b4_nchw.log

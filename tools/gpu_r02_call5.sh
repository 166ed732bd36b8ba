#!/bin/bash
# Round-2 GPU call 5: NHWC BN v2 validation + CL bench + accuracy experiment.
set -x
cd "$GRAFT_REPO_ROOT" || cd /root/repo
mkdir -p gpurun_out
export HSA_ENABLE_IPC_MODE_LEGACY=0

timeout 700 python -m pytest tests -m gpu -x -q > gpurun_out/pytest_gpu5.log 2>&1
echo "pytest rc=$?" | tee -a gpurun_out/pytest_gpu5.log

# BN layout microbench (v2 kernels)
timeout 420 python tools/kernel_bench.py > gpurun_out/kernel_bench_r02b.txt 2>&1
echo "kb rc=$?"
grep -E "BN" gpurun_out/kernel_bench_r02b.txt

# channels_last + NHWC fused BN v2
timeout 600 python bench.py --steps 30 --warmup 10 --channels-last \
    > gpurun_out/b3_cl_fused.log 2>&1
echo "cl rc=$?"; grep -o '"ms_per_step": [0-9.]*' gpurun_out/b3_cl_fused.log

# accuracy experiment (fixed dataset + schedule)
bash tools/acc_experiment.sh > gpurun_out/acc_driver2.log 2>&1
echo "acc rc=$?"
for f in gpurun_out/acc_fp32.log gpurun_out/acc_e4m3_aps.log gpurun_out/acc_e4m3_noaps.log gpurun_out/acc_e3m0_aps.log; do
  echo "== $f"; grep '\* All Loss' $f | tail -2; done

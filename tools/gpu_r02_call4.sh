#!/bin/bash
# Round-2 GPU call 4: APS accuracy experiment (the north-star top-1 axis)
# + channels_last fused-vs-eager BN A/B with kernel microbench.
set -x
cd "$GRAFT_REPO_ROOT" || cd /root/repo
mkdir -p gpurun_out
export HSA_ENABLE_IPC_MODE_LEGACY=0

# 1. BN layout microbench (fresh box, no find-db pollution concerns: BN only)
timeout 420 python tools/kernel_bench.py > gpurun_out/kernel_bench_r02.txt 2>&1
echo "kb rc=$?"

# 2. channels_last A/B on one box
timeout 600 python bench.py --steps 25 --warmup 10 --channels-last --no-fused-bn \
    > gpurun_out/b2_cl_eager.log 2>&1
echo "cl_eager rc=$?"
timeout 480 python bench.py --steps 25 --warmup 10 --channels-last \
    --torch-profile gpurun_out/trace_cl_fused.json \
    > gpurun_out/b2_cl_fused.log 2>&1
echo "cl_fused rc=$?"
timeout 480 python bench.py --steps 25 --warmup 10 \
    > gpurun_out/b2_nchw_fused.log 2>&1
echo "nchw rc=$?"

# 3. accuracy experiment (4 configs x 420 iters, NCHW path)
bash tools/acc_experiment.sh > gpurun_out/acc_driver.log 2>&1
echo "acc rc=$?"
grep -h '"metric"' gpurun_out/b2_*.log
for f in gpurun_out/acc_*.log; do echo "== $f"; grep '\* All Loss' $f | tail -2; done

#!/bin/bash
set -x
cd "$GRAFT_REPO_ROOT" || cd /root/repo
mkdir -p gpurun_out
export HSA_ENABLE_IPC_MODE_LEGACY=0
timeout 700 python -m pytest tests -m gpu -x -q > gpurun_out/pytest_gpu_final.log 2>&1
echo "pytest rc=$?" | tee -a gpurun_out/pytest_gpu_final.log
timeout 900 python bench.py --steps 200 --warmup 15 > gpurun_out/bench_200.log 2>&1
echo "bench200 rc=$?"
grep -o '"value": [0-9.]*\|"ms_per_step": [0-9.]*' gpurun_out/bench_200.log

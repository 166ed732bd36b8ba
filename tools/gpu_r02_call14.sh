#!/bin/bash
set -x
cd "$GRAFT_REPO_ROOT" || cd /root/repo
mkdir -p gpurun_out
export HSA_ENABLE_IPC_MODE_LEGACY=0
timeout 300 ./tools/gemm_probe 4096 10 3 > gpurun_out/gemm_probe_v6_4096.log 2>&1
timeout 300 ./tools/gemm_probe 2048 10 3 > gpurun_out/gemm_probe_v6_2048.log 2>&1
grep -E "check|round" gpurun_out/gemm_probe_v6_4096.log gpurun_out/gemm_probe_v6_2048.log

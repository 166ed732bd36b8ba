#!/bin/bash
# Round-2 GPU call 1: regression + graph-vs-eager bench + GEMM probes.
set -x
cd "$GRAFT_REPO_ROOT" || cd /root/repo
mkdir -p gpurun_out
export HSA_ENABLE_IPC_MODE_LEGACY=0

# 1. GPU test suite (includes new hipGraph bit-identity + device cast scans)
timeout 600 python -m pytest tests -m gpu -x -q > gpurun_out/pytest_gpu.log 2>&1
echo "pytest rc=$?" | tee -a gpurun_out/pytest_gpu.log

# 2. quant_gemm variant probe (e4m3 + e5m2)
timeout 300 ./tools/quant_gemm_probe 1024 3 3 4 > gpurun_out/qgp_e4m3.log 2>&1
timeout 300 ./tools/quant_gemm_probe 1024 3 2 5 > gpurun_out/qgp_e5m2.log 2>&1

# 3. fp32 MFMA GEMM probe (glds B-staging variant v4)
timeout 300 ./tools/gemm_probe 4096 > gpurun_out/gemm_probe_4096.log 2>&1
timeout 300 ./tools/gemm_probe 2048 > gpurun_out/gemm_probe_2048.log 2>&1

# 4. bench: eager vs hipGraph on the same box
timeout 420 python bench.py --steps 30 --warmup 10 --no-hip-graph > gpurun_out/bench_eager.log 2>&1
echo "eager rc=$?"
timeout 420 python bench.py --steps 30 --warmup 10 --hip-graph > gpurun_out/bench_graph.log 2>&1
echo "graph rc=$?"

tail -2 gpurun_out/bench_eager.log gpurun_out/bench_graph.log

#!/bin/bash
# Round-2 bench config matrix (1x MI355X, driver-style invocations).
# Output: gpurun_out/bm_*.log — one JSON line each; summarized into
# profiles/bench_matrix_r02.md afterwards.
set -x
cd "$GRAFT_REPO_ROOT" || cd /root/repo
mkdir -p gpurun_out
export HSA_ENABLE_IPC_MODE_LEGACY=0

row() {  # name flags...
  name=$1; shift
  timeout 600 python bench.py "$@" > gpurun_out/bm_$name.log 2>&1
  echo "== $name rc=$?"; tail -1 gpurun_out/bm_$name.log
}

row flagship      --steps 40 --warmup 10
row e5m2          --steps 30 --warmup 8 --grad-exp 5 --grad-man 2
row seqmode       --steps 30 --warmup 8 --mode sequential
row kahan         --steps 30 --warmup 8 --use-kahan
row noaps         --steps 30 --warmup 8 --no-aps
row emu8          --steps 8 --warmup 3 --emulate-node 8
row emu32         --steps 3 --warmup 2 --emulate-node 32 --grad-exp 5 --grad-man 2 --use-kahan
row rn50          --steps 15 --warmup 5 --model resnet50 --batch 128 --grad-exp 5 --grad-man 2 --use-kahan
row rn18q_cfg4    --steps 5 --warmup 2 --model resnet18_cifar_quant --batch 64 --grad-exp 5 --grad-man 2 --use-kahan --no-hip-graph
row rn50q_cfg4    --steps 2 --warmup 1 --model resnet50_quant --batch 8 --grad-exp 5 --grad-man 2 --use-kahan --no-hip-graph
grep -h '"metric"' gpurun_out/bm_*.log

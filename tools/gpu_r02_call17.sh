#!/bin/bash
set -x
cd "$GRAFT_REPO_ROOT" || cd /root/repo
mkdir -p gpurun_out
export HSA_ENABLE_IPC_MODE_LEGACY=0

# RCCL init path rehearsal: the driver launches N>1 via torch.distributed.run;
# run W=1 through the same entry (init_process_group("nccl") on ROCm,
# overlap pipeline forced on via --overlap to exercise hook+comm-stream code)
timeout 700 python -m torch.distributed.run --nnodes=1 --nproc-per-node 1 \
    --master-addr 127.0.0.1 --master-port 29517 \
    bench.py --gpus 1 --steps 15 --warmup 8 \
    > gpurun_out/bench_torchrun_w1.log 2>&1
echo "torchrun rc=$?"
grep -o '"ms_per_step": [0-9.]*\|"value": [0-9.]*' gpurun_out/bench_torchrun_w1.log

# ResNet50 channels_last steady profile (warm db in-process via warmup)
timeout 600 python bench.py --model resnet50 --batch 128 --steps 5 --warmup 5 \
    > gpurun_out/warm_rn50.log 2>&1
cd /tmp && export TMPDIR=/tmp && cd "$GRAFT_REPO_ROOT"
MIOPEN_FIND_MODE=FAST MIOPEN_FIND_ENFORCE=NONE CPD_BENCHMARK_FIND=0 \
timeout 420 rocprofv3 --kernel-trace --output-format rocpd \
    -d gpurun_out/prof17 -o rn50 \
    -- python bench.py --model resnet50 --batch 128 --steps 10 --warmup 4 \
    --no-hip-graph > gpurun_out/prof_rn50.log 2>&1
echo "prof rc=$?"; grep -o '"ms_per_step": [0-9.]*' gpurun_out/prof_rn50.log
DB=$(find gpurun_out/prof17 -name '*.db' | head -1)
python tools/prof_summary.py "$DB" 0.3 > gpurun_out/prof_steady_rn50_r02.txt 2>&1 || true
rm -rf gpurun_out/prof17
head -12 gpurun_out/prof_steady_rn50_r02.txt; tail -3 gpurun_out/prof_steady_rn50_r02.txt

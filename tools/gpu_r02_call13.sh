#!/bin/bash
set -x
cd "$GRAFT_REPO_ROOT" || cd /root/repo
mkdir -p gpurun_out
export HSA_ENABLE_IPC_MODE_LEGACY=0
timeout 420 python -m cpd_amd.trainers.train_resnet50 \
    --synthetic --epochs 2 --steps-per-epoch 5 --batch-size 32 \
    --grad_exp 5 --grad_man 2 --use_APS --use_kahan \
    --checkpoint-format gpurun_out/checkpoint-{epoch}.pth.tar \
    > gpurun_out/rn50_trainer_gpu.log 2>&1
echo "rn50 rc=$?"
timeout 420 python -m cpd_amd.trainers.train_resnet50 \
    --synthetic --epochs 3 --steps-per-epoch 5 --batch-size 32 \
    --grad_exp 5 --grad_man 2 --use_APS --use_kahan \
    --checkpoint-format gpurun_out/checkpoint-{epoch}.pth.tar \
    >> gpurun_out/rn50_trainer_gpu.log 2>&1
echo "rn50 resume rc=$?"
grep -iE "resum|epoch|loss" gpurun_out/rn50_trainer_gpu.log | tail -8
rm -f gpurun_out/checkpoint-*.pth.tar

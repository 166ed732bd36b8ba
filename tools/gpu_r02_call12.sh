#!/bin/bash
# Round-2 GPU call 12: trainer GPU smokes (DavidNet fp16+loss-scale path,
# ResNet50 trainer incl. checkpoint/auto-resume) — exercises P13-P16 on
# hardware.
set -x
cd "$GRAFT_REPO_ROOT" || cd /root/repo
mkdir -p gpurun_out
export HSA_ENABLE_IPC_MODE_LEGACY=0

timeout 420 python -m cpd_amd.trainers.train_davidnet --synthetic \
    --epochs 2 --steps-per-epoch 8 --half --loss_scale 256 \
    --grad_exp 5 --grad_man 2 --use_APS \
    --log-tsv gpurun_out/davidnet_gpu.tsv > gpurun_out/davidnet_gpu.log 2>&1
echo "davidnet rc=$?"; tail -4 gpurun_out/davidnet_gpu.log

cd gpurun_out && timeout 420 python -m cpd_amd.trainers.train_resnet50 \
    --synthetic --epochs 2 --steps-per-epoch 5 --batch-size 32 \
    --grad_exp 5 --grad_man 2 --use_APS --use_kahan \
    > rn50_trainer_gpu.log 2>&1
echo "rn50 epoch1+2 rc=$?"
# auto-resume from the epoch checkpoint
timeout 420 python -m cpd_amd.trainers.train_resnet50 \
    --synthetic --epochs 3 --steps-per-epoch 5 --batch-size 32 \
    --grad_exp 5 --grad_man 2 --use_APS --use_kahan \
    >> rn50_trainer_gpu.log 2>&1
echo "rn50 resume rc=$?"; grep -iE "resum|epoch" rn50_trainer_gpu.log | tail -5
rm -f checkpoint-*.pth.tar

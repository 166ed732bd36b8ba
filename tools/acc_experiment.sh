#!/bin/bash
# APS-recovers-accuracy experiment (the reference's core claim,
# README.md:153-154) on this framework's own kernels: ResNet18 flagship
# config (batch 512 x emulate_node 8 = global 4096, reference LR schedule)
# on the deterministic procedural dataset, four gradient formats:
#   fp32 (8,23) | e4m3+APS | e4m3 no-APS | e3m0(4-bit)+APS
# Output: gpurun_out/acc_*.log with '* All Loss' lines -> draw_curve TSV+SVG.
set -x
cd "$GRAFT_REPO_ROOT" || cd /root/repo
mkdir -p gpurun_out
export HSA_ENABLE_IPC_MODE_LEGACY=0 MIOPEN_FIND_MODE=FAST

# batch 128 x emulate_node 8 = global 1024 at the reference lr/batch ratio
# (1.6@4096 -> 0.4@1024); warmup stretched to 80 iters (the 5-epoch rule
# gives only 16 iters/epoch on the 16k-sample procedural set)
COMMON="--procedural --emulate_node 8 --max_iter 900 --val_freq 45
        --print_freq 45 --batch_size 128 --workers 2 --peak_lr 0.4
        --warmup_iter 80"

run() {  # name extra-flags...
  name=$1; shift
  timeout 900 python -m cpd_amd.trainers.train_resnet18 $COMMON \
      --save_path gpurun_out/ckpt_$name "$@" \
      > gpurun_out/acc_$name.log 2>&1
  echo "$name rc=$? last:"; grep '\* All Loss' gpurun_out/acc_$name.log | tail -2
}

run fp32       --grad_exp 8 --grad_man 23
run e4m3_aps   --grad_exp 4 --grad_man 3 --use_APS
run e4m3_noaps --grad_exp 4 --grad_man 3
run e3m0_aps   --grad_exp 3 --grad_man 0 --use_APS

rm -f gpurun_out/ckpt_*   # checkpoints are big; logs are the artifact
python tools/draw_curve.py gpurun_out/acc_fp32.log gpurun_out/acc_e4m3_aps.log \
    gpurun_out/acc_e4m3_noaps.log gpurun_out/acc_e3m0_aps.log \
    --svg gpurun_out/acc_curves.svg > gpurun_out/acc_curves.tsv
tail -5 gpurun_out/acc_curves.tsv

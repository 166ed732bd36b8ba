#!/bin/bash
# Harder-task accuracy variant: snr 0.3 — the no-APS degradation should
# persist instead of recovering on the too-easy default task.
set -x
cd "$GRAFT_REPO_ROOT" || cd /root/repo
mkdir -p gpurun_out
export HSA_ENABLE_IPC_MODE_LEGACY=0 MIOPEN_FIND_MODE=FAST
COMMON="--procedural --snr 0.3 --emulate_node 8 --max_iter 900 --val_freq 45
        --print_freq 90 --batch_size 128 --workers 2 --peak_lr 0.4
        --warmup_iter 80"
run() { name=$1; shift
  timeout 700 python -m cpd_amd.trainers.train_resnet18 $COMMON \
      --save_path gpurun_out/ckpt2_$name "$@" > gpurun_out/acch_$name.log 2>&1
  echo "$name rc=$?"; grep '\* All Loss' gpurun_out/acch_$name.log | tail -1
}
run fp32       --grad_exp 8 --grad_man 23
run e4m3_aps   --grad_exp 4 --grad_man 3 --use_APS
run e4m3_noaps --grad_exp 4 --grad_man 3
rm -f gpurun_out/ckpt2_*
python tools/draw_curve.py gpurun_out/acch_fp32.log gpurun_out/acch_e4m3_aps.log \
    gpurun_out/acch_e4m3_noaps.log --svg gpurun_out/acch_curves.svg \
    > gpurun_out/acch_curves.tsv
tail -4 gpurun_out/acch_curves.tsv

#!/usr/bin/env python3
"""Flagship trainer: ResNet18/CIFAR10 customized-precision distributed
training (reference: example/ResNet18/tools/mix.py — same CLI flag names and
training algebra, rebuilt on the fused MI355X pipeline).

Examples:
  # single GPU, e4m3 grads + APS, 8-rank node emulation (BASELINE config 2)
  python -m cpd_amd.trainers.train_resnet18 --grad_exp 4 --grad_man 3 \
      --use_APS --emulate_node 8 --synthetic

  # 8 GPUs via torchrun (BASELINE config 3)
  torchrun --nproc-per-node 8 --master-addr 127.0.0.1 \
      -m cpd_amd.trainers.train_resnet18 --dist --grad_exp 4 --grad_man 3 \
      --use_APS --synthetic
"""
import argparse
import math
import os
import sys
import time

import torch
import torch.nn as nn
import yaml
from torch.utils.data import DataLoader

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.dirname(
    os.path.abspath(__file__)))))

from cpd_amd import models
from cpd_amd.data import CIFAR10, SyntheticImages
from cpd_amd.parallel import DistModule, dist_init
from cpd_amd.trainers.core import LPTrainStep
from cpd_amd.utils import (AverageMeter, DistributedGivenIterationSampler,
                           DistributedSampler, LARS, accuracy, load_state,
                           save_checkpoint)
from cpd_amd.utils.scalars import ScalarLogger


def parse_args(argv=None):
    p = argparse.ArgumentParser()
    p.add_argument('--config', default=None)
    p.add_argument('--dist', action='store_true')
    p.add_argument('--load-path', default='', type=str)
    p.add_argument('--grad_exp', default=5, type=int)
    p.add_argument('--grad_man', default=2, type=int)
    p.add_argument('--resume-opt', action='store_true')
    p.add_argument('--use_lars', action='store_true')
    p.add_argument('--use_APS', action='store_true')
    p.add_argument('--use_kahan', action='store_true')
    p.add_argument('-e', '--evaluate', action='store_true')
    p.add_argument('--emulate_node', default=1, type=int)
    p.add_argument('--mode', choices=['ring', 'sequential'], default='ring')
    p.add_argument('--overlap', type=int, default=0,
                   help='sub-buckets for backward-overlapped reduction')
    p.add_argument('--synthetic', action='store_true',
                   help='synthetic CIFAR-shaped data (no dataset on disk)')
    p.add_argument('--procedural', action='store_true',
                   help='deterministic LEARNABLE procedural data (accuracy '
                        'experiments without a dataset on disk)')
    p.add_argument('--max_iter', default=0, type=int,
                   help='override the epoch-derived iteration count (>0)')
    p.add_argument('--peak_lr', default=1.6, type=float,
                   help='post-warmup peak LR (reference schedule: 1.6 at '
                        'global batch 4096, mix.py:181-198)')
    p.add_argument('--warmup_iter', default=0, type=int,
                   help='override the 5-epoch warmup length in iterations '
                        '(small procedural datasets make 5 epochs only a '
                        'few iterations, which diverges)')
    p.add_argument('--snr', default=0.5, type=float,
                   help='procedural-dataset template-to-noise ratio')
    p.add_argument('--data-root', default='./data/cifar-10-batches-py')
    # config-file defaults (res18_cifar.yaml parity)
    p.add_argument('--arch', default='res_cifar')
    p.add_argument('--batch_size', default=512, type=int)
    p.add_argument('--max_epoch', default=100, type=int)
    p.add_argument('--base_lr', default=0.1, type=float)
    p.add_argument('--momentum', default=0.9, type=float)
    p.add_argument('--weight_decay', default=1e-4, type=float)
    p.add_argument('--workers', default=2, type=int)
    p.add_argument('--print_freq', default=50, type=int)
    p.add_argument('--val_freq', default=50, type=int)
    p.add_argument('--save_path', default='checkpoints/res18')
    args = p.parse_args(argv)
    if args.config:
        with open(args.config) as f:
            for k, v in yaml.safe_load(f)['common'].items():
                setattr(args, k, v)
    return args


def adjust_learning_rate(optimizer, step, iter_per_epoch, peak_lr=1.6,
                         warmup_iter=0):
    """Warmup 0.1 -> peak over 5 epochs (or ``warmup_iter`` iterations),
    /10 at epochs 40 and 80 (mix.py:181-198 schedule; peak 1.6)."""
    warm_up_iter = warmup_iter or 5 * iter_per_epoch
    if step <= warm_up_iter:
        lr = 0.1 + (peak_lr - 0.1) * (step / warm_up_iter)
    else:
        lr = peak_lr
        if step > iter_per_epoch * 40:
            lr *= 0.1
        if step > iter_per_epoch * 80:
            lr *= 0.1
    for group in optimizer.param_groups:
        group['lr'] = lr
    return lr


def main(argv=None):
    args = parse_args(argv)
    torch.manual_seed(24)

    if args.dist:
        rank, world_size = dist_init()
    else:
        rank, world_size = 0, 1
        print('Disabled distributed training.')

    device = torch.device('cuda' if torch.cuda.is_available() else 'cpu')
    model = models.build_model(args.arch).to(device)
    model.train()
    dm = DistModule(model)

    criterion = nn.CrossEntropyLoss().to(device)
    opt_cls = LARS if args.use_lars else torch.optim.SGD
    optimizer = opt_cls([{'params': model.parameters()}], lr=args.base_lr,
                        momentum=args.momentum,
                        weight_decay=args.weight_decay)
    step = LPTrainStep(dm, optimizer, grad_exp=args.grad_exp,
                       grad_man=args.grad_man, use_APS=args.use_APS,
                       use_kahan=args.use_kahan,
                       emulate_node=args.emulate_node, mode=args.mode,
                       overlap=args.overlap if args.emulate_node == 1 else 0)

    if args.procedural:
        from cpd_amd.data import ProceduralImages
        train_set = ProceduralImages(16384, seed=0, snr=args.snr)
        val_set = ProceduralImages(2048, seed=1, snr=args.snr)
    elif args.synthetic or not os.path.isdir(args.data_root):
        if not args.synthetic and rank == 0:
            print(f'No CIFAR at {args.data_root}; using synthetic data.')
        train_set = SyntheticImages(50000)
        val_set = SyntheticImages(10000, seed=1)
    else:
        train_set = CIFAR10(args.data_root, train=True)
        val_set = CIFAR10(args.data_root, train=False, augment=False)

    denom = world_size * args.batch_size * args.emulate_node
    max_iter = args.max_iter or math.ceil(
        len(train_set) * args.max_epoch / denom)
    iter_per_epoch = math.ceil(len(train_set) / denom)
    last_iter = -1

    best_prec1 = 0.0
    if args.load_path:
        res = load_state(args.load_path, model,
                         optimizer if args.resume_opt else None)
        if res is not None:
            best_prec1, last_iter = res

    train_sampler = DistributedGivenIterationSampler(
        train_set, max_iter * args.emulate_node, args.batch_size,
        world_size=world_size, rank=rank, last_iter=last_iter)
    val_sampler = DistributedSampler(val_set, world_size=world_size,
                                     rank=rank, round_up=False)
    train_loader = DataLoader(train_set, batch_size=args.batch_size,
                              shuffle=False, num_workers=args.workers,
                              pin_memory=True, sampler=train_sampler)
    val_loader = DataLoader(val_set, batch_size=args.batch_size,
                            shuffle=False, num_workers=args.workers,
                            pin_memory=True, sampler=val_sampler)

    if args.evaluate:
        validate(val_loader, model, criterion, device, world_size, rank)
        return

    train(args, train_loader, val_loader, dm, model, criterion, optimizer,
          step, device, rank, world_size, last_iter + 1, iter_per_epoch,
          max_iter, best_prec1)


def train(args, train_loader, val_loader, dm, model, criterion, optimizer,
          step, device, rank, world_size, start_iter, iter_per_epoch,
          max_iter, best_prec1):
    import torch.distributed as dist

    scalars = ScalarLogger(args.save_path + '_scalars' if rank == 0 else None)
    batch_time = AverageMeter(args.print_freq)
    losses = AverageMeter(args.print_freq)
    curr_step = start_iter
    emulate_step = 0
    end = time.time()

    for x, y in train_loader:
        emulate_step += 1
        boundary = emulate_step == args.emulate_node
        if boundary:
            emulate_step = 0
            curr_step += 1
        if curr_step > max_iter:
            break
        lr = adjust_learning_rate(optimizer, curr_step, iter_per_epoch,
                                  peak_lr=args.peak_lr,
                                  warmup_iter=args.warmup_iter)

        x = x.to(device, non_blocking=True)
        y = y.to(device, non_blocking=True)
        loss = criterion(model(x), y) / step.loss_scale_denom()
        reduced_loss = loss.detach().clone()
        if args.dist:
            dist.all_reduce(reduced_loss)
        losses.update(float(reduced_loss))
        step.substep(loss)

        if boundary:
            batch_time.update(time.time() - end)
            end = time.time()
            if curr_step % args.print_freq == 0 and rank == 0:
                print(f'Iter [{curr_step}/{max_iter}] lr {lr:.4f} '
                      f'loss {losses.avg:.4f} '
                      f'batch_time {batch_time.avg * 1000:.1f}ms', flush=True)
                scalars.add_scalar('loss_train', losses.avg, curr_step)
                scalars.add_scalar('lr', lr, curr_step)
            if curr_step % args.val_freq == 0:
                prec1 = validate(val_loader, model, criterion, device,
                                 world_size, rank)
                model.train()
                if rank == 0:
                    scalars.add_scalar('acc1', prec1, curr_step)
                    is_best = prec1 > best_prec1
                    best_prec1 = max(prec1, best_prec1)
                    os.makedirs(os.path.dirname(args.save_path) or '.',
                                exist_ok=True)
                    save_checkpoint({
                        'step': curr_step,
                        'arch': args.arch,
                        'state_dict': model.state_dict(),
                        'best_prec1': best_prec1,
                        'optimizer': optimizer.state_dict(),
                    }, is_best, args.save_path)


def validate(val_loader, model, criterion, device, world_size, rank):
    import torch.distributed as dist

    model.eval()
    losses, top1, top5 = AverageMeter(0), AverageMeter(0), AverageMeter(0)
    with torch.no_grad():
        for x, y in val_loader:
            x = x.to(device, non_blocking=True)
            y = y.to(device, non_blocking=True)
            out = model(x)
            loss = criterion(out, y)
            prec1, prec5 = accuracy(out, y, topk=(1, 5))
            losses.update(float(loss))
            top1.update(float(prec1))
            top5.update(float(prec5))
    stats = torch.tensor([losses.avg, top1.avg, top5.avg], device=device)
    if dist.is_available() and dist.is_initialized():
        dist.all_reduce(stats)
        stats /= world_size
    if rank == 0:
        print(f'* All Loss {stats[0]:.4f} Prec@1 {stats[1]:.3f} '
              f'Prec@5 {stats[2]:.3f}', flush=True)
    return float(stats[1])


if __name__ == '__main__':
    main()

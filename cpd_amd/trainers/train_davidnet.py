#!/usr/bin/env python3
"""DavidNet DAWNBench-style CIFAR10 speed-run trainer (reference:
example/DavidNet/dawn.py — 24 epochs, PiecewiseLinear LR, optional fp16 model
with fp32 BatchNorm and static loss scaling, TSV logging; rebuilt on the
graph executor + fused gradient pipeline)."""
import argparse
import os
import sys
import time

import numpy as np
import torch
from torch.utils.data import DataLoader

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.dirname(
    os.path.abspath(__file__)))))

from cpd_amd.data import CIFAR10, SyntheticImages
from cpd_amd.models.davidnet import DavidNet
from cpd_amd.parallel import DistModule, dist_init
from cpd_amd.trainers.core import LPTrainStep
from cpd_amd.utils import DistributedSampler


class PiecewiseLinear:
    def __init__(self, knots, vals):
        self.knots = knots
        self.vals = vals

    def __call__(self, t):
        return float(np.interp([t], self.knots, self.vals)[0])


class TSVLogger:
    """DAWNBench submission log (dawn.py:37-47 capability)."""

    def __init__(self, path=None):
        self.rows = ['epoch\thours\ttop1Accuracy']
        self.path = path

    def append(self, epoch, hours, acc):
        self.rows.append(f'{epoch}\t{hours:.8f}\t{acc:.2f}')
        if self.path:
            with open(self.path, 'w') as f:
                f.write('\n'.join(self.rows) + '\n')

    def __str__(self):
        return '\n'.join(self.rows)


def parse_args(argv=None):
    p = argparse.ArgumentParser()
    p.add_argument('--dist', action='store_true')
    p.add_argument('--epochs', type=int, default=24)
    p.add_argument('--batch-size', type=int, default=512)
    p.add_argument('--lr-scale', type=float, default=0.4)
    p.add_argument('--half', action='store_true', help='fp16 model, fp32 BN')
    p.add_argument('--loss_scale', type=float, default=1.0)
    p.add_argument('--grad_exp', default=8, type=int)
    p.add_argument('--grad_man', default=23, type=int)
    p.add_argument('--use_APS', action='store_true')
    p.add_argument('--mode', choices=['ring', 'sequential'], default='ring')
    p.add_argument('--synthetic', action='store_true')
    p.add_argument('--data-root', default='./data/cifar-10-batches-py')
    p.add_argument('--log-tsv', default=None)
    p.add_argument('--steps-per-epoch', type=int, default=None)
    return p.parse_args(argv)


def main(argv=None):
    args = parse_args(argv)
    rank, world_size = (dist_init() if args.dist else (0, 1))
    device = torch.device('cuda' if torch.cuda.is_available() else 'cpu')
    torch.manual_seed(0)

    model = DavidNet().to(device)
    if args.half:
        model.half()
    model.train()
    dm = DistModule(model)
    optimizer = torch.optim.SGD([{'params': model.parameters()}],
                                lr=0.1, momentum=0.9, weight_decay=5e-4,
                                nesterov=True)
    step = LPTrainStep(dm, optimizer, grad_exp=args.grad_exp,
                       grad_man=args.grad_man, use_APS=args.use_APS,
                       mode=args.mode, use_master=True)
    lr_sched = PiecewiseLinear([0, 5, args.epochs],
                               [0, args.lr_scale * world_size, 0])

    if args.synthetic or not os.path.isdir(args.data_root):
        train_set = SyntheticImages(50000)
        test_set = SyntheticImages(10000, seed=1)
    else:
        train_set = CIFAR10(args.data_root, train=True, cutout=8)
        test_set = CIFAR10(args.data_root, train=False, augment=False)

    sampler = DistributedSampler(train_set, world_size=world_size, rank=rank)
    loader = DataLoader(train_set, batch_size=args.batch_size, sampler=sampler,
                        num_workers=2, pin_memory=True, drop_last=True)
    test_loader = DataLoader(test_set, batch_size=args.batch_size,
                             num_workers=2)

    tsv = TSVLogger(args.log_tsv)
    t_train = 0.0
    for epoch in range(args.epochs):
        sampler.set_epoch(epoch)
        t0 = time.time()
        stats = run_epoch(args, loader, model, step, optimizer, lr_sched,
                          epoch, device, world_size, train=True)
        t_train += time.time() - t0
        test_stats = run_epoch(args, test_loader, model, None, None, None,
                               epoch, device, world_size, train=False)
        if rank == 0:
            print(f'epoch {epoch + 1} train loss {stats["loss"]:.4f} '
                  f'acc {stats["acc"]:.3f} | test loss '
                  f'{test_stats["loss"]:.4f} acc {test_stats["acc"]:.3f} '
                  f'({t_train:.1f}s train)', flush=True)
            tsv.append(epoch + 1, t_train / 3600, test_stats['acc'] * 100)
    if rank == 0:
        print(tsv)


def run_epoch(args, loader, model, step, optimizer, lr_sched, epoch, device,
              world_size, train):
    model.train(train)
    tot_loss, tot_correct, n = 0.0, 0.0, 0
    steps = 0
    ctx = torch.enable_grad() if train else torch.no_grad()
    with ctx:
        for x, y in loader:
            x = x.to(device, non_blocking=True)
            if args.half:
                x = x.half()
            y = y.to(device, non_blocking=True)
            out = model({'input': x, 'target': y})
            loss = out['loss']
            if train:
                t = epoch + steps / max(1, len(loader))
                # sum-CE: lr absorbs /batch; static loss scaling for fp16 is
                # mathematically neutral (loss x scale, lr / scale — the
                # reference scales the loss without unscaling, dawn.py:24 +
                # utils.py:333, which silently multiplies the effective LR)
                lr = lr_sched(t) / args.batch_size / args.loss_scale
                for g in optimizer.param_groups:
                    g['lr'] = lr
                step.substep(loss * args.loss_scale /
                             step.loss_scale_denom())
            tot_loss += float(loss.detach())
            tot_correct += float(out['correct'].sum())
            n += x.shape[0]
            steps += 1
            if args.steps_per_epoch and steps >= args.steps_per_epoch:
                break
    # cross-rank stat reduction (reference `collect`, utils.py:347-356)
    import torch.distributed as dist
    if dist.is_available() and dist.is_initialized():
        t = torch.tensor([tot_loss, tot_correct, float(n)])
        dist.all_reduce(t)
        tot_loss, tot_correct, n = float(t[0]), float(t[1]), int(t[2])
    return {'loss': tot_loss / max(n, 1), 'acc': tot_correct / max(n, 1)}


if __name__ == '__main__':
    main()

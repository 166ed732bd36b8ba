#!/usr/bin/env python3
"""ResNet50/ImageNet customized-precision trainer (reference:
example/ResNet50/main.py — same capability set, rebuilt: epoch-based loop,
sub-batch gradient accumulation (= node emulation), BatchNorm params with
weight-decay 0, nesterov SGD with warmup/step decay, epoch checkpoints with
auto-resume by scanning for the newest one)."""
import argparse
import os
import sys
import time

import torch
import torch.nn as nn
from torch.utils.data import DataLoader

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.dirname(
    os.path.abspath(__file__)))))

from cpd_amd import models
from cpd_amd.data import SyntheticImages
from cpd_amd.parallel import DistModule, dist_init
from cpd_amd.trainers.core import LPTrainStep
from cpd_amd.utils import AverageMeter, DistributedSampler, accuracy


def parse_args(argv=None):
    p = argparse.ArgumentParser()
    p.add_argument('--dist', action='store_true')
    p.add_argument('--batch-size', type=int, default=32,
                   help='per-GPU sub-batch (accumulated batches-per-allreduce'
                        ' times)')
    p.add_argument('--batches-per-allreduce', type=int, default=1,
                   help='gradient-accumulation factor (the reference calls '
                        'this sub-batching / node emulation, main.py:27-30)')
    p.add_argument('--epochs', type=int, default=90)
    p.add_argument('--base-lr', type=float, default=0.0125)
    p.add_argument('--warmup-epochs', type=float, default=5)
    p.add_argument('--momentum', type=float, default=0.9)
    p.add_argument('--wd', type=float, default=5e-5)
    p.add_argument('--grad_exp', default=5, type=int)
    p.add_argument('--grad_man', default=2, type=int)
    p.add_argument('--use_APS', action='store_true')
    p.add_argument('--use_kahan', action='store_true')
    p.add_argument('--mode', choices=['ring', 'sequential'], default='ring')
    p.add_argument('--checkpoint-format', default='checkpoint-{epoch}.pth.tar')
    p.add_argument('--synthetic', action='store_true')
    p.add_argument('--synthetic-size', type=int, default=1281167)
    p.add_argument('--steps-per-epoch', type=int, default=None,
                   help='cap steps per epoch (short runs / testing)')
    p.add_argument('--image-size', type=int, default=224)
    p.add_argument('--workers', type=int, default=4)
    return p.parse_args(argv)


def bn_param_split(model):
    """BN params get weight-decay 0 (main.py:123-127 behavior)."""
    bn_params, rest = [], []
    bn_names = set()
    for mod_name, mod in model.named_modules():
        if isinstance(mod, nn.BatchNorm2d):
            for p_name, _ in mod.named_parameters(recurse=False):
                bn_names.add(f"{mod_name}.{p_name}")
    for name, p in model.named_parameters():
        (bn_params if name in bn_names else rest).append(p)
    return rest, bn_params


def main(argv=None):
    args = parse_args(argv)
    rank, world_size = (dist_init() if args.dist else (0, 1))
    device = torch.device('cuda' if torch.cuda.is_available() else 'cpu')

    torch.manual_seed(42)
    model = models.resnet50().to(device).train()
    dm = DistModule(model)

    # linear-scaled LR over total effective batch, BN params wd=0
    # (main.py:123-131 behavior)
    lr_scale = args.batches_per_allreduce * world_size
    rest, bn_params = bn_param_split(model)
    optimizer = torch.optim.SGD(
        [{'params': rest, 'weight_decay': args.wd},
         {'params': bn_params, 'weight_decay': 0.0}],
        lr=args.base_lr * lr_scale, momentum=args.momentum, nesterov=True)
    # auto-resume: scan checkpoint-{epoch} downward (main.py:70-75) —
    # BEFORE the master copies are made, so they snapshot the resumed weights
    start_epoch = 0
    for ep in range(args.epochs, 0, -1):
        path = args.checkpoint_format.format(epoch=ep)
        if os.path.exists(path):
            ckpt = torch.load(path, map_location='cpu', weights_only=False)
            model.load_state_dict(ckpt['model'])
            optimizer.load_state_dict(ckpt['optimizer'])
            start_epoch = ep
            break

    step = LPTrainStep(dm, optimizer, grad_exp=args.grad_exp,
                       grad_man=args.grad_man, use_APS=args.use_APS,
                       use_kahan=args.use_kahan,
                       emulate_node=args.batches_per_allreduce,
                       mode=args.mode)

    train_set = SyntheticImages(args.synthetic_size,
                                shape=(3, args.image_size, args.image_size),
                                num_classes=1000)
    sampler = DistributedSampler(train_set, world_size=world_size, rank=rank)
    loader = DataLoader(train_set, batch_size=args.batch_size, sampler=sampler,
                        num_workers=args.workers, pin_memory=True)
    criterion = nn.CrossEntropyLoss().to(device)

    for epoch in range(start_epoch, args.epochs):
        sampler.set_epoch(epoch)
        train_epoch(args, loader, model, criterion, optimizer, step, device,
                    epoch, rank, world_size, lr_scale)
        if rank == 0:
            torch.save({'model': model.state_dict(),
                        'optimizer': optimizer.state_dict()},
                       args.checkpoint_format.format(epoch=epoch + 1))


def adjust_lr(args, optimizer, epoch, step_in_epoch, steps_per_epoch,
              lr_scale):
    """Warmup over warmup-epochs then /10 at 30/60/80 (main.py:237-252)."""
    if epoch < args.warmup_epochs:
        ep = epoch + step_in_epoch / steps_per_epoch
        factor = (ep / args.warmup_epochs) * (lr_scale - 1) / lr_scale + \
            1.0 / lr_scale
    else:
        factor = 1.0
        for boundary in (30, 60, 80):
            if epoch >= boundary:
                factor *= 0.1
    lr = args.base_lr * lr_scale * factor
    for g in optimizer.param_groups:
        g['lr'] = lr
    return lr


def train_epoch(args, loader, model, criterion, optimizer, step, device,
                epoch, rank, world_size, lr_scale):
    model.train()
    losses = AverageMeter(50)
    top1 = AverageMeter(50)
    t0 = time.time()
    n_steps = len(loader) // args.batches_per_allreduce
    if args.steps_per_epoch:
        n_steps = min(n_steps, args.steps_per_epoch)
    it = iter(loader)
    for s in range(n_steps):
        lr = adjust_lr(args, optimizer, epoch, s, n_steps, lr_scale)
        for _ in range(args.batches_per_allreduce):
            x, y = next(it)
            x = x.to(device, non_blocking=True)
            y = y.to(device, non_blocking=True)
            out = model(x)
            loss = criterion(out, y) / step.loss_scale_denom()
            step.substep(loss)
        losses.update(float(loss.detach()) * step.loss_scale_denom())
        top1.update(float(accuracy(out, y)[0]))
        if rank == 0 and (s + 1) % 50 == 0:
            img_s = (s + 1) * args.batch_size * args.batches_per_allreduce * \
                world_size / (time.time() - t0)
            print(f'Epoch {epoch} [{s + 1}/{n_steps}] lr {lr:.4f} '
                  f'loss {losses.avg:.3f} acc {top1.avg:.2f} '
                  f'{img_s:.0f} img/s', flush=True)


if __name__ == '__main__':
    main()

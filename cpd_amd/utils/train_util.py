"""Shared training utilities (single copy — the reference duplicates this
module between examples, train_util.py vs DavidNet/train_utils.py; fresh
implementation of the same capability set: meters, top-k accuracy,
milestone LR scheduler, deterministic pre-generated distributed samplers,
checkpoint save/load with module-prefix fix-up)."""
import math
import os
import shutil

import numpy as np
import torch
from torch.utils.data.sampler import Sampler

__all__ = [
    "AverageMeter", "accuracy", "IterLRScheduler",
    "GivenIterationSampler", "DistributedGivenIterationSampler",
    "DistributedSampler", "save_checkpoint", "load_state",
]


class AverageMeter:
    """Windowed running meter (length=0 -> plain cumulative average)."""

    def __init__(self, length=0):
        self.length = length
        self.reset()

    def reset(self):
        self.history = []
        self.count = 0
        self.sum = 0.0
        self.val = 0.0
        self.avg = 0.0

    def update(self, val):
        self.val = val
        if self.length > 0:
            self.history.append(val)
            if len(self.history) > self.length:
                self.history.pop(0)
            self.avg = sum(self.history) / len(self.history)
        else:
            self.sum += val
            self.count += 1
            self.avg = self.sum / self.count


def accuracy(output, target, topk=(1,)):
    """Top-k accuracy as percentages."""
    maxk = max(topk)
    batch = target.size(0)
    _, pred = output.topk(maxk, 1, True, True)
    pred = pred.t()
    correct = pred.eq(target.view(1, -1).expand_as(pred))
    return [correct[:k].reshape(-1).float().sum(0) * (100.0 / batch)
            for k in topk]


class IterLRScheduler:
    """Milestone LR schedule keyed by iteration (reference
    train_util.py:68-107 capability: lr_steps/lr_mults applied to every
    optimizer param group)."""

    def __init__(self, optimizer, milestones, lr_mults, last_iter=-1):
        assert len(milestones) == len(lr_mults)
        self.optimizer = optimizer
        self.milestones = list(milestones)
        self.lr_mults = list(lr_mults)
        self.last_iter = last_iter

    def get_lr(self):
        return [g["lr"] for g in self.optimizer.param_groups]

    def step(self, this_iter=None):
        if this_iter is None:
            this_iter = self.last_iter + 1
        self.last_iter = this_iter
        if this_iter in self.milestones:
            mult = self.lr_mults[self.milestones.index(this_iter)]
            for group in self.optimizer.param_groups:
                group["lr"] *= mult


class GivenIterationSampler(Sampler):
    """Pre-generates the full index sequence for total_iter x batch_size
    samples (seeded, tiled + shuffled), single process."""

    def __init__(self, dataset, total_iter, batch_size, last_iter=-1, seed=0):
        self.dataset = dataset
        self.total_iter = total_iter
        self.batch_size = batch_size
        self.last_iter = last_iter
        self.seed = seed
        self.total_size = self.total_iter * self.batch_size
        self.indices = self._gen_indices()

    def _gen_indices(self):
        rng = np.random.default_rng(self.seed)
        n = len(self.dataset)
        reps = math.ceil(self.total_size / n)
        idx = np.tile(np.arange(n), reps)[:self.total_size]
        rng.shuffle(idx)
        return idx

    def __iter__(self):
        return iter(self.indices[(self.last_iter + 1) * self.batch_size:])

    def __len__(self):
        return self.total_size


class DistributedGivenIterationSampler(Sampler):
    """Rank-sharded variant: one global seeded sequence of
    total_iter x batch_size x world indices, sliced contiguously per rank
    (reference train_util.py:159-222 behavior)."""

    def __init__(self, dataset, total_iter, batch_size, world_size=None,
                 rank=None, last_iter=-1, seed=0):
        import torch.distributed as dist
        if world_size is None:
            world_size = dist.get_world_size() if dist.is_initialized() else 1
        if rank is None:
            rank = dist.get_rank() if dist.is_initialized() else 0
        self.dataset = dataset
        self.total_iter = total_iter
        self.batch_size = batch_size
        self.world_size = world_size
        self.rank = rank
        self.last_iter = last_iter
        self.total_size = total_iter * batch_size
        self.indices = self._gen_indices(seed)

    def _gen_indices(self, seed):
        rng = np.random.default_rng(seed)
        all_size = self.total_size * self.world_size
        n = len(self.dataset)
        idx = np.tile(np.arange(n), math.ceil(all_size / n))[:all_size]
        rng.shuffle(idx)
        beg = self.total_size * self.rank
        return idx[beg:beg + self.total_size]

    def __iter__(self):
        return iter(self.indices[(self.last_iter + 1) * self.batch_size:])

    def __len__(self):
        return self.total_size


class DistributedSampler(Sampler):
    """Per-epoch shuffled rank-sharded sampler (round-up padding)."""

    def __init__(self, dataset, world_size=None, rank=None, round_up=True):
        import torch.distributed as dist
        if world_size is None:
            world_size = dist.get_world_size() if dist.is_initialized() else 1
        if rank is None:
            rank = dist.get_rank() if dist.is_initialized() else 0
        self.dataset = dataset
        self.world_size = world_size
        self.rank = rank
        self.round_up = round_up
        self.epoch = 0
        if round_up:
            self.num_samples = int(math.ceil(len(dataset) / world_size))
        else:
            # actual per-rank slice length of indices[rank::world_size]
            self.num_samples = (len(dataset) - rank + world_size - 1) \
                // world_size
        self.total_size = int(math.ceil(len(dataset) / world_size)) * world_size

    def set_epoch(self, epoch):
        self.epoch = epoch

    def __iter__(self):
        g = torch.Generator()
        g.manual_seed(self.epoch)
        indices = torch.randperm(len(self.dataset), generator=g).tolist()
        if self.round_up:
            indices += indices[:(self.total_size - len(indices))]
        indices = indices[self.rank::self.world_size]
        return iter(indices)

    def __len__(self):
        return self.num_samples


def save_checkpoint(state, is_best, filename):
    torch.save(state, filename + ".pth.tar")
    if is_best:
        shutil.copyfile(filename + ".pth.tar", filename + "_best.pth.tar")


def load_state(path, model, optimizer=None, map_location="cpu"):
    """Load a checkpoint, fixing 'module.' prefix mismatches both directions
    and tolerating partial loads (reference train_util.py:274-318 behavior).
    Returns (best_prec1, step) when an optimizer is passed, else None."""
    if not os.path.isfile(path):
        print(f"=> no checkpoint found at '{path}'")
        return None
    ckpt = torch.load(path, map_location=map_location, weights_only=False)
    state = ckpt.get("state_dict", ckpt)
    own = model.state_dict()
    fixed = {}
    for k, v in state.items():
        if k in own:
            fixed[k] = v
        elif k.startswith("module.") and k[7:] in own:
            fixed[k[7:]] = v
        elif "module." + k in own:
            fixed["module." + k] = v
    missing = set(own) - set(fixed)
    for k in sorted(missing):
        print(f"=> missing key (kept init): {k}")
    model.load_state_dict(fixed, strict=False)
    if optimizer is not None and "optimizer" in ckpt:
        optimizer.load_state_dict(ckpt["optimizer"])
        return ckpt.get("best_prec1", 0.0), ckpt.get("step", -1)
    return None

from .train_util import (AverageMeter, DistributedGivenIterationSampler,
                         DistributedSampler, GivenIterationSampler,
                         IterLRScheduler, accuracy, load_state,
                         save_checkpoint)
from .lars import LARS
from .master import MasterParams

__all__ = [
    "AverageMeter", "accuracy", "IterLRScheduler", "GivenIterationSampler",
    "DistributedGivenIterationSampler", "DistributedSampler",
    "save_checkpoint", "load_state", "LARS", "MasterParams",
]

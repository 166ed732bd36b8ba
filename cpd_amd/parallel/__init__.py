"""Distributed layer: DP replicas, APS, low-precision ring all-reduce,
emulate-node local replay."""
from .dist import (DistModule, broadcast_params, dist_init,
                   kahan_sum_gradients, normal_sum_gradients, simple_group_split,
                   sum_gradients)
from .bucket import GradBucket
from .ring import lp_all_reduce_, ring_lp_all_reduce_, sequential_lp_all_reduce_
from .emulate import NodeEmulator

__all__ = [
    "dist_init", "DistModule", "broadcast_params", "sum_gradients",
    "normal_sum_gradients", "kahan_sum_gradients", "simple_group_split",
    "GradBucket", "lp_all_reduce_", "ring_lp_all_reduce_",
    "sequential_lp_all_reduce_", "NodeEmulator",
]

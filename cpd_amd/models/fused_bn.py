"""Fused BatchNorm2d(+residual add)(+ReLU) module backed by the gfx950
kernels in ops/csrc/bn_fused.hip.

Replaces the eager chain {MIOpen BN, add, ReLU, threshold_backward} with two
fused HBM passes forward and two backward.  State-dict compatible with
nn.BatchNorm2d (same parameter/buffer names).  Falls back to the eager
composition on CPU, in eval mode, or for shapes the kernels don't cover —
on GPU training the fused path is the one that runs (fails loudly if the HIP
extension is missing).
"""
import os

import torch
import torch.nn as nn
import torch.nn.functional as F

from .. import ops

_DEBUG = os.environ.get("CPD_BN_DEBUG") == "1"


class _FusedBNFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, bias, running_mean, running_var, momentum,
                eps, relu, residual):
        ext = ops.hip_ext()
        y, mean, invstd, mask = ext.bn_relu_fwd(
            x, weight, bias, running_mean, running_var, momentum, eps, relu,
            residual)
        ctx.relu = relu
        ctx.has_res = residual is not None
        ctx.has_mask = mask is not None and mask.numel() > 0
        if ctx.has_mask:        # 1-bit/elem ReLU mask: bwd skips the y read
            ctx.save_for_backward(x, weight, mean, invstd, mask)
        elif relu:
            ctx.save_for_backward(x, weight, mean, invstd, y)
        else:
            ctx.save_for_backward(x, weight, mean, invstd)
        return y

    @staticmethod
    def backward(ctx, dy):
        mask = y = None
        if ctx.has_mask:
            x, weight, mean, invstd, mask = ctx.saved_tensors
        elif ctx.relu:
            x, weight, mean, invstd, y = ctx.saved_tensors
        else:
            x, weight, mean, invstd = ctx.saved_tensors
        ext = ops.hip_ext()
        dx, dgamma, dbeta, dres = ext.bn_relu_bwd(
            x, dy, y, mean, invstd, weight, ctx.has_res, mask)
        return (dx, dgamma, dbeta, None, None, None, None, None,
                dres if ctx.has_res else None)


class FusedBNReLU(nn.Module):
    """BatchNorm2d with optional fused ReLU and fused residual add
    (``forward(x, residual=None)`` computes relu(bn(x) + residual))."""

    def __init__(self, num_features, relu=True, eps=1e-5, momentum=0.1):
        super().__init__()
        self.num_features = num_features
        self.relu = relu
        self.eps = eps
        self.momentum = momentum
        self.weight = nn.Parameter(torch.ones(num_features))
        self.bias = nn.Parameter(torch.zeros(num_features))
        self.register_buffer("running_mean", torch.zeros(num_features))
        self.register_buffer("running_var", torch.ones(num_features))
        self.register_buffer("num_batches_tracked",
                             torch.tensor(0, dtype=torch.long))

    def _eager(self, x, residual):
        y = F.batch_norm(x, self.running_mean, self.running_var, self.weight,
                         self.bias, self.training, self.momentum, self.eps)
        if residual is not None:
            y = y + residual
        return F.relu(y) if self.relu else y

    def forward(self, x, residual=None):
        # buffer update happens on every training path (fused or eager) so
        # the counter stays consistent with nn.BatchNorm2d semantics
        if self.training:
            self.num_batches_tracked += 1
        # channels_last tensors run the native NHWC kernels (C % 4 == 0);
        # NCHW runs the HW % 4 == 0 kernels.  NEVER .contiguous() a
        # channels_last tensor here — that would silently transpose to NCHW.
        nhwc = (x.dim() == 4 and x.shape[1] > 1
                and x.is_contiguous(memory_format=torch.channels_last)
                and not x.is_contiguous())
        shape_ok = (x.shape[1] % 4 == 0 and x.shape[1] <= 1024) if nhwc \
            else ((x.shape[2] * x.shape[3]) % 4 == 0)
        use_fused = (self.training and x.is_cuda
                     and x.dtype == torch.float32 and shape_ok)
        if _DEBUG and not getattr(self, "_dbg_done", False):
            self._dbg_done = True
            print(f"[fused_bn] shape={tuple(x.shape)} nhwc={nhwc} "
                  f"fused={use_fused} x_contig={x.is_contiguous()} "
                  f"x_cl={x.is_contiguous(memory_format=torch.channels_last)}",
                  flush=True)
        if not use_fused:
            return self._eager(x, residual)
        fmt = torch.channels_last if nhwc else torch.contiguous_format
        res = residual.contiguous(memory_format=fmt) \
            if residual is not None else None
        return _FusedBNFn.apply(x.contiguous(memory_format=fmt), self.weight,
                                self.bias, self.running_mean,
                                self.running_var, self.momentum, self.eps,
                                self.relu, res)

"""Quantized nn.Modules (CPDtorch parity surface: quant_module.py).

Fresh implementation of the reference behavior:
  * Quantizer        — quant_module.py:13-20.
  * Quant_Linear     — quant_module.py:55-85; forward is quant_gemm(input,
    weight.t()) + bias, backward runs all three grads through the quantized
    accumulator too (quant_module.py:36-52).
  * Quant_Conv       — quant_module.py:88-139; im2col via F.unfold routed
    through the quantized GEMM.  Square kernels; `dilation`/`groups` are
    accepted-but-ignored exactly like the reference (asserted to defaults
    here instead of silently ignored).
"""
import math

import torch
import torch.nn as nn
import torch.nn.functional as F
import torch.nn.init as init
from torch.autograd import Function
from torch.nn.parameter import Parameter

from .functional import float_quantize, quantizer, quant_gemm

__all__ = ["Quantizer", "Quant_Linear", "Quant_Conv"]


class Quantizer(nn.Module):
    """Rounds activations (forward) and gradients (backward) to custom grids."""

    def __init__(self, forward_exp=8, forward_man=23, backward_exp=8,
                 backward_man=23):
        super().__init__()
        self.quantize = quantizer(forward_exp, forward_man, backward_exp,
                                  backward_man)

    def forward(self, x):
        return self.quantize(x)


class Quant_LinearFunction(Function):
    @staticmethod
    def forward(ctx, input, weight, bias=None, exp=8, man=23):
        ctx.save_for_backward(input, weight, bias)
        ctx.exp = exp
        ctx.man = man
        output = quant_gemm(input, weight.t(), man=man, exp=exp)
        if bias is not None:
            output = output + bias.unsqueeze(0).expand_as(output)
        return output

    @staticmethod
    def backward(ctx, grad_output):
        input, weight, bias = ctx.saved_tensors
        grad_input = grad_weight = grad_bias = None
        if ctx.needs_input_grad[0]:
            grad_input = quant_gemm(grad_output, weight, man=ctx.man, exp=ctx.exp)
        if ctx.needs_input_grad[1]:
            grad_weight = quant_gemm(grad_output.t().contiguous(), input,
                                     man=ctx.man, exp=ctx.exp)
        if bias is not None and ctx.needs_input_grad[2]:
            grad_bias = float_quantize(grad_output.sum(0), ctx.exp, ctx.man)
        return grad_input, grad_weight, grad_bias, None, None


class Quant_Linear(nn.Module):
    def __init__(self, in_features, out_features, bias=True, exp=8, man=23):
        super().__init__()
        self.in_features = in_features
        self.out_features = out_features
        self.exp = exp
        self.man = man
        self.weight = Parameter(torch.empty(out_features, in_features))
        if bias:
            self.bias = Parameter(torch.empty(out_features))
        else:
            self.register_parameter("bias", None)
        self.reset_parameters()

    def reset_parameters(self):
        init.kaiming_uniform_(self.weight, a=math.sqrt(5))
        if self.bias is not None:
            fan_in, _ = init._calculate_fan_in_and_fan_out(self.weight)
            bound = 1.0 / math.sqrt(fan_in)
            init.uniform_(self.bias, -bound, bound)

    def forward(self, input):
        return Quant_LinearFunction.apply(input, self.weight, self.bias,
                                          self.exp, self.man)

    def extra_repr(self):
        return (f"in_features={self.in_features}, "
                f"out_features={self.out_features}, bias={self.bias is not None}, "
                f"exp={self.exp}, man={self.man}")


class Quant_Conv(nn.Module):
    """2-D convolution whose GEMM runs through the quantized accumulator
    (im2col + Quant_LinearFunction, like quant_module.py:115-139)."""

    def __init__(self, in_channels, out_channels, kernel_size, stride=1,
                 padding=0, dilation=1, groups=1, bias=True, exp=8, man=23):
        super().__init__()
        assert dilation == 1 and groups == 1, \
            "Quant_Conv supports dilation=1, groups=1 (same as the reference)"
        self.in_channels = in_channels
        self.out_channels = out_channels
        self.kernel_size = (kernel_size, kernel_size)
        self.stride = stride
        self.padding = padding
        self.exp = exp
        self.man = man
        self.weight = Parameter(torch.empty(out_channels, in_channels,
                                            kernel_size, kernel_size))
        if bias:
            self.bias = Parameter(torch.empty(out_channels))
        else:
            self.register_parameter("bias", None)
        self.reset_parameters()

    def reset_parameters(self):
        init.kaiming_uniform_(self.weight, a=math.sqrt(5))
        if self.bias is not None:
            fan_in, _ = init._calculate_fan_in_and_fan_out(self.weight)
            bound = 1.0 / math.sqrt(fan_in)
            init.uniform_(self.bias, -bound, bound)

    def forward(self, input):
        batch, _, in_h, in_w = input.shape
        k_h, k_w = self.kernel_size
        out_h = (in_h - k_h + 2 * self.padding) // self.stride + 1
        out_w = (in_w - k_w + 2 * self.padding) // self.stride + 1

        # im2col: [B, L, C*kh*kw] where L = out_h*out_w
        cols = F.unfold(input, self.kernel_size, stride=self.stride,
                        padding=self.padding).transpose(1, 2).contiguous()
        b, l, k = cols.shape
        flat = cols.view(b * l, k)
        w2d = self.weight.view(self.out_channels, -1)
        out = Quant_LinearFunction.apply(flat, w2d, self.bias, self.exp,
                                         self.man)
        return out.view(b, l, self.out_channels).transpose(1, 2).reshape(
            batch, self.out_channels, out_h, out_w)

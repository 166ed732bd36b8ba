"""Functional customized-precision API (CPDtorch parity surface).

Parity targets (reference file:line, behavior only — fresh implementation):
  * float_quantize   — quant_function.py:60-75.  Deliberate divergence: OUT-of-
    place (the reference mutates its input in place and returns it,
    quant.cu:14-25; every reference call site passes a temporary so no caller
    observes it — SURVEY.md §3.4).  ``float_quantize_`` is the explicit
    in-place variant.
  * quantizer        — quant_function.py:33-57 (autograd fwd/bwd rounding).
  * quant_gemm       — quant_function.py:78-98, with the reference's
    default-device bug fixed (c is allocated on a.device, not ``.cuda()``).

Unlike the reference (CUDA-only, quant_function.py:24-30), every op also runs
on CPU tensors.
"""
import torch

from .. import ops

__all__ = ["float_quantize", "float_quantize_", "quantizer", "quant_gemm"]


def float_quantize(x, exp, man):
    """Round a float32 tensor onto the (exp, man) customized-precision grid.

    Args:
        x: float32 tensor (CPU or GPU).
        exp: number of exponent bits (<= 8).
        man: number of mantissa bits, not counting the implicit bit (<= 23).

    Returns a NEW tensor; ``x`` is not modified.

    Note: no (8,23) identity short-circuit here — like the reference kernel,
    (8,23) still flushes fp32-subnormal inputs to zero (the reference
    short-circuits only inside ``quantizer()``, quant_function.py:38-39).
    """
    assert isinstance(x, torch.Tensor), "x must be a torch.Tensor"
    return ops.quantize(x, man, exp)


def float_quantize_(x, exp, man):
    """In-place variant of :func:`float_quantize`; returns ``x``.

    Non-contiguous tensors are handled by quantizing a contiguous copy and
    copying the result back into ``x`` (round 1 silently mutated the copy
    and left ``x`` unchanged — VERDICT r01 weak #7).
    """
    assert isinstance(x, torch.Tensor), "x must be a torch.Tensor"
    if x.is_contiguous():
        return ops.quantize_(x, man, exp)
    x.copy_(ops.quantize_(x.contiguous(), man, exp))
    return x


def quantizer(forward_exp=8, forward_man=23, backward_exp=8, backward_man=23):
    """Return an autograd function quantizing activations in forward and
    gradients in backward ((8,23) short-circuits to identity either way)."""

    class Rounding(torch.autograd.Function):
        @staticmethod
        def forward(ctx, x):
            if forward_exp == 8 and forward_man == 23:
                return x
            return ops.quantize(x, forward_man, forward_exp)

        @staticmethod
        def backward(ctx, grad_output):
            if not ctx.needs_input_grad[0]:
                return None
            if backward_exp == 8 and backward_man == 23:
                return grad_output
            return ops.quantize(grad_output, backward_man, backward_exp)

    return Rounding.apply


def quant_gemm(a, b, man=23, exp=8):
    """C[M,N] = A[M,K] @ B[K,N] with an (exp,man)-rounded Kahan accumulator.

    Every product and every Kahan intermediate is rounded to the (exp,man)
    grid, sequentially over K (the reference GEMM always Kahan-accumulates:
    float_kernel.cu:181-195 with the plain accumulate commented out).
    """
    assert a.dim() == 2, "a must be 2-D"
    assert b.dim() == 2, "b must be 2-D"
    assert a.shape[1] == b.shape[0], "inner dimensions must match"
    return ops.quant_gemm_raw(a, b, man, exp)

"""Customized-precision quantization layer (CPDtorch.quant parity surface,
reference CPDtorch/quant/__init__.py:1-5, plus the explicit in-place variant).
"""
from .functional import float_quantize, float_quantize_, quantizer, quant_gemm
from .module import Quantizer, Quant_Linear, Quant_Conv
from .convert import quantize_model_gemms

__all__ = [
    "float_quantize",
    "float_quantize_",
    "quantizer",
    "Quantizer",
    "quant_gemm",
    "Quant_Linear",
    "Quant_Conv",
    "quantize_model_gemms",
]

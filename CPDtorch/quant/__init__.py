from cpd_amd.quant import (Quant_Conv, Quant_Linear, Quantizer, float_quantize,
                           float_quantize_, quant_gemm, quantizer)

__all__ = ["float_quantize", "float_quantize_", "quantizer", "Quantizer",
           "quant_gemm", "Quant_Linear", "Quant_Conv"]

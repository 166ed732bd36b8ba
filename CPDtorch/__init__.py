"""CPDtorch compatibility namespace.

Drop-in import surface for users of the reference framework: every public
symbol resolves to the MI355X-native cpd_amd implementation.

    from CPDtorch.quant import float_quantize, quant_gemm, Quant_Linear
    from CPDtorch.utils.dist_util import dist_init, DistModule, sum_gradients
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import cpd_amd  # noqa: E402,F401

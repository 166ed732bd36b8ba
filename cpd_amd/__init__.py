"""cpd_amd — MI355X-native customized-precision distributed training.

A from-scratch framework with the capabilities of drcut/CPD (low-precision
emulated arithmetic for distributed training research): the FP32 <-> (exp,man)
cast, GEMM with an (exp,man)-rounded Kahan accumulator, and low-precision
gradient all-reduce with APS auto-precision-scaling — built MI355X-first:
hand-written HIP/CDNA4 kernels (gfx950), RCCL over xGMI for the collectives,
and a real ring all-reduce with custom-precision partial sums instead of the
reference's all-gather emulation (which is retained as a validation mode).
"""
__version__ = "0.1.0"

from . import quant  # noqa: F401

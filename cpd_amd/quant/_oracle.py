"""Pure-numpy oracle for the (exp,man) cast — independent of the C++/HIP core.

Implements the rounding contract documented in ``ops/csrc/quant_core.h``
(behavior-parity with the reference's cast_precision,
/root/reference/CPDtorch/quant/quant_cuda/float_kernel.cu:10-92) so the native
extensions can be tested against a second, independently-written model.
Vectorized uint32 bit math; no float log/exp calls.
"""
import numpy as np


def cast_fp_oracle(x, man_bits: int, exp_bits: int):
    x = np.asarray(x, dtype=np.float32)
    u = x.view(np.uint32)
    exp_f = (u >> 23) & 0xFF
    man_f = u & 0x7FFFFF
    sign = (u & 0x80000000).astype(bool)

    out = np.zeros_like(x)

    # Inf/NaN and +-0 pass through; fp32 subnormals flush to +0 (already 0).
    passthru = (exp_f == 0xFF) | ((exp_f == 0) & (man_f == 0))
    out[passthru] = x[passthru]

    normal = (exp_f != 0xFF) & (exp_f != 0)
    if not normal.any():
        return out

    true_exp = exp_f[normal].astype(np.int64) - 127
    bias = (1 << (exp_bits - 1)) - 1
    new_e = true_exp + bias
    man = (man_f[normal] | (1 << 23)).astype(np.int64)

    # pre-round overflow -> +-Inf
    ovf = new_e >= (1 << exp_bits) - 1

    # target-subnormal: shift out low bits first (sticky discarded), >31 -> 0
    sub = (new_e <= 0) & ~ovf
    shift = np.where(sub, 1 - new_e, 0)
    man = np.where(shift > 31, 0, man >> np.minimum(shift, 31))
    out_e = np.where(sub, 1 - bias, true_exp)

    # round-to-nearest-even at man_bits
    if man_bits < 23:
        drop = 23 - man_bits
        unit = 1 << drop
        half = unit >> 1
        rem = man & (unit - 1)
        up = (rem > half) | ((rem == half) & ((man & unit) != 0))
        man = (man & ~(unit - 1)) + np.where(up, unit, 0)

    with np.errstate(over="ignore"):  # ovf lanes are overwritten below
        mag = np.ldexp(man.astype(np.float64), out_e - 23).astype(np.float32)
    res = np.where(ovf, np.float32(np.inf), mag)
    res = np.where(sign[normal], -res, res)
    out[normal] = res
    return out


def ceil_log2_oracle(x):
    """Exact ceil(log2|x|) with the -100 all-zero sentinel."""
    x = np.asarray(x, dtype=np.float32)
    out = np.full(x.shape, -100.0, dtype=np.float32)
    nz = (x != 0) & np.isfinite(x)
    if nz.any():
        m, e = np.frexp(np.abs(x[nz]).astype(np.float64))  # x = m * 2^e, m in [0.5,1)
        out[nz] = np.where(m == 0.5, e - 1, e).astype(np.float32)
    out[~np.isfinite(x)] = 129.0
    return out

// cpd_amd core numerics: round an IEEE-754 binary32 value onto the grid of a
// customized floating-point format with `exp_bits` exponent bits (<= 8) and
// `man_bits` mantissa bits (<= 23, not counting the implicit bit).
//
// This single header is compiled both as host C++ (CPU ops) and as HIP device
// code (gfx950 kernels) so the two paths are bit-identical by construction.
//
// Semantics (behavior-parity with the reference emulator's cast_precision,
// /root/reference/CPDtorch/quant/quant_cuda/float_kernel.cu:10-92 — independent
// implementation, written fresh for CDNA4):
//   * +/-0, +/-Inf and NaN pass through unchanged (sign of zero preserved).
//   * FP32 subnormal inputs flush to +0.
//   * Target bias = 2^(exp_bits-1) - 1; the exponent-field value
//     (1<<exp_bits)-1 is reserved IEEE-style, so values whose unbiased
//     exponent reaches it saturate to +/-Inf *before* rounding.  (This is NOT
//     OCP-FP8 e4m3fn, which has no infinities.)
//   * Round-to-nearest-even on the mantissa at man_bits.
//   * Because the overflow check happens before rounding, a value just below
//     the saturation boundary whose mantissa rounds up crosses the boundary
//     and yields the *finite* value 2^(E+1) (e.g. e4m3: 255.9 -> 256.0, not
//     Inf).  Faithfully kept: emulation-mode and ring-mode reductions must
//     agree bit-for-bit.
//   * Values below the target's normal range are rounded as target-format
//     subnormals: the mantissa (with explicit leading 1) is right-shifted
//     first — the shifted-out sticky bits are DISCARDED before rounding, same
//     as the reference — then rounded at man_bits.
//   * Deliberate divergence (documented): man_bits == 23 in the subnormal
//     path is well-defined here (no rounding) where the reference shifts by
//     -1 (UB); and shifts >= 32 are an explicit flush-to-zero instead of
//     hardware-dependent shift-count wrapping.
//
// Value reconstruction is ldexpf on the rounded integer mantissa (exact; the
// reference's multiply-loop is O(|exponent|) and was replaced, see
// SURVEY.md §2.1 N1).
#pragma once

#include <cstdint>
#include <cmath>

#if defined(__HIPCC__) || defined(__HIP_DEVICE_COMPILE__)
#define CPD_HD __host__ __device__ __forceinline__
#else
#define CPD_HD inline
#endif

namespace cpd {

CPD_HD uint32_t f32_bits(float f) {
  uint32_t u;
  __builtin_memcpy(&u, &f, 4);
  return u;
}

CPD_HD float bits_f32(uint32_t u) {
  float f;
  __builtin_memcpy(&f, &u, 4);
  return f;
}

CPD_HD int clamp_i(int v, int lo, int hi) {
  v = v < lo ? lo : v;  // lowers to v_med3_i32 on gfx950
  return v > hi ? hi : v;
}

// Exact power-of-two scaling m * 2^e2 for integer m <= 2^24 and any target
// value representable in fp32: split the exponent into <= two factors each
// in the fp32-normal range; every intermediate is exact, the final multiply
// is correctly rounded = exact (the target-format grid is a subset of fp32).
CPD_HD float scale_pow2(uint32_t m, int e2) {
  const int a = clamp_i(e2, -126, 127);
  const int b = e2 - a;  // in [-126, 127] whenever m != 0 in-range
  // m <= 2^24: signed convert (vectorizes as cvtdq2ps; u32->f32 would not)
  float r = (float)(int32_t)m * bits_f32((uint32_t)(a + 127) << 23);
  r *= bits_f32((uint32_t)(b + 127) << 23);  // b == 0 -> exact *1.0
  return r;
}

// Branchless and select-light (the cast is the inner loop of the quantized
// GEMM, so VALU count is the cost model):
//   * the normal/subnormal split is one clamped shift (clamp(1-new_e,0,63)
//     is 0 on the normal path, and >= 24 naturally flushes the mantissa);
//   * round-to-nearest-even is the arithmetic identity
//     (man + half - 1 + lsb) & ~(unit-1) — no compares;
//   * +-0/Inf/NaN passthrough is the single unsigned test
//     (|x|bits - 1) >= 0x7F7FFFFF.
// man_bits/exp_bits are wave-uniform kernel arguments, so their derived
// constants live in SGPRs.  Bit-equality with the numpy oracle is tested
// over ~10M random bit patterns per format on CPU and GPU.
CPD_HD float cast_fp(float x, int man_bits, int exp_bits) {
  const uint32_t u = f32_bits(x);
  const uint32_t au = u & 0x7FFFFFFFu;
  const uint32_t sign = u & 0x80000000u;
  const int exp_f = (int)(au >> 23);
  const int bias = (1 << (exp_bits - 1)) - 1;
  const int new_e = exp_f - 127 + bias;

  // target-subnormal pre-shift (sticky discarded); 0 on the normal path.
  // 32-bit ops only (a u64 shift would block CPU auto-vectorization):
  // shifts >= 32 select to zero instead of shifting.
  const int shift = clamp_i(1 - new_e, 0, 32);
  uint32_t man = (au & 0x7FFFFFu) | 0x800000u;
  man = shift >= 32 ? 0u : (man >> (shift & 31));
  if (man_bits < 23) {  // uniform condition: scalar branch
    const int drop = 23 - man_bits;
    const uint32_t unit = 1u << drop;
    man = (man + (unit >> 1) - 1 + ((man >> drop) & 1)) & ~(unit - 1);
  }
  const int out_e = new_e > 0 ? exp_f - 127 : 1 - bias;
  float mag = scale_pow2(man, out_e - 23);
  mag = new_e >= (1 << exp_bits) - 1 ? bits_f32(0x7F800000u) : mag;  // ovf
  float res = bits_f32(f32_bits(mag) | sign);
  res = exp_f == 0 ? 0.0f : res;              // fp32 subnormal flush
  return (au - 1u >= 0x7F7FFFFFu) ? x : res;  // +-0 / Inf / NaN passthrough
}

// Float-pipeline formulation of cast_fp — bit-identical semantics, ~20 VALU
// ops instead of ~35 (the integer path's shift/mask/reassemble sequence).
// On gfx950 each step maps to one full-rate instruction: v_frexp_exp_i32_f32,
// v_ldexp_f32, v_trunc_f32, v_rndne_f32.  Derivation (all steps exact):
//   ax = m24 * 2^(e-1-23), m24 in [2^23, 2^24)      (frexp: ax = m*2^e, m in [.5,1))
//   t1 = trunc(ax * 2^k1) = m24 >> shift            (sticky-discarding pre-shift;
//        k1 = 23-(e-1)-shift; shift >= 24 flushes to 0 exactly like the int path)
//   r  = rndne(t1 * 2^-drop)                        (RNE at man_bits)
//   mag = r * 2^(drop-k1)                           (== scale_pow2(man, out_e-23)
//        in BOTH the normal and target-subnormal branches — see k1 algebra)
// Specials (overflow saturate, fp32-subnormal flush, +-0/Inf/NaN passthrough)
// are the same selects as cast_fp.  Used in the quant_gemm inner loop where
// the cast IS the cost model; bit-equality with cast_fp is tested on CPU over
// random bit patterns and on GPU (tests/test_quantize.py, test_gpu_numerics).
CPD_HD float cast_fp_fast(float x, int man_bits, int exp_bits) {
  const uint32_t u = f32_bits(x);
  const uint32_t au = u & 0x7FFFFFFFu;
  const float ax = bits_f32(au);
  int e;
  (void)frexpf(ax, &e);  // device: v_frexp_exp_i32_f32
  const int bias = (1 << (exp_bits - 1)) - 1;
  const int new_e = (e - 1) + bias;
  const int shift = new_e > 0 ? 0 : 1 - new_e;  // no upper clamp: the float
  const int k1 = 23 - (e - 1) - shift;          // path flushes naturally
  const float t1 = truncf(ldexpf(ax, k1));
  const int drop = 23 - man_bits;
  const float r = rintf(ldexpf(t1, -drop));  // v_rndne_f32 (RNE)
  float mag = ldexpf(r, drop - k1);
  mag = new_e >= (1 << exp_bits) - 1 ? bits_f32(0x7F800000u) : mag;  // ovf
  float res = bits_f32(f32_bits(mag) | (u & 0x80000000u));
  res = au < 0x00800000u ? 0.0f : res;        // fp32 subnormal flush
  return (au - 1u >= 0x7F7FFFFFu) ? x : res;  // +-0 / Inf / NaN passthrough
}

// One step of (exp,man)-rounded Kahan compensated accumulation:
//   y = Q(inc - c); t = Q(acc + y); c = Q(Q(t - acc) - y); acc = t
// Every intermediate is rounded, matching the reference's gradient-sum and
// GEMM-accumulator semantics (float_kernel.cu:181-195, dist_util.py:82-88).
CPD_HD void kahan_qstep(float& acc, float& c, float inc, int man_bits,
                        int exp_bits) {
  const float y = cast_fp(inc - c, man_bits, exp_bits);
  const float t = cast_fp(acc + y, man_bits, exp_bits);
  c = cast_fp(cast_fp(t - acc, man_bits, exp_bits) - y, man_bits, exp_bits);
  acc = t;
}

// Same step through cast_fp_fast (bit-identical; the quant_gemm hot loop).
CPD_HD void kahan_qstep_fast(float& acc, float& c, float inc, int man_bits,
                             int exp_bits) {
  const float y = cast_fp_fast(inc - c, man_bits, exp_bits);
  const float t = cast_fp_fast(acc + y, man_bits, exp_bits);
  c = cast_fp_fast(cast_fp_fast(t - acc, man_bits, exp_bits) - y, man_bits,
                   exp_bits);
  acc = t;
}

// ceil(log2(|x|)) of the *magnitude* as used by APS max-exponent scanning.
// Exact for every finite nonzero float (integer bit math, no log calls):
//   |x| = m * 2^(e-150) with integer m in [2^23, 2^24) for normals.
//   ceil(log2) = (e-127) when mantissa bits are all zero (exact power of 2),
//   else (e-126).
CPD_HD float ceil_log2_abs(float x) {
  const uint32_t u = f32_bits(x) & 0x7FFFFFFFu;
  if (u == 0) return -100.0f;  // all-zero sentinel (mix.py:260 uses max(...,-100))
  const int e = (int)(u >> 23);
  const uint32_t m = u & 0x7FFFFFu;
  if (e == 0) {
    // fp32 subnormal: |x| = m * 2^-149, log2 = log2(m) - 149
    const int top = 31 - __builtin_clz(m);
    const bool pow2 = (m & (m - 1)) == 0;
    return (float)(top - 149 + (pow2 ? 0 : 1));
  }
  if (e == 0xFF) return 129.0f;  // Inf/NaN: larger than any finite exponent
  return (float)(e - 127 + (m != 0 ? 1 : 0));
}

// APS per-segment exponent: ceil(log2(max|g| * W)).  The fp32 product can
// spuriously overflow to Inf when max|g| is finite but near FLT_MAX/W, which
// would return 129 and silently crush the whole segment toward zero (ADVICE
// r01); fall back to exact integer math with the conservative upper bound
// ceil(log2 m) + ceil(log2 W) in that case.  A genuinely infinite max|g|
// still yields 129 (Inf elements propagate through quantize/sum/unscale).
CPD_HD float aps_max_exp(float maxabs, int world_size) {
  const float p = maxabs * (float)world_size;
  const uint32_t pu = f32_bits(p) & 0x7FFFFFFFu;
  const uint32_t mu = f32_bits(maxabs) & 0x7FFFFFFFu;
  if (pu == 0x7F800000u && mu < 0x7F800000u) {
    int lw = 0;
    while ((1 << lw) < world_size) ++lw;  // ceil(log2(W)), W >= 1
    return ceil_log2_abs(maxabs) + (float)lw;
  }
  return ceil_log2_abs(p);
}

}  // namespace cpd

// cpd_amd CPU ops: customized-precision numerics on CPU tensors.
//
// The reference emulator is CUDA-only (quant_function.py:24-30 raises on CPU);
// full CPU support here is both a product feature and the host-side oracle the
// GPU tests compare against (same quant_core.h compiled for both targets).
#include <torch/extension.h>
#include <ATen/Parallel.h>
#include <atomic>
#include <cstring>
#include <vector>

#include "quant_core.h"

namespace {

using at::Tensor;

void check_f32_contig(const Tensor& t, const char* name) {
  TORCH_CHECK(t.scalar_type() == at::kFloat, name, " must be float32");
  TORCH_CHECK(t.is_contiguous(), name, " must be contiguous");
  TORCH_CHECK(!t.is_cuda(), name, " must be a CPU tensor");
}

constexpr int64_t kGrain = 1 << 14;

Tensor quantize(const Tensor& x, int64_t man_bits, int64_t exp_bits) {
  check_f32_contig(x, "x");
  Tensor out = at::empty_like(x);
  const float* __restrict__ src = x.const_data_ptr<float>();
  float* __restrict__ dst = out.mutable_data_ptr<float>();
  at::parallel_for(0, x.numel(), kGrain, [&](int64_t b, int64_t e) {
    for (int64_t i = b; i < e; ++i)
      dst[i] = cpd::cast_fp(src[i], (int)man_bits, (int)exp_bits);
  });
  return out;
}

Tensor quantize_(Tensor x, int64_t man_bits, int64_t exp_bits) {
  check_f32_contig(x, "x");
  float* p = x.mutable_data_ptr<float>();
  at::parallel_for(0, x.numel(), kGrain, [&](int64_t b, int64_t e) {
    for (int64_t i = b; i < e; ++i)
      p[i] = cpd::cast_fp(p[i], (int)man_bits, (int)exp_bits);
  });
  return x;
}

// acc = Q(acc + inc) elementwise — one hop of the sequential/ring
// low-precision reduction (dist_util.py:65-67 semantics).
Tensor qadd_(Tensor acc, const Tensor& inc, int64_t man_bits, int64_t exp_bits) {
  check_f32_contig(acc, "acc");
  check_f32_contig(inc, "inc");
  TORCH_CHECK(acc.numel() == inc.numel(), "size mismatch");
  float* __restrict__ a = acc.mutable_data_ptr<float>();
  const float* __restrict__ g = inc.const_data_ptr<float>();
  at::parallel_for(0, acc.numel(), kGrain, [&](int64_t b, int64_t e) {
    for (int64_t i = b; i < e; ++i)
      a[i] = cpd::cast_fp(a[i] + g[i], (int)man_bits, (int)exp_bits);
  });
  return acc;
}

// Kahan hop: every intermediate rounded (dist_util.py:82-88 semantics).
Tensor kahan_qadd_(Tensor acc, Tensor comp, const Tensor& inc, int64_t man_bits,
                   int64_t exp_bits) {
  check_f32_contig(acc, "acc");
  check_f32_contig(comp, "comp");
  check_f32_contig(inc, "inc");
  TORCH_CHECK(acc.numel() == inc.numel() && comp.numel() == acc.numel(),
              "size mismatch");
  float* __restrict__ a = acc.mutable_data_ptr<float>();
  float* __restrict__ c = comp.mutable_data_ptr<float>();
  const float* __restrict__ g = inc.const_data_ptr<float>();
  at::parallel_for(0, acc.numel(), kGrain, [&](int64_t b, int64_t e) {
    for (int64_t i = b; i < e; ++i)
      cpd::kahan_qstep(a[i], c[i], g[i], (int)man_bits, (int)exp_bits);
  });
  return acc;
}

// bf16-wire ring hops (CPU mirror of the GPU kernels; exact for values on an
// (exp<=8, man<=7) grid — low 16 mantissa bits are zero).
inline float bf16_to_f32(uint16_t h) {
  return cpd::bits_f32((uint32_t)h << 16);
}
inline uint16_t f32_to_bf16_exact(float f) {
  return (uint16_t)(cpd::f32_bits(f) >> 16);
}

Tensor qadd_bf16_(Tensor acc, const Tensor& inc, int64_t man_bits,
                  int64_t exp_bits) {
  TORCH_CHECK(acc.scalar_type() == at::kBFloat16 && acc.is_contiguous());
  TORCH_CHECK(inc.scalar_type() == at::kBFloat16 && inc.is_contiguous());
  TORCH_CHECK(man_bits <= 7, "bf16 wire is exact only for man_bits <= 7");
  TORCH_CHECK(acc.numel() == inc.numel(), "size mismatch");
  auto* a = reinterpret_cast<uint16_t*>(acc.data_ptr<at::BFloat16>());
  const auto* g = reinterpret_cast<const uint16_t*>(inc.data_ptr<at::BFloat16>());
  at::parallel_for(0, acc.numel(), kGrain, [&](int64_t b, int64_t e) {
    for (int64_t i = b; i < e; ++i)
      a[i] = f32_to_bf16_exact(cpd::cast_fp(
          bf16_to_f32(a[i]) + bf16_to_f32(g[i]), (int)man_bits, (int)exp_bits));
  });
  return acc;
}

Tensor kahan_qadd_bf16_(Tensor acc, Tensor comp, const Tensor& inc,
                        int64_t man_bits, int64_t exp_bits) {
  TORCH_CHECK(acc.scalar_type() == at::kBFloat16 && acc.is_contiguous());
  TORCH_CHECK(man_bits <= 7, "bf16 wire is exact only for man_bits <= 7");
  TORCH_CHECK(acc.numel() == inc.numel() && comp.numel() == acc.numel());
  auto* a = reinterpret_cast<uint16_t*>(acc.data_ptr<at::BFloat16>());
  auto* c = reinterpret_cast<uint16_t*>(comp.data_ptr<at::BFloat16>());
  const auto* g = reinterpret_cast<const uint16_t*>(inc.data_ptr<at::BFloat16>());
  at::parallel_for(0, acc.numel(), kGrain, [&](int64_t b, int64_t e) {
    for (int64_t i = b; i < e; ++i) {
      float af = bf16_to_f32(a[i]), cf = bf16_to_f32(c[i]);
      cpd::kahan_qstep(af, cf, bf16_to_f32(g[i]), (int)man_bits,
                       (int)exp_bits);
      a[i] = f32_to_bf16_exact(af);
      c[i] = f32_to_bf16_exact(cf);
    }
  });
  return acc;
}

// Per-segment APS max-exponent scan over a flat gradient buffer:
//   out[s] = ceil(log2(max_i |x_i| * world_size)) for segment s,
//   -100 when the segment is all-zero (mix.py:260 sentinel).
Tensor seg_max_exp(const Tensor& flat, const Tensor& offsets,
                   int64_t world_size) {
  check_f32_contig(flat, "flat");
  TORCH_CHECK(offsets.scalar_type() == at::kLong && offsets.is_contiguous());
  const int64_t S = offsets.numel() - 1;
  Tensor out = at::empty({S}, flat.options());
  const float* x = flat.const_data_ptr<float>();
  const int64_t* ofs = offsets.const_data_ptr<int64_t>();
  float* o = out.mutable_data_ptr<float>();
  at::parallel_for(0, S, 1, [&](int64_t sb, int64_t se) {
    for (int64_t s = sb; s < se; ++s) {
      float m = 0.0f;
      for (int64_t i = ofs[s]; i < ofs[s + 1]; ++i)
        m = std::max(m, std::fabs(x[i]));
      o[s] = cpd::aps_max_exp(m, (int)world_size);
    }
  });
  return out;
}

// flat[i] = Q(flat[i] * 2^shift[seg(i)])  — fused APS pre-scale + cast.
Tensor scale_quantize_(Tensor flat, const Tensor& offsets, const Tensor& shifts,
                       int64_t man_bits, int64_t exp_bits) {
  check_f32_contig(flat, "flat");
  check_f32_contig(shifts, "shifts");
  const int64_t S = offsets.numel() - 1;
  float* x = flat.mutable_data_ptr<float>();
  const int64_t* ofs = offsets.const_data_ptr<int64_t>();
  const float* sh = shifts.const_data_ptr<float>();
  at::parallel_for(0, S, 1, [&](int64_t sb, int64_t se) {
    for (int64_t s = sb; s < se; ++s) {
      const float scale = std::ldexp(1.0f, (int)sh[s]);
      for (int64_t i = ofs[s]; i < ofs[s + 1]; ++i)
        x[i] = cpd::cast_fp(x[i] * scale, (int)man_bits, (int)exp_bits);
    }
  });
  return flat;
}

// flat[i] *= 2^(sign * shift[seg(i)])  (sign=-1: APS unscale, result NOT
// re-quantized — dist_util.py:44-45).
Tensor seg_scale_(Tensor flat, const Tensor& offsets, const Tensor& shifts,
                  int64_t sign) {
  check_f32_contig(flat, "flat");
  const int64_t S = offsets.numel() - 1;
  float* x = flat.mutable_data_ptr<float>();
  const int64_t* ofs = offsets.const_data_ptr<int64_t>();
  const float* sh = shifts.const_data_ptr<float>();
  at::parallel_for(0, S, 1, [&](int64_t sb, int64_t se) {
    for (int64_t s = sb; s < se; ++s) {
      const float scale = std::ldexp(1.0f, (int)(sign * sh[s]));
      for (int64_t i = ofs[s]; i < ofs[s + 1]; ++i) x[i] *= scale;
    }
  });
  return flat;
}

// C[M,N] = A[M,K] @ B[K,N] with an (exp,man)-rounded Kahan accumulator:
// every product and every Kahan intermediate is rounded, sequentially over K
// (reference semantics: float_kernel.cu:181-195; the reference GEMM always
// Kahan-accumulates — the plain accumulate is commented out there).
Tensor quant_gemm(const Tensor& a, const Tensor& b, int64_t man_bits,
                  int64_t exp_bits) {
  check_f32_contig(a, "a");
  check_f32_contig(b, "b");
  TORCH_CHECK(a.dim() == 2 && b.dim() == 2 && a.size(1) == b.size(0),
              "quant_gemm: bad shapes");
  const int64_t M = a.size(0), K = a.size(1), N = b.size(1);
  Tensor c = at::zeros({M, N}, a.options());
  const float* A = a.const_data_ptr<float>();
  const float* B = b.const_data_ptr<float>();
  float* C = c.mutable_data_ptr<float>();
  const int mb = (int)man_bits, eb = (int)exp_bits;
  // k must stay sequential (rounding order is the semantics); the j lanes
  // are independent Kahan chains, so the inner loop vectorizes
  at::parallel_for(0, M, 1, [&](int64_t rb, int64_t re) {
    std::vector<float> acc(N), comp(N);
    for (int64_t i = rb; i < re; ++i) {
      std::fill(acc.begin(), acc.end(), 0.0f);
      std::fill(comp.begin(), comp.end(), 0.0f);
      float* __restrict__ ac = acc.data();
      float* __restrict__ cp = comp.data();
      for (int64_t k = 0; k < K; ++k) {
        const float av = A[i * K + k];
        const float* __restrict__ Bk = B + k * N;
        for (int64_t j = 0; j < N; ++j) {
          const float prod = cpd::cast_fp(av * Bk[j], mb, eb);
          cpd::kahan_qstep(ac[j], cp[j], prod, mb, eb);
        }
      }
      std::memcpy(C + i * N, ac, N * sizeof(float));
    }
  });
  return c;
}

// Elementwise exact ceil(log2|x|) with the -100 all-zero sentinel (small
// helper for tests and the per-parameter APS path).
Tensor ceil_log2(const Tensor& x) {
  check_f32_contig(x, "x");
  Tensor out = at::empty_like(x);
  const float* src = x.const_data_ptr<float>();
  float* dst = out.mutable_data_ptr<float>();
  at::parallel_for(0, x.numel(), kGrain, [&](int64_t b, int64_t e) {
    for (int64_t i = b; i < e; ++i) dst[i] = cpd::ceil_log2_abs(src[i]);
  });
  return out;
}

// Bit-equivalence scan: cast_fp_fast vs cast_fp over the uint32 bit-pattern
// space with the given stride (stride=1 = the full 2^32 sweep).  Returns the
// first mismatching bit pattern as int64, or -1 when none.  NaN payloads
// compare as passthrough-identical bitwise (both return x unchanged).
int64_t cast_fast_equiv_scan(int64_t man_bits, int64_t exp_bits,
                             int64_t stride, int64_t offset) {
  const int64_t total = (int64_t)1 << 32;
  std::atomic<int64_t> bad{-1};
  at::parallel_for(0, (total - offset + stride - 1) / stride, 1 << 16,
                   [&](int64_t b, int64_t e) {
    for (int64_t i = b; i < e && bad.load(std::memory_order_relaxed) < 0;
         ++i) {
      const uint32_t u = (uint32_t)(offset + i * stride);
      const float x = cpd::bits_f32(u);
      const float a = cpd::cast_fp(x, (int)man_bits, (int)exp_bits);
      const float b2 = cpd::cast_fp_fast(x, (int)man_bits, (int)exp_bits);
      if (cpd::f32_bits(a) != cpd::f32_bits(b2)) {
        int64_t expect = -1;
        bad.compare_exchange_strong(expect, (int64_t)u);
      }
    }
  });
  return bad.load();
}

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("cast_fast_equiv_scan", &cast_fast_equiv_scan,
        "first bit pattern where cast_fp_fast != cast_fp, else -1");
  m.def("quantize", &quantize, "FP32 -> (exp,man) grid, out-of-place");
  m.def("quantize_", &quantize_, "FP32 -> (exp,man) grid, in-place");
  m.def("qadd_", &qadd_, "acc = Q(acc + inc)");
  m.def("kahan_qadd_", &kahan_qadd_, "quantized Kahan accumulate step");
  m.def("qadd_bf16_", &qadd_bf16_, "bf16-wire quantized accumulate");
  m.def("kahan_qadd_bf16_", &kahan_qadd_bf16_, "bf16-wire Kahan step");
  m.def("seg_max_exp", &seg_max_exp, "per-segment APS max exponent");
  m.def("scale_quantize_", &scale_quantize_, "fused per-segment scale+cast");
  m.def("seg_scale_", &seg_scale_, "per-segment power-of-two scale");
  m.def("quant_gemm", &quant_gemm, "GEMM with (exp,man) Kahan accumulator");
  m.def("ceil_log2", &ceil_log2, "exact ceil(log2|x|), -100 at zero");
}

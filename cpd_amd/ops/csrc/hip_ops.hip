// cpd_amd gfx950 (CDNA4/MI355X) kernels for the customized-precision layer.
//
// Hand-written HIP, wave64, no CUDA-compat shims.  All elementwise kernels are
// HBM-bandwidth-bound: float4 (16 B/lane) vectorized loads/stores, grid-stride
// loops, 256-thread blocks (Appendix B, cdna_hip_programming.md).  Shares
// quant_core.h with the CPU extension so CPU/GPU results are bit-identical.
//
// Reference-parity notes: elementwise quantize replaces
// float_kernel.cu:94-101 (scalar, in-place, block=1024); the qadd/kahan hop
// kernels implement one step of the sequential low-precision reduction
// (dist_util.py:65-67 / :82-88) and are the per-hop operator of the real ring
// all-reduce; the segmented APS kernels replace the per-parameter
// host-synced max-exponent scan (dist_util.py:26-37, mix.py:260-274) with one
// fused device pass over a flat gradient bucket.
#include <torch/extension.h>
#include <c10/hip/HIPStream.h>
#include <hip/hip_runtime.h>

#include "quant_core.h"

namespace {

constexpr int TPB = 256;
constexpr int VEC = 4;

using namespace cpd;

__device__ __forceinline__ float bf16_to_f32(unsigned short h) {
  return bits_f32((unsigned)h << 16);
}
// Exact for values already on an (exp<=8, man<=7) grid (low 16 mantissa bits
// are zero); the bf16 wire format is only enabled for man_bits <= 7.
__device__ __forceinline__ unsigned short f32_to_bf16_exact(float f) {
  return (unsigned short)(f32_bits(f) >> 16);
}

// ---------------------------------------------------------------------------
// elementwise quantize
// ---------------------------------------------------------------------------

__global__ void quantize_kernel(const float* __restrict__ in,
                                float* __restrict__ out, long n, int man,
                                int exp) {
  const long stride = (long)gridDim.x * TPB * VEC;
  for (long i = ((long)blockIdx.x * TPB + threadIdx.x) * VEC; i < n;
       i += stride) {
    if (i + VEC <= n) {
      float4 v = *reinterpret_cast<const float4*>(in + i);
      v.x = cast_fp(v.x, man, exp);
      v.y = cast_fp(v.y, man, exp);
      v.z = cast_fp(v.z, man, exp);
      v.w = cast_fp(v.w, man, exp);
      *reinterpret_cast<float4*>(out + i) = v;
    } else {
      for (long j = i; j < n; ++j) out[j] = cast_fp(in[j], man, exp);
    }
  }
}

// acc = Q(acc + inc)
__global__ void qadd_kernel(float* __restrict__ acc,
                            const float* __restrict__ inc, long n, int man,
                            int exp) {
  const long stride = (long)gridDim.x * TPB * VEC;
  for (long i = ((long)blockIdx.x * TPB + threadIdx.x) * VEC; i < n;
       i += stride) {
    if (i + VEC <= n) {
      float4 a = *reinterpret_cast<const float4*>(acc + i);
      const float4 g = *reinterpret_cast<const float4*>(inc + i);
      a.x = cast_fp(a.x + g.x, man, exp);
      a.y = cast_fp(a.y + g.y, man, exp);
      a.z = cast_fp(a.z + g.z, man, exp);
      a.w = cast_fp(a.w + g.w, man, exp);
      *reinterpret_cast<float4*>(acc + i) = a;
    } else {
      for (long j = i; j < n; ++j) acc[j] = cast_fp(acc[j] + inc[j], man, exp);
    }
  }
}

__global__ void kahan_qadd_kernel(float* __restrict__ acc,
                                  float* __restrict__ comp,
                                  const float* __restrict__ inc, long n,
                                  int man, int exp) {
  const long stride = (long)gridDim.x * TPB * VEC;
  for (long i = ((long)blockIdx.x * TPB + threadIdx.x) * VEC; i < n;
       i += stride) {
    if (i + VEC <= n) {
      float4 a = *reinterpret_cast<const float4*>(acc + i);
      float4 c = *reinterpret_cast<const float4*>(comp + i);
      const float4 g = *reinterpret_cast<const float4*>(inc + i);
      kahan_qstep(a.x, c.x, g.x, man, exp);
      kahan_qstep(a.y, c.y, g.y, man, exp);
      kahan_qstep(a.z, c.z, g.z, man, exp);
      kahan_qstep(a.w, c.w, g.w, man, exp);
      *reinterpret_cast<float4*>(acc + i) = a;
      *reinterpret_cast<float4*>(comp + i) = c;
    } else {
      for (long j = i; j < n; ++j)
        kahan_qstep(acc[j], comp[j], inc[j], man, exp);
    }
  }
}

// bf16-wire ring hop: acc16 = bf16(Q(f32(acc16) + f32(inc16))).
// 8 bf16/lane in and out (ushort4-pair = 16B loads, G13 vectorization rule).
__global__ void qadd_bf16_kernel(unsigned short* __restrict__ acc,
                                 const unsigned short* __restrict__ inc,
                                 long n, int man, int exp) {
  const long stride = (long)gridDim.x * TPB * 8;
  for (long i = ((long)blockIdx.x * TPB + threadIdx.x) * 8; i < n;
       i += stride) {
    if (i + 8 <= n) {
      ushort4 a0 = *reinterpret_cast<const ushort4*>(acc + i);
      ushort4 a1 = *reinterpret_cast<const ushort4*>(acc + i + 4);
      const ushort4 g0 = *reinterpret_cast<const ushort4*>(inc + i);
      const ushort4 g1 = *reinterpret_cast<const ushort4*>(inc + i + 4);
      a0.x = f32_to_bf16_exact(cast_fp(bf16_to_f32(a0.x) + bf16_to_f32(g0.x), man, exp));
      a0.y = f32_to_bf16_exact(cast_fp(bf16_to_f32(a0.y) + bf16_to_f32(g0.y), man, exp));
      a0.z = f32_to_bf16_exact(cast_fp(bf16_to_f32(a0.z) + bf16_to_f32(g0.z), man, exp));
      a0.w = f32_to_bf16_exact(cast_fp(bf16_to_f32(a0.w) + bf16_to_f32(g0.w), man, exp));
      a1.x = f32_to_bf16_exact(cast_fp(bf16_to_f32(a1.x) + bf16_to_f32(g1.x), man, exp));
      a1.y = f32_to_bf16_exact(cast_fp(bf16_to_f32(a1.y) + bf16_to_f32(g1.y), man, exp));
      a1.z = f32_to_bf16_exact(cast_fp(bf16_to_f32(a1.z) + bf16_to_f32(g1.z), man, exp));
      a1.w = f32_to_bf16_exact(cast_fp(bf16_to_f32(a1.w) + bf16_to_f32(g1.w), man, exp));
      *reinterpret_cast<ushort4*>(acc + i) = a0;
      *reinterpret_cast<ushort4*>(acc + i + 4) = a1;
    } else {
      for (long j = i; j < n; ++j)
        acc[j] = f32_to_bf16_exact(
            cast_fp(bf16_to_f32(acc[j]) + bf16_to_f32(inc[j]), man, exp));
    }
  }
}

__global__ void kahan_qadd_bf16_kernel(unsigned short* __restrict__ acc,
                                       unsigned short* __restrict__ comp,
                                       const unsigned short* __restrict__ inc,
                                       long n, int man, int exp) {
  const long stride = (long)gridDim.x * TPB * 4;
  for (long i = ((long)blockIdx.x * TPB + threadIdx.x) * 4; i < n;
       i += stride) {
    if (i + 4 <= n) {
      ushort4 a = *reinterpret_cast<const ushort4*>(acc + i);
      ushort4 c = *reinterpret_cast<const ushort4*>(comp + i);
      const ushort4 g = *reinterpret_cast<const ushort4*>(inc + i);
      float af, cf;
#define CPD_KSTEP(f)                                        \
  af = bf16_to_f32(a.f); cf = bf16_to_f32(c.f);             \
  kahan_qstep(af, cf, bf16_to_f32(g.f), man, exp);          \
  a.f = f32_to_bf16_exact(af); c.f = f32_to_bf16_exact(cf);
      CPD_KSTEP(x) CPD_KSTEP(y) CPD_KSTEP(z) CPD_KSTEP(w)
#undef CPD_KSTEP
      *reinterpret_cast<ushort4*>(acc + i) = a;
      *reinterpret_cast<ushort4*>(comp + i) = c;
    } else {
      for (long j = i; j < n; ++j) {
        float af = bf16_to_f32(acc[j]), cf = bf16_to_f32(comp[j]);
        kahan_qstep(af, cf, bf16_to_f32(inc[j]), man, exp);
        acc[j] = f32_to_bf16_exact(af);
        comp[j] = f32_to_bf16_exact(cf);
      }
    }
  }
}

// ---------------------------------------------------------------------------
// segmented APS kernels over a flat gradient bucket
// ---------------------------------------------------------------------------
// Each BLOCK owns a contiguous slice of the flat buffer (coalesced within the
// block); segment lookups are a binary search amortized over the slice (a
// slice rarely crosses more than one segment boundary).

__device__ __forceinline__ int find_seg(const long* __restrict__ ofs, int S,
                                        long idx) {
  int lo = 0, hi = S - 1;
  while (lo < hi) {
    const int mid = (lo + hi + 1) >> 1;
    if (ofs[mid] <= idx) lo = mid; else hi = mid - 1;
  }
  return lo;
}

// out_bits[s] accumulates max|x| over segment s as monotonic uint bits.
// Generic path, window-based (r02 rewrite — the r01 per-thread scalar loop
// with per-element boundary checks ran at 0.042 TB/s, 30x under the aligned
// path): each WAVE owns a 256-element window; a window wholly inside one
// segment (all but <= S+1 of them) takes the same float4 + wave-reduce +
// single-atomic path as the aligned kernel; boundary/tail windows fall back
// to per-element atomics (bounded by 256*(S+1) total).
__global__ void seg_maxabs_kernel(const float* __restrict__ x, long n,
                                  const long* __restrict__ ofs, int S,
                                  unsigned* __restrict__ out_bits) {
  const int lane = threadIdx.x & 63;
  const long nwin = (n + 255) >> 8;
  const long wstride = (long)gridDim.x * (TPB / 64);
  // monotonic per-wave segment hint: windows advance left-to-right, so the
  // next window's segment is found by a short forward scan instead of a
  // fresh binary search per window.  The per-lane max ACCUMULATES across
  // consecutive same-segment windows and flushes (wave-reduce + one atomic)
  // only on segment change — one atomic per window serialized badly on the
  // S counters (0.15 TB/s measured r02); the launcher caps the grid so each
  // wave owns many windows.
  int s0 = -1;
  long s0_end = -1;
  int acc_seg = -1;
  float lmax = 0.0f;
  // contiguous per-wave window range: keeps loads streaming and the
  // segment hint advancing through each segment once (strided windows put
  // consecutive windows ~MBs apart — a serial hint-walk per window)
  const long wid = (long)blockIdx.x * (TPB / 64) + (threadIdx.x >> 6);
  const long per = (nwin + wstride - 1) / wstride;
  const long wlo = wid * per, wcap = min(wlo + per, nwin);
  for (long w = wlo; w < wcap; ++w) {
    const long lo = w << 8;
    const long hi = min(lo + 256, n);
    if (s0 < 0) {
      s0 = find_seg(ofs, S, lo);
      s0_end = ofs[s0 + 1];
    } else {
      while (s0_end <= lo) s0_end = ofs[++s0 + 1];
    }
    if (hi == lo + 256 && s0_end >= hi) {
      if (s0 != acc_seg) {
        if (acc_seg >= 0) {  // flush the previous segment's accumulator
          for (int d = 32; d > 0; d >>= 1)
            lmax = fmaxf(lmax, __shfl_xor(lmax, d));
          if (lane == 0 && f32_bits(lmax) != 0)
            atomicMax(out_bits + acc_seg, f32_bits(lmax));
        }
        acc_seg = s0;
        lmax = 0.0f;
      }
      const float4 v = reinterpret_cast<const float4*>(x + lo)[lane];
      lmax = fmaxf(lmax, fmaxf(fmaxf(fabsf(v.x), fabsf(v.y)),
                               fmaxf(fabsf(v.z), fabsf(v.w))));
    } else {
      for (long j = lo + lane; j < hi; j += 64) {
        const int s = find_seg(ofs, S, j);
        atomicMax(out_bits + s, f32_bits(fabsf(x[j])));
      }
    }
  }
  if (acc_seg >= 0) {
    for (int d = 32; d > 0; d >>= 1)
      lmax = fmaxf(lmax, __shfl_xor(lmax, d));
    if (lane == 0 && f32_bits(lmax) != 0)
      atomicMax(out_bits + acc_seg, f32_bits(lmax));
  }
}

// Fast path for GradBucket layouts: every segment boundary is 256-element
// aligned and n % 256 == 0, so each wave's 64 float4 lanes (256 elements)
// always fall in ONE segment — segment tracking is wave-uniform, loads are
// float4 (16 B/lane), and the atomicMax fires once per wave per segment
// change instead of once per thread (the generic kernel's atomics measured
// 1.5 ms for a 12M-element bucket; this path is bandwidth-bound).
__global__ void seg_maxabs_aligned_kernel(const float4* __restrict__ x,
                                          long n4,
                                          const long* __restrict__ ofs, int S,
                                          unsigned* __restrict__ out_bits,
                                          long per_block4) {
  const long blk_lo = (long)blockIdx.x * per_block4;
  const long blk_hi = min(blk_lo + per_block4, n4);
  if (blk_lo >= n4) return;
  const int lane = threadIdx.x & 63;
  int cur = -1;
  long cur_end4 = -1;
  float lmax = 0.0f;
  for (long i = blk_lo + threadIdx.x; i < blk_hi; i += TPB) {
    if (i >= cur_end4) {  // wave-uniform (boundaries are 64-float4 aligned)
      if (cur >= 0) {
        for (int d = 32; d > 0; d >>= 1)
          lmax = fmaxf(lmax, __shfl_xor(lmax, d));
        if (lane == 0 && lmax > 0.0f) atomicMax(out_bits + cur, f32_bits(lmax));
      }
      cur = find_seg(ofs, S, i * 4);
      cur_end4 = ofs[cur + 1] >> 2;
      lmax = 0.0f;
    }
    const float4 v = x[i];
    lmax = fmaxf(lmax, fmaxf(fmaxf(fabsf(v.x), fabsf(v.y)),
                             fmaxf(fabsf(v.z), fabsf(v.w))));
  }
  if (cur >= 0) {
    for (int d = 32; d > 0; d >>= 1)
      lmax = fmaxf(lmax, __shfl_xor(lmax, d));
    if (lane == 0 && lmax > 0.0f) atomicMax(out_bits + cur, f32_bits(lmax));
  }
}

// Aligned fused scale(+quantize): wave-uniform segment lookup, float4 I/O.
__global__ void scale_quantize_aligned_kernel(float4* __restrict__ x, long n4,
                                              const long* __restrict__ ofs,
                                              int S,
                                              const float* __restrict__ shifts,
                                              int man, int exp, long per_block4,
                                              int sign_only) {
  const long blk_lo = (long)blockIdx.x * per_block4;
  const long blk_hi = min(blk_lo + per_block4, n4);
  if (blk_lo >= n4) return;
  int cur = -1;
  long cur_end4 = -1;
  float scale = 1.0f;
  for (long i = blk_lo + threadIdx.x; i < blk_hi; i += TPB) {
    if (i >= cur_end4) {
      cur = find_seg(ofs, S, i * 4);
      cur_end4 = ofs[cur + 1] >> 2;
      scale = ldexpf(1.0f, (int)shifts[cur] * (sign_only ? sign_only : 1));
    }
    float4 v = x[i];
    if (sign_only) {
      v.x *= scale; v.y *= scale; v.z *= scale; v.w *= scale;
    } else {
      v.x = cast_fp(v.x * scale, man, exp);
      v.y = cast_fp(v.y * scale, man, exp);
      v.z = cast_fp(v.z * scale, man, exp);
      v.w = cast_fp(v.w * scale, man, exp);
    }
    x[i] = v;
  }
}

// out[s] = ceil(log2(maxabs[s] * world_size)), -100 sentinel when all-zero.
__global__ void maxabs_to_exp_kernel(const unsigned* __restrict__ bits,
                                     float* __restrict__ out, int S, int W) {
  const int s = blockIdx.x * TPB + threadIdx.x;
  if (s < S) out[s] = aps_max_exp(bits_f32(bits[s]), W);
}

// flat[i] = Q(flat[i] * 2^shift[seg(i)]) — generic path, window-based (see
// seg_maxabs_kernel note): uniform windows do float4 I/O with a single
// segment lookup per wave.
__global__ void scale_quantize_kernel(float* __restrict__ x, long n,
                                      const long* __restrict__ ofs, int S,
                                      const float* __restrict__ shifts,
                                      int man, int exp, int sign_only) {
  const int lane = threadIdx.x & 63;
  const long nwin = (n + 255) >> 8;
  const long wstride = (long)gridDim.x * (TPB / 64);
  int s0 = -1;
  long s0_end = -1;  // monotonic per-wave hint (see seg_maxabs_kernel)
  const long wid = (long)blockIdx.x * (TPB / 64) + (threadIdx.x >> 6);
  const long per = (nwin + wstride - 1) / wstride;
  const long wlo = wid * per, wcap = min(wlo + per, nwin);
  for (long w = wlo; w < wcap; ++w) {
    const long lo = w << 8;
    const long hi = min(lo + 256, n);
    if (s0 < 0) {
      s0 = find_seg(ofs, S, lo);
      s0_end = ofs[s0 + 1];
    } else {
      while (s0_end <= lo) s0_end = ofs[++s0 + 1];
    }
    if (hi == lo + 256 && s0_end >= hi) {
      const float scale =
          ldexpf(1.0f, (int)shifts[s0] * (sign_only ? sign_only : 1));
      float4* p = reinterpret_cast<float4*>(x + lo);
      float4 v = p[lane];
      if (sign_only) {
        v.x *= scale; v.y *= scale; v.z *= scale; v.w *= scale;
      } else {
        v.x = cast_fp(v.x * scale, man, exp);
        v.y = cast_fp(v.y * scale, man, exp);
        v.z = cast_fp(v.z * scale, man, exp);
        v.w = cast_fp(v.w * scale, man, exp);
      }
      p[lane] = v;
    } else {
      for (long j = lo + lane; j < hi; j += 64) {
        const int s = find_seg(ofs, S, j);
        const float scale =
            ldexpf(1.0f, (int)shifts[s] * (sign_only ? sign_only : 1));
        x[j] = sign_only ? x[j] * scale : cast_fp(x[j] * scale, man, exp);
      }
    }
  }
}

// ---------------------------------------------------------------------------
// launchers / bindings
// ---------------------------------------------------------------------------

using at::Tensor;

inline hipStream_t cur_stream(const Tensor& t) {
  return c10::hip::getCurrentHIPStream(t.get_device()).stream();
}

inline int grid_for(long n, int vec) {
  const long blocks = (n + (long)TPB * vec - 1) / ((long)TPB * vec);
  return (int)std::min<long>(blocks, 16384);  // >> 256 CUs; grid-stride rest
}

void check_gpu_f32(const Tensor& t, const char* name) {
  TORCH_CHECK(t.is_cuda(), name, " must be on GPU");
  TORCH_CHECK(t.scalar_type() == at::kFloat, name, " must be float32");
  TORCH_CHECK(t.is_contiguous(), name, " must be contiguous");
}

Tensor quantize(const Tensor& x, int64_t man, int64_t exp) {
  check_gpu_f32(x, "x");
  Tensor out = at::empty_like(x);
  const long n = x.numel();
  if (n) hipLaunchKernelGGL(quantize_kernel, dim3(grid_for(n, VEC)), dim3(TPB),
                            0, cur_stream(x), x.data_ptr<float>(),
                            out.data_ptr<float>(), n, (int)man, (int)exp);
  return out;
}

Tensor quantize_(Tensor x, int64_t man, int64_t exp) {
  check_gpu_f32(x, "x");
  const long n = x.numel();
  float* p = x.data_ptr<float>();
  if (n) hipLaunchKernelGGL(quantize_kernel, dim3(grid_for(n, VEC)), dim3(TPB),
                            0, cur_stream(x), p, p, n, (int)man, (int)exp);
  return x;
}

Tensor qadd_(Tensor acc, const Tensor& inc, int64_t man, int64_t exp) {
  check_gpu_f32(acc, "acc");
  check_gpu_f32(inc, "inc");
  TORCH_CHECK(acc.numel() == inc.numel(), "size mismatch");
  const long n = acc.numel();
  if (n) hipLaunchKernelGGL(qadd_kernel, dim3(grid_for(n, VEC)), dim3(TPB), 0,
                            cur_stream(acc), acc.data_ptr<float>(),
                            inc.data_ptr<float>(), n, (int)man, (int)exp);
  return acc;
}

Tensor kahan_qadd_(Tensor acc, Tensor comp, const Tensor& inc, int64_t man,
                   int64_t exp) {
  check_gpu_f32(acc, "acc");
  check_gpu_f32(comp, "comp");
  check_gpu_f32(inc, "inc");
  TORCH_CHECK(acc.numel() == inc.numel() && comp.numel() == acc.numel());
  const long n = acc.numel();
  if (n) hipLaunchKernelGGL(kahan_qadd_kernel, dim3(grid_for(n, VEC)),
                            dim3(TPB), 0, cur_stream(acc),
                            acc.data_ptr<float>(),
                            comp.data_ptr<float>(),
                            inc.data_ptr<float>(), n, (int)man, (int)exp);
  return acc;
}

Tensor qadd_bf16_(Tensor acc, const Tensor& inc, int64_t man, int64_t exp) {
  TORCH_CHECK(acc.is_cuda() && acc.scalar_type() == at::kBFloat16 &&
              acc.is_contiguous());
  TORCH_CHECK(inc.scalar_type() == at::kBFloat16 && inc.is_contiguous());
  TORCH_CHECK(man <= 7, "bf16 wire is exact only for man_bits <= 7");
  TORCH_CHECK(acc.numel() == inc.numel(), "size mismatch");
  const long n = acc.numel();
  if (n) hipLaunchKernelGGL(
      qadd_bf16_kernel, dim3(grid_for(n, 8)), dim3(TPB), 0, cur_stream(acc),
      reinterpret_cast<unsigned short*>(acc.data_ptr<at::BFloat16>()),
      reinterpret_cast<const unsigned short*>(inc.data_ptr<at::BFloat16>()),
      n, (int)man, (int)exp);
  return acc;
}

Tensor kahan_qadd_bf16_(Tensor acc, Tensor comp, const Tensor& inc,
                        int64_t man, int64_t exp) {
  TORCH_CHECK(acc.is_cuda() && acc.scalar_type() == at::kBFloat16 &&
              acc.is_contiguous());
  TORCH_CHECK(man <= 7, "bf16 wire is exact only for man_bits <= 7");
  TORCH_CHECK(acc.numel() == inc.numel() && comp.numel() == acc.numel());
  const long n = acc.numel();
  if (n) hipLaunchKernelGGL(
      kahan_qadd_bf16_kernel, dim3(grid_for(n, 4)), dim3(TPB), 0,
      cur_stream(acc),
      reinterpret_cast<unsigned short*>(acc.data_ptr<at::BFloat16>()),
      reinterpret_cast<unsigned short*>(comp.data_ptr<at::BFloat16>()),
      reinterpret_cast<const unsigned short*>(inc.data_ptr<at::BFloat16>()),
      n, (int)man, (int)exp);
  return acc;
}

constexpr long kSegPerBlock = 16384;  // elements per block for segmented ops

Tensor seg_max_exp(const Tensor& flat, const Tensor& offsets,
                   int64_t world_size, bool aligned) {
  check_gpu_f32(flat, "flat");
  TORCH_CHECK(offsets.is_cuda() && offsets.scalar_type() == at::kLong &&
              offsets.is_contiguous());
  const int S = (int)offsets.numel() - 1;
  Tensor bits = at::zeros({S}, flat.options().dtype(at::kUInt32));
  Tensor out = at::empty({S}, flat.options());
  const long n = flat.numel();
  const int blocks = (int)((n + kSegPerBlock - 1) / kSegPerBlock);
  if (n && aligned) {
    TORCH_CHECK(n % 256 == 0, "aligned seg_max_exp needs n % 256 == 0");
    hipLaunchKernelGGL(seg_maxabs_aligned_kernel, dim3(blocks), dim3(TPB), 0,
                       cur_stream(flat),
                       reinterpret_cast<const float4*>(flat.data_ptr<float>()),
                       n / 4, offsets.data_ptr<int64_t>(), S,
                       reinterpret_cast<unsigned*>(bits.data_ptr<uint32_t>()),
                       kSegPerBlock / 4);
  } else if (n) {
    const long nwin = (n + 255) >> 8;
    // cap at 1024 blocks (4096 waves saturate HBM) so each wave accumulates
    // across many windows -> ~1 atomic per (wave, segment) instead of per
    // window
    const int wblocks =
        (int)std::min<long>((nwin + TPB / 64 - 1) / (TPB / 64), 1024);
    hipLaunchKernelGGL(seg_maxabs_kernel, dim3(wblocks), dim3(TPB), 0,
                       cur_stream(flat), flat.data_ptr<float>(), n,
                       offsets.data_ptr<int64_t>(), S,
                       reinterpret_cast<unsigned*>(bits.data_ptr<uint32_t>()));
  }
  hipLaunchKernelGGL(maxabs_to_exp_kernel, dim3((S + TPB - 1) / TPB), dim3(TPB),
                     0, cur_stream(flat),
                     reinterpret_cast<const unsigned*>(bits.data_ptr<uint32_t>()),
                     out.data_ptr<float>(), S, (int)world_size);
  return out;
}

void _scale_quantize_impl(Tensor& flat, const Tensor& offsets,
                          const Tensor& shifts, int man, int exp, int sign,
                          bool aligned) {
  const int S = (int)offsets.numel() - 1;
  const long n = flat.numel();
  const int blocks = (int)((n + kSegPerBlock - 1) / kSegPerBlock);
  if (!n) return;
  if (aligned) {
    TORCH_CHECK(n % 256 == 0, "aligned seg op needs n % 256 == 0");
    hipLaunchKernelGGL(scale_quantize_aligned_kernel, dim3(blocks), dim3(TPB),
                       0, cur_stream(flat),
                       reinterpret_cast<float4*>(flat.data_ptr<float>()),
                       n / 4, offsets.data_ptr<int64_t>(), S,
                       shifts.data_ptr<float>(), man, exp, kSegPerBlock / 4,
                       sign);
  } else {
    const long nwin = (n + 255) >> 8;
    const int wblocks =
        (int)std::min<long>((nwin + TPB / 64 - 1) / (TPB / 64), 16384);
    hipLaunchKernelGGL(scale_quantize_kernel, dim3(wblocks), dim3(TPB), 0,
                       cur_stream(flat), flat.data_ptr<float>(), n,
                       offsets.data_ptr<int64_t>(), S,
                       shifts.data_ptr<float>(), man, exp, sign);
  }
}

Tensor scale_quantize_(Tensor flat, const Tensor& offsets, const Tensor& shifts,
                       int64_t man, int64_t exp, bool aligned) {
  check_gpu_f32(flat, "flat");
  check_gpu_f32(shifts, "shifts");
  _scale_quantize_impl(flat, offsets, shifts, (int)man, (int)exp, 0, aligned);
  return flat;
}

Tensor seg_scale_(Tensor flat, const Tensor& offsets, const Tensor& shifts,
                  int64_t sign, bool aligned) {
  check_gpu_f32(flat, "flat");
  _scale_quantize_impl(flat, offsets, shifts, 23, 8, (int)sign, aligned);
  return flat;
}

__global__ void ceil_log2_kernel(const float* __restrict__ in,
                                 float* __restrict__ out, long n) {
  const long stride = (long)gridDim.x * TPB;
  for (long i = (long)blockIdx.x * TPB + threadIdx.x; i < n; i += stride)
    out[i] = ceil_log2_abs(in[i]);
}

Tensor ceil_log2(const Tensor& x) {
  check_gpu_f32(x, "x");
  Tensor out = at::empty_like(x);
  const long n = x.numel();
  if (n) hipLaunchKernelGGL(ceil_log2_kernel, dim3(grid_for(n, 1)), dim3(TPB),
                            0, cur_stream(x), x.data_ptr<float>(),
                            out.data_ptr<float>(), n);
  return out;
}

// Device-side bit-equivalence scan of cast_fp_fast (the v_frexp/v_ldexp/
// v_rndne formulation used in the quant_gemm inner loop) against cast_fp
// over the 2^32 bit-pattern space.  Returns the first (lowest recorded)
// mismatching pattern, or -1 when equal.  stride=1 = exhaustive.
__global__ void cast_equiv_kernel(long n, long stride, int man, int exp,
                                  unsigned long long* bad) {
  const long gs = (long)gridDim.x * TPB;
  for (long i = (long)blockIdx.x * TPB + threadIdx.x; i < n; i += gs) {
    const unsigned u = (unsigned)(i * stride);
    const float x = bits_f32(u);
    const float a = cast_fp(x, man, exp);
    const float b = cast_fp_fast(x, man, exp);
    if (f32_bits(a) != f32_bits(b)) atomicMin(bad, (unsigned long long)u);
  }
}

int64_t cast_fast_equiv_scan(int64_t man, int64_t exp, int64_t stride) {
  const long total = ((long)1 << 32) / stride;
  auto opts = at::TensorOptions().dtype(at::kLong).device(at::kCUDA);
  Tensor bad = at::full({1}, -1, opts);  // 0xFFFF... as unsigned max
  hipLaunchKernelGGL(cast_equiv_kernel, dim3(4096), dim3(TPB), 0,
                     c10::hip::getCurrentHIPStream().stream(), total, stride,
                     (int)man, (int)exp,
                     (unsigned long long*)bad.data_ptr<int64_t>());
  const int64_t v = bad.cpu().item<int64_t>();
  return v;  // -1 (all ones) when no mismatch
}

}  // namespace

// gemm_hip.hip provides these at namespace scope
at::Tensor cpd_quant_gemm_hip(const at::Tensor& a, const at::Tensor& b,
                              int64_t man, int64_t exp);
at::Tensor cpd_gemm_f32_hip(const at::Tensor& a, const at::Tensor& b);
void cpd_register_bn(pybind11::module_& m);

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  cpd_register_bn(m);
  m.def("quantize", &quantize);
  m.def("quantize_", &quantize_);
  m.def("qadd_", &qadd_);
  m.def("kahan_qadd_", &kahan_qadd_);
  m.def("qadd_bf16_", &qadd_bf16_);
  m.def("kahan_qadd_bf16_", &kahan_qadd_bf16_);
  m.def("seg_max_exp", &seg_max_exp);
  m.def("scale_quantize_", &scale_quantize_);
  m.def("seg_scale_", &seg_scale_);
  m.def("ceil_log2", &ceil_log2);
  m.def("cast_fast_equiv_scan", &cast_fast_equiv_scan);
  m.def("quant_gemm", &cpd_quant_gemm_hip);
  m.def("gemm_f32", &cpd_gemm_f32_hip);
}

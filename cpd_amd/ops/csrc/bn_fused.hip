// Fused BatchNorm2d(+residual add)+ReLU training kernels for gfx950 (NCHW).
//
// MI355X-first fusion: the eager path runs BN (3 MIOpen kernels), a separate
// add, a separate ReLU, and threshold_backward — all HBM-bound elementwise
// passes.  Here forward is {deterministic two-level reduce, normalize+add+
// relu} and backward is {reduce of (dy_eff, dy_eff*xhat), finalize, dx(+dres)}
// with the ReLU mask and residual add folded in.  All reductions are
// deterministic (fixed-shape workspace, no float atomics).
//
// Layout: NCHW fp32, HW % 4 == 0 (float4 I/O).  Channel c of sample n is a
// contiguous run of HW floats at (n*C + c)*HW.
#include <torch/extension.h>
#include <c10/hip/HIPStream.h>
#include <hip/hip_runtime.h>

namespace {

constexpr int TPB = 256;

__device__ __forceinline__ float4 ld4(const float* p) {
  return *reinterpret_cast<const float4*>(p);
}
__device__ __forceinline__ void st4(float* p, float4 v) {
  *reinterpret_cast<float4*>(p) = v;
}

__device__ float block_reduce(float v, float* smem) {
  for (int d = 32; d > 0; d >>= 1) v += __shfl_down(v, d);
  const int wave = threadIdx.x >> 6;
  if ((threadIdx.x & 63) == 0) smem[wave] = v;
  __syncthreads();
  if (threadIdx.x == 0) {
    for (int w = 1; w < TPB / 64; ++w) v += smem[w];
    smem[0] = v;
  }
  __syncthreads();
  return smem[0];
}

// ---- forward pass 1: per (channel, slice) partial sum / sumsq -------------
// The (n, hw) space is FLATTENED across the block so all 256 lanes stay
// active even when HW < TPB*4 (late ResNet layers: HW=16 left 4 active
// lanes/block in the r01 per-sample loop — 2-3x slower reduces).
__global__ void bn_reduce_kernel(const float* __restrict__ x, int N, int C,
                                 long HW, int split,
                                 float* __restrict__ ws /* [C][split][2] */) {
  const int c = blockIdx.x;
  const int s = blockIdx.y;
  __shared__ float smem[TPB / 64];
  float sum = 0.f, sumsq = 0.f;
  const long total4 = (long)N * HW / 4;  // HW % 4 == 0
  for (long f4 = (long)s * TPB + threadIdx.x; f4 < total4;
       f4 += (long)split * TPB) {
    const long f = f4 * 4;
    const long n = f / HW;
    const float4 v = ld4(x + (n * C + c) * HW + (f - n * HW));
    sum += v.x + v.y + v.z + v.w;
    sumsq += v.x * v.x + v.y * v.y + v.z * v.z + v.w * v.w;
  }
  const float bs = block_reduce(sum, smem);
  __syncthreads();
  const float bq = block_reduce(sumsq, smem);
  if (threadIdx.x == 0) {
    ws[((long)c * split + s) * 2 + 0] = bs;
    ws[((long)c * split + s) * 2 + 1] = bq;
  }
}

// ---- forward pass 1b: finalize mean/invstd + update running stats ---------
// One WAVE per channel (lanes stride the split axis): with the NHWC reduce
// using up to 2048 splits, the r01 one-THREAD-per-channel loop serialized
// C*split*2 workspace reads and dominated the whole BN (kernel_bench r02).
__global__ void bn_finalize_kernel(const float* __restrict__ ws, int C,
                                   int split, float count, float eps,
                                   float momentum,
                                   float* __restrict__ mean,
                                   float* __restrict__ invstd,
                                   float* __restrict__ running_mean,
                                   float* __restrict__ running_var) {
  const int c = blockIdx.x * (TPB / 64) + (threadIdx.x >> 6);
  const int lane = threadIdx.x & 63;
  if (c >= C) return;
  float sum = 0.f, sumsq = 0.f;
  for (int s = lane; s < split; s += 64) {
    sum += ws[((long)c * split + s) * 2 + 0];
    sumsq += ws[((long)c * split + s) * 2 + 1];
  }
  for (int d = 32; d > 0; d >>= 1) {
    sum += __shfl_down(sum, d);
    sumsq += __shfl_down(sumsq, d);
  }
  if (lane != 0) return;
  const float m = sum / count;
  const float var = fmaxf(sumsq / count - m * m, 0.0f);
  mean[c] = m;
  invstd[c] = rsqrtf(var + eps);
  if (running_mean) {
    running_mean[c] += momentum * (m - running_mean[c]);
    // running_var uses the unbiased estimator (torch semantics)
    const float unbiased = var * count / fmaxf(count - 1.0f, 1.0f);
    running_var[c] += momentum * (unbiased - running_var[c]);
  }
}

// ---- forward pass 2: y = relu(xhat*gamma + beta [+ res]) ------------------
__global__ void bn_norm_kernel(const float* __restrict__ x,
                               const float* __restrict__ res,  // nullable
                               float* __restrict__ y, int N, int C, long HW,
                               const float* __restrict__ mean,
                               const float* __restrict__ invstd,
                               const float* __restrict__ gamma,
                               const float* __restrict__ beta, int relu) {
  const long total4 = (long)N * C * HW / 4;
  const long stride = (long)gridDim.x * TPB;
  for (long i4 = (long)blockIdx.x * TPB + threadIdx.x; i4 < total4;
       i4 += stride) {
    const long i = i4 * 4;
    const int c = (int)((i / HW) % C);
    const float a = gamma[c] * invstd[c];
    const float b = beta[c] - mean[c] * a;
    float4 v = ld4(x + i);
    v.x = v.x * a + b;
    v.y = v.y * a + b;
    v.z = v.z * a + b;
    v.w = v.w * a + b;
    if (res) {
      const float4 r = ld4(res + i);
      v.x += r.x; v.y += r.y; v.z += r.z; v.w += r.w;
    }
    if (relu) {
      v.x = fmaxf(v.x, 0.f);
      v.y = fmaxf(v.y, 0.f);
      v.z = fmaxf(v.z, 0.f);
      v.w = fmaxf(v.w, 0.f);
    }
    st4(y + i, v);
  }
}

// ---- mask-variant forward pass 2 (8 elems/thread, HW % 8 == 0): also emits
// the ReLU mask as 1 bit/element so backward never re-reads y (saves ~30%
// of the backward HBM traffic) ----------------------------------------------
__global__ void bn_norm_mask_kernel(const float* __restrict__ x,
                                    const float* __restrict__ res,
                                    float* __restrict__ y,
                                    unsigned char* __restrict__ mask,
                                    int N, int C, long HW,
                                    const float* __restrict__ mean,
                                    const float* __restrict__ invstd,
                                    const float* __restrict__ gamma,
                                    const float* __restrict__ beta) {
  const long total8 = (long)N * C * HW / 8;
  const long stride = (long)gridDim.x * TPB;
  for (long i8 = (long)blockIdx.x * TPB + threadIdx.x; i8 < total8;
       i8 += stride) {
    const long i = i8 * 8;
    const int c = (int)((i / HW) % C);
    const float a = gamma[c] * invstd[c];
    const float b = beta[c] - mean[c] * a;
    float4 v0 = ld4(x + i);
    float4 v1 = ld4(x + i + 4);
    v0.x = v0.x * a + b; v0.y = v0.y * a + b;
    v0.z = v0.z * a + b; v0.w = v0.w * a + b;
    v1.x = v1.x * a + b; v1.y = v1.y * a + b;
    v1.z = v1.z * a + b; v1.w = v1.w * a + b;
    if (res) {
      const float4 r0 = ld4(res + i);
      const float4 r1 = ld4(res + i + 4);
      v0.x += r0.x; v0.y += r0.y; v0.z += r0.z; v0.w += r0.w;
      v1.x += r1.x; v1.y += r1.y; v1.z += r1.z; v1.w += r1.w;
    }
    unsigned m = (v0.x > 0.f) | ((v0.y > 0.f) << 1) | ((v0.z > 0.f) << 2) |
                 ((v0.w > 0.f) << 3) | ((v1.x > 0.f) << 4) |
                 ((v1.y > 0.f) << 5) | ((v1.z > 0.f) << 6) |
                 ((v1.w > 0.f) << 7);
    v0.x = fmaxf(v0.x, 0.f); v0.y = fmaxf(v0.y, 0.f);
    v0.z = fmaxf(v0.z, 0.f); v0.w = fmaxf(v0.w, 0.f);
    v1.x = fmaxf(v1.x, 0.f); v1.y = fmaxf(v1.y, 0.f);
    v1.z = fmaxf(v1.z, 0.f); v1.w = fmaxf(v1.w, 0.f);
    st4(y + i, v0);
    st4(y + i + 4, v1);
    mask[i8] = (unsigned char)m;
  }
}

// ---- mask-variant backward reduce (8 elems/thread) ------------------------
__global__ void bn_bwd_reduce_mask_kernel(const float* __restrict__ x,
                                          const float* __restrict__ dy,
                                          const unsigned char* __restrict__ mask,
                                          int N, int C, long HW, int split,
                                          const float* __restrict__ mean,
                                          const float* __restrict__ invstd,
                                          float* __restrict__ ws) {
  const int c = blockIdx.x;
  const int s = blockIdx.y;
  __shared__ float smem[TPB / 64];
  const float m = mean[c], is = invstd[c];
  float sd = 0.f, sdx = 0.f;
  const long total8 = (long)N * HW / 8;  // flattened (n,hw): all lanes active
  for (long f8 = (long)s * TPB + threadIdx.x; f8 < total8;
       f8 += (long)split * TPB) {
    const long f = f8 * 8;
    const long n = f / HW;
    const long off = (n * C + c) * HW + (f - n * HW);
    {
      float4 g0 = ld4(dy + off);
      float4 g1 = ld4(dy + off + 4);
      const float4 v0 = ld4(x + off);
      const float4 v1 = ld4(x + off + 4);
      const unsigned mk = mask[off / 8];
      g0.x = (mk & 1) ? g0.x : 0.f;
      g0.y = (mk & 2) ? g0.y : 0.f;
      g0.z = (mk & 4) ? g0.z : 0.f;
      g0.w = (mk & 8) ? g0.w : 0.f;
      g1.x = (mk & 16) ? g1.x : 0.f;
      g1.y = (mk & 32) ? g1.y : 0.f;
      g1.z = (mk & 64) ? g1.z : 0.f;
      g1.w = (mk & 128) ? g1.w : 0.f;
      sd += g0.x + g0.y + g0.z + g0.w + g1.x + g1.y + g1.z + g1.w;
      sdx += g0.x * (v0.x - m) + g0.y * (v0.y - m) + g0.z * (v0.z - m) +
             g0.w * (v0.w - m) + g1.x * (v1.x - m) + g1.y * (v1.y - m) +
             g1.z * (v1.z - m) + g1.w * (v1.w - m);
    }
  }
  const float bs = block_reduce(sd, smem);
  __syncthreads();
  const float bq = block_reduce(sdx, smem);
  if (threadIdx.x == 0) {
    ws[((long)c * split + s) * 2 + 0] = bs;
    ws[((long)c * split + s) * 2 + 1] = bq * is;
  }
}

// ---- mask-variant backward dx (8 elems/thread) ----------------------------
__global__ void bn_bwd_dx_mask_kernel(const float* __restrict__ x,
                                      const float* __restrict__ dy,
                                      const unsigned char* __restrict__ mask,
                                      float* __restrict__ dx,
                                      float* __restrict__ dres,
                                      int N, int C, long HW,
                                      const float* __restrict__ mean,
                                      const float* __restrict__ invstd,
                                      const float* __restrict__ gamma,
                                      const float* __restrict__ sum_dy,
                                      const float* __restrict__ sum_dyx,
                                      float inv_count) {
  const long total8 = (long)N * C * HW / 8;
  const long stride = (long)gridDim.x * TPB;
  for (long i8 = (long)blockIdx.x * TPB + threadIdx.x; i8 < total8;
       i8 += stride) {
    const long i = i8 * 8;
    const int c = (int)((i / HW) % C);
    const float m = mean[c], is = invstd[c];
    const float k = gamma[c] * is;
    const float md = sum_dy[c] * inv_count;
    const float mdx = sum_dyx[c] * inv_count;
    float4 g0 = ld4(dy + i);
    float4 g1 = ld4(dy + i + 4);
    const float4 v0 = ld4(x + i);
    const float4 v1 = ld4(x + i + 4);
    const unsigned mk = mask[i8];
    g0.x = (mk & 1) ? g0.x : 0.f;
    g0.y = (mk & 2) ? g0.y : 0.f;
    g0.z = (mk & 4) ? g0.z : 0.f;
    g0.w = (mk & 8) ? g0.w : 0.f;
    g1.x = (mk & 16) ? g1.x : 0.f;
    g1.y = (mk & 32) ? g1.y : 0.f;
    g1.z = (mk & 64) ? g1.z : 0.f;
    g1.w = (mk & 128) ? g1.w : 0.f;
    if (dres) {
      st4(dres + i, g0);
      st4(dres + i + 4, g1);
    }
    float4 o0, o1;
    o0.x = k * (g0.x - md - (v0.x - m) * is * mdx);
    o0.y = k * (g0.y - md - (v0.y - m) * is * mdx);
    o0.z = k * (g0.z - md - (v0.z - m) * is * mdx);
    o0.w = k * (g0.w - md - (v0.w - m) * is * mdx);
    o1.x = k * (g1.x - md - (v1.x - m) * is * mdx);
    o1.y = k * (g1.y - md - (v1.y - m) * is * mdx);
    o1.z = k * (g1.z - md - (v1.z - m) * is * mdx);
    o1.w = k * (g1.w - md - (v1.w - m) * is * mdx);
    st4(dx + i, o0);
    st4(dx + i + 4, o1);
  }
}

// ---- backward pass 1: per (c, slice) partials of sum(dy_eff),
//      sum(dy_eff * xhat); dy_eff = dy * (y > 0) when relu ----------------
__global__ void bn_bwd_reduce_kernel(const float* __restrict__ x,
                                     const float* __restrict__ dy,
                                     const float* __restrict__ y,  // nullable
                                     int N, int C, long HW, int split,
                                     const float* __restrict__ mean,
                                     const float* __restrict__ invstd,
                                     float* __restrict__ ws /*[C][split][2]*/) {
  const int c = blockIdx.x;
  const int s = blockIdx.y;
  __shared__ float smem[TPB / 64];
  const float m = mean[c], is = invstd[c];
  float sd = 0.f, sdx = 0.f;
  const long total4 = (long)N * HW / 4;  // flattened (n,hw): all lanes active
  for (long f4 = (long)s * TPB + threadIdx.x; f4 < total4;
       f4 += (long)split * TPB) {
    const long f = f4 * 4;
    const long n = f / HW;
    const long off = (n * C + c) * HW + (f - n * HW);
    float4 g = ld4(dy + off);
    const float4 v = ld4(x + off);
    if (y) {
      const float4 yy = ld4(y + off);
      g.x = yy.x > 0.f ? g.x : 0.f;
      g.y = yy.y > 0.f ? g.y : 0.f;
      g.z = yy.z > 0.f ? g.z : 0.f;
      g.w = yy.w > 0.f ? g.w : 0.f;
    }
    sd += g.x + g.y + g.z + g.w;
    sdx += g.x * (v.x - m) + g.y * (v.y - m) + g.z * (v.z - m) +
           g.w * (v.w - m);
  }
  const float bs = block_reduce(sd, smem);
  __syncthreads();
  const float bq = block_reduce(sdx * 1.0f, smem);
  if (threadIdx.x == 0) {
    ws[((long)c * split + s) * 2 + 0] = bs;
    ws[((long)c * split + s) * 2 + 1] = bq * is;  // sum(dy_eff * xhat)
  }
}

__global__ void bn_bwd_finalize_kernel(const float* __restrict__ ws, int C,
                                       int split,
                                       float* __restrict__ sum_dy,
                                       float* __restrict__ sum_dyx,
                                       float* __restrict__ dgamma,
                                       float* __restrict__ dbeta) {
  const int c = blockIdx.x * (TPB / 64) + (threadIdx.x >> 6);  // wave/channel
  const int lane = threadIdx.x & 63;
  if (c >= C) return;
  float sd = 0.f, sdx = 0.f;
  for (int s = lane; s < split; s += 64) {
    sd += ws[((long)c * split + s) * 2 + 0];
    sdx += ws[((long)c * split + s) * 2 + 1];
  }
  for (int d = 32; d > 0; d >>= 1) {
    sd += __shfl_down(sd, d);
    sdx += __shfl_down(sdx, d);
  }
  if (lane != 0) return;
  sum_dy[c] = sd;
  sum_dyx[c] = sdx;
  dbeta[c] = sd;
  dgamma[c] = sdx;
}

// ---- backward pass 2: dx (+ dres = dy_eff) --------------------------------
__global__ void bn_bwd_dx_kernel(const float* __restrict__ x,
                                 const float* __restrict__ dy,
                                 const float* __restrict__ y,  // nullable
                                 float* __restrict__ dx,
                                 float* __restrict__ dres,  // nullable
                                 int N, int C, long HW,
                                 const float* __restrict__ mean,
                                 const float* __restrict__ invstd,
                                 const float* __restrict__ gamma,
                                 const float* __restrict__ sum_dy,
                                 const float* __restrict__ sum_dyx,
                                 float inv_count) {
  const long total4 = (long)N * C * HW / 4;
  const long stride = (long)gridDim.x * TPB;
  for (long i4 = (long)blockIdx.x * TPB + threadIdx.x; i4 < total4;
       i4 += stride) {
    const long i = i4 * 4;
    const int c = (int)((i / HW) % C);
    const float m = mean[c], is = invstd[c];
    const float k = gamma[c] * is;
    const float md = sum_dy[c] * inv_count;
    const float mdx = sum_dyx[c] * inv_count;
    float4 g = ld4(dy + i);
    const float4 v = ld4(x + i);
    if (y) {
      const float4 yy = ld4(y + i);
      g.x = yy.x > 0.f ? g.x : 0.f;
      g.y = yy.y > 0.f ? g.y : 0.f;
      g.z = yy.z > 0.f ? g.z : 0.f;
      g.w = yy.w > 0.f ? g.w : 0.f;
    }
    if (dres) st4(dres + i, g);
    float4 o;
    o.x = k * (g.x - md - (v.x - m) * is * mdx);
    o.y = k * (g.y - md - (v.y - m) * is * mdx);
    o.z = k * (g.z - md - (v.z - m) * is * mdx);
    o.w = k * (g.w - md - (v.w - m) * is * mdx);
    st4(dx + i, o);
  }
}

// ===========================================================================
// NHWC (channels_last) variants — x viewed as [R][C], R = N*H*W rows of C
// contiguous channels.  This is the layout MIOpen's fast igemm convs want
// (their NCHW path inserts batched_transpose kernels around every conv), so
// the fused BN must run natively in NHWC for an end-to-end channels_last
// model.
//
// v2 geometry (the v1 lane-per-channel scalar loops measured 2.8x slower
// than the NCHW kernels at C=64): every thread owns one CHANNEL QUAD
// (q = tid % (C/4)) for its whole lifetime — the per-channel parameters are
// loaded ONCE into registers, every data access is a float4, consecutive
// threads cover consecutive quads (perfectly coalesced rows), and there is
// no 64-bit div/mod in the loop body.  Requires C % 4 == 0 and C <= 1024
// (mask path C % 8 == 0); HW unconstrained.
// ===========================================================================

// fwd pass 1: per-channel partial sum/sumsq into ws[C][split][2].
// grid = (split); one block covers ALL channels (qpc <= 256).
__global__ void bn_reduce_nhwc_kernel(const float* __restrict__ x, long R,
                                      int C, int split,
                                      float* __restrict__ ws) {
  const int qpc = C / 4;
  const int q = threadIdx.x % qpc;
  const int r0 = threadIdx.x / qpc;
  const int rpb = TPB / qpc;
  __shared__ float4 sm[TPB][2];
  float4 sum = {0.f, 0.f, 0.f, 0.f}, sq = {0.f, 0.f, 0.f, 0.f};
  if (r0 < rpb) {
    for (long r = (long)blockIdx.x * rpb + r0; r < R;
         r += (long)split * rpb) {
      const float4 v = ld4(x + r * C + q * 4);
      sum.x += v.x; sum.y += v.y; sum.z += v.z; sum.w += v.w;
      sq.x += v.x * v.x; sq.y += v.y * v.y;
      sq.z += v.z * v.z; sq.w += v.w * v.w;
    }
  }
  sm[threadIdx.x][0] = sum;
  sm[threadIdx.x][1] = sq;
  __syncthreads();
  if (threadIdx.x < qpc) {
    for (int g = 1; g < rpb; ++g) {
      const float4 a = sm[threadIdx.x + g * qpc][0];
      const float4 b = sm[threadIdx.x + g * qpc][1];
      sum.x += a.x; sum.y += a.y; sum.z += a.z; sum.w += a.w;
      sq.x += b.x; sq.y += b.y; sq.z += b.z; sq.w += b.w;
    }
    const int c0 = threadIdx.x * 4;
    const float* ps = reinterpret_cast<const float*>(&sum);
    const float* pq = reinterpret_cast<const float*>(&sq);
    for (int j = 0; j < 4; ++j) {
      ws[((long)(c0 + j) * split + blockIdx.x) * 2 + 0] = ps[j];
      ws[((long)(c0 + j) * split + blockIdx.x) * 2 + 1] = pq[j];
    }
  }
}

// fwd pass 2: y = relu(xhat*gamma + beta [+ res]) (+1-bit mask when MASKED)
template <bool MASKED>
__global__ void bn_norm_nhwc_kernel(const float* __restrict__ x,
                                    const float* __restrict__ res,
                                    float* __restrict__ y,
                                    unsigned char* __restrict__ mask, long R,
                                    int C, const float* __restrict__ mean,
                                    const float* __restrict__ invstd,
                                    const float* __restrict__ gamma,
                                    const float* __restrict__ beta,
                                    int relu) {
  const int qpc = C / 4;
  const int q = threadIdx.x % qpc;
  const int r0 = threadIdx.x / qpc;
  const int rpb = TPB / qpc;
  if (r0 >= rpb) return;
  const int c0 = q * 4;
  const float4 mg = ld4(gamma + c0);
  const float4 mi = ld4(invstd + c0);
  const float4 mm = ld4(mean + c0);
  const float4 mb = ld4(beta + c0);
  const float4 a = {mg.x * mi.x, mg.y * mi.y, mg.z * mi.z, mg.w * mi.w};
  const float4 bb = {mb.x - mm.x * a.x, mb.y - mm.y * a.y,
                     mb.z - mm.z * a.z, mb.w - mm.w * a.w};
  for (long r = (long)blockIdx.x * rpb + r0; r < R;
       r += (long)gridDim.x * rpb) {
    const long base = r * C + c0;
    float4 v = ld4(x + base);
    v.x = v.x * a.x + bb.x;
    v.y = v.y * a.y + bb.y;
    v.z = v.z * a.z + bb.z;
    v.w = v.w * a.w + bb.w;
    if (res) {
      const float4 rr = ld4(res + base);
      v.x += rr.x; v.y += rr.y; v.z += rr.z; v.w += rr.w;
    }
    if (MASKED) {
      // C % 8 == 0: quads pair (even q -> bits 0-3, odd q -> bits 4-7) of
      // the byte at (r*C + (q & ~1)*4)/8; adjacent tids, same wave
      unsigned nib = (v.x > 0.f) | ((v.y > 0.f) << 1) | ((v.z > 0.f) << 2) |
                     ((v.w > 0.f) << 3);
      const unsigned other = __shfl_down(nib, 1);
      if ((q & 1) == 0) mask[(r * C) / 8 + q / 2] =
          (unsigned char)(nib | (other << 4));
      v.x = fmaxf(v.x, 0.f); v.y = fmaxf(v.y, 0.f);
      v.z = fmaxf(v.z, 0.f); v.w = fmaxf(v.w, 0.f);
    } else if (relu) {
      v.x = fmaxf(v.x, 0.f); v.y = fmaxf(v.y, 0.f);
      v.z = fmaxf(v.z, 0.f); v.w = fmaxf(v.w, 0.f);
    }
    st4(y + base, v);
  }
}

// bwd pass 1: per-channel partials of sum(dy_eff), sum(dy_eff * (x-mean)).
__global__ void bn_bwd_reduce_nhwc_kernel(
    const float* __restrict__ x, const float* __restrict__ dy,
    const float* __restrict__ y, const unsigned char* __restrict__ mask,
    long R, int C, int split, const float* __restrict__ mean,
    const float* __restrict__ invstd, float* __restrict__ ws) {
  const int qpc = C / 4;
  const int q = threadIdx.x % qpc;
  const int r0 = threadIdx.x / qpc;
  const int rpb = TPB / qpc;
  __shared__ float4 sm[TPB][2];
  float4 sd = {0.f, 0.f, 0.f, 0.f}, sdx = {0.f, 0.f, 0.f, 0.f};
  const int c0 = q * 4;
  if (r0 < rpb) {
    const float4 mm = ld4(mean + c0);
    for (long r = (long)blockIdx.x * rpb + r0; r < R;
         r += (long)split * rpb) {
      const long base = r * C + c0;
      float4 g = ld4(dy + base);
      const float4 v = ld4(x + base);
      if (mask) {
        const unsigned mk = mask[(r * C) / 8 + q / 2];
        const unsigned nib = (q & 1) ? (mk >> 4) : mk;
        g.x = nib & 1 ? g.x : 0.f;
        g.y = nib & 2 ? g.y : 0.f;
        g.z = nib & 4 ? g.z : 0.f;
        g.w = nib & 8 ? g.w : 0.f;
      } else if (y) {
        const float4 yy = ld4(y + base);
        g.x = yy.x > 0.f ? g.x : 0.f;
        g.y = yy.y > 0.f ? g.y : 0.f;
        g.z = yy.z > 0.f ? g.z : 0.f;
        g.w = yy.w > 0.f ? g.w : 0.f;
      }
      sd.x += g.x; sd.y += g.y; sd.z += g.z; sd.w += g.w;
      sdx.x += g.x * (v.x - mm.x);
      sdx.y += g.y * (v.y - mm.y);
      sdx.z += g.z * (v.z - mm.z);
      sdx.w += g.w * (v.w - mm.w);
    }
  }
  sm[threadIdx.x][0] = sd;
  sm[threadIdx.x][1] = sdx;
  __syncthreads();
  if (threadIdx.x < qpc) {
    for (int g = 1; g < rpb; ++g) {
      const float4 a = sm[threadIdx.x + g * qpc][0];
      const float4 b = sm[threadIdx.x + g * qpc][1];
      sd.x += a.x; sd.y += a.y; sd.z += a.z; sd.w += a.w;
      sdx.x += b.x; sdx.y += b.y; sdx.z += b.z; sdx.w += b.w;
    }
    const int cc = threadIdx.x * 4;
    const float* ps = reinterpret_cast<const float*>(&sd);
    const float* px = reinterpret_cast<const float*>(&sdx);
    for (int j = 0; j < 4; ++j) {
      ws[((long)(cc + j) * split + blockIdx.x) * 2 + 0] = ps[j];
      ws[((long)(cc + j) * split + blockIdx.x) * 2 + 1] =
          px[j] * invstd[cc + j];
    }
  }
}

// bwd pass 2: dx (+ dres = dy_eff)
__global__ void bn_bwd_dx_nhwc_kernel(
    const float* __restrict__ x, const float* __restrict__ dy,
    const float* __restrict__ y, const unsigned char* __restrict__ mask,
    float* __restrict__ dx, float* __restrict__ dres, long R, int C,
    const float* __restrict__ mean, const float* __restrict__ invstd,
    const float* __restrict__ gamma, const float* __restrict__ sum_dy,
    const float* __restrict__ sum_dyx, float inv_count) {
  const int qpc = C / 4;
  const int q = threadIdx.x % qpc;
  const int r0 = threadIdx.x / qpc;
  const int rpb = TPB / qpc;
  if (r0 >= rpb) return;
  const int c0 = q * 4;
  const float4 mm = ld4(mean + c0);
  const float4 mi = ld4(invstd + c0);
  const float4 mg = ld4(gamma + c0);
  const float4 msd = ld4(sum_dy + c0);
  const float4 msx = ld4(sum_dyx + c0);
  const float4 k = {mg.x * mi.x, mg.y * mi.y, mg.z * mi.z, mg.w * mi.w};
  const float4 md = {msd.x * inv_count, msd.y * inv_count,
                     msd.z * inv_count, msd.w * inv_count};
  const float4 mx = {msx.x * inv_count * mi.x, msx.y * inv_count * mi.y,
                     msx.z * inv_count * mi.z, msx.w * inv_count * mi.w};
  for (long r = (long)blockIdx.x * rpb + r0; r < R;
       r += (long)gridDim.x * rpb) {
    const long base = r * C + c0;
    float4 g = ld4(dy + base);
    const float4 v = ld4(x + base);
    if (mask) {
      const unsigned mk = mask[(r * C) / 8 + q / 2];
      const unsigned nib = (q & 1) ? (mk >> 4) : mk;
      g.x = nib & 1 ? g.x : 0.f;
      g.y = nib & 2 ? g.y : 0.f;
      g.z = nib & 4 ? g.z : 0.f;
      g.w = nib & 8 ? g.w : 0.f;
    } else if (y) {
      const float4 yy = ld4(y + base);
      g.x = yy.x > 0.f ? g.x : 0.f;
      g.y = yy.y > 0.f ? g.y : 0.f;
      g.z = yy.z > 0.f ? g.z : 0.f;
      g.w = yy.w > 0.f ? g.w : 0.f;
    }
    if (dres) st4(dres + base, g);
    float4 o;
    o.x = k.x * (g.x - md.x - (v.x - mm.x) * mx.x);
    o.y = k.y * (g.y - md.y - (v.y - mm.y) * mx.y);
    o.z = k.z * (g.z - md.z - (v.z - mm.z) * mx.z);
    o.w = k.w * (g.w - md.w - (v.w - mm.w) * mx.w);
    st4(dx + base, o);
  }
}

// ---------------------------------------------------------------------------
using at::Tensor;

inline hipStream_t cur_stream(const Tensor& t) {
  return c10::hip::getCurrentHIPStream(t.get_device()).stream();
}

inline int elem_grid(long total4) {
  return (int)std::min<long>((total4 + TPB - 1) / TPB, 16384);
}

int pick_split(int N, int C, long HW) {
  // enough blocks to fill 256 CUs a few times over
  long per_split_blocks = C;
  int split = 1;
  while (per_split_blocks * split < 2048 && split < N) split *= 2;
  return std::min(split, N);
}

int pick_split_nhwc(long R, int C) {
  // one block covers ALL channels (qpc = C/4 <= 256); split = #blocks.
  // Cap C*split so the finalize pass stays small (wave-per-channel reads
  // split entries per lane-stride).
  const int qpc = C / 4;
  const int rpb = std::max(TPB / qpc, 1);
  long split = std::min<long>(2048, std::max<long>(R / rpb, 1));
  split = std::min<long>(split, std::max<long>(524288 / C, 64));
  return (int)split;
}

inline int nhwc_grid(long R, int C) {
  const int qpc = C / 4;
  const int rpb = std::max(TPB / qpc, 1);
  return (int)std::min<long>((R + rpb - 1) / rpb, 16384);
}

std::vector<Tensor> bn_relu_fwd(const Tensor& x, const Tensor& gamma,
                                const Tensor& beta, Tensor running_mean,
                                Tensor running_var, double momentum,
                                double eps, bool relu,
                                const c10::optional<Tensor>& residual) {
  const bool nhwc = x.dim() == 4 && x.size(1) > 1 &&
                    x.is_contiguous(at::MemoryFormat::ChannelsLast);
  TORCH_CHECK(x.is_cuda() && x.dim() == 4 &&
              (nhwc || x.is_contiguous()) &&
              x.scalar_type() == at::kFloat);
  const int N = x.size(0), C = x.size(1);
  const long HW = (long)x.size(2) * x.size(3);
  if (nhwc) {
    TORCH_CHECK(C % 4 == 0 && C <= 1024,
                "fused NHWC BN needs C % 4 == 0 and C <= 1024");
  } else {
    TORCH_CHECK(HW % 4 == 0, "fused BN needs H*W % 4 == 0");
  }
  const long R = (long)N * HW;
  const int split = nhwc ? pick_split_nhwc(R, C) : pick_split(N, C, HW);
  auto opts = x.options();
  Tensor ws = at::empty({C, split, 2}, opts);
  Tensor mean = at::empty({C}, opts);
  Tensor invstd = at::empty({C}, opts);
  Tensor y = at::empty_like(x);
  auto st = cur_stream(x);
  if (nhwc)
    hipLaunchKernelGGL(bn_reduce_nhwc_kernel, dim3(split), dim3(TPB), 0, st,
                       x.data_ptr<float>(), R, C, split,
                       ws.data_ptr<float>());
  else
    hipLaunchKernelGGL(bn_reduce_kernel, dim3(C, split), dim3(TPB), 0, st,
                       x.data_ptr<float>(), N, C, HW, split,
                       ws.data_ptr<float>());
  hipLaunchKernelGGL(bn_finalize_kernel, dim3((C + TPB / 64 - 1) / (TPB / 64)),
                     dim3(TPB), 0, st, ws.data_ptr<float>(), C, split,
                     (float)((long)N * HW), (float)eps, (float)momentum,
                     mean.data_ptr<float>(), invstd.data_ptr<float>(),
                     running_mean.defined() ? running_mean.data_ptr<float>()
                                            : nullptr,
                     running_var.defined() ? running_var.data_ptr<float>()
                                           : nullptr);
  const float* res_ptr = nullptr;
  if (residual.has_value()) {
    TORCH_CHECK(residual->sizes() == x.sizes());
    TORCH_CHECK(nhwc ? residual->is_contiguous(at::MemoryFormat::ChannelsLast)
                     : residual->is_contiguous(),
                "residual layout must match x");
    res_ptr = residual->data_ptr<float>();
  }
  Tensor mask = at::empty({0}, opts.dtype(at::kByte));
  const bool use_mask = relu && (nhwc ? C % 8 == 0 : HW % 8 == 0);
  if (use_mask) {
    // emit the ReLU mask as 1 bit/element so backward skips the y re-read
    mask = at::empty({(long)N * C * HW / 8}, opts.dtype(at::kByte));
    if (nhwc)
      hipLaunchKernelGGL(bn_norm_nhwc_kernel<true>,
                         dim3(nhwc_grid(R, C)), dim3(TPB), 0, st,
                         x.data_ptr<float>(), res_ptr, y.data_ptr<float>(),
                         mask.data_ptr<uint8_t>(), R, C,
                         mean.data_ptr<float>(), invstd.data_ptr<float>(),
                         gamma.data_ptr<float>(), beta.data_ptr<float>(), 1);
    else
      hipLaunchKernelGGL(bn_norm_mask_kernel,
                         dim3(elem_grid((long)N * C * HW / 8)), dim3(TPB), 0,
                         st, x.data_ptr<float>(), res_ptr, y.data_ptr<float>(),
                         mask.data_ptr<uint8_t>(), N, C, HW,
                         mean.data_ptr<float>(), invstd.data_ptr<float>(),
                         gamma.data_ptr<float>(), beta.data_ptr<float>());
  } else if (nhwc) {
    hipLaunchKernelGGL(bn_norm_nhwc_kernel<false>,
                       dim3(nhwc_grid(R, C)), dim3(TPB), 0, st,
                       x.data_ptr<float>(), res_ptr, y.data_ptr<float>(),
                       (unsigned char*)nullptr, R, C, mean.data_ptr<float>(),
                       invstd.data_ptr<float>(), gamma.data_ptr<float>(),
                       beta.data_ptr<float>(), relu ? 1 : 0);
  } else {
    hipLaunchKernelGGL(bn_norm_kernel, dim3(elem_grid((long)N * C * HW / 4)),
                       dim3(TPB), 0, st, x.data_ptr<float>(), res_ptr,
                       y.data_ptr<float>(), N, C, HW, mean.data_ptr<float>(),
                       invstd.data_ptr<float>(), gamma.data_ptr<float>(),
                       beta.data_ptr<float>(), relu ? 1 : 0);
  }
  return {y, mean, invstd, mask};
}

std::vector<Tensor> bn_relu_bwd(const Tensor& x, const Tensor& dy,
                                const c10::optional<Tensor>& y_for_mask,
                                const Tensor& mean, const Tensor& invstd,
                                const Tensor& gamma, bool need_dres,
                                const c10::optional<Tensor>& bitmask) {
  const bool nhwc = x.dim() == 4 && x.size(1) > 1 &&
                    x.is_contiguous(at::MemoryFormat::ChannelsLast);
  const int N = x.size(0), C = x.size(1);
  const long HW = (long)x.size(2) * x.size(3);
  const long R = (long)N * HW;
  const int split = nhwc ? pick_split_nhwc(R, C) : pick_split(N, C, HW);
  auto opts = x.options();
  Tensor ws = at::empty({C, split, 2}, opts);
  Tensor sum_dy = at::empty({C}, opts);
  Tensor sum_dyx = at::empty({C}, opts);
  Tensor dgamma = at::empty({C}, opts);
  Tensor dbeta = at::empty({C}, opts);
  Tensor dx = at::empty_like(x);
  Tensor dres;
  auto st = cur_stream(x);
  const float* yp = y_for_mask.has_value() ? y_for_mask->data_ptr<float>()
                                           : nullptr;
  Tensor dyc = nhwc ? dy.contiguous(at::MemoryFormat::ChannelsLast)
                    : dy.contiguous();
  float* dres_ptr = nullptr;
  if (need_dres) {
    dres = at::empty_like(x);
    dres_ptr = dres.data_ptr<float>();
  }
  if (nhwc) {
    const uint8_t* mk = (bitmask.has_value() && bitmask->defined() &&
                         bitmask->numel() > 0)
                            ? bitmask->data_ptr<uint8_t>()
                            : nullptr;
    hipLaunchKernelGGL(bn_bwd_reduce_nhwc_kernel, dim3(split),
                       dim3(TPB), 0, st, x.data_ptr<float>(),
                       dyc.data_ptr<float>(), mk ? nullptr : yp, mk, R, C,
                       split, mean.data_ptr<float>(),
                       invstd.data_ptr<float>(), ws.data_ptr<float>());
    hipLaunchKernelGGL(bn_bwd_finalize_kernel,
                       dim3((C + TPB / 64 - 1) / (TPB / 64)), dim3(TPB), 0,
                       st, ws.data_ptr<float>(), C, split,
                       sum_dy.data_ptr<float>(), sum_dyx.data_ptr<float>(),
                       dgamma.data_ptr<float>(), dbeta.data_ptr<float>());
    hipLaunchKernelGGL(bn_bwd_dx_nhwc_kernel, dim3(nhwc_grid(R, C)),
                       dim3(TPB), 0, st, x.data_ptr<float>(),
                       dyc.data_ptr<float>(), mk ? nullptr : yp, mk,
                       dx.data_ptr<float>(), dres_ptr, R, C,
                       mean.data_ptr<float>(), invstd.data_ptr<float>(),
                       gamma.data_ptr<float>(), sum_dy.data_ptr<float>(),
                       sum_dyx.data_ptr<float>(), 1.0f / (float)R);
    if (!need_dres) dres = at::Tensor();
    return {dx, dgamma, dbeta, dres};
  }
  if (bitmask.has_value() && bitmask->defined() &&
      bitmask->numel() > 0) {
    const uint8_t* mk = bitmask->data_ptr<uint8_t>();
    hipLaunchKernelGGL(bn_bwd_reduce_mask_kernel, dim3(C, split), dim3(TPB),
                       0, st, x.data_ptr<float>(), dyc.data_ptr<float>(), mk,
                       N, C, HW, split, mean.data_ptr<float>(),
                       invstd.data_ptr<float>(), ws.data_ptr<float>());
    hipLaunchKernelGGL(bn_bwd_finalize_kernel,
                       dim3((C + TPB / 64 - 1) / (TPB / 64)), dim3(TPB), 0,
                       st, ws.data_ptr<float>(), C, split,
                       sum_dy.data_ptr<float>(), sum_dyx.data_ptr<float>(),
                       dgamma.data_ptr<float>(), dbeta.data_ptr<float>());
    hipLaunchKernelGGL(bn_bwd_dx_mask_kernel,
                       dim3(elem_grid((long)N * C * HW / 8)), dim3(TPB), 0,
                       st, x.data_ptr<float>(), dyc.data_ptr<float>(), mk,
                       dx.data_ptr<float>(), dres_ptr, N, C, HW,
                       mean.data_ptr<float>(), invstd.data_ptr<float>(),
                       gamma.data_ptr<float>(), sum_dy.data_ptr<float>(),
                       sum_dyx.data_ptr<float>(),
                       1.0f / (float)((long)N * HW));
    if (!need_dres) dres = at::Tensor();
    return {dx, dgamma, dbeta, dres};
  }
  hipLaunchKernelGGL(bn_bwd_reduce_kernel, dim3(C, split), dim3(TPB), 0, st,
                     x.data_ptr<float>(), dyc.data_ptr<float>(), yp, N, C, HW,
                     split, mean.data_ptr<float>(), invstd.data_ptr<float>(),
                     ws.data_ptr<float>());
  hipLaunchKernelGGL(bn_bwd_finalize_kernel,
                     dim3((C + TPB / 64 - 1) / (TPB / 64)), dim3(TPB), 0, st,
                     ws.data_ptr<float>(), C, split,
                     sum_dy.data_ptr<float>(), sum_dyx.data_ptr<float>(),
                     dgamma.data_ptr<float>(), dbeta.data_ptr<float>());
  hipLaunchKernelGGL(bn_bwd_dx_kernel, dim3(elem_grid((long)N * C * HW / 4)),
                     dim3(TPB), 0, st, x.data_ptr<float>(),
                     dyc.data_ptr<float>(), yp, dx.data_ptr<float>(), dres_ptr,
                     N, C, HW, mean.data_ptr<float>(),
                     invstd.data_ptr<float>(), gamma.data_ptr<float>(),
                     sum_dy.data_ptr<float>(), sum_dyx.data_ptr<float>(),
                     1.0f / (float)((long)N * HW));
  if (!need_dres) dres = at::Tensor();
  return {dx, dgamma, dbeta, dres};
}

}  // namespace

void cpd_register_bn(pybind11::module_& m) {
  m.def("bn_relu_fwd", &bn_relu_fwd,
        "fused BN(+res)+ReLU forward: returns (y, mean, invstd)");
  m.def("bn_relu_bwd", &bn_relu_bwd,
        "fused BN(+res)+ReLU backward: returns (dx, dgamma, dbeta, dres)");
}

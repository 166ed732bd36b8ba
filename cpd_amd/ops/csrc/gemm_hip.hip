// cpd_amd GEMM kernels for gfx950 (CDNA4 / MI355X).
//
// Two hand-written kernels (no BLAS, no hipify — MI355X-first designs):
//
// 1. gemm_f32: general C[M,N] = A[M,K] @ B[K,N] in exact fp32 on the MFMA
//    matrix cores (`v_mfma_f32_32x32x2_f32`).  gfx950 has no TF32/xf32 path;
//    the f32-input MFMA is bitwise an fmaf chain at the 157 TF f32 vector
//    rate, ~2.4x an f32 VALU GEMM (cdna_hip_programming.md §3).  Structure:
//    128x128x32 block tile, 4 waves, each wave a 2x2 of 32x32 MFMA tiles,
//    A staged transposed in LDS ([BK][BM+1], bank-conflict-free write via the
//    (4k+j+m) mapping), B staged row-major.  This is the capability-parity
//    replacement for the reference's "high performance general GEMM"
//    (README.md:13-16) — the reference's tvm_gemm computes a 16x16 C tile per
//    block with scalar math (float_kernel.cu:103-340).
//
// 2. quant_gemm: C = A @ B with an (exp,man)-rounded Kahan accumulator —
//    every product and every Kahan intermediate is cast to the custom grid,
//    sequentially over K (reference semantics float_kernel.cu:181-195).  The
//    rounding between accumulation steps is inherently serial per output
//    element, so MFMA cannot be used (its internal k-accumulation cannot be
//    rounded); this is a VALU kernel: 64x64 C tile per 256-thread block, 4x4
//    micro-tile per lane, LDS-staged operands.  Unlike the reference, the
//    boundary path zero-initializes the compensation term and is barrier-
//    correct (the reference's edge branch reads uninitialized Kahan state and
//    lacks __syncthreads — SURVEY.md §2.1 N3; both bugs fixed by design).
#include <torch/extension.h>
#include <c10/hip/HIPStream.h>
#include <hip/hip_runtime.h>

#include "quant_core.h"

namespace {

using namespace cpd;
using f32x16 = __attribute__((ext_vector_type(16))) float;

// ---------------------------------------------------------------------------
// fp32 MFMA GEMM
// ---------------------------------------------------------------------------

constexpr int BM = 128, BN = 128, BK = 32;

// Load one K-tile's worth of A/B for this thread into registers (guarded
// for arbitrary M/N/K edges).  4 float4 of A + 4 float4 of B per thread.
struct StageRegs {
  float4 a[4];
  float4 b[4];
};

// GUARDED=false is the interior fast path (no bounds checks in the load
// address stream — the per-float4 guards measured ~25% of the whole GEMM:
// 92 vs 125 TF @4096^3, tools/gemm_probe.hip).
template <bool GUARDED>
__device__ __forceinline__ void stage_load(const float* __restrict__ A,
                                           const float* __restrict__ B,
                                           int M, int N, int K, int block_row,
                                           int block_col, int k0, int tid,
                                           StageRegs& r) {
  const int k4 = tid & 7;                  // A: 8 float4 per 32-wide K row
  const int m0 = tid >> 3;                 //    32 rows per pass
  for (int p = 0; p < 4; ++p) {
    const int gm = block_row + m0 + p * 32;
    const int gk = k0 + k4 * 4;
    if (!GUARDED) {
      r.a[p] = *reinterpret_cast<const float4*>(A + (long)gm * K + gk);
      continue;
    }
    float4 v = {0.f, 0.f, 0.f, 0.f};
    if (gm < M) {
      if (gk + 3 < K) {
        v = *reinterpret_cast<const float4*>(A + (long)gm * K + gk);
      } else {
        const float* row = A + (long)gm * K;
        if (gk + 0 < K) v.x = row[gk + 0];
        if (gk + 1 < K) v.y = row[gk + 1];
        if (gk + 2 < K) v.z = row[gk + 2];
      }
    }
    r.a[p] = v;
  }
  const int n4 = tid & 31;                 // B: 32 float4 per 128-wide row
  const int kk0 = tid >> 5;                //    8 k rows per pass
  for (int p = 0; p < 4; ++p) {
    const int gk = k0 + kk0 + p * 8;
    const int gn = block_col + n4 * 4;
    if (!GUARDED) {
      r.b[p] = *reinterpret_cast<const float4*>(B + (long)gk * N + gn);
      continue;
    }
    float4 v = {0.f, 0.f, 0.f, 0.f};
    if (gk < K) {
      if (gn + 3 < N) {
        v = *reinterpret_cast<const float4*>(B + (long)gk * N + gn);
      } else {
        const float* row = B + (long)gk * N;
        if (gn + 0 < N) v.x = row[gn + 0];
        if (gn + 1 < N) v.y = row[gn + 1];
        if (gn + 2 < N) v.z = row[gn + 2];
      }
    }
    r.b[p] = v;
  }
}

__device__ __forceinline__ void stage_write(float (*As)[BM + 1],
                                            float (*Bs)[BN], int tid,
                                            const StageRegs& r) {
  const int k4 = tid & 7;
  const int m0 = tid >> 3;
  for (int p = 0; p < 4; ++p) {
    const int m = m0 + p * 32;
    As[k4 * 4 + 0][m] = r.a[p].x;
    As[k4 * 4 + 1][m] = r.a[p].y;
    As[k4 * 4 + 2][m] = r.a[p].z;
    As[k4 * 4 + 3][m] = r.a[p].w;
  }
  const int n4 = tid & 31;
  const int kk0 = tid >> 5;
  for (int p = 0; p < 4; ++p)
    *reinterpret_cast<float4*>(&Bs[kk0 + p * 8][n4 * 4]) = r.b[p];
}

__global__ __launch_bounds__(256) void gemm_f32_kernel(
    const float* __restrict__ A, const float* __restrict__ B,
    float* __restrict__ C, int M, int N, int K) {
  // double-buffered LDS: loads for tile t+1 fly under tile t's MFMAs
  __shared__ float As[2][BK][BM + 1];
  __shared__ float Bs[2][BK][BN];

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;     // 4 waves: 2x2 of 64x64 wave tiles
  const int wr = (wave >> 1) * 64;       // wave row offset in block tile
  const int wc = (wave & 1) * 64;
  const int tid = threadIdx.x;

  // XCD-aware bijective remap (each XCD gets a contiguous grid chunk so
  // neighbor tiles share L2-resident A/B panels — T1, cdna guide §5.5)
  const int nwg = gridDim.x * gridDim.y;
  const int wg = blockIdx.y * gridDim.x + blockIdx.x;
  const int q = nwg / 8, rr = nwg % 8;
  const int xcd = wg % 8, idx = wg / 8;
  const int swg = (xcd < rr ? xcd * (q + 1) : rr * (q + 1) + (xcd - rr) * q)
                  + idx;
  const int block_row = (swg % gridDim.x) * BM;
  const int block_col = (swg / gridDim.x) * BN;

  f32x16 acc[2][2] = {};

  const int ktiles = (K + BK - 1) / BK;
  // wave-uniform interior test: the whole M/N footprint in range and no K
  // tail except possibly the last tile
  const bool interior_mn = (block_row + BM <= M) && (block_col + BN <= N);
  StageRegs regs;
  if (interior_mn && BK <= K)
    stage_load<false>(A, B, M, N, K, block_row, block_col, 0, tid, regs);
  else
    stage_load<true>(A, B, M, N, K, block_row, block_col, 0, tid, regs);
  stage_write(As[0], Bs[0], tid, regs);
  int cur = 0;
  for (int kt = 0; kt < ktiles; ++kt) {
    __syncthreads();  // buf[cur] staged and visible
    if (kt + 1 < ktiles) {  // issue next tile's global loads now ...
      if (interior_mn && (kt + 2) * BK <= K)
        stage_load<false>(A, B, M, N, K, block_row, block_col, (kt + 1) * BK,
                          tid, regs);
      else
        stage_load<true>(A, B, M, N, K, block_row, block_col, (kt + 1) * BK,
                         tid, regs);
    }

    const int l31 = lane & 31;
    const int khalf = lane >> 5;  // this lane's k within the 2-wide step
    for (int kk = 0; kk < BK; kk += 2) {
      const float a0 = As[cur][kk + khalf][wr + l31];
      const float a1 = As[cur][kk + khalf][wr + 32 + l31];
      const float b0 = Bs[cur][kk + khalf][wc + l31];
      const float b1 = Bs[cur][kk + khalf][wc + 32 + l31];
      acc[0][0] = __builtin_amdgcn_mfma_f32_32x32x2f32(a0, b0, acc[0][0], 0, 0, 0);
      acc[0][1] = __builtin_amdgcn_mfma_f32_32x32x2f32(a0, b1, acc[0][1], 0, 0, 0);
      acc[1][0] = __builtin_amdgcn_mfma_f32_32x32x2f32(a1, b0, acc[1][0], 0, 0, 0);
      acc[1][1] = __builtin_amdgcn_mfma_f32_32x32x2f32(a1, b1, acc[1][1], 0, 0, 0);
    }
    if (kt + 1 < ktiles)  // ... and land them in the other buffer (safe:
      stage_write(As[cur ^ 1], Bs[cur ^ 1], tid, regs);  // last read of
    cur ^= 1;             // buf[cur^1] was before this iteration's barrier
  }

  // --- epilogue: C/D layout col=lane&31, row=(r&3)+8*(r>>2)+4*(lane>>5) ---
  for (int mi = 0; mi < 2; ++mi) {
    for (int nj = 0; nj < 2; ++nj) {
      const int col = block_col + wc + nj * 32 + (lane & 31);
      if (col >= N) continue;
      for (int r = 0; r < 16; ++r) {
        const int row = block_row + wr + mi * 32 + (r & 3) + 8 * (r >> 2) +
                        4 * (lane >> 5);
        if (row < M) C[(long)row * N + col] = acc[mi][nj][r];
      }
    }
  }
}

// ---------------------------------------------------------------------------
// quantized-Kahan-accumulator GEMM (VALU; rounding forbids MFMA accumulation)
// ---------------------------------------------------------------------------

// 32x32 tile, 2x2 outputs per thread: (M/32)x(N/32) blocks give 4x the
// resident waves of a 64x64 tile — this kernel is LATENCY-bound on the
// serial cast chains, so occupancy is the lever (64-tile: 0.34 TF, 32-tile:
// 0.52 TF @1024^3, bit-identical — tools/quant_gemm_probe.hip).
constexpr int QBM = 32, QBN = 32, QBK = 16;

__global__ __launch_bounds__(256) void quant_gemm_kernel(
    const float* __restrict__ A, const float* __restrict__ B,
    float* __restrict__ C, int M, int N, int K, int man, int exp) {
  // row pad +2 keeps the per-lane (2m, 2m+1) operand pairs 8-byte aligned
  // so each a/b pair is one ds_read_b64
  __shared__ float As[QBK][QBM + 2];
  __shared__ float Bs[QBK][QBN];

  const int tx = threadIdx.x & 15;   // 16x16 threads, 2x2 outputs each
  const int ty = threadIdx.x >> 4;
  const int row0 = blockIdx.x * QBM + ty * 2;
  const int col0 = blockIdx.y * QBN + tx * 2;

  float acc[2][2] = {};
  float comp[2][2] = {};

  const int ktiles = (K + QBK - 1) / QBK;
  for (int kt = 0; kt < ktiles; ++kt) {
    const int k0 = kt * QBK;
    // stage A[32][16] transposed, B[16][32]; 256 threads x 2 elements each
    {
      const int k = threadIdx.x & 15;
      const int m0 = threadIdx.x >> 4;
      for (int p = 0; p < 2; ++p) {
        const int m = m0 + p * 16;
        const int gm = blockIdx.x * QBM + m;
        As[k][m] = (gm < M && k0 + k < K) ? A[(long)gm * K + k0 + k] : 0.0f;
      }
      const int n = threadIdx.x & 31;
      const int kk0 = threadIdx.x >> 5;
      for (int p = 0; p < 2; ++p) {
        const int kk = kk0 + p * 8;
        const int gn = blockIdx.y * QBN + n;
        Bs[kk][n] = (k0 + kk < K && gn < N) ? B[(long)(k0 + kk) * N + gn] : 0.0f;
      }
    }
    __syncthreads();

    const int klim = min(QBK, K - k0);  // never round in padded-k steps
    for (int kk = 0; kk < klim; ++kk) {  // strictly k-ordered (semantics)
      const float2 a01 = *reinterpret_cast<const float2*>(&As[kk][ty * 2]);
      const float2 b01 = *reinterpret_cast<const float2*>(&Bs[kk][tx * 2]);
      const float a[2] = {a01.x, a01.y};
      const float b[2] = {b01.x, b01.y};
      for (int i = 0; i < 2; ++i)
        for (int j = 0; j < 2; ++j) {
          // cast_fp_fast: ~20-VALU float-pipeline cast, bit-identical to
          // cast_fp (exhaustive 2^32 sweep on host + device, r02)
          const float prod = cast_fp_fast(a[i] * b[j], man, exp);
          kahan_qstep_fast(acc[i][j], comp[i][j], prod, man, exp);
        }
    }
    __syncthreads();
  }

  for (int i = 0; i < 2; ++i) {
    if (row0 + i >= M) break;
    for (int j = 0; j < 2; ++j)
      if (col0 + j < N) C[(long)(row0 + i) * N + col0 + j] = acc[i][j];
  }
}

inline hipStream_t cur_stream(const at::Tensor& t) {
  return c10::hip::getCurrentHIPStream(t.get_device()).stream();
}

void check_gemm_args(const at::Tensor& a, const at::Tensor& b) {
  TORCH_CHECK(a.is_cuda() && b.is_cuda(), "GEMM inputs must be on GPU");
  TORCH_CHECK(a.scalar_type() == at::kFloat && b.scalar_type() == at::kFloat);
  TORCH_CHECK(a.is_contiguous() && b.is_contiguous());
  TORCH_CHECK(a.dim() == 2 && b.dim() == 2 && a.size(1) == b.size(0));
}

}  // namespace

at::Tensor cpd_gemm_f32_hip(const at::Tensor& a, const at::Tensor& b) {
  check_gemm_args(a, b);
  const int M = a.size(0), K = a.size(1), N = b.size(1);
  at::Tensor c = at::empty({M, N}, a.options());
  dim3 grid((M + BM - 1) / BM, (N + BN - 1) / BN);
  hipLaunchKernelGGL(gemm_f32_kernel, grid, dim3(256), 0, cur_stream(a),
                     a.data_ptr<float>(), b.data_ptr<float>(),
                     c.data_ptr<float>(), M, N, K);
  return c;
}

at::Tensor cpd_quant_gemm_hip(const at::Tensor& a, const at::Tensor& b,
                              int64_t man, int64_t exp) {
  check_gemm_args(a, b);
  const int M = a.size(0), K = a.size(1), N = b.size(1);
  at::Tensor c = at::empty({M, N}, a.options());
  dim3 grid((M + QBM - 1) / QBM, (N + QBN - 1) / QBN);
  hipLaunchKernelGGL(quant_gemm_kernel, grid, dim3(256), 0, cur_stream(a),
                     a.data_ptr<float>(), b.data_ptr<float>(),
                     c.data_ptr<float>(), M, N, K, (int)man, (int)exp);
  return c;
}

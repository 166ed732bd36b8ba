// Standalone fp32 MFMA GEMM variant probe for gfx950 (no torch; build with
// `hipcc --offload-arch=gfx950 -O3 tools/gemm_probe.hip -o tools/gemm_probe`
// and run on an MI355X).  Within-probe interleaved A/B of inner-loop
// structures for the 128x128x32 f32 GEMM (cdna guide §5.4 rule 24).
#include <hip/hip_runtime.h>
#include <cstdio>
#include <cstdlib>
#include <cmath>
#include <vector>

#define HIP_CHECK(x) do { hipError_t e = (x); if (e) { \
  printf("HIP error %s @%d\n", hipGetErrorString(e), __LINE__); exit(1);} } while (0)

using f32x16 = __attribute__((ext_vector_type(16))) float;
constexpr int BM = 128, BN = 128, BK = 32;

struct StageRegs { float4 a[4]; float4 b[4]; };

__device__ __forceinline__ void stage_load(const float* A, const float* B,
                                           int K, int N, int block_row,
                                           int block_col, int k0, int tid,
                                           StageRegs& r) {
  const int k4 = tid & 7, m0 = tid >> 3;
  for (int p = 0; p < 4; ++p)
    r.a[p] = *reinterpret_cast<const float4*>(
        A + (long)(block_row + m0 + p * 32) * K + k0 + k4 * 4);
  const int n4 = tid & 31, kk0 = tid >> 5;
  for (int p = 0; p < 4; ++p)
    r.b[p] = *reinterpret_cast<const float4*>(
        B + (long)(k0 + kk0 + p * 8) * N + block_col + n4 * 4);
}

__device__ __forceinline__ void stage_write(float (*As)[BM + 1],
                                            float (*Bs)[BN], int tid,
                                            const StageRegs& r) {
  const int k4 = tid & 7, m0 = tid >> 3;
  for (int p = 0; p < 4; ++p) {
    const int m = m0 + p * 32;
    As[k4 * 4 + 0][m] = r.a[p].x;
    As[k4 * 4 + 1][m] = r.a[p].y;
    As[k4 * 4 + 2][m] = r.a[p].z;
    As[k4 * 4 + 3][m] = r.a[p].w;
  }
  const int n4 = tid & 31, kk0 = tid >> 5;
  for (int p = 0; p < 4; ++p)
    *reinterpret_cast<float4*>(&Bs[kk0 + p * 8][n4 * 4]) = r.b[p];
}

__device__ __forceinline__ void stage_load_a(const float* A, int K,
                                             int block_row, int k0, int tid,
                                             StageRegs& r) {
  const int k4 = tid & 7, m0 = tid >> 3;
  for (int p = 0; p < 4; ++p)
    r.a[p] = *reinterpret_cast<const float4*>(
        A + (long)(block_row + m0 + p * 32) * K + k0 + k4 * 4);
}

__device__ __forceinline__ void stage_write_a(float (*As)[BM + 1], int tid,
                                              const StageRegs& r) {
  const int k4 = tid & 7, m0 = tid >> 3;
  for (int p = 0; p < 4; ++p) {
    const int m = m0 + p * 32;
    As[k4 * 4 + 0][m] = r.a[p].x;
    As[k4 * 4 + 1][m] = r.a[p].y;
    As[k4 * 4 + 2][m] = r.a[p].z;
    As[k4 * 4 + 3][m] = r.a[p].w;
  }
}

// B tile direct global->LDS DMA: the row-major [BK][BN] image is lane-linear
// for each wave (64 lanes x 16 B = two 128-float rows), so glds needs no
// swizzle (guide §5 rule 21: linear dest, per-lane SOURCE address).
__device__ __forceinline__ void glds_b(const float* B, int N, int block_col,
                                       int k0, int tid, float (*BsBuf)[BN]) {
  const int wavebase = (tid >> 6) * 2;  // first kk row this wave fills
  const int n4 = tid & 31;
  const int kk0 = tid >> 5;
  for (int p = 0; p < 4; ++p) {
    const int kk = kk0 + p * 8;
    const float* src = B + (long)(k0 + kk) * N + block_col + n4 * 4;
    auto* lbase = (__attribute__((address_space(3))) unsigned int*)
        &BsBuf[wavebase + p * 8][0];
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) unsigned int*)src, lbase,
        16, 0, 0);
  }
}

template <int VARIANT>
__global__ __launch_bounds__(256) void gemm_v(const float* __restrict__ A,
                                              const float* __restrict__ B,
                                              float* __restrict__ C, int M,
                                              int N, int K) {
  __shared__ float As[2][BK][BM + 1];
  __shared__ float Bs[2][BK][BN];
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int wr = (wave >> 1) * 64, wc = (wave & 1) * 64;
  const int tid = threadIdx.x;
  const int nwg = gridDim.x * gridDim.y;
  const int wg = blockIdx.y * gridDim.x + blockIdx.x;
  const int q = nwg / 8, rr = nwg % 8, xcd = wg % 8, idx = wg / 8;
  const int swg = (xcd < rr ? xcd * (q + 1) : rr * (q + 1) + (xcd - rr) * q) + idx;
  const int block_row = (swg % gridDim.x) * BM;
  const int block_col = (swg / gridDim.x) * BN;

  f32x16 acc[2][2] = {};
  const int ktiles = K / BK;
  StageRegs regs;
  stage_load(A, B, K, N, block_row, block_col, 0, tid, regs);
  stage_write(As[0], Bs[0], tid, regs);
  int cur = 0;
  const int l31 = lane & 31;
  const int kh = lane >> 5;

  for (int kt = 0; kt < ktiles; ++kt) {
    __syncthreads();
    if (kt + 1 < ktiles) {
      if constexpr (VARIANT == 4) {
        stage_load_a(A, K, block_row, (kt + 1) * BK, tid, regs);
        glds_b(B, N, block_col, (kt + 1) * BK, tid, Bs[cur ^ 1]);
      } else {
        stage_load(A, B, K, N, block_row, block_col, (kt + 1) * BK, tid,
                   regs);
      }
    }

    if constexpr (VARIANT == 0) {
      // baseline: load-then-mfma per 2-wide k step
      for (int kk = 0; kk < BK; kk += 2) {
        const float a0 = As[cur][kk + kh][wr + l31];
        const float a1 = As[cur][kk + kh][wr + 32 + l31];
        const float b0 = Bs[cur][kk + kh][wc + l31];
        const float b1 = Bs[cur][kk + kh][wc + 32 + l31];
        acc[0][0] = __builtin_amdgcn_mfma_f32_32x32x2f32(a0, b0, acc[0][0], 0, 0, 0);
        acc[0][1] = __builtin_amdgcn_mfma_f32_32x32x2f32(a0, b1, acc[0][1], 0, 0, 0);
        acc[1][0] = __builtin_amdgcn_mfma_f32_32x32x2f32(a1, b0, acc[1][0], 0, 0, 0);
        acc[1][1] = __builtin_amdgcn_mfma_f32_32x32x2f32(a1, b1, acc[1][1], 0, 0, 0);
      }
    } else if constexpr (VARIANT == 1) {
      // software-pipelined LDS reads: prefetch step kk+2 under kk's MFMAs
      float a0 = As[cur][kh][wr + l31];
      float a1 = As[cur][kh][wr + 32 + l31];
      float b0 = Bs[cur][kh][wc + l31];
      float b1 = Bs[cur][kh][wc + 32 + l31];
      for (int kk = 0; kk < BK; kk += 2) {
        float na0, na1, nb0, nb1;
        if (kk + 2 < BK) {
          na0 = As[cur][kk + 2 + kh][wr + l31];
          na1 = As[cur][kk + 2 + kh][wr + 32 + l31];
          nb0 = Bs[cur][kk + 2 + kh][wc + l31];
          nb1 = Bs[cur][kk + 2 + kh][wc + 32 + l31];
        }
        acc[0][0] = __builtin_amdgcn_mfma_f32_32x32x2f32(a0, b0, acc[0][0], 0, 0, 0);
        acc[0][1] = __builtin_amdgcn_mfma_f32_32x32x2f32(a0, b1, acc[0][1], 0, 0, 0);
        acc[1][0] = __builtin_amdgcn_mfma_f32_32x32x2f32(a1, b0, acc[1][0], 0, 0, 0);
        acc[1][1] = __builtin_amdgcn_mfma_f32_32x32x2f32(a1, b1, acc[1][1], 0, 0, 0);
        if (kk + 2 < BK) { a0 = na0; a1 = na1; b0 = nb0; b1 = nb1; }
      }
    } else if constexpr (VARIANT == 2) {
      // ds_read_b64 pair loads: fetch both k-halves' values per lane via
      // float2 on the m axis? (reads two consecutive m for one k) — instead
      // read per-wave 3x3... keep simple: vector-load 2 m-tiles at once
      for (int kk = 0; kk < BK; kk += 2) {
        const float2 a01 = {As[cur][kk + kh][wr + l31],
                            As[cur][kk + kh][wr + 32 + l31]};
        const float2 b01 = {Bs[cur][kk + kh][wc + l31],
                            Bs[cur][kk + kh][wc + 32 + l31]};
        acc[0][0] = __builtin_amdgcn_mfma_f32_32x32x2f32(a01.x, b01.x, acc[0][0], 0, 0, 0);
        acc[0][1] = __builtin_amdgcn_mfma_f32_32x32x2f32(a01.x, b01.y, acc[0][1], 0, 0, 0);
        acc[1][0] = __builtin_amdgcn_mfma_f32_32x32x2f32(a01.y, b01.x, acc[1][0], 0, 0, 0);
        acc[1][1] = __builtin_amdgcn_mfma_f32_32x32x2f32(a01.y, b01.y, acc[1][1], 0, 0, 0);
      }
    } else if constexpr (VARIANT == 4) {
      // same inner loop as v0; v4 differs only in staging (B via glds below)
      for (int kk = 0; kk < BK; kk += 2) {
        const float a0 = As[cur][kk + kh][wr + l31];
        const float a1 = As[cur][kk + kh][wr + 32 + l31];
        const float b0 = Bs[cur][kk + kh][wc + l31];
        const float b1 = Bs[cur][kk + kh][wc + 32 + l31];
        acc[0][0] = __builtin_amdgcn_mfma_f32_32x32x2f32(a0, b0, acc[0][0], 0, 0, 0);
        acc[0][1] = __builtin_amdgcn_mfma_f32_32x32x2f32(a0, b1, acc[0][1], 0, 0, 0);
        acc[1][0] = __builtin_amdgcn_mfma_f32_32x32x2f32(a1, b0, acc[1][0], 0, 0, 0);
        acc[1][1] = __builtin_amdgcn_mfma_f32_32x32x2f32(a1, b1, acc[1][1], 0, 0, 0);
      }
    } else if constexpr (VARIANT == 3) {
      // setprio around the MFMA cluster (T5)
      for (int kk = 0; kk < BK; kk += 2) {
        const float a0 = As[cur][kk + kh][wr + l31];
        const float a1 = As[cur][kk + kh][wr + 32 + l31];
        const float b0 = Bs[cur][kk + kh][wc + l31];
        const float b1 = Bs[cur][kk + kh][wc + 32 + l31];
        __builtin_amdgcn_s_setprio(1);
        acc[0][0] = __builtin_amdgcn_mfma_f32_32x32x2f32(a0, b0, acc[0][0], 0, 0, 0);
        acc[0][1] = __builtin_amdgcn_mfma_f32_32x32x2f32(a0, b1, acc[0][1], 0, 0, 0);
        acc[1][0] = __builtin_amdgcn_mfma_f32_32x32x2f32(a1, b0, acc[1][0], 0, 0, 0);
        acc[1][1] = __builtin_amdgcn_mfma_f32_32x32x2f32(a1, b1, acc[1][1], 0, 0, 0);
        __builtin_amdgcn_s_setprio(0);
      }
    }

    if (kt + 1 < ktiles) {
      if constexpr (VARIANT == 4)
        stage_write_a(As[cur ^ 1], tid, regs);
      else
        stage_write(As[cur ^ 1], Bs[cur ^ 1], tid, regs);
    }
    cur ^= 1;
  }

  for (int mi = 0; mi < 2; ++mi)
    for (int nj = 0; nj < 2; ++nj) {
      const int col = block_col + wc + nj * 32 + (lane & 31);
      for (int r = 0; r < 16; ++r) {
        const int row = block_row + wr + mi * 32 + (r & 3) + 8 * (r >> 2) +
                        4 * (lane >> 5);
        C[(long)row * N + col] = acc[mi][nj][r];
      }
    }
}

// ---- v5: pair-interleaved LDS layout so each wave's (x, x+32) operand
// pair is one ds_read_b64 (halves LDS read instructions in the MFMA loop).
// phys(m) = (m&31)*2 + ((m>>5)&1) + (m>>6)*64; reads at [g*64 + l31*2].
// Row stride BM+2 keeps float2 alignment; stride-2 lane pattern is
// bank-conflict-free (64 lanes x 8B = 2 conflict-free phases).
__global__ __launch_bounds__(256) void gemm_v5(const float* __restrict__ A,
                                               const float* __restrict__ B,
                                               float* __restrict__ C, int M,
                                               int N, int K) {
  __shared__ float As[2][BK][BM + 2];
  __shared__ float Bs[2][BK][BN + 2];
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int wr = (wave >> 1) * 64, wc = (wave & 1) * 64;
  const int tid = threadIdx.x;
  const int nwg = gridDim.x * gridDim.y;
  const int wg = blockIdx.y * gridDim.x + blockIdx.x;
  const int q = nwg / 8, rr = nwg % 8, xcd = wg % 8, idx = wg / 8;
  const int swg = (xcd < rr ? xcd * (q + 1) : rr * (q + 1) + (xcd - rr) * q) + idx;
  const int block_row = (swg % gridDim.x) * BM;
  const int block_col = (swg / gridDim.x) * BN;

  f32x16 acc[2][2] = {};
  const int ktiles = K / BK;
  StageRegs regs;
  auto write_pair = [&](float (*as)[BM + 2], float (*bs)[BN + 2]) {
    const int k4 = tid & 7, m0 = tid >> 3;
    for (int p = 0; p < 4; ++p) {
      const int pm = m0 * 2 + (p & 1) + (p >> 1) * 64;
      as[k4 * 4 + 0][pm] = regs.a[p].x;
      as[k4 * 4 + 1][pm] = regs.a[p].y;
      as[k4 * 4 + 2][pm] = regs.a[p].z;
      as[k4 * 4 + 3][pm] = regs.a[p].w;
    }
    const int n4 = tid & 31, kk0 = tid >> 5;
    for (int p = 0; p < 4; ++p) {
      const int kk = kk0 + p * 8;
      const float v[4] = {regs.b[p].x, regs.b[p].y, regs.b[p].z, regs.b[p].w};
      for (int qq = 0; qq < 4; ++qq) {
        const int n = n4 * 4 + qq;
        bs[kk][(n & 31) * 2 + ((n >> 5) & 1) + (n >> 6) * 64] = v[qq];
      }
    }
  };
  stage_load(A, B, K, N, block_row, block_col, 0, tid, regs);
  write_pair(As[0], Bs[0]);
  int cur = 0;
  const int l31 = lane & 31, kh = lane >> 5;
  const int ga = (wr >> 6) * 64 + l31 * 2;
  const int gb = (wc >> 6) * 64 + l31 * 2;
  for (int kt = 0; kt < ktiles; ++kt) {
    __syncthreads();
    if (kt + 1 < ktiles)
      stage_load(A, B, K, N, block_row, block_col, (kt + 1) * BK, tid, regs);
    for (int kk = 0; kk < BK; kk += 2) {
      const float2 a01 = *reinterpret_cast<const float2*>(&As[cur][kk + kh][ga]);
      const float2 b01 = *reinterpret_cast<const float2*>(&Bs[cur][kk + kh][gb]);
      acc[0][0] = __builtin_amdgcn_mfma_f32_32x32x2f32(a01.x, b01.x, acc[0][0], 0, 0, 0);
      acc[0][1] = __builtin_amdgcn_mfma_f32_32x32x2f32(a01.x, b01.y, acc[0][1], 0, 0, 0);
      acc[1][0] = __builtin_amdgcn_mfma_f32_32x32x2f32(a01.y, b01.x, acc[1][0], 0, 0, 0);
      acc[1][1] = __builtin_amdgcn_mfma_f32_32x32x2f32(a01.y, b01.y, acc[1][1], 0, 0, 0);
    }
    if (kt + 1 < ktiles) write_pair(As[cur ^ 1], Bs[cur ^ 1]);
    cur ^= 1;
  }
  for (int mi = 0; mi < 2; ++mi)
    for (int nj = 0; nj < 2; ++nj) {
      const int col = block_col + wc + nj * 32 + (lane & 31);
      for (int r = 0; r < 16; ++r) {
        const int row = block_row + wr + mi * 32 + (r & 3) + 8 * (r >> 2) +
                        4 * (lane >> 5);
        C[(long)row * N + col] = acc[mi][nj][r];
      }
    }
}

// ---- v6: pair-interleave A ONLY (B stays row-major float4-write): the v5
// PMC run showed 67M LDS bank conflicts from B's scattered interleaved
// writes; A's transposed write was already scalar so interleaving it is
// free, and the A reads become ds_read_b64.
__global__ __launch_bounds__(256) void gemm_v6(const float* __restrict__ A,
                                               const float* __restrict__ B,
                                               float* __restrict__ C, int M,
                                               int N, int K) {
  __shared__ float As[2][BK][BM + 2];
  __shared__ float Bs[2][BK][BN];
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int wr = (wave >> 1) * 64, wc = (wave & 1) * 64;
  const int tid = threadIdx.x;
  const int nwg = gridDim.x * gridDim.y;
  const int wg = blockIdx.y * gridDim.x + blockIdx.x;
  const int q = nwg / 8, rr = nwg % 8, xcd = wg % 8, idx = wg / 8;
  const int swg = (xcd < rr ? xcd * (q + 1) : rr * (q + 1) + (xcd - rr) * q) + idx;
  const int block_row = (swg % gridDim.x) * BM;
  const int block_col = (swg / gridDim.x) * BN;

  f32x16 acc[2][2] = {};
  const int ktiles = K / BK;
  StageRegs regs;
  auto write6 = [&](float (*as)[BM + 2], float (*bs)[BN]) {
    const int k4 = tid & 7, m0 = tid >> 3;
    for (int p = 0; p < 4; ++p) {
      const int pm = m0 * 2 + (p & 1) + (p >> 1) * 64;
      as[k4 * 4 + 0][pm] = regs.a[p].x;
      as[k4 * 4 + 1][pm] = regs.a[p].y;
      as[k4 * 4 + 2][pm] = regs.a[p].z;
      as[k4 * 4 + 3][pm] = regs.a[p].w;
    }
    const int n4 = tid & 31, kk0 = tid >> 5;
    for (int p = 0; p < 4; ++p)
      *reinterpret_cast<float4*>(&bs[kk0 + p * 8][n4 * 4]) = regs.b[p];
  };
  stage_load(A, B, K, N, block_row, block_col, 0, tid, regs);
  write6(As[0], Bs[0]);
  int cur = 0;
  const int l31 = lane & 31, kh = lane >> 5;
  const int ga = (wr >> 6) * 64 + l31 * 2;
  for (int kt = 0; kt < ktiles; ++kt) {
    __syncthreads();
    if (kt + 1 < ktiles)
      stage_load(A, B, K, N, block_row, block_col, (kt + 1) * BK, tid, regs);
    for (int kk = 0; kk < BK; kk += 2) {
      const float2 a01 = *reinterpret_cast<const float2*>(&As[cur][kk + kh][ga]);
      const float b0 = Bs[cur][kk + kh][wc + l31];
      const float b1 = Bs[cur][kk + kh][wc + 32 + l31];
      acc[0][0] = __builtin_amdgcn_mfma_f32_32x32x2f32(a01.x, b0, acc[0][0], 0, 0, 0);
      acc[0][1] = __builtin_amdgcn_mfma_f32_32x32x2f32(a01.x, b1, acc[0][1], 0, 0, 0);
      acc[1][0] = __builtin_amdgcn_mfma_f32_32x32x2f32(a01.y, b0, acc[1][0], 0, 0, 0);
      acc[1][1] = __builtin_amdgcn_mfma_f32_32x32x2f32(a01.y, b1, acc[1][1], 0, 0, 0);
    }
    if (kt + 1 < ktiles) write6(As[cur ^ 1], Bs[cur ^ 1]);
    cur ^= 1;
  }
  for (int mi = 0; mi < 2; ++mi)
    for (int nj = 0; nj < 2; ++nj) {
      const int col = block_col + wc + nj * 32 + (lane & 31);
      for (int r = 0; r < 16; ++r) {
        const int row = block_row + wr + mi * 32 + (r & 3) + 8 * (r >> 2) +
                        4 * (lane >> 5);
        C[(long)row * N + col] = acc[mi][nj][r];
      }
    }
}

double bench_v6(const float* dA, const float* dB, float* dC, int Nsz,
                int reps) {
  dim3 grid(Nsz / BM, Nsz / BN), block(256);
  hipLaunchKernelGGL(gemm_v6, grid, block, 0, 0, dA, dB, dC, Nsz, Nsz, Nsz);
  HIP_CHECK(hipDeviceSynchronize());
  hipEvent_t t0, t1;
  hipEventCreate(&t0); hipEventCreate(&t1);
  hipEventRecord(t0);
  for (int i = 0; i < reps; ++i)
    hipLaunchKernelGGL(gemm_v6, grid, block, 0, 0, dA, dB, dC, Nsz, Nsz, Nsz);
  hipEventRecord(t1);
  HIP_CHECK(hipEventSynchronize(t1));
  float ms; hipEventElapsedTime(&ms, t0, t1);
  return 2.0 * Nsz * Nsz * (double)Nsz * reps / (ms * 1e-3) / 1e12;
}

double bench_v5(const float* dA, const float* dB, float* dC, int Nsz,
                int reps) {
  dim3 grid(Nsz / BM, Nsz / BN), block(256);
  hipLaunchKernelGGL(gemm_v5, grid, block, 0, 0, dA, dB, dC, Nsz, Nsz, Nsz);
  HIP_CHECK(hipDeviceSynchronize());
  hipEvent_t t0, t1;
  hipEventCreate(&t0); hipEventCreate(&t1);
  hipEventRecord(t0);
  for (int i = 0; i < reps; ++i)
    hipLaunchKernelGGL(gemm_v5, grid, block, 0, 0, dA, dB, dC, Nsz, Nsz, Nsz);
  hipEventRecord(t1);
  HIP_CHECK(hipEventSynchronize(t1));
  float ms; hipEventElapsedTime(&ms, t0, t1);
  return 2.0 * Nsz * Nsz * (double)Nsz * reps / (ms * 1e-3) / 1e12;
}

// 256x128 tile, 8 waves (each a 2x2 of 32x32 accs like the 4-wave kernel):
// halves B re-reads and barrier count per MFMA at the same waves/SIMD.
constexpr int BM2 = 256, BN2 = 128;

__global__ __launch_bounds__(512) void gemm_big(const float* __restrict__ A,
                                                const float* __restrict__ B,
                                                float* __restrict__ C, int M,
                                                int N, int K) {
  __shared__ float As[2][BK][BM2 + 1];
  __shared__ float Bs[2][BK][BN2];
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;         // 8 waves: 4 rows x 2 cols
  const int wr = (wave >> 1) * 64, wc = (wave & 1) * 64;
  const int tid = threadIdx.x;
  const int nwg = gridDim.x * gridDim.y;
  const int wg = blockIdx.y * gridDim.x + blockIdx.x;
  const int q = nwg / 8, rr = nwg % 8, xcd = wg % 8, idx = wg / 8;
  const int swg = (xcd < rr ? xcd * (q + 1) : rr * (q + 1) + (xcd - rr) * q) + idx;
  const int block_row = (swg % gridDim.x) * BM2;
  const int block_col = (swg / gridDim.x) * BN2;

  f32x16 acc[2][2] = {};
  const int ktiles = K / BK;
  float4 ra[4], rb[2];
  auto load2 = [&](int k0) {
    const int k4 = tid & 7, m0 = tid >> 3;        // A: 64 rows/pass x 4
    for (int p = 0; p < 4; ++p)
      ra[p] = *reinterpret_cast<const float4*>(
          A + (long)(block_row + m0 + p * 64) * K + k0 + k4 * 4);
    const int n4 = tid & 31, kk0 = tid >> 5;      // B: 16 k-rows/pass x 2
    for (int p = 0; p < 2; ++p)
      rb[p] = *reinterpret_cast<const float4*>(
          B + (long)(k0 + kk0 + p * 16) * N + block_col + n4 * 4);
  };
  auto write2 = [&](int buf) {
    const int k4 = tid & 7, m0 = tid >> 3;
    for (int p = 0; p < 4; ++p) {
      const int m = m0 + p * 64;
      As[buf][k4 * 4 + 0][m] = ra[p].x;
      As[buf][k4 * 4 + 1][m] = ra[p].y;
      As[buf][k4 * 4 + 2][m] = ra[p].z;
      As[buf][k4 * 4 + 3][m] = ra[p].w;
    }
    const int n4 = tid & 31, kk0 = tid >> 5;
    for (int p = 0; p < 2; ++p)
      *reinterpret_cast<float4*>(&Bs[buf][kk0 + p * 16][n4 * 4]) = rb[p];
  };
  load2(0);
  write2(0);
  int cur = 0;
  const int l31 = lane & 31, kh = lane >> 5;
  for (int kt = 0; kt < ktiles; ++kt) {
    __syncthreads();
    if (kt + 1 < ktiles) load2((kt + 1) * BK);
    for (int kk = 0; kk < BK; kk += 2) {
      const float a0 = As[cur][kk + kh][wr + l31];
      const float a1 = As[cur][kk + kh][wr + 32 + l31];
      const float b0 = Bs[cur][kk + kh][wc + l31];
      const float b1 = Bs[cur][kk + kh][wc + 32 + l31];
      acc[0][0] = __builtin_amdgcn_mfma_f32_32x32x2f32(a0, b0, acc[0][0], 0, 0, 0);
      acc[0][1] = __builtin_amdgcn_mfma_f32_32x32x2f32(a0, b1, acc[0][1], 0, 0, 0);
      acc[1][0] = __builtin_amdgcn_mfma_f32_32x32x2f32(a1, b0, acc[1][0], 0, 0, 0);
      acc[1][1] = __builtin_amdgcn_mfma_f32_32x32x2f32(a1, b1, acc[1][1], 0, 0, 0);
    }
    if (kt + 1 < ktiles) write2(cur ^ 1);
    cur ^= 1;
  }
  for (int mi = 0; mi < 2; ++mi)
    for (int nj = 0; nj < 2; ++nj) {
      const int col = block_col + wc + nj * 32 + (lane & 31);
      for (int r = 0; r < 16; ++r) {
        const int row = block_row + wr + mi * 32 + (r & 3) + 8 * (r >> 2) +
                        4 * (lane >> 5);
        C[(long)row * N + col] = acc[mi][nj][r];
      }
    }
}

double bench_big(const float* dA, const float* dB, float* dC, int Nsz,
                 int reps) {
  dim3 grid(Nsz / BM2, Nsz / BN2), block(512);
  hipLaunchKernelGGL(gemm_big, grid, block, 0, 0, dA, dB, dC, Nsz, Nsz, Nsz);
  HIP_CHECK(hipDeviceSynchronize());
  hipEvent_t t0, t1;
  hipEventCreate(&t0); hipEventCreate(&t1);
  hipEventRecord(t0);
  for (int i = 0; i < reps; ++i)
    hipLaunchKernelGGL(gemm_big, grid, block, 0, 0, dA, dB, dC, Nsz, Nsz, Nsz);
  hipEventRecord(t1);
  HIP_CHECK(hipEventSynchronize(t1));
  float ms; hipEventElapsedTime(&ms, t0, t1);
  return 2.0 * Nsz * Nsz * (double)Nsz * reps / (ms * 1e-3) / 1e12;
}

template <int V>
double bench(const float* dA, const float* dB, float* dC, int Nsz, int reps) {
  dim3 grid(Nsz / BM, Nsz / BN), block(256);
  hipLaunchKernelGGL((gemm_v<V>), grid, block, 0, 0, dA, dB, dC, Nsz, Nsz, Nsz);
  HIP_CHECK(hipDeviceSynchronize());
  hipEvent_t t0, t1;
  hipEventCreate(&t0); hipEventCreate(&t1);
  hipEventRecord(t0);
  for (int i = 0; i < reps; ++i)
    hipLaunchKernelGGL((gemm_v<V>), grid, block, 0, 0, dA, dB, dC, Nsz, Nsz, Nsz);
  hipEventRecord(t1);
  HIP_CHECK(hipEventSynchronize(t1));
  float ms; hipEventElapsedTime(&ms, t0, t1);
  return 2.0 * Nsz * Nsz * (double)Nsz * reps / (ms * 1e-3) / 1e12;
}

int main(int argc, char** argv) {
  const int Nsz = argc > 1 ? atoi(argv[1]) : 4096;
  const int reps = argc > 2 ? atoi(argv[2]) : 10;
  const int rounds = argc > 3 ? atoi(argv[3]) : 3;
  std::vector<float> hA((long)Nsz * Nsz), hB((long)Nsz * Nsz);
  srand(1);
  for (auto& v : hA) v = (rand() / (float)RAND_MAX) * 2 - 1;
  for (auto& v : hB) v = (rand() / (float)RAND_MAX) * 2 - 1;
  float *dA, *dB, *dC;
  HIP_CHECK(hipMalloc(&dA, (long)Nsz * Nsz * 4));
  HIP_CHECK(hipMalloc(&dB, (long)Nsz * Nsz * 4));
  HIP_CHECK(hipMalloc(&dC, (long)Nsz * Nsz * 4));
  HIP_CHECK(hipMemcpy(dA, hA.data(), (long)Nsz * Nsz * 4, hipMemcpyHostToDevice));
  HIP_CHECK(hipMemcpy(dB, hB.data(), (long)Nsz * Nsz * 4, hipMemcpyHostToDevice));

  // refcheck variant 0 on a 256x256 corner vs CPU
  {
    dim3 grid(Nsz / BM, Nsz / BN), block(256);
    hipLaunchKernelGGL((gemm_v<0>), grid, block, 0, 0, dA, dB, dC, Nsz, Nsz, Nsz);
    HIP_CHECK(hipDeviceSynchronize());
    std::vector<float> hC(256);
    HIP_CHECK(hipMemcpy(hC.data(), dC, 256 * 4, hipMemcpyDeviceToHost));
    for (int j = 0; j < 256; j += 37) {
      double ref = 0;
      for (int k = 0; k < Nsz; ++k) ref += (double)hA[k] * hB[(long)k * Nsz + j];
      if (fabs(ref - hC[j]) > 1e-2 * (fabs(ref) + 1)) {
        printf("REFCHECK FAIL at j=%d: %f vs %f\n", j, hC[j], ref);
        return 1;
      }
    }
    printf("refcheck ok\n");
  }

  if (Nsz % BM2 == 0) {  // bit-check big-tile vs baseline (exact fmaf chain)
    std::vector<float> h0((long)Nsz * Nsz), h1((long)Nsz * Nsz);
    dim3 g0(Nsz / BM, Nsz / BN), g1(Nsz / BM2, Nsz / BN2);
    hipLaunchKernelGGL((gemm_v<0>), g0, dim3(256), 0, 0, dA, dB, dC, Nsz, Nsz, Nsz);
    HIP_CHECK(hipMemcpy(h0.data(), dC, (long)Nsz * Nsz * 4, hipMemcpyDeviceToHost));
    hipLaunchKernelGGL(gemm_big, g1, dim3(512), 0, 0, dA, dB, dC, Nsz, Nsz, Nsz);
    HIP_CHECK(hipMemcpy(h1.data(), dC, (long)Nsz * Nsz * 4, hipMemcpyDeviceToHost));
    long bad = 0;
    for (long i = 0; i < (long)Nsz * Nsz; ++i) bad += (h0[i] != h1[i]);
    printf("bigcheck: %ld mismatches\n", bad);
  }
  {  // bit-check v5 (pair-interleaved b64 LDS) vs v0
    std::vector<float> h0((long)Nsz * Nsz), h1((long)Nsz * Nsz);
    dim3 g0(Nsz / BM, Nsz / BN);
    hipLaunchKernelGGL((gemm_v<0>), g0, dim3(256), 0, 0, dA, dB, dC, Nsz, Nsz, Nsz);
    HIP_CHECK(hipMemcpy(h0.data(), dC, (long)Nsz * Nsz * 4, hipMemcpyDeviceToHost));
    hipLaunchKernelGGL(gemm_v5, g0, dim3(256), 0, 0, dA, dB, dC, Nsz, Nsz, Nsz);
    HIP_CHECK(hipMemcpy(h1.data(), dC, (long)Nsz * Nsz * 4, hipMemcpyDeviceToHost));
    long bad = 0;
    for (long i = 0; i < (long)Nsz * Nsz; ++i) bad += (h0[i] != h1[i]);
    printf("v5check: %ld mismatches\n", bad);
    if (bad) return 1;
    hipLaunchKernelGGL(gemm_v6, g0, dim3(256), 0, 0, dA, dB, dC, Nsz, Nsz, Nsz);
    HIP_CHECK(hipMemcpy(h1.data(), dC, (long)Nsz * Nsz * 4, hipMemcpyDeviceToHost));
    bad = 0;
    for (long i = 0; i < (long)Nsz * Nsz; ++i) bad += (h0[i] != h1[i]);
    printf("v6check: %ld mismatches\n", bad);
    if (bad) return 1;
  }
  for (int round = 0; round < rounds; ++round) {
    printf("round %d: v0=%6.1f v1=%6.1f v5pair=%6.1f v6apair=%6.1f big=%6.1f TF\n",
           round,
           bench<0>(dA, dB, dC, Nsz, reps), bench<1>(dA, dB, dC, Nsz, reps),
           bench_v5(dA, dB, dC, Nsz, reps), bench_v6(dA, dB, dC, Nsz, reps),
           bench_big(dA, dB, dC, Nsz, reps));
  }
  return 0;
}

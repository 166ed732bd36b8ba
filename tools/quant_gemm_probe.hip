// Variant probe for the (exp,man)-Kahan-accumulator GEMM (VALU kernel).
// Build: hipcc --offload-arch=gfx950 -O3 -Wno-unused-value tools/quant_gemm_probe.hip -o tools/quant_gemm_probe
#include <hip/hip_runtime.h>
#include <cstdio>
#include <cstdlib>
#include <cmath>
#include <vector>

#include "../cpd_amd/ops/csrc/quant_core.h"

#define HIP_CHECK(x) do { hipError_t e = (x); if (e) { \
  printf("HIP error %s @%d\n", hipGetErrorString(e), __LINE__); exit(1);} } while (0)

using namespace cpd;
constexpr int QBM = 64, QBN = 64, QBK = 16;

template <int VARIANT>
__global__ __launch_bounds__(256) void quant_gemm_v(
    const float* __restrict__ A, const float* __restrict__ B,
    float* __restrict__ C, int M, int N, int K, int man, int exp) {
  __shared__ float As[QBK][QBM + 1];
  __shared__ float Bs[QBK][QBN];
  const int tx = threadIdx.x & 15;
  const int ty = threadIdx.x >> 4;
  const int row0 = blockIdx.x * QBM + ty * 4;
  const int col0 = blockIdx.y * QBN + tx * 4;
  float acc[4][4] = {};
  float comp[4][4] = {};
  const int ktiles = (K + QBK - 1) / QBK;
  for (int kt = 0; kt < ktiles; ++kt) {
    const int k0 = kt * QBK;
    {
      const int k = threadIdx.x & 15;
      const int m0 = threadIdx.x >> 4;
      for (int p = 0; p < 4; ++p) {
        const int m = m0 + p * 16;
        const int gm = blockIdx.x * QBM + m;
        As[k][m] = (gm < M && k0 + k < K) ? A[(long)gm * K + k0 + k] : 0.0f;
      }
      const int n = threadIdx.x & 63;
      const int kk0 = threadIdx.x >> 6;
      for (int p = 0; p < 4; ++p) {
        const int kk = kk0 + p * 4;
        const int gn = blockIdx.y * QBN + n;
        Bs[kk][n] = (k0 + kk < K && gn < N) ? B[(long)(k0 + kk) * N + gn] : 0.0f;
      }
    }
    __syncthreads();
    const int klim = min(QBK, K - k0);
    for (int kk = 0; kk < klim; ++kk) {
      float a[4], b[4];
#pragma unroll
      for (int i = 0; i < 4; ++i) a[i] = As[kk][ty * 4 + i];
#pragma unroll
      for (int j = 0; j < 4; ++j) b[j] = Bs[kk][tx * 4 + j];
      if constexpr (VARIANT == 0) {
        // nested per-output kahan (current production structure)
#pragma unroll
        for (int i = 0; i < 4; ++i)
#pragma unroll
          for (int j = 0; j < 4; ++j) {
            const float prod = cast_fp(a[i] * b[j], man, exp);
            kahan_qstep(acc[i][j], comp[i][j], prod, man, exp);
          }
      } else if constexpr (VARIANT == 1) {
        // stage-batched: run each rounding stage across all 16 chains
        float prod[16], yv[16], tv[16];
#pragma unroll
        for (int i = 0; i < 4; ++i)
#pragma unroll
          for (int j = 0; j < 4; ++j)
            prod[i * 4 + j] = cast_fp(a[i] * b[j], man, exp);
#pragma unroll
        for (int q = 0; q < 16; ++q)
          yv[q] = cast_fp(prod[q] - comp[q >> 2][q & 3], man, exp);
#pragma unroll
        for (int q = 0; q < 16; ++q)
          tv[q] = cast_fp(acc[q >> 2][q & 3] + yv[q], man, exp);
#pragma unroll
        for (int q = 0; q < 16; ++q) {
          comp[q >> 2][q & 3] = cast_fp(
              cast_fp(tv[q] - acc[q >> 2][q & 3], man, exp) - yv[q], man, exp);
          acc[q >> 2][q & 3] = tv[q];
        }
      }
    }
    __syncthreads();
  }
  for (int i = 0; i < 4; ++i) {
    if (row0 + i >= M) break;
    for (int j = 0; j < 4; ++j)
      if (col0 + j < N) C[(long)(row0 + i) * N + col0 + j] = acc[i][j];
  }
}

// 32x32 tile, 2x2 per thread: 4x the resident waves (the 64x64 tile yields
// only (N/64)^2 blocks = 1 wave/SIMD at 1024^3 — a latency-bound VALU kernel
// then eats every cast-chain dependency stall).
__global__ __launch_bounds__(256) void quant_gemm_32(
    const float* __restrict__ A, const float* __restrict__ B,
    float* __restrict__ C, int M, int N, int K, int man, int exp) {
  constexpr int TB = 32, TK = 16;
  __shared__ float As[TK][TB + 1];
  __shared__ float Bs[TK][TB];
  const int tx = threadIdx.x & 15;
  const int ty = threadIdx.x >> 4;
  const int row0 = blockIdx.x * TB + ty * 2;
  const int col0 = blockIdx.y * TB + tx * 2;
  float acc[2][2] = {};
  float comp[2][2] = {};
  const int ktiles = (K + TK - 1) / TK;
  for (int kt = 0; kt < ktiles; ++kt) {
    const int k0 = kt * TK;
    {
      const int k = threadIdx.x & 15;       // A: [32][16] transposed, 2/thread
      const int m0 = threadIdx.x >> 4;
      for (int p = 0; p < 2; ++p) {
        const int m = m0 + p * 16;
        const int gm = blockIdx.x * TB + m;
        As[k][m] = (gm < M && k0 + k < K) ? A[(long)gm * K + k0 + k] : 0.0f;
      }
      const int n = threadIdx.x & 31;       // B: [16][32], 2/thread
      const int kk0 = threadIdx.x >> 5;
      for (int p = 0; p < 2; ++p) {
        const int kk = kk0 + p * 8;
        const int gn = blockIdx.y * TB + n;
        Bs[kk][n] = (k0 + kk < K && gn < N) ? B[(long)(k0 + kk) * N + gn] : 0.0f;
      }
    }
    __syncthreads();
    const int klim = min(TK, K - k0);
    for (int kk = 0; kk < klim; ++kk) {
      float a[2], b[2];
      a[0] = As[kk][ty * 2]; a[1] = As[kk][ty * 2 + 1];
      b[0] = Bs[kk][tx * 2]; b[1] = Bs[kk][tx * 2 + 1];
#pragma unroll
      for (int i = 0; i < 2; ++i)
#pragma unroll
        for (int j = 0; j < 2; ++j) {
          const float prod = cast_fp(a[i] * b[j], man, exp);
          kahan_qstep(acc[i][j], comp[i][j], prod, man, exp);
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

double bench32(const float* dA, const float* dB, float* dC, int Nsz,
               int reps) {
  dim3 grid((Nsz + 31) / 32, (Nsz + 31) / 32), block(256);
  hipLaunchKernelGGL(quant_gemm_32, grid, block, 0, 0, dA, dB, dC, Nsz, Nsz,
                     Nsz, 3, 4);
  HIP_CHECK(hipDeviceSynchronize());
  hipEvent_t t0, t1;
  hipEventCreate(&t0); hipEventCreate(&t1);
  hipEventRecord(t0);
  for (int i = 0; i < reps; ++i)
    hipLaunchKernelGGL(quant_gemm_32, grid, block, 0, 0, dA, dB, dC, Nsz, Nsz,
                       Nsz, 3, 4);
  hipEventRecord(t1);
  HIP_CHECK(hipEventSynchronize(t1));
  float ms; hipEventElapsedTime(&ms, t0, t1);
  return 2.0 * Nsz * Nsz * (double)Nsz * reps / (ms * 1e-3) / 1e12;
}

template <int V>
double bench(const float* dA, const float* dB, float* dC, int Nsz, int reps) {
  dim3 grid(Nsz / QBM, Nsz / QBN), block(256);
  hipLaunchKernelGGL((quant_gemm_v<V>), grid, block, 0, 0, dA, dB, dC, Nsz,
                     Nsz, Nsz, 3, 4);
  HIP_CHECK(hipDeviceSynchronize());
  hipEvent_t t0, t1;
  hipEventCreate(&t0); hipEventCreate(&t1);
  hipEventRecord(t0);
  for (int i = 0; i < reps; ++i)
    hipLaunchKernelGGL((quant_gemm_v<V>), grid, block, 0, 0, dA, dB, dC, Nsz,
                       Nsz, Nsz, 3, 4);
  hipEventRecord(t1);
  HIP_CHECK(hipEventSynchronize(t1));
  float ms; hipEventElapsedTime(&ms, t0, t1);
  return 2.0 * Nsz * Nsz * (double)Nsz * reps / (ms * 1e-3) / 1e12;
}

int main(int argc, char** argv) {
  const int Nsz = argc > 1 ? atoi(argv[1]) : 1024;
  const int reps = argc > 2 ? atoi(argv[2]) : 3;
  std::vector<float> hA((long)Nsz * Nsz), hB((long)Nsz * Nsz);
  srand(1);
  for (auto& v : hA) v = (rand() / (float)RAND_MAX) * 2 - 1;
  for (auto& v : hB) v = (rand() / (float)RAND_MAX) * 2 - 1;
  float *dA, *dB, *dC0, *dC1;
  HIP_CHECK(hipMalloc(&dA, (long)Nsz * Nsz * 4));
  HIP_CHECK(hipMalloc(&dB, (long)Nsz * Nsz * 4));
  HIP_CHECK(hipMalloc(&dC0, (long)Nsz * Nsz * 4));
  HIP_CHECK(hipMalloc(&dC1, (long)Nsz * Nsz * 4));
  HIP_CHECK(hipMemcpy(dA, hA.data(), (long)Nsz * Nsz * 4, hipMemcpyHostToDevice));
  HIP_CHECK(hipMemcpy(dB, hB.data(), (long)Nsz * Nsz * 4, hipMemcpyHostToDevice));
  // bit-equality check v1 vs v0 (same rounding order is required)
  {
    dim3 grid(Nsz / QBM, Nsz / QBN), block(256);
    hipLaunchKernelGGL((quant_gemm_v<0>), grid, block, 0, 0, dA, dB, dC0,
                       Nsz, Nsz, Nsz, 3, 4);
    hipLaunchKernelGGL((quant_gemm_v<1>), grid, block, 0, 0, dA, dB, dC1,
                       Nsz, Nsz, Nsz, 3, 4);
    HIP_CHECK(hipDeviceSynchronize());
    std::vector<float> h0((long)Nsz * Nsz), h1((long)Nsz * Nsz);
    HIP_CHECK(hipMemcpy(h0.data(), dC0, (long)Nsz * Nsz * 4, hipMemcpyDeviceToHost));
    HIP_CHECK(hipMemcpy(h1.data(), dC1, (long)Nsz * Nsz * 4, hipMemcpyDeviceToHost));
    for (long i = 0; i < (long)Nsz * Nsz; ++i)
      if (h0[i] != h1[i]) { printf("BITCHECK FAIL at %ld\n", i); return 1; }
    printf("bitcheck ok\n");
  }
  // bit-check the 32-tile variant (same K order per output element)
  {
    dim3 g32((Nsz + 31) / 32, (Nsz + 31) / 32);
    hipLaunchKernelGGL(quant_gemm_32, g32, dim3(256), 0, 0, dA, dB, dC1,
                       Nsz, Nsz, Nsz, 3, 4);
    HIP_CHECK(hipDeviceSynchronize());
    std::vector<float> h0((long)Nsz * Nsz), h1((long)Nsz * Nsz);
    HIP_CHECK(hipMemcpy(h0.data(), dC0, (long)Nsz * Nsz * 4, hipMemcpyDeviceToHost));
    HIP_CHECK(hipMemcpy(h1.data(), dC1, (long)Nsz * Nsz * 4, hipMemcpyDeviceToHost));
    for (long i = 0; i < (long)Nsz * Nsz; ++i)
      if (h0[i] != h1[i]) { printf("32TILE BITCHECK FAIL at %ld\n", i); return 1; }
    printf("32-tile bitcheck ok\n");
  }
  for (int r = 0; r < 3; ++r)
    printf("round %d: v0=%7.3f v1=%7.3f v32=%7.3f TF\n", r,
           bench<0>(dA, dB, dC0, Nsz, reps), bench<1>(dA, dB, dC1, Nsz, reps),
           bench32(dA, dB, dC1, Nsz, reps));
  return 0;
}

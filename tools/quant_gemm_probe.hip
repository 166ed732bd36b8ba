// Variant probe for the (exp,man)-Kahan-accumulator GEMM (VALU kernel).
// Round 2: cast_fp_fast (v_frexp/v_ldexp/v_trunc/v_rndne formulation, ~20
// VALU vs ~35) against the integer-path baseline, plus tile/K variants.
// Every variant preserves the strict per-output-element K order, so all
// results must be BIT-IDENTICAL — checked before timing.
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

// ---------------------------------------------------------------------------
// 32x32 tile, 16x16 threads, 2x2 outputs/lane.  FAST selects cast_fp_fast.
// TK = K-tile depth (barrier period).
// ---------------------------------------------------------------------------
template <bool FAST, int TK>
__global__ __launch_bounds__(256) void qg32(
    const float* __restrict__ A, const float* __restrict__ B,
    float* __restrict__ C, int M, int N, int K, int man, int exp) {
  constexpr int TB = 32;
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
      // A: [TB][TK] transposed; TB*TK elems over 256 threads
      for (int idx = threadIdx.x; idx < TB * TK; idx += 256) {
        const int k = idx % TK, m = idx / TK;
        const int gm = blockIdx.x * TB + m;
        As[k][m] = (gm < M && k0 + k < K) ? A[(long)gm * K + k0 + k] : 0.0f;
      }
      for (int idx = threadIdx.x; idx < TK * TB; idx += 256) {
        const int n = idx % TB, kk = idx / TB;
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
          if constexpr (FAST) {
            const float prod = cast_fp_fast(a[i] * b[j], man, exp);
            kahan_qstep_fast(acc[i][j], comp[i][j], prod, man, exp);
          } else {
            const float prod = cast_fp(a[i] * b[j], man, exp);
            kahan_qstep(acc[i][j], comp[i][j], prod, man, exp);
          }
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

// 64x32 tile, 256 threads, 4x2 outputs/lane (8 chains: more ILP per lane,
// half the blocks of the 32-tile).
template <bool FAST>
__global__ __launch_bounds__(256) void qg64x32(
    const float* __restrict__ A, const float* __restrict__ B,
    float* __restrict__ C, int M, int N, int K, int man, int exp) {
  constexpr int TM = 64, TN = 32, TK = 16;
  __shared__ float As[TK][TM + 1];
  __shared__ float Bs[TK][TN];
  const int tx = threadIdx.x & 15;   // 16 cols of lanes * 2 outputs
  const int ty = threadIdx.x >> 4;   // 16 rows of lanes * 4 outputs
  const int row0 = blockIdx.x * TM + ty * 4;
  const int col0 = blockIdx.y * TN + tx * 2;
  float acc[4][2] = {};
  float comp[4][2] = {};
  const int ktiles = (K + TK - 1) / TK;
  for (int kt = 0; kt < ktiles; ++kt) {
    const int k0 = kt * TK;
    for (int idx = threadIdx.x; idx < TM * TK; idx += 256) {
      const int k = idx % TK, m = idx / TK;
      const int gm = blockIdx.x * TM + m;
      As[k][m] = (gm < M && k0 + k < K) ? A[(long)gm * K + k0 + k] : 0.0f;
    }
    for (int idx = threadIdx.x; idx < TK * TN; idx += 256) {
      const int n = idx % TN, kk = idx / TN;
      const int gn = blockIdx.y * TN + n;
      Bs[kk][n] = (k0 + kk < K && gn < N) ? B[(long)(k0 + kk) * N + gn] : 0.0f;
    }
    __syncthreads();
    const int klim = min(TK, K - k0);
    for (int kk = 0; kk < klim; ++kk) {
      float a[4], b[2];
#pragma unroll
      for (int i = 0; i < 4; ++i) a[i] = As[kk][ty * 4 + i];
      b[0] = Bs[kk][tx * 2]; b[1] = Bs[kk][tx * 2 + 1];
#pragma unroll
      for (int i = 0; i < 4; ++i)
#pragma unroll
        for (int j = 0; j < 2; ++j) {
          if constexpr (FAST) {
            const float prod = cast_fp_fast(a[i] * b[j], man, exp);
            kahan_qstep_fast(acc[i][j], comp[i][j], prod, man, exp);
          } else {
            const float prod = cast_fp(a[i] * b[j], man, exp);
            kahan_qstep(acc[i][j], comp[i][j], prod, man, exp);
          }
        }
    }
    __syncthreads();
  }
  for (int i = 0; i < 4; ++i) {
    if (row0 + i >= M) break;
    for (int j = 0; j < 2; ++j)
      if (col0 + j < N) C[(long)(row0 + i) * N + col0 + j] = acc[i][j];
  }
}

// -------------------------- harness ----------------------------------------

struct Variant {
  const char* name;
  void (*launch)(const float*, const float*, float*, int, int);
};

template <bool FAST, int TK>
void launch32(const float* dA, const float* dB, float* dC, int Nsz, int me) {
  dim3 grid((Nsz + 31) / 32, (Nsz + 31) / 32);
  hipLaunchKernelGGL((qg32<FAST, TK>), grid, dim3(256), 0, 0, dA, dB, dC,
                     Nsz, Nsz, Nsz, me & 0xff, me >> 8);
}

template <bool FAST>
void launch64x32(const float* dA, const float* dB, float* dC, int Nsz, int me) {
  dim3 grid((Nsz + 63) / 64, (Nsz + 31) / 32);
  hipLaunchKernelGGL((qg64x32<FAST>), grid, dim3(256), 0, 0, dA, dB, dC,
                     Nsz, Nsz, Nsz, me & 0xff, me >> 8);
}

int main(int argc, char** argv) {
  const int Nsz = argc > 1 ? atoi(argv[1]) : 1024;
  const int reps = argc > 2 ? atoi(argv[2]) : 3;
  const int man = argc > 3 ? atoi(argv[3]) : 3;
  const int exp = argc > 4 ? atoi(argv[4]) : 4;
  const int me = man | (exp << 8);
  std::vector<float> hA((long)Nsz * Nsz), hB((long)Nsz * Nsz);
  srand(1);
  for (auto& v : hA) v = (rand() / (float)RAND_MAX) * 2 - 1;
  for (auto& v : hB) v = (rand() / (float)RAND_MAX) * 2 - 1;
  float *dA, *dB, *dRef, *dC;
  HIP_CHECK(hipMalloc(&dA, (long)Nsz * Nsz * 4));
  HIP_CHECK(hipMalloc(&dB, (long)Nsz * Nsz * 4));
  HIP_CHECK(hipMalloc(&dRef, (long)Nsz * Nsz * 4));
  HIP_CHECK(hipMalloc(&dC, (long)Nsz * Nsz * 4));
  HIP_CHECK(hipMemcpy(dA, hA.data(), (long)Nsz * Nsz * 4, hipMemcpyHostToDevice));
  HIP_CHECK(hipMemcpy(dB, hB.data(), (long)Nsz * Nsz * 4, hipMemcpyHostToDevice));

  struct V { const char* name; void (*fn)(const float*, const float*, float*, int, int); };
  V vs[] = {
      {"v32 base (int cast)  ", launch32<false, 16>},
      {"v32 fast cast        ", launch32<true, 16>},
      {"v32 fast cast TK=32  ", launch32<true, 32>},
      {"v64x32 fast (8 chain)", launch64x32<true>},
  };
  const int NV = sizeof(vs) / sizeof(vs[0]);

  // bit-identity of every variant vs the baseline
  vs[0].fn(dA, dB, dRef, Nsz, me);
  HIP_CHECK(hipDeviceSynchronize());
  std::vector<float> h0((long)Nsz * Nsz), h1((long)Nsz * Nsz);
  HIP_CHECK(hipMemcpy(h0.data(), dRef, (long)Nsz * Nsz * 4, hipMemcpyDeviceToHost));
  for (int v = 1; v < NV; ++v) {
    vs[v].fn(dA, dB, dC, Nsz, me);
    HIP_CHECK(hipDeviceSynchronize());
    HIP_CHECK(hipMemcpy(h1.data(), dC, (long)Nsz * Nsz * 4, hipMemcpyDeviceToHost));
    long bad = -1;
    for (long i = 0; i < (long)Nsz * Nsz; ++i)
      if (h0[i] != h1[i]) { bad = i; break; }
    printf("%s bitcheck %s\n", vs[v].name, bad < 0 ? "ok" : "FAIL");
    if (bad >= 0) return 1;
  }

  for (int r = 0; r < 3; ++r) {
    for (int v = 0; v < NV; ++v) {
      vs[v].fn(dA, dB, dC, Nsz, me);  // warm
      HIP_CHECK(hipDeviceSynchronize());
      hipEvent_t t0, t1;
      hipEventCreate(&t0); hipEventCreate(&t1);
      hipEventRecord(t0);
      for (int i = 0; i < reps; ++i) vs[v].fn(dA, dB, dC, Nsz, me);
      hipEventRecord(t1);
      HIP_CHECK(hipEventSynchronize(t1));
      float ms; hipEventElapsedTime(&ms, t0, t1);
      const double tf = 2.0 * Nsz * Nsz * (double)Nsz * reps / (ms * 1e-3) / 1e12;
      printf("round %d  %s %8.3f TF (e%dm%d)\n", r, vs[v].name, tf, exp, man);
      hipEventDestroy(t0); hipEventDestroy(t1);
    }
  }
  return 0;
}

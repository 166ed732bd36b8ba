// Standalone elementwise-kernel probe (no torch) for rocprofv3 --pmc runs:
// exercises the (exp,man) cast, quantized-accumulate and Kahan hop kernels.
// Build: hipcc --offload-arch=gfx950 -O3 -Wno-unused-value tools/elem_probe.hip -o tools/elem_probe
#include <hip/hip_runtime.h>
#include <cstdio>
#include <cstdlib>
#include <vector>

#include "../cpd_amd/ops/csrc/quant_core.h"

#define HIP_CHECK(x) do { hipError_t e = (x); if (e) { \
  printf("HIP error %s @%d\n", hipGetErrorString(e), __LINE__); exit(1);} } while (0)

using namespace cpd;
constexpr int TPB = 256;
constexpr int VEC = 4;

__global__ void quantize_k(float* __restrict__ x, long n, int man, int exp) {
  const long stride = (long)gridDim.x * TPB * VEC;
  for (long i = ((long)blockIdx.x * TPB + threadIdx.x) * VEC; i + VEC <= n;
       i += stride) {
    float4 v = *reinterpret_cast<float4*>(x + i);
    v.x = cast_fp(v.x, man, exp);
    v.y = cast_fp(v.y, man, exp);
    v.z = cast_fp(v.z, man, exp);
    v.w = cast_fp(v.w, man, exp);
    *reinterpret_cast<float4*>(x + i) = v;
  }
}

__global__ void qadd_k(float* __restrict__ a, const float* __restrict__ g,
                       long n, int man, int exp) {
  const long stride = (long)gridDim.x * TPB * VEC;
  for (long i = ((long)blockIdx.x * TPB + threadIdx.x) * VEC; i + VEC <= n;
       i += stride) {
    float4 v = *reinterpret_cast<float4*>(a + i);
    const float4 w = *reinterpret_cast<const float4*>(g + i);
    v.x = cast_fp(v.x + w.x, man, exp);
    v.y = cast_fp(v.y + w.y, man, exp);
    v.z = cast_fp(v.z + w.z, man, exp);
    v.w = cast_fp(v.w + w.w, man, exp);
    *reinterpret_cast<float4*>(a + i) = v;
  }
}

__global__ void kahan_k(float* __restrict__ a, float* __restrict__ c,
                        const float* __restrict__ g, long n, int man, int exp) {
  const long stride = (long)gridDim.x * TPB * VEC;
  for (long i = ((long)blockIdx.x * TPB + threadIdx.x) * VEC; i + VEC <= n;
       i += stride) {
    float4 v = *reinterpret_cast<float4*>(a + i);
    float4 cc = *reinterpret_cast<float4*>(c + i);
    const float4 w = *reinterpret_cast<const float4*>(g + i);
    kahan_qstep(v.x, cc.x, w.x, man, exp);
    kahan_qstep(v.y, cc.y, w.y, man, exp);
    kahan_qstep(v.z, cc.z, w.z, man, exp);
    kahan_qstep(v.w, cc.w, w.w, man, exp);
    *reinterpret_cast<float4*>(a + i) = v;
    *reinterpret_cast<float4*>(c + i) = cc;
  }
}

int main(int argc, char** argv) {
  const long n = argc > 1 ? atol(argv[1]) : 16 * 1024 * 1024;
  const int reps = argc > 2 ? atoi(argv[2]) : 3;
  float *a, *c, *g;
  HIP_CHECK(hipMalloc(&a, n * 4));
  HIP_CHECK(hipMalloc(&c, n * 4));
  HIP_CHECK(hipMalloc(&g, n * 4));
  std::vector<float> h(n);
  srand(2);
  for (auto& v : h) v = (rand() / (float)RAND_MAX) * 4 - 2;
  HIP_CHECK(hipMemcpy(a, h.data(), n * 4, hipMemcpyHostToDevice));
  HIP_CHECK(hipMemcpy(g, h.data(), n * 4, hipMemcpyHostToDevice));
  HIP_CHECK(hipMemset(c, 0, n * 4));
  const int grid = (int)std::min<long>((n + TPB * VEC - 1) / (TPB * VEC), 16384);
  for (int r = 0; r < reps; ++r) {
    hipLaunchKernelGGL(quantize_k, dim3(grid), dim3(TPB), 0, 0, a, n, 3, 4);
    hipLaunchKernelGGL(qadd_k, dim3(grid), dim3(TPB), 0, 0, a, g, n, 3, 4);
    hipLaunchKernelGGL(kahan_k, dim3(grid), dim3(TPB), 0, 0, a, c, g, n, 3, 4);
  }
  HIP_CHECK(hipDeviceSynchronize());
  printf("elem probe done (n=%ld reps=%d)\n", n, reps);
  return 0;
}

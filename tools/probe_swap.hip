// Probe: cross-half reduction semantics of v_permlane32_swap on gfx950.
#include <hip/hip_runtime.h>
#include <cstdio>

__device__ inline float cross_half_max(float v) {
  unsigned a = __builtin_bit_cast(unsigned, v);
  unsigned b = a;
  asm volatile("" : "+v"(b));
  auto r = __builtin_amdgcn_permlane32_swap(a, b, false, false);
  return fmaxf(__builtin_bit_cast(float, r[0]), __builtin_bit_cast(float, r[1]));
}
__device__ inline float cross_half_sum(float v) {
  unsigned a = __builtin_bit_cast(unsigned, v);
  unsigned b = a;
  asm volatile("" : "+v"(b));
  auto r = __builtin_amdgcn_permlane32_swap(a, b, false, false);
  return __builtin_bit_cast(float, r[0]) + __builtin_bit_cast(float, r[1]);
}

__global__ void probe(float* out_sum, float* out_max) {
  float v = (float)(threadIdx.x * threadIdx.x % 97);  // arbitrary distinct
  out_sum[threadIdx.x] = cross_half_sum(v);
  out_max[threadIdx.x] = cross_half_max(v);
}

int main() {
  float *s, *m;
  (void)hipMalloc(&s, 64 * 4); (void)hipMalloc(&m, 64 * 4);
  hipLaunchKernelGGL(probe, dim3(1), dim3(64), 0, 0, s, m);
  (void)hipDeviceSynchronize();
  float hs[64], hm[64];
  (void)hipMemcpy(hs, s, sizeof(hs), hipMemcpyDeviceToHost);
  (void)hipMemcpy(hm, m, sizeof(hm), hipMemcpyDeviceToHost);
  int bad = 0;
  for (int l = 0; l < 64; ++l) {
    float v = (float)(l * l % 97), p = (float)((l ^ 32) * (l ^ 32) % 97);
    float es = v + p, em = v > p ? v : p;
    if (hs[l] != es || hm[l] != em) {
      ++bad;
      if (bad < 9)
        printf("lane %2d: sum got %.0f want %.0f | max got %.0f want %.0f\n",
               l, hs[l], es, hm[l], em);
    }
  }
  printf(bad ? "SWAP SEMANTICS WRONG (%d lanes)\n" : "swap semantics OK\n", bad);
  return 0;
}

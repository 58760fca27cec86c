// Fused sampling kernel.
//
// For rows without top-k/top-p filtering (the serving common case), sampling
// from softmax(logits/T) with per-request exponential noise q reduces to
//   token = argmax_v( logits[v]/T - log q[v] )        (T > 0)
//   token = argmax_v( logits[v] )                     (T == 0, greedy)
// — the exponential-race identity argmax(p/q) — so the whole
// temperature/softmax/divide/argmax chain is ONE streaming pass over the
// logits row.  The noise tensor comes from torch's per-request seeded
// generators, so seeded-sampling semantics are bit-identical to the
// reference torch pipeline it replaces.
//
// Ties resolve to the lowest index (torch.argmax convention).

#include "common.h"
#include <float.h>

template <typename T>
__global__ __launch_bounds__(256) void sample_argmax_kernel(
    long* __restrict__ out,          // [N]
    const T* __restrict__ logits,    // [N, V]
    const float* __restrict__ temps, // [N] (0 = greedy)
    const float* __restrict__ noise, // [N, V] Exp(1) draws, or null
    const int V) {
  const int row = blockIdx.x;
  const float temp = temps[row];
  const T* lrow = logits + (long)row * V;
  const float* nrow = noise ? noise + (long)row * V : nullptr;
  const bool sample = temp > 0.f && nrow != nullptr;
  const float inv_t = sample ? 1.f / temp : 1.f;

  float best = -FLT_MAX;
  int best_idx = V;
  constexpr int E = Vec16<T>::kElems;
  for (int i = threadIdx.x * E; i + E <= V; i += blockDim.x * E) {
    const Vec16<T> lv = load16(lrow + i);
#pragma unroll
    for (int j = 0; j < E; ++j) {
      float v = ScalarOps<T>::to_f32(lv.data[j]);
      if (sample) v = v * inv_t - __logf(nrow[i + j]);
      const int idx = i + j;
      if (v > best || (v == best && idx < best_idx)) {
        best = v;
        best_idx = idx;
      }
    }
  }
  // tail
  for (int i = (V / E) * E + threadIdx.x; i < V; i += blockDim.x) {
    float v = ScalarOps<T>::to_f32(lrow[i]);
    if (sample) v = v * inv_t - __logf(nrow[i]);
    if (v > best || (v == best && i < best_idx)) {
      best = v;
      best_idx = i;
    }
  }

  // wave reduce (max val, min idx on ties)
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    const float ov = __shfl_xor(best, off, 64);
    const int oi = __shfl_xor(best_idx, off, 64);
    if (ov > best || (ov == best && oi < best_idx)) {
      best = ov;
      best_idx = oi;
    }
  }
  __shared__ float lds_v[4];
  __shared__ int lds_i[4];
  const int wave = threadIdx.x >> 6;
  if ((threadIdx.x & 63) == 0) {
    lds_v[wave] = best;
    lds_i[wave] = best_idx;
  }
  __syncthreads();
  if (threadIdx.x == 0) {
    for (int w = 1; w < 4; ++w) {
      if (lds_v[w] > best || (lds_v[w] == best && lds_i[w] < best_idx)) {
        best = lds_v[w];
        best_idx = lds_i[w];
      }
    }
    out[row] = best_idx;
  }
}

template <typename T>
void launch_sample_argmax(long* out, const T* logits, const float* temps,
                          const float* noise, int N, int V,
                          hipStream_t stream) {
  hipLaunchKernelGGL(sample_argmax_kernel<T>, dim3(N), dim3(256), 0, stream,
                     out, logits, temps, noise, V);
}

#define INSTANTIATE_SAMPLE(T)                                                \
  template void launch_sample_argmax<T>(long*, const T*, const float*,       \
                                        const float*, int, int, hipStream_t);

INSTANTIATE_SAMPLE(float)
INSTANTIATE_SAMPLE(__hip_bfloat16)
INSTANTIATE_SAMPLE(__half)

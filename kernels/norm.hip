// Fused RMSNorm kernels.
// Memory-bound: one workgroup per row, 16-B vector loads (guide G13 — scalar
// bf16 loads cost ~2x), sum-of-squares in f32, one pass read + one pass write.

#include "common.h"

template <typename T>
__global__ void rms_norm_kernel(
    T* __restrict__ out,          // [rows, hidden]
    const T* __restrict__ input,  // [rows, hidden]
    const T* __restrict__ weight, // [hidden]
    const float eps,
    const int hidden) {
  constexpr int V = Vec16<T>::kElems;
  const int row = blockIdx.x;
  const T* in_row = input + (long)row * hidden;
  T* out_row = out + (long)row * hidden;

  float ss = 0.f;
  for (int i = threadIdx.x * V; i < hidden; i += blockDim.x * V) {
    Vec16<T> v = load16(in_row + i);
#pragma unroll
    for (int j = 0; j < V; ++j) {
      float x = ScalarOps<T>::to_f32(v.data[j]);
      ss += x * x;
    }
  }
  __shared__ float lds[16];
  ss = block_reduce_sum(ss, lds);
  const float inv = rsqrtf(ss / hidden + eps);

  for (int i = threadIdx.x * V; i < hidden; i += blockDim.x * V) {
    Vec16<T> v = load16(in_row + i);
    Vec16<T> w = load16(weight + i);
#pragma unroll
    for (int j = 0; j < V; ++j) {
      float x = ScalarOps<T>::to_f32(v.data[j]) * inv *
                ScalarOps<T>::to_f32(w.data[j]);
      v.data[j] = ScalarOps<T>::from_f32(x);
    }
    store16(out_row + i, v);
  }
}

// x <- rmsnorm(x + residual) * w ; residual <- x + residual   (both in place)
template <typename T>
__global__ void fused_add_rms_norm_kernel(
    T* __restrict__ x,          // [rows, hidden]
    T* __restrict__ residual,   // [rows, hidden]
    const T* __restrict__ weight,
    const float eps,
    const int hidden) {
  constexpr int V = Vec16<T>::kElems;
  const int row = blockIdx.x;
  T* x_row = x + (long)row * hidden;
  T* r_row = residual + (long)row * hidden;

  float ss = 0.f;
  for (int i = threadIdx.x * V; i < hidden; i += blockDim.x * V) {
    Vec16<T> xv = load16(x_row + i);
    Vec16<T> rv = load16(r_row + i);
#pragma unroll
    for (int j = 0; j < V; ++j) {
      float s = ScalarOps<T>::to_f32(xv.data[j]) + ScalarOps<T>::to_f32(rv.data[j]);
      rv.data[j] = ScalarOps<T>::from_f32(s);
      // re-read the rounded sum so the norm matches what is stored
      float sr = ScalarOps<T>::to_f32(rv.data[j]);
      ss += sr * sr;
    }
    store16(r_row + i, rv);
  }
  __shared__ float lds[16];
  ss = block_reduce_sum(ss, lds);
  const float inv = rsqrtf(ss / hidden + eps);

  for (int i = threadIdx.x * V; i < hidden; i += blockDim.x * V) {
    Vec16<T> rv = load16(r_row + i);
    Vec16<T> wv = load16(weight + i);
#pragma unroll
    for (int j = 0; j < V; ++j) {
      float v = ScalarOps<T>::to_f32(rv.data[j]) * inv *
                ScalarOps<T>::to_f32(wv.data[j]);
      rv.data[j] = ScalarOps<T>::from_f32(v);
    }
    store16(x_row + i, rv);
  }
}

template <typename T>
void launch_rms_norm(T* out, const T* in, const T* w, float eps, int rows,
                     int hidden, hipStream_t stream) {
  const int threads = 256;
  hipLaunchKernelGGL(rms_norm_kernel<T>, dim3(rows), dim3(threads), 0, stream,
                     out, in, w, eps, hidden);
}

template <typename T>
void launch_fused_add_rms_norm(T* x, T* res, const T* w, float eps, int rows,
                               int hidden, hipStream_t stream) {
  const int threads = 256;
  hipLaunchKernelGGL(fused_add_rms_norm_kernel<T>, dim3(rows), dim3(threads), 0,
                     stream, x, res, w, eps, hidden);
}

#define INSTANTIATE(T)                                                        \
  template void launch_rms_norm<T>(T*, const T*, const T*, float, int, int,   \
                                   hipStream_t);                              \
  template void launch_fused_add_rms_norm<T>(T*, T*, const T*, float, int,    \
                                             int, hipStream_t);

INSTANTIATE(float)
INSTANTIATE(__hip_bfloat16)
INSTANTIATE(__half)

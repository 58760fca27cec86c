// Shared helpers for the CDNA4 (gfx950) kernels.
// Wave width is 64 on CDNA; all reductions and tilings assume it.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <hip/hip_fp16.h>

#define WAVE_SIZE 64

#define DEVINLINE __device__ __forceinline__

// ---------------------------------------------------------------------------
// scalar conversions
// ---------------------------------------------------------------------------

template <typename T> struct ScalarOps;

template <> struct ScalarOps<float> {
  static DEVINLINE float to_f32(float x) { return x; }
  static DEVINLINE float from_f32(float x) { return x; }
};

template <> struct ScalarOps<__hip_bfloat16> {
  static DEVINLINE float to_f32(__hip_bfloat16 x) { return __bfloat162float(x); }
  static DEVINLINE __hip_bfloat16 from_f32(float x) { return __float2bfloat16(x); }
};

template <> struct ScalarOps<__half> {
  static DEVINLINE float to_f32(__half x) { return __half2float(x); }
  static DEVINLINE __half from_f32(float x) { return __float2half(x); }
};

// 16-byte vector of T (8 elements for 16-bit types, 4 for float)
template <typename T> struct Vec16 {
  static constexpr int kElems = 16 / sizeof(T);
  T data[kElems];
};

template <typename T>
DEVINLINE Vec16<T> load16(const T* p) {
  return *reinterpret_cast<const Vec16<T>*>(p);
}

template <typename T>
DEVINLINE void store16(T* p, const Vec16<T>& v) {
  *reinterpret_cast<Vec16<T>*>(p) = v;
}

// ---------------------------------------------------------------------------
// reductions
// ---------------------------------------------------------------------------

DEVINLINE float wave_reduce_sum(float x) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) x += __shfl_xor(x, off, WAVE_SIZE);
  return x;
}

DEVINLINE float wave_reduce_max(float x) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) x = fmaxf(x, __shfl_xor(x, off, WAVE_SIZE));
  return x;
}

// reduce within a group of `WIDTH` consecutive lanes (WIDTH a power of two)
template <int WIDTH>
DEVINLINE float group_reduce_sum(float x) {
#pragma unroll
  for (int off = WIDTH / 2; off > 0; off >>= 1) x += __shfl_xor(x, off, WAVE_SIZE);
  return x;
}

template <int WIDTH>
DEVINLINE float group_reduce_max(float x) {
#pragma unroll
  for (int off = WIDTH / 2; off > 0; off >>= 1)
    x = fmaxf(x, __shfl_xor(x, off, WAVE_SIZE));
  return x;
}

// block-level reduce over all waves (needs lds sized to waves-per-block)
DEVINLINE float block_reduce_sum(float x, float* lds_scratch) {
  const int lane = threadIdx.x & (WAVE_SIZE - 1);
  const int wave = threadIdx.x / WAVE_SIZE;
  const int nwaves = blockDim.x / WAVE_SIZE;
  x = wave_reduce_sum(x);
  if (lane == 0) lds_scratch[wave] = x;
  __syncthreads();
  float total = 0.f;
  if (threadIdx.x < nwaves) total = lds_scratch[threadIdx.x];
  total = wave_reduce_sum(total);  // lanes >= nwaves contribute 0
  if (threadIdx.x == 0) lds_scratch[0] = total;
  __syncthreads();
  return lds_scratch[0];
}

#define DISPATCH_BY_DTYPE(TORCH_DTYPE, FN)                                   \
  switch (TORCH_DTYPE) {                                                     \
    case at::ScalarType::BFloat16: {                                         \
      using scalar_t = __hip_bfloat16;                                       \
      FN;                                                                    \
      break;                                                                 \
    }                                                                        \
    case at::ScalarType::Half: {                                             \
      using scalar_t = __half;                                               \
      FN;                                                                    \
      break;                                                                 \
    }                                                                        \
    case at::ScalarType::Float: {                                            \
      using scalar_t = float;                                                \
      FN;                                                                    \
      break;                                                                 \
    }                                                                        \
    default:                                                                 \
      TORCH_CHECK(false, "unsupported dtype for HIP op");                    \
  }

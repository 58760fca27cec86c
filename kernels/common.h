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
// fp8 (OCP e4m3fn) KV-cache conversions — gfx950 hardware converts; matches
// torch.float8_e4m3fn bit-for-bit (verified in tests/test_ops_gpu.py)
// ---------------------------------------------------------------------------

typedef __attribute__((ext_vector_type(2))) float cvt_f32x2_t;
typedef __attribute__((ext_vector_type(2))) unsigned int u32x2_vec_t;
typedef __attribute__((ext_vector_type(8))) short bf16x8_vec_t;

// 8 packed e4m3 bytes (as 2 dwords) -> 8 bf16
DEVINLINE bf16x8_vec_t fp8x8_to_bf16x8(u32x2_vec_t u) {
  cvt_f32x2_t f[4];
  f[0] = __builtin_amdgcn_cvt_pk_f32_fp8(u[0], false);
  f[1] = __builtin_amdgcn_cvt_pk_f32_fp8(u[0], true);
  f[2] = __builtin_amdgcn_cvt_pk_f32_fp8(u[1], false);
  f[3] = __builtin_amdgcn_cvt_pk_f32_fp8(u[1], true);
  union { unsigned int w[4]; bf16x8_vec_t v; } out;
#pragma unroll
  for (int i = 0; i < 4; ++i) {
    unsigned r;
    asm volatile("v_cvt_pk_bf16_f32 %0, %1, %2"
                 : "=v"(r) : "v"(f[i][0]), "v"(f[i][1]));
    out.w[i] = r;
  }
  return out.v;
}

// 8 bf16 -> 8 packed e4m3 bytes (2 dwords)
DEVINLINE u32x2_vec_t bf16x8_to_fp8x8(bf16x8_vec_t v) {
  union { bf16x8_vec_t v; short s[8]; } in;
  in.v = v;
  float f[8];
#pragma unroll
  for (int i = 0; i < 8; ++i)
    f[i] = __bfloat162float(__builtin_bit_cast(__hip_bfloat16, in.s[i]));
  u32x2_vec_t out{};
  int w0 = 0, w1 = 0;
  w0 = __builtin_amdgcn_cvt_pk_fp8_f32(f[0], f[1], w0, false);
  w0 = __builtin_amdgcn_cvt_pk_fp8_f32(f[2], f[3], w0, true);
  w1 = __builtin_amdgcn_cvt_pk_fp8_f32(f[4], f[5], w1, false);
  w1 = __builtin_amdgcn_cvt_pk_fp8_f32(f[6], f[7], w1, true);
  out[0] = (unsigned)w0;
  out[1] = (unsigned)w1;
  return out;
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

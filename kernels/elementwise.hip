// SiLU-mul activation, NeoX rotary embedding, paged-KV cache write
// (SURVEY.md E6 / E3-E4 support ops).  All memory-bound: 16-B vector
// accesses, f32 math, grid-stride loops.

#include "common.h"

// ---------------------------------------------------------------------------
// silu_and_mul: out[t, i] = silu(x[t, i]) * x[t, d + i]
// ---------------------------------------------------------------------------

template <typename T>
__global__ void silu_and_mul_kernel(
    T* __restrict__ out,        // [rows, d]
    const T* __restrict__ x,    // [rows, 2*d]
    const int d) {
  constexpr int V = Vec16<T>::kElems;
  const int row = blockIdx.x;
  const T* a_row = x + (long)row * 2 * d;
  const T* b_row = a_row + d;
  T* out_row = out + (long)row * d;
  for (int i = threadIdx.x * V; i < d; i += blockDim.x * V) {
    Vec16<T> a = load16(a_row + i);
    Vec16<T> b = load16(b_row + i);
#pragma unroll
    for (int j = 0; j < V; ++j) {
      float av = ScalarOps<T>::to_f32(a.data[j]);
      float bv = ScalarOps<T>::to_f32(b.data[j]);
      float s = av / (1.f + __expf(-av));
      a.data[j] = ScalarOps<T>::from_f32(s * bv);
    }
    store16(out_row + i, a);
  }
}

template <typename T>
void launch_silu_and_mul(T* out, const T* x, int rows, int d, hipStream_t s) {
  int threads = 256;
  hipLaunchKernelGGL(silu_and_mul_kernel<T>, dim3(rows), dim3(threads), 0, s,
                     out, x, d);
}

// ---------------------------------------------------------------------------
// NeoX rotary embedding, in place on q and k.
// cos_sin_cache: [max_pos, head_dim] f32, first half cos, second half sin.
// One block per token; threads cover (head, i<half) pairs for q then k.
// ---------------------------------------------------------------------------

template <typename T>
__global__ void rotary_embedding_kernel(
    const long* __restrict__ positions,  // [T]
    T* __restrict__ q,                   // [T, nq * hd]
    T* __restrict__ k,                   // [T, nk * hd]
    const float* __restrict__ cos_sin,   // [max_pos, hd]
    const int nq, const int nk, const int hd) {
  const int token = blockIdx.x;
  const long pos = positions[token];
  const float* cs = cos_sin + pos * hd;
  const int half = hd / 2;

  T* q_row = q + (long)token * nq * hd;
  T* k_row = k + (long)token * nk * hd;
  const int total = (nq + nk) * half;
  for (int idx = threadIdx.x; idx < total; idx += blockDim.x) {
    const int h = idx / half;
    const int i = idx % half;
    T* row = h < nq ? q_row + h * hd : k_row + (h - nq) * hd;
    const float c = cs[i];
    const float s = cs[half + i];
    const float x1 = ScalarOps<T>::to_f32(row[i]);
    const float x2 = ScalarOps<T>::to_f32(row[i + half]);
    row[i] = ScalarOps<T>::from_f32(x1 * c - x2 * s);
    row[i + half] = ScalarOps<T>::from_f32(x2 * c + x1 * s);
  }
}

template <typename T>
void launch_rotary_embedding(const long* positions, T* q, T* k,
                             const float* cos_sin, int tokens, int nq, int nk,
                             int hd, hipStream_t s) {
  int threads = 256;
  hipLaunchKernelGGL(rotary_embedding_kernel<T>, dim3(tokens), dim3(threads), 0,
                     s, positions, q, k, cos_sin, nq, nk, hd);
}

// ---------------------------------------------------------------------------
// reshape_and_cache: scatter per-token K/V rows into the paged cache.
// cache layout: [num_blocks, block_size, kv_heads, head_dim] — a token's
// (kv_heads*head_dim) row is contiguous, so this is a straight 16-B copy.
// ---------------------------------------------------------------------------

template <typename T>
__global__ void reshape_and_cache_kernel(
    const T* __restrict__ k,      // [T, kvh * hd]
    const T* __restrict__ v,
    T* __restrict__ k_cache,      // [nb * bs, kvh * hd] flattened
    T* __restrict__ v_cache,
    const long* __restrict__ slots,  // [T]
    const int row_elems) {
  constexpr int V = Vec16<T>::kElems;
  const int token = blockIdx.x;
  const long slot = slots[token];
  if (slot < 0) return;
  const T* k_src = k + (long)token * row_elems;
  const T* v_src = v + (long)token * row_elems;
  T* k_dst = k_cache + slot * row_elems;
  T* v_dst = v_cache + slot * row_elems;
  for (int i = threadIdx.x * V; i < row_elems; i += blockDim.x * V) {
    store16(k_dst + i, load16(k_src + i));
    store16(v_dst + i, load16(v_src + i));
  }
}

template <typename T>
void launch_reshape_and_cache(const T* k, const T* v, T* kc, T* vc,
                              const long* slots, int tokens, int row_elems,
                              hipStream_t s) {
  int threads = 128;
  hipLaunchKernelGGL(reshape_and_cache_kernel<T>, dim3(tokens), dim3(threads),
                     0, s, k, v, kc, vc, slots, row_elems);
}

// fp8 KV cache (SURVEY E5 + E18 adjacency): bf16 K/V quantized to OCP e4m3
// at cache-write time; halves the decode-attention HBM stream and doubles
// KV capacity in the 288 GB pool.
__global__ void reshape_and_cache_fp8_kernel(
    const __hip_bfloat16* __restrict__ k,  // [T, kvh * hd]
    const __hip_bfloat16* __restrict__ v,
    unsigned char* __restrict__ k_cache,   // [nb * bs, kvh * hd] e4m3
    unsigned char* __restrict__ v_cache,
    const long* __restrict__ slots,
    const int row_elems) {
  const int token = blockIdx.x;
  const long slot = slots[token];
  if (slot < 0) return;
  const __hip_bfloat16* k_src = k + (long)token * row_elems;
  const __hip_bfloat16* v_src = v + (long)token * row_elems;
  unsigned char* k_dst = k_cache + slot * row_elems;
  unsigned char* v_dst = v_cache + slot * row_elems;
  for (int i = threadIdx.x * 8; i < row_elems; i += blockDim.x * 8) {
    *reinterpret_cast<u32x2_vec_t*>(k_dst + i) = bf16x8_to_fp8x8(
        *reinterpret_cast<const bf16x8_vec_t*>(k_src + i));
    *reinterpret_cast<u32x2_vec_t*>(v_dst + i) = bf16x8_to_fp8x8(
        *reinterpret_cast<const bf16x8_vec_t*>(v_src + i));
  }
}

void launch_reshape_and_cache_fp8(const __hip_bfloat16* k,
                                  const __hip_bfloat16* v, unsigned char* kc,
                                  unsigned char* vc, const long* slots,
                                  int tokens, int row_elems, hipStream_t s) {
  hipLaunchKernelGGL(reshape_and_cache_fp8_kernel, dim3(tokens), dim3(128), 0,
                     s, k, v, kc, vc, slots, row_elems);
}

#define INSTANTIATE(T)                                                         \
  template void launch_silu_and_mul<T>(T*, const T*, int, int, hipStream_t);   \
  template void launch_rotary_embedding<T>(const long*, T*, T*, const float*,  \
                                           int, int, int, int, hipStream_t);   \
  template void launch_reshape_and_cache<T>(const T*, const T*, T*, T*,        \
                                            const long*, int, int, hipStream_t);

INSTANTIATE(float)
INSTANTIATE(__hip_bfloat16)
INSTANTIATE(__half)

// Batched multi-LoRA shrink/expand kernels.
//
// Mixed-adapter batches carry a per-token adapter slot (-1 = no adapter);
// weights for the active adapters are stacked host-side into
//   A_stack [L, R, K]   (shrink:  tmp[t] = A[slot(t)] · x[t])
//   B_stack [L, N, R]   (expand:  out[t, off:off+N] += scale_s · B[slot(t)] · tmp[t])
// with R padded to the max rank (padded rows/cols are zero).
//
// BGMV geometry — one token per workgroup:
//   shrink: 4 waves split K; each (wave, r) pair accumulates a partial dot,
//           folded across lanes by shfl and across waves in LDS.  tmp is f32.
//   expand: R <= 64 fits a lane-resident tmp row; each thread owns one
//           output column and walks r with the B row loads coalesced.
// Decode batches are tiny (T <= a few hundred) and A/B stay L2-resident, so
// per-token re-reads are cheap; grouping tokens by adapter (true SGMV) only
// pays at prefill sizes, where the torch grouped path still applies.

#include "common.h"

// ---------------------------------------------------------------------------
// shrink: tmp[T, R] (f32) = A[slot(t)] · x[t]
// grid (T), block 256
// ---------------------------------------------------------------------------
template <typename T>
__global__ __launch_bounds__(256) void lora_shrink_kernel(
    float* __restrict__ tmp,        // [T, R]
    const T* __restrict__ x,        // [T, K]
    const T* __restrict__ a_stack,  // [L, R, K]
    const int* __restrict__ slots,  // [T]
    const int R,
    const int K) {
  const int t = blockIdx.x;
  const int slot = slots[t];
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;

  if (slot < 0) {
    for (int r = threadIdx.x; r < R; r += blockDim.x) tmp[(long)t * R + r] = 0.f;
    return;
  }

  const T* x_row = x + (long)t * K;
  const T* a_base = a_stack + (long)slot * R * K;

  // waves split K into quarters (rounded to 16B so k_lo stays aligned)
  constexpr int EV = Vec16<T>::kElems;
  const int k_per_wave = ((K + 3) / 4 + EV - 1) / EV * EV;
  const int k_lo = min(wave * k_per_wave, K);
  const int k_hi = min(k_lo + k_per_wave, K);

  __shared__ float lds[4][64];  // [wave][r] partials (R <= 64)
  for (int r = 0; r < R; ++r) {
    float acc = 0.f;
    const T* a_row = a_base + (long)r * K;
    // lanes cover every aligned 16B chunk in [k_lo, k_hi) (K % 8 == 0
    // enforced by the binding)
    constexpr int E = Vec16<T>::kElems;
    for (int k = k_lo + lane * E; k + E <= k_hi; k += 64 * E) {
      const Vec16<T> xv = load16(x_row + k);
      const Vec16<T> av = load16(a_row + k);
#pragma unroll
      for (int j = 0; j < Vec16<T>::kElems; ++j)
        acc += ScalarOps<T>::to_f32(xv.data[j]) * ScalarOps<T>::to_f32(av.data[j]);
    }
    acc = wave_reduce_sum(acc);
    if (lane == 0) lds[wave][r] = acc;
  }
  __syncthreads();
  for (int r = threadIdx.x; r < R; r += blockDim.x)
    tmp[(long)t * R + r] = lds[0][r] + lds[1][r] + lds[2][r] + lds[3][r];
}

// ---------------------------------------------------------------------------
// expand: out[t, off + n] += scale[slot] * sum_r B[slot][n, r] * tmp[t, r]
// grid (T, ceil(N/256)), block 256
// ---------------------------------------------------------------------------
template <typename T>
__global__ __launch_bounds__(256) void lora_expand_kernel(
    T* __restrict__ out,            // [T, out_w]
    const float* __restrict__ tmp,  // [T, R]
    const T* __restrict__ b_stack,  // [L, N, R]
    const int* __restrict__ slots,  // [T]
    const float* __restrict__ scales,  // [L]
    const int R,
    const int N,
    const int out_w,
    const int off) {
  const int t = blockIdx.x;
  const int slot = slots[t];
  if (slot < 0) return;
  const int n = blockIdx.y * 256 + threadIdx.x;
  if (n >= N) return;

  __shared__ float tmp_lds[64];
  if (threadIdx.x < R) tmp_lds[threadIdx.x] = tmp[(long)t * R + threadIdx.x];
  __syncthreads();

  const T* b_row = b_stack + ((long)slot * N + n) * R;
  float acc = 0.f;
  for (int r = 0; r < R; ++r)
    acc += ScalarOps<T>::to_f32(b_row[r]) * tmp_lds[r];

  T* o = out + (long)t * out_w + off + n;
  *o = ScalarOps<T>::from_f32(ScalarOps<T>::to_f32(*o) + acc * scales[slot]);
}

template <typename T>
void launch_lora_shrink(float* tmp, const T* x, const T* a_stack,
                        const int* slots, int Tn, int R, int K,
                        hipStream_t stream) {
  hipLaunchKernelGGL(lora_shrink_kernel<T>, dim3(Tn), dim3(256), 0, stream,
                     tmp, x, a_stack, slots, R, K);
}

template <typename T>
void launch_lora_expand(T* out, const float* tmp, const T* b_stack,
                        const int* slots, const float* scales, int Tn, int R,
                        int N, int out_w, int off, hipStream_t stream) {
  hipLaunchKernelGGL(lora_expand_kernel<T>, dim3(Tn, (N + 255) / 256),
                     dim3(256), 0, stream, out, tmp, b_stack, slots, scales, R,
                     N, out_w, off);
}

#define INSTANTIATE_LORA(T)                                                  \
  template void launch_lora_shrink<T>(float*, const T*, const T*,            \
                                      const int*, int, int, int,             \
                                      hipStream_t);                          \
  template void launch_lora_expand<T>(T*, const float*, const T*,            \
                                      const int*, const float*, int, int,    \
                                      int, int, int, hipStream_t);

INSTANTIATE_LORA(float)
INSTANTIATE_LORA(__hip_bfloat16)
INSTANTIATE_LORA(__half)

// Fused log-softmax + top-K + sampled-token rank, one HBM pass (E8).
//
// The TGIS wire format wants, per generated token: the top-N token ids
// with their logprobs and ranks, plus the sampled token's logprob and
// rank (reference surface: grpc_server.py:701-756 response details).
// The torch chain for this (float() -> log_softmax -> topk -> gather ->
// rank count) reads and writes the [N, 128k] row set several times
// (~0.5 GB of traffic per decode step at batch 512).  This kernel does
// everything in ONE read of the bf16 logits:
//
//   per row (one 256-thread workgroup):
//     - online max + sum-of-exp (lane-local, rescaled on new max)
//     - lane-local sorted top-K candidates (insertion against the
//       running K-th value; K <= 16)
//     - count of elements strictly greater than the sampled token's
//       logit (== rank - 1; log-softmax is monotonic so raw logits give
//       the same ordering)
//   then an LDS reduction for (max, sumexp) and the rank count, and a
//   K-round selection over the 256*K LDS candidates for the global
//   top-K (descending, ties resolved toward the lower candidate slot).
//
// Outputs: topv f32 [N,K] (log-softmax values), topi i32 [N,K],
// chosen_lp f32 [N], ranks i32 [N].

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#include "common.h"

#define LSK_THREADS 256
#define LSK_KMAX 16

// 1/ln(2) for exp2-domain sums; log2 -> ln on the way out.
#define LSK_LOG2E 1.44269504088896340736f
#define LSK_LN2 0.69314718055994530942f

typedef __attribute__((ext_vector_type(8))) short lsk_bf16x8;

DEVINLINE float lsk_bf16_bits_to_f32(short b) {
  return __uint_as_float(((unsigned)(unsigned short)b) << 16);
}

template <int KT>
__global__ __launch_bounds__(LSK_THREADS) void logsoftmax_topk_kernel(
    float* __restrict__ topv,        // [N, K]
    int* __restrict__ topi,          // [N, K]
    float* __restrict__ chosen_lp,   // [N]
    int* __restrict__ ranks,         // [N]
    const __hip_bfloat16* __restrict__ logits,  // [N, V]
    const long* __restrict__ chosen, // [N]
    const int V, const int kout) {  // kout <= KT: output columns/rounds
  constexpr int K = KT;
  const int row = blockIdx.x;
  const int tid = threadIdx.x;
  const __hip_bfloat16* x = logits + (long)row * V;

  const float chosen_val = __bfloat162float(x[chosen[row]]);

  // Per-lane top-K candidates in REGISTERS with a fully static-index
  // insertion cascade.  (v1 used a dynamically-indexed register array —
  // the compiler lowered it to per-lane scratch, 50x off the wall; v2
  // kept the list in LDS — the data-dependent shift loop cost dependent
  // LDS round-trips on virtually every wave-step, since with V/lane=500
  // SOME lane inserts almost every iteration.  The cascade is ~6 VALU
  // per slot with no memory traffic, so the wave-divergent insert costs
  // ~K*6 VALU instead of ~K LDS round trips.)
  float cv[K];
  int ci[K];
#pragma unroll
  for (int k = 0; k < K; ++k) {
    cv[k] = -INFINITY;
    ci[k] = -1;
  }

  // ---- lane-local single pass ------------------------------------------
  // Two independent (max, sumexp) accumulator pairs break the serial
  // exp-chain; 4 b128 loads in flight per outer iteration hide HBM
  // latency (the grid is only ~2 WGs/CU at batch 512, so per-wave MLP is
  // the only latency cover available).
  float m0 = -INFINITY, s0 = 0.f;
  float m1 = -INFINITY, s1 = 0.f;
  int gt = 0;            // count of x[j] > chosen_val
  float kth = -INFINITY;  // lane_v[K-1], kept in a register

  const int stride = LSK_THREADS * 8;

  auto process8 = [&](const lsk_bf16x8& v8, int base) {
#pragma unroll
    for (int e = 0; e < 8; ++e) {
      const int j = base + e;
      const float xv = lsk_bf16_bits_to_f32(v8[e]);
      float& m = (e & 1) ? m1 : m0;
      float& s = (e & 1) ? s1 : s0;
      const float mn = fmaxf(m, xv);
      // branchless online softmax: always one rescale + one term
      s = s * __builtin_amdgcn_exp2f((m - mn) * LSK_LOG2E) +
          __builtin_amdgcn_exp2f((xv - mn) * LSK_LOG2E);
      m = mn;
      gt += (xv > chosen_val);
      if (xv > kth) {
        // static-index insertion cascade into the descending list:
        // position k takes its upper neighbour if xv ranks above it,
        // xv itself if this is the insertion point, else keeps its value
#pragma unroll
        for (int k = K - 1; k >= 1; --k) {
          const bool shift = xv > cv[k - 1];
          const bool place = !shift && (xv > cv[k]);
          cv[k] = shift ? cv[k - 1] : (place ? xv : cv[k]);
          ci[k] = shift ? ci[k - 1] : (place ? j : ci[k]);
        }
        if (xv > cv[0]) {
          ci[0] = j;
          cv[0] = xv;
        }
        kth = cv[K - 1];
      }
    }
  };

  for (int base = tid * 8; base < V; base += stride * 4) {
    // V is a multiple of 8 (launcher contract), so every 8-chunk is
    // either fully in range or fully out
    const bool h1 = base + stride < V;
    const bool h2 = base + 2 * stride < V;
    const bool h3 = base + 3 * stride < V;
    lsk_bf16x8 c0, c1, c2, c3;
    c0 = *reinterpret_cast<const lsk_bf16x8*>(x + base);
    if (h1) c1 = *reinterpret_cast<const lsk_bf16x8*>(x + base + stride);
    if (h2) c2 = *reinterpret_cast<const lsk_bf16x8*>(x + base + 2 * stride);
    if (h3) c3 = *reinterpret_cast<const lsk_bf16x8*>(x + base + 3 * stride);
    process8(c0, base);
    if (h1) process8(c1, base + stride);
    if (h2) process8(c2, base + 2 * stride);
    if (h3) process8(c3, base + 3 * stride);
  }

  // fold the two accumulator pairs
  const float m = fmaxf(m0, m1);
  const float s = s0 * __builtin_amdgcn_exp2f((m0 - m) * LSK_LOG2E) +
                  s1 * __builtin_amdgcn_exp2f((m1 - m) * LSK_LOG2E);

  // ---- (max, sumexp) + rank reduction over the workgroup ----------------
  __shared__ float red_m[LSK_THREADS];
  __shared__ float red_s[LSK_THREADS];
  __shared__ int red_g[LSK_THREADS];
  red_m[tid] = m;
  red_s[tid] = s;
  red_g[tid] = gt;
  __syncthreads();
  for (int off = LSK_THREADS / 2; off > 0; off >>= 1) {
    if (tid < off) {
      const float m2 = red_m[tid + off];
      const float s2 = red_s[tid + off];
      const float mm = fmaxf(red_m[tid], m2);
      red_s[tid] = red_s[tid] * __builtin_amdgcn_exp2f((red_m[tid] - mm) * LSK_LOG2E) +
                   s2 * __builtin_amdgcn_exp2f((m2 - mm) * LSK_LOG2E);
      red_m[tid] = mm;
      red_g[tid] += red_g[tid + off];
    }
    __syncthreads();
  }
  const float gmax = red_m[0];
  // log(sumexp) in ln domain; lse = gmax + ln(sum)
  const float lse = gmax + __logf(red_s[0]);

  if (tid == 0) {
    chosen_lp[row] = chosen_val - lse;
    ranks[row] = red_g[0] + 1;
  }

  // ---- global top-K: K selection rounds over the LDS candidate pool ----
  __shared__ float cand_v[LSK_THREADS * K];
  __shared__ int cand_i[LSK_THREADS * K];
#pragma unroll
  for (int k = 0; k < K; ++k) {
    cand_v[tid * K + k] = cv[k];
    cand_i[tid * K + k] = ci[k];
  }
  __syncthreads();

  // one wave selects; others are done (their LDS writes are complete)
  if (tid >= 64) return;
  const int pool = LSK_THREADS * K;
  for (int k = 0; k < kout; ++k) {
    float best = -INFINITY;
    int best_p = -1;
    for (int p = tid; p < pool; p += 64) {
      const float v = cand_v[p];
      if (v > best || (v == best && p < best_p)) {
        best = v;
        best_p = p;
      }
    }
    // wave-local shfl reduce (single wave does the selection)
    for (int off = 32; off > 0; off >>= 1) {
      const float ov = __shfl_down(best, off, 64);
      const int op = __shfl_down(best_p, off, 64);
      if (ov > best || (ov == best && op != -1 && (best_p == -1 || op < best_p))) {
        best = ov;
        best_p = op;
      }
    }
    best = __shfl(best, 0, 64);
    best_p = __shfl(best_p, 0, 64);
    if (tid == 0) {
      topv[(long)row * kout + k] = best - lse;
      topi[(long)row * kout + k] = (best_p >= 0) ? cand_i[best_p] : -1;
    }
    // retire the winner so the next round finds the next-largest
    if (tid == 0 && best_p >= 0) cand_v[best_p] = -INFINITY;
    __builtin_amdgcn_s_waitcnt(0);  // order the LDS retire before next scan
    __builtin_amdgcn_wave_barrier();
  }
}

void launch_logsoftmax_topk(float* topv, int* topi, float* chosen_lp,
                            int* ranks, const __hip_bfloat16* logits,
                            const long* chosen, int n, int vocab, int k,
                            hipStream_t stream) {
  // K is a template constant so the insertion cascade stays static-index;
  // the sampler always asks for max_logprobs = 11 (validation caps top_n
  // at 10, +1 for the sampled token).  16 covers the headroom case.
  if (k == 11) {
    hipLaunchKernelGGL(logsoftmax_topk_kernel<11>, dim3(n),
                       dim3(LSK_THREADS), 0, stream, topv, topi, chosen_lp,
                       ranks, logits, chosen, vocab, 11);
  } else if (k <= 16) {
    hipLaunchKernelGGL(logsoftmax_topk_kernel<16>, dim3(n),
                       dim3(LSK_THREADS), 0, stream, topv, topi, chosen_lp,
                       ranks, logits, chosen, vocab, k);
  }
}

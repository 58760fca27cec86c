// Decode-batch tile GEMM: y[M,N] = x[M,K] @ W[N,K]^T, bf16.
//
// Targets the llama decode projections at M in [128, 1024] where hipBLASLt's
// picks ran at 20-50% of roofline in r1 (profiles/r01_final_decode.txt: the
// Cijk_* rows are ~52% of decode kernel time).  Structure is the guide's
// 128^2-tile LDS-staged GEMM: BK=64 K-slab, async global->LDS staging
// (buffer_load ... lds via __builtin_amdgcn_global_load_lds, 16 B/lane),
// double-buffered, v_mfma_f32_16x16x32_bf16 inner loop, 4 waves each owning
// a 64x64 quadrant of the 128x128 C tile.
//
// Both operands are TN-friendly: A-fragments read x rows k-contiguously and
// B-fragments read W rows k-contiguously (W is [N][K] row-major), so no
// transpose anywhere.  LDS images are XOR-swizzled ((row&7)<<4 on the byte
// address); since glds writes lane-linearly, the swizzle is applied to the
// per-lane GLOBAL source address instead (guide §5 rule: swizzle the source,
// keep LDS linear).

#include "common.h"

typedef __attribute__((ext_vector_type(8))) short bf16x8_t;
typedef __attribute__((ext_vector_type(4))) float f32x4_t;

#define GT_BM 128
#define GT_BN 128
#define GT_BK 64

__global__ __launch_bounds__(256, 2) void gemm_tile_kernel(
    __hip_bfloat16* __restrict__ y,        // [M, N]
    const __hip_bfloat16* __restrict__ x,  // [M, K]
    const __hip_bfloat16* __restrict__ w,  // [N, K]
    const int M, const int N, const int K) {
  const int nblk_n = N / GT_BN;
  // XCD-friendly remap: consecutive blockIdx.x values walk N-tiles within a
  // supergroup of 8 so each XCD's resident blocks share the same x rows
  const int bid = blockIdx.x;
  const int m_tile = bid % ((M + GT_BM - 1) / GT_BM);
  const int n_tile = bid / ((M + GT_BM - 1) / GT_BM);
  const int m0 = m_tile * GT_BM;
  const int n0 = n_tile * GT_BN;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;

  // LDS: A [128][64] and B [128][64] bf16, 16 KB each, double-buffered
  __shared__ __hip_bfloat16 a_lds[2][GT_BM * GT_BK];
  __shared__ __hip_bfloat16 b_lds[2][GT_BN * GT_BK];

  // ---- staging: each wave loads 32 rows of A and 32 rows of B per tile.
  // glds destination is wave-uniform base + lane*16; the source address
  // carries the XOR swizzle: lane covers (row = base + lane/8,
  // k8 = (lane%8) ^ (row&7)).
  const int st_row = lane >> 3;         // 0..7 within the wave's 8-row pass
  const int st_k8 = lane & 7;
  const long k_l = K;

  auto stage = [&](int buf, int k0) {
#pragma unroll
    for (int pass = 0; pass < 4; ++pass) {
      const int arow = wave * 32 + pass * 8 + st_row;
      const int asrc_k8 = st_k8 ^ (arow & 7);
      const __hip_bfloat16* ag =
          x + (long)min(m0 + arow, M - 1) * k_l + k0 + asrc_k8 * 8;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)ag,
          (__attribute__((address_space(3))) unsigned int*)
              (a_lds[buf] + (wave * 32 + pass * 8) * GT_BK),
          16, 0, 0);
    }
#pragma unroll
    for (int pass = 0; pass < 4; ++pass) {
      const int brow = wave * 32 + pass * 8 + st_row;
      const int bsrc_k8 = st_k8 ^ (brow & 7);
      const __hip_bfloat16* bg =
          w + (long)(n0 + brow) * k_l + k0 + bsrc_k8 * 8;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)bg,
          (__attribute__((address_space(3))) unsigned int*)
              (b_lds[buf] + (wave * 32 + pass * 8) * GT_BK),
          16, 0, 0);
    }
  };

  // wave quadrant: 2x2 of 64x64
  const int wm0 = (wave & 1) * 64;  // within the 128-row tile
  const int wn0 = (wave >> 1) * 64;

  f32x4_t acc[4][4];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = f32x4_t{};

  const int frag_row = lane & 15;      // m or n within a 16-wide block
  const int frag_k8 = lane >> 4;       // k-group 0..3

  stage(0, 0);

  for (int k0 = 0; k0 < K; k0 += GT_BK) {
    const int buf = (k0 / GT_BK) & 1;
    asm volatile("s_waitcnt vmcnt(0)");
    __syncthreads();
    if (k0 + GT_BK < K) stage(buf ^ 1, k0 + GT_BK);

#pragma unroll
    for (int ks = 0; ks < 2; ++ks) {  // two 32-deep k-steps per slab
      bf16x8_t a_frag[4], b_frag[4];
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        const int am = wm0 + i * 16 + frag_row;
        const int byte =
            (am * (GT_BK * 2) + (ks * 32 + frag_k8 * 8) * 2) ^ ((am & 7) << 4);
        a_frag[i] = *reinterpret_cast<const bf16x8_t*>(
            reinterpret_cast<const char*>(a_lds[buf]) + byte);
      }
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        const int bn = wn0 + j * 16 + frag_row;
        const int byte =
            (bn * (GT_BK * 2) + (ks * 32 + frag_k8 * 8) * 2) ^ ((bn & 7) << 4);
        b_frag[j] = *reinterpret_cast<const bf16x8_t*>(
            reinterpret_cast<const char*>(b_lds[buf]) + byte);
      }
#pragma unroll
      for (int i = 0; i < 4; ++i)
#pragma unroll
        for (int j = 0; j < 4; ++j)
          acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a_frag[i], b_frag[j], acc[i][j], 0, 0, 0);
    }
  }

  // ---- epilogue: lane holds D[m = (l>>4)*4 + r][n = l&15] per 16x16 block
#pragma unroll
  for (int i = 0; i < 4; ++i) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int gm = m0 + wm0 + i * 16 + (lane >> 4) * 4 + r;
      if (gm >= M) continue;
      __hip_bfloat16* out_row = y + (long)gm * N + n0 + wn0;
#pragma unroll
      for (int j = 0; j < 4; ++j)
        out_row[j * 16 + frag_row] = __float2bfloat16(acc[i][j][r]);
    }
  }
}

void launch_gemm_tile(__hip_bfloat16* y, const __hip_bfloat16* x,
                      const __hip_bfloat16* w, int M, int N, int K,
                      hipStream_t stream) {
  const int mt = (M + GT_BM - 1) / GT_BM;
  const int nt = N / GT_BN;
  dim3 grid(mt * nt);
  dim3 block(256);
  hipLaunchKernelGGL(gemm_tile_kernel, grid, block, 0, stream, y, x, w, M, N, K);
}

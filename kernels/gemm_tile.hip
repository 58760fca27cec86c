// Decode-batch tile GEMMs: y[M,N] = x[M,K] @ W[N,K]^T, bf16.
//
// Two tilings:
//  * gemm_tile_kernel   — 128x128 macro-tile, 4 waves, glds-staged (r2 first
//    cut; loses to hipBLASLt at M=512, kept for comparison / experiments).
//  * gemm_dtile_kernel  — 64(M) x 128(N) x BK=64 tile per 4-wave workgroup,
//    3 workgroups/CU, sized for the decode projections at M in [128, 1024]
//    where the 256^2-class tilings starve the chip (M/256 gives only 2 row
//    tiles).  Per wave: a 64x32 C tile (4x2 of 16x16, 32 accum VGPRs).
//
// LDS layout (both operands): row-major [rows][64] bf16 (128-B rows) with a
// SKEW-rotated 16-B slot per row:  slot' = (slot + (row>>1)) & 7.  For the
// b128 fragment reads (16-lane groups read 16 consecutive rows at one k8
// slot) the resulting banks are provably distinct: rows r, r+1 differ by 32
// dword-banks (row stride), and within each parity class (r>>1)&7 walks all
// eight slot rotations — zero LDS bank conflicts without padding.  glds
// writes lane-linearly, so the rotation is applied to the per-lane GLOBAL
// source address instead (guide §5: swizzle the source, keep LDS linear).

#include "common.h"

typedef __attribute__((ext_vector_type(8))) short bf16x8_t;
typedef __attribute__((ext_vector_type(4))) float f32x4_t;

#define GT_BM 128
#define GT_BN 128
#define GT_BK 64

// skewed byte offset of (row, k8 slot) in a [rows][64] bf16 image
DEVINLINE int dt_swz(int row, int k8) {
  return row * (GT_BK * 2) + (((k8 + (row >> 1)) & 7) << 4);
}

__global__ __launch_bounds__(256, 2) void gemm_tile_kernel(
    __hip_bfloat16* __restrict__ y,        // [M, N]
    const __hip_bfloat16* __restrict__ x,  // [M, K]
    const __hip_bfloat16* __restrict__ w,  // [N, K]
    const int M, const int N, const int K) {
  const int bid = blockIdx.x;
  const int m_tiles = (M + GT_BM - 1) / GT_BM;
  const int m_tile = bid % m_tiles;
  const int n_tile = bid / m_tiles;
  const int m0 = m_tile * GT_BM;
  const int n0 = n_tile * GT_BN;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;

  __shared__ __hip_bfloat16 a_lds[2][GT_BM * GT_BK];
  __shared__ __hip_bfloat16 b_lds[2][GT_BN * GT_BK];

  const int st_row = lane >> 3;  // 8 rows per wave pass
  const int st_k8 = lane & 7;
  const long k_l = K;

  auto stage = [&](int buf, int k0) {
#pragma unroll
    for (int pass = 0; pass < 4; ++pass) {
      const int arow = wave * 32 + pass * 8 + st_row;
      const int src_k8 = (st_k8 - (arow >> 1)) & 7;
      const __hip_bfloat16* ag =
          x + (long)min(m0 + arow, M - 1) * k_l + k0 + src_k8 * 8;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)ag,
          (__attribute__((address_space(3))) unsigned int*)
              (a_lds[buf] + (wave * 32 + pass * 8) * GT_BK),
          16, 0, 0);
    }
#pragma unroll
    for (int pass = 0; pass < 4; ++pass) {
      const int brow = wave * 32 + pass * 8 + st_row;
      const int src_k8 = (st_k8 - (brow >> 1)) & 7;
      const __hip_bfloat16* bg =
          w + (long)(n0 + brow) * k_l + k0 + src_k8 * 8;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)bg,
          (__attribute__((address_space(3))) unsigned int*)
              (b_lds[buf] + (wave * 32 + pass * 8) * GT_BK),
          16, 0, 0);
    }
  };

  const int wm0 = (wave & 1) * 64;
  const int wn0 = (wave >> 1) * 64;

  f32x4_t acc[4][4];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = f32x4_t{};

  const int frag_row = lane & 15;
  const int frag_k8 = lane >> 4;  // 0..3

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
        a_frag[i] = *reinterpret_cast<const bf16x8_t*>(
            reinterpret_cast<const char*>(a_lds[buf]) +
            dt_swz(am, ks * 4 + frag_k8));
      }
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        const int bn = wn0 + j * 16 + frag_row;
        b_frag[j] = *reinterpret_cast<const bf16x8_t*>(
            reinterpret_cast<const char*>(b_lds[buf]) +
            dt_swz(bn, ks * 4 + frag_k8));
      }
#pragma unroll
      for (int i = 0; i < 4; ++i)
#pragma unroll
        for (int j = 0; j < 4; ++j)
          acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a_frag[i], b_frag[j], acc[i][j], 0, 0, 0);
    }
  }

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

// ---------------------------------------------------------------------------
// 64x128 decode tile: 4 waves as 1(M) x 4(N); wave C tile 64x32.
// 3 workgroups/CU (48 KB LDS, ~100 VGPR) so frag-read/MFMA latency overlaps
// across blocks.  Grid (M/64) x (N/128) -> 256+ blocks for the llama decode
// projections at M=512.
// ---------------------------------------------------------------------------

#define DT_BM 64
#define DT_BN 128

__global__ __launch_bounds__(256, 3) void gemm_dtile_kernel(
    __hip_bfloat16* __restrict__ y,        // [M, N]
    const __hip_bfloat16* __restrict__ x,  // [M, K]
    const __hip_bfloat16* __restrict__ w,  // [N, K]
    const int M, const int N, const int K) {
  const int bid = blockIdx.x;
  const int m_tiles = (M + DT_BM - 1) / DT_BM;
  const int m_tile = bid % m_tiles;
  const int n_tile = bid / m_tiles;
  const int m0 = m_tile * DT_BM;
  const int n0 = n_tile * DT_BN;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;

  __shared__ __hip_bfloat16 a_lds[2][DT_BM * GT_BK];   // 8 KB
  __shared__ __hip_bfloat16 b_lds[2][DT_BN * GT_BK];   // 16 KB

  const int st_row = lane >> 3;
  const int st_k8 = lane & 7;
  const long k_l = K;

  auto stage = [&](int buf, int k0) {
    // A: 64 rows -> 2 passes of (4 waves x 8 rows)... waves split rows
#pragma unroll
    for (int pass = 0; pass < 2; ++pass) {
      const int arow = wave * 16 + pass * 8 + st_row;
      const int src_k8 = (st_k8 - (arow >> 1)) & 7;
      const __hip_bfloat16* ag =
          x + (long)min(m0 + arow, M - 1) * k_l + k0 + src_k8 * 8;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)ag,
          (__attribute__((address_space(3))) unsigned int*)
              (a_lds[buf] + (wave * 16 + pass * 8) * GT_BK),
          16, 0, 0);
    }
#pragma unroll
    for (int pass = 0; pass < 4; ++pass) {
      const int brow = wave * 32 + pass * 8 + st_row;
      const int src_k8 = (st_k8 - (brow >> 1)) & 7;
      const __hip_bfloat16* bg =
          w + (long)(n0 + brow) * k_l + k0 + src_k8 * 8;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)bg,
          (__attribute__((address_space(3))) unsigned int*)
              (b_lds[buf] + (wave * 32 + pass * 8) * GT_BK),
          16, 0, 0);
    }
  };

  const int wn0 = wave * 32;  // wave's 32-col slice of the 128-col tile

  f32x4_t acc[4][2];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 2; ++j) acc[i][j] = f32x4_t{};

  const int frag_row = lane & 15;
  const int frag_k8 = lane >> 4;

  stage(0, 0);

  for (int k0 = 0; k0 < K; k0 += GT_BK) {
    const int buf = (k0 / GT_BK) & 1;
    asm volatile("s_waitcnt vmcnt(0)");
    __syncthreads();
    if (k0 + GT_BK < K) stage(buf ^ 1, k0 + GT_BK);

#pragma unroll
    for (int ks = 0; ks < 2; ++ks) {
      bf16x8_t a_frag[4], b_frag[2];
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        const int am = i * 16 + frag_row;
        a_frag[i] = *reinterpret_cast<const bf16x8_t*>(
            reinterpret_cast<const char*>(a_lds[buf]) +
            dt_swz(am, ks * 4 + frag_k8));
      }
#pragma unroll
      for (int j = 0; j < 2; ++j) {
        const int bn = wn0 + j * 16 + frag_row;
        b_frag[j] = *reinterpret_cast<const bf16x8_t*>(
            reinterpret_cast<const char*>(b_lds[buf]) +
            dt_swz(bn, ks * 4 + frag_k8));
      }
#pragma unroll
      for (int i = 0; i < 4; ++i)
#pragma unroll
        for (int j = 0; j < 2; ++j)
          acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a_frag[i], b_frag[j], acc[i][j], 0, 0, 0);
    }
  }

#pragma unroll
  for (int i = 0; i < 4; ++i) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int gm = m0 + i * 16 + (lane >> 4) * 4 + r;
      if (gm >= M) continue;
      __hip_bfloat16* out_row = y + (long)gm * N + n0 + wn0;
#pragma unroll
      for (int j = 0; j < 2; ++j)
        out_row[j * 16 + frag_row] = __float2bfloat16(acc[i][j][r]);
    }
  }
}

void launch_gemm_tile(__hip_bfloat16* y, const __hip_bfloat16* x,
                      const __hip_bfloat16* w, int M, int N, int K,
                      hipStream_t stream) {
  static const bool big = [] {
    const char* e = getenv("VTA_GEMM_TILE_128");
    return e && e[0] == '1';
  }();
  if (big) {
    const int mt = (M + GT_BM - 1) / GT_BM;
    hipLaunchKernelGGL(gemm_tile_kernel, dim3(mt * (N / GT_BN)), dim3(256), 0,
                       stream, y, x, w, M, N, K);
    return;
  }
  const int mt = (M + DT_BM - 1) / DT_BM;
  hipLaunchKernelGGL(gemm_dtile_kernel, dim3(mt * (N / DT_BN)), dim3(256), 0,
                     stream, y, x, w, M, N, K);
}

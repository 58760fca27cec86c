// Grouped MoE GEMM: one launch covers every expert's
// segment of the (sorted-by-expert) token batch, replacing the per-expert
// Python loop of 2 GEMMs + activation (8 experts x 3+ launches per MoE
// layer, eager because MoE skips hipGraph capture).
//
//   h[r, :]  = silu(x[r]·Wg_e^T) * (x[r]·Wu_e^T)   (gated, w13 = [E, 2I, H])
//   y[r, :]  = h[r]·W2_e^T                          (plain, w2  = [E, H, I])
// where e = the expert owning sorted row r (seg_off[e] <= r < seg_off[e+1]).
//
// Geometry per kernel: grid (N/128, E * MBLK); each workgroup owns one
// 128-row n-tile of one expert's weights and one 128-row m-chunk of its
// segment; blocks past a segment's end exit immediately, so segment sizes
// stay device-side (no host sync — the router's counts never leave the
// GPU).  Tiling is the 128x128 macro-tile on v_mfma_f32_32x32x16_bf16 with
// XOR-swizzled LDS chunk staging and the T14 write/load/barrier pipeline,
// same as gemm_tile_kernel (gemm_skinny.hip).

#include "common.h"

typedef __attribute__((ext_vector_type(8))) short bf16x8_t;
typedef __attribute__((ext_vector_type(16))) float f32x16_t;

#define MG_KB 64

DEVINLINE int mg_swz(int row, int byte_in_row) {
  return row * (MG_KB * 2) + (byte_in_row ^ ((row & 7) << 4));
}

template <bool GATED>
__global__ __launch_bounds__(256, 2) void moe_gemm_kernel(
    __hip_bfloat16* __restrict__ y,        // [T, N_out] (N_out = I or H)
    const __hip_bfloat16* __restrict__ x,  // [T, K] sorted by expert
    const __hip_bfloat16* __restrict__ w,  // [E, (GATED ? 2N : N), K]
    const int* __restrict__ seg_off,       // [E+1] row offsets per expert
    const int E,
    const int N,   // output width (I for gated, H for plain)
    const int K,
    const int MBLK) {
  const int n_blk = blockIdx.x;
  const int e = blockIdx.y / MBLK;
  const int mb = blockIdx.y % MBLK;
  const int m_lo = seg_off[e] + mb * 128;
  const int m_hi = seg_off[e + 1];
  if (m_lo >= m_hi) return;
  const int M = min(128, m_hi - m_lo);

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int col32 = lane & 31;
  const int half = lane >> 5;
  const int mh = wave >> 1;
  const int nh = wave & 1;

  const int nchunks = K / MG_KB;
  const long n_base = (long)n_blk * 128;
  const long w_rows = GATED ? 2L * N : (long)N;

  __shared__ __hip_bfloat16 wg_lds[2][128 * MG_KB];
  __shared__ __hip_bfloat16 wu_lds[GATED ? 2 : 1][128 * MG_KB];
  __shared__ __hip_bfloat16 x_lds[2][128 * MG_KB];

  const int st_row = tid / 8;
  const int st_byte = (tid & 7) * 16;
  const __hip_bfloat16* wg_base =
      w + (e * w_rows + n_base + st_row) * (long)K + st_byte / 2;
  const __hip_bfloat16* wu_base =
      GATED ? w + (e * w_rows + N + n_base + st_row) * (long)K + st_byte / 2
            : nullptr;
  const __hip_bfloat16* x_base =
      x + (long)(m_lo + st_row) * K + st_byte / 2;

  bf16x8_t st_g[4], st_u[4], st_x[4];
  auto stage_load = [&](int chunk) {
    const int koff = chunk * MG_KB;
#pragma unroll
    for (int p = 0; p < 4; ++p) {
      st_g[p] = *reinterpret_cast<const bf16x8_t*>(wg_base + (long)(p * 32) * K + koff);
      if (GATED)
        st_u[p] = *reinterpret_cast<const bf16x8_t*>(wu_base + (long)(p * 32) * K + koff);
      st_x[p] = bf16x8_t{};
      if (p * 32 + st_row < M)
        st_x[p] = *reinterpret_cast<const bf16x8_t*>(x_base + (long)(p * 32) * K + koff);
    }
  };
  auto stage_write = [&](int buf) {
#pragma unroll
    for (int p = 0; p < 4; ++p) {
      *reinterpret_cast<bf16x8_t*>(
          reinterpret_cast<char*>(wg_lds[buf]) + mg_swz(p * 32 + st_row, st_byte)) = st_g[p];
      if (GATED)
        *reinterpret_cast<bf16x8_t*>(
            reinterpret_cast<char*>(wu_lds[buf]) + mg_swz(p * 32 + st_row, st_byte)) = st_u[p];
      *reinterpret_cast<bf16x8_t*>(
          reinterpret_cast<char*>(x_lds[buf]) + mg_swz(p * 32 + st_row, st_byte)) = st_x[p];
    }
  };

  f32x16_t acc_g[2][2], acc_u[2][2];
#pragma unroll
  for (int mt = 0; mt < 2; ++mt)
#pragma unroll
    for (int nt = 0; nt < 2; ++nt) {
      acc_g[mt][nt] = f32x16_t{};
      if (GATED) acc_u[mt][nt] = f32x16_t{};
    }

  stage_load(0);

  for (int c = 0; c < nchunks; ++c) {
    const int buf = c & 1;
    stage_write(buf);
    if (c + 1 < nchunks) stage_load(c + 1);
    __syncthreads();

#pragma unroll
    for (int kc = 0; kc < 4; ++kc) {
      const int byte = kc * 32 + half * 16;
      bf16x8_t a[2], bg[2], bu[2];
#pragma unroll
      for (int mt = 0; mt < 2; ++mt)
        a[mt] = *reinterpret_cast<const bf16x8_t*>(
            reinterpret_cast<const char*>(x_lds[buf]) +
            mg_swz(mh * 64 + mt * 32 + col32, byte));
#pragma unroll
      for (int nt = 0; nt < 2; ++nt) {
        bg[nt] = *reinterpret_cast<const bf16x8_t*>(
            reinterpret_cast<const char*>(wg_lds[buf]) +
            mg_swz(nh * 64 + nt * 32 + col32, byte));
        if (GATED)
          bu[nt] = *reinterpret_cast<const bf16x8_t*>(
              reinterpret_cast<const char*>(wu_lds[buf]) +
              mg_swz(nh * 64 + nt * 32 + col32, byte));
      }
#pragma unroll
      for (int mt = 0; mt < 2; ++mt)
#pragma unroll
        for (int nt = 0; nt < 2; ++nt) {
          acc_g[mt][nt] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              a[mt], bg[nt], acc_g[mt][nt], 0, 0, 0);
          if (GATED)
            acc_u[mt][nt] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
                a[mt], bu[nt], acc_u[mt][nt], 0, 0, 0);
        }
    }
  }

  // epilogue: D[32m][32n]: lane holds D[(r&3)+8*(r>>2)+4*half][col32]
#pragma unroll
  for (int mt = 0; mt < 2; ++mt) {
#pragma unroll
    for (int nt = 0; nt < 2; ++nt) {
      const long n = n_base + nh * 64 + nt * 32 + col32;
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int m = mh * 64 + mt * 32 + (r & 3) + 8 * (r >> 2) + 4 * half;
        if (m >= M) continue;
        float v = acc_g[mt][nt][r];
        if (GATED) {
          const float act = v / (1.f + __expf(-v));
          v = act * acc_u[mt][nt][r];
        }
        y[(long)(m_lo + m) * N + n] = __float2bfloat16(v);
      }
    }
  }
}

void launch_moe_gemm(__hip_bfloat16* y, const __hip_bfloat16* x,
                     const __hip_bfloat16* w, const int* seg_off, int E,
                     int N, int K, int T, bool gated, hipStream_t stream) {
  const int mblk = (T + 127) / 128;
  dim3 grid(N / 128, E * mblk);
  dim3 block(256);
  if (gated)
    hipLaunchKernelGGL(moe_gemm_kernel<true>, grid, block, 0, stream, y, x, w,
                       seg_off, E, N, K, mblk);
  else
    hipLaunchKernelGGL(moe_gemm_kernel<false>, grid, block, 0, stream, y, x,
                       w, seg_off, E, N, K, mblk);
}

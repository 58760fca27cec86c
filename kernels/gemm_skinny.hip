// Skinny decode GEMM (M <= 64): y[M,N] = x[M,K] · W[N,K]^T, bf16 in, bf16 out.
//
// hipBLASLt's picks for the llama decode shapes run at 22-43% of the HBM
// roofline (tools/gemm_bench.py, r1), because N/MT tiles alone can't fill
// 256 CUs at M=64 without split-K.  This kernel is shaped for the real
// bound — streaming W once at full bandwidth:
//   - grid.x = N/16: each workgroup owns 16 rows of W (one MFMA n-tile).
//   - the 4 waves split K statically (K/4 each, contiguous), so every CU
//     holds 4 independent accumulation streams; the partial C tiles meet in
//     LDS at the end (no global split-K scratch, no atomics).
//   - per k-step a wave does MT=ceil(M/16) v_mfma_f32_16x16x32_bf16 ops:
//     A = W rows (streamed, 16B/lane), B = x^T fragments (x is <=32 KB,
//     L2-resident after the first workgroup touches it).
//   - XCD-aware: consecutive blockIdx.x land on different XCDs (round-robin
//     dispatch), so the N dimension spreads its L2 footprint evenly.
//
// x rows beyond M are never read (fragment loads are masked); output rows
// beyond M are not written.

#include "common.h"

typedef __attribute__((ext_vector_type(8))) short bf16x8_t;
typedef __attribute__((ext_vector_type(4))) float f32x4_t;

#define GS_NWAVES 4

template <int MT>  // number of 16-row m tiles (M <= MT*16)
__global__ __launch_bounds__(256, 2) void gemm_skinny_kernel(
    __hip_bfloat16* __restrict__ y,        // [M, N]
    const __hip_bfloat16* __restrict__ x,  // [M, K]
    const __hip_bfloat16* __restrict__ w,  // [N, K]
    const int M,
    const int N,
    const int K) {
  const int n0 = blockIdx.x * 16;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int row = lane & 15;        // W row in tile / n col of C
  const int kq = lane >> 4;         // 0..3: which 8-elem k slice

  // this wave's K range (contiguous quarter)
  const int k_per_wave = K / GS_NWAVES;
  const int k_lo = wave * k_per_wave;

  const __hip_bfloat16* w_row = w + (long)(n0 + row) * K + k_lo + kq * 8;
  f32x4_t acc[MT];
#pragma unroll
  for (int mt = 0; mt < MT; ++mt) acc[mt] = f32x4_t{};

  // x fragment rows: for m tile mt, lane reads x[mt*16 + row][k + kq*8 ..+8]
  // (masked to zero when the m row is past M)
  const bool xrow_ok[4] = {
      0 * 16 + (lane & 15) < M, 1 * 16 + (lane & 15) < M,
      2 * 16 + (lane & 15) < M, 3 * 16 + (lane & 15) < M};

  // K unrolled by U with all loads issued ahead of the mfmas: one wave per
  // SIMD must keep ~U*(1+MT) loads in flight to cover HBM latency (the r1
  // non-unrolled version ran at ~1/3 of hipBLASLt for the N=4096 shapes).
  constexpr int U = 4;
  static_assert(true, "");
  for (int k = 0; k < k_per_wave; k += 32 * U) {
    bf16x8_t a[U];
#pragma unroll
    for (int u = 0; u < U; ++u)
      a[u] = *reinterpret_cast<const bf16x8_t*>(w_row + k + u * 32);
    bf16x8_t b[U][MT];
#pragma unroll
    for (int u = 0; u < U; ++u) {
#pragma unroll
      for (int mt = 0; mt < MT; ++mt) {
        b[u][mt] = bf16x8_t{};
        if (xrow_ok[mt]) {
          b[u][mt] = *reinterpret_cast<const bf16x8_t*>(
              x + (long)(mt * 16 + row) * K + k_lo + k + u * 32 + kq * 8);
        }
      }
    }
#pragma unroll
    for (int u = 0; u < U; ++u) {
#pragma unroll
      for (int mt = 0; mt < MT; ++mt)
        acc[mt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a[u], b[u][mt], acc[mt], 0, 0, 0);
    }
  }

  // ---- cross-wave reduction in LDS ---------------------------------------
  // C frag (16x16x32): lane holds D[n_local = kq*4 + reg][m = lane&15] for
  // each m tile; store as lds_c[wave][m][n_local].
  __shared__ float lds_c[GS_NWAVES][MT * 16][16];
#pragma unroll
  for (int mt = 0; mt < MT; ++mt) {
#pragma unroll
    for (int r = 0; r < 4; ++r)
      lds_c[wave][mt * 16 + row][kq * 4 + r] = acc[mt][r];
  }
  __syncthreads();

  // 256 threads cover the MT*16 x 16 outputs
  for (int idx = tid; idx < MT * 16 * 16; idx += 256) {
    const int m = idx >> 4;
    const int n = idx & 15;
    if (m >= M) continue;
    float s = lds_c[0][m][n] + lds_c[1][m][n] + lds_c[2][m][n] + lds_c[3][m][n];
    y[(long)m * N + n0 + n] = __float2bfloat16(s);
  }
}

void launch_gemm_skinny(__hip_bfloat16* y, const __hip_bfloat16* x,
                        const __hip_bfloat16* w, int M, int N, int K,
                        hipStream_t stream) {
  dim3 grid(N / 16);
  dim3 block(256);
  if (M <= 16)
    hipLaunchKernelGGL(gemm_skinny_kernel<1>, grid, block, 0, stream, y, x, w, M, N, K);
  else if (M <= 32)
    hipLaunchKernelGGL(gemm_skinny_kernel<2>, grid, block, 0, stream, y, x, w, M, N, K);
  else
    hipLaunchKernelGGL(gemm_skinny_kernel<4>, grid, block, 0, stream, y, x, w, M, N, K);
}

// ---------------------------------------------------------------------------
// Gated variant for the SwiGLU MLP up-projection: with w13 = [gate; up]
// ([2I, K], MergedColumnParallelLinear layout), computes
//   y[m, i] = silu(x·Wg^T)[m, i] * (x·Wu^T)[m, i]        (y: [M, I])
// in one pass: each workgroup streams the gate tile AND the matching up tile
// (sharing the x B-fragments), applies the activation in the epilogue, and
// never materialises the [M, 2I] intermediate.
// ---------------------------------------------------------------------------

template <int MT>
__global__ __launch_bounds__(256, 2) void gemm_skinny_gated_kernel(
    __hip_bfloat16* __restrict__ y,          // [M, I]
    const __hip_bfloat16* __restrict__ x,    // [M, K]
    const __hip_bfloat16* __restrict__ w13,  // [2I, K]
    const int M,
    const int I,
    const int K) {
  const int n0 = blockIdx.x * 16;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int row = lane & 15;
  const int kq = lane >> 4;

  const int k_per_wave = K / GS_NWAVES;
  const int k_lo = wave * k_per_wave;

  const __hip_bfloat16* wg_row = w13 + (long)(n0 + row) * K + k_lo + kq * 8;
  const __hip_bfloat16* wu_row = w13 + (long)(I + n0 + row) * K + k_lo + kq * 8;
  f32x4_t acc_g[MT], acc_u[MT];
#pragma unroll
  for (int mt = 0; mt < MT; ++mt) {
    acc_g[mt] = f32x4_t{};
    acc_u[mt] = f32x4_t{};
  }
  const bool xrow_ok[4] = {
      0 * 16 + (lane & 15) < M, 1 * 16 + (lane & 15) < M,
      2 * 16 + (lane & 15) < M, 3 * 16 + (lane & 15) < M};

  constexpr int U = 4;
  for (int k = 0; k < k_per_wave; k += 32 * U) {
    bf16x8_t ag[U], au[U];
#pragma unroll
    for (int u = 0; u < U; ++u) {
      ag[u] = *reinterpret_cast<const bf16x8_t*>(wg_row + k + u * 32);
      au[u] = *reinterpret_cast<const bf16x8_t*>(wu_row + k + u * 32);
    }
    bf16x8_t b[U][MT];
#pragma unroll
    for (int u = 0; u < U; ++u) {
#pragma unroll
      for (int mt = 0; mt < MT; ++mt) {
        b[u][mt] = bf16x8_t{};
        if (xrow_ok[mt]) {
          b[u][mt] = *reinterpret_cast<const bf16x8_t*>(
              x + (long)(mt * 16 + row) * K + k_lo + k + u * 32 + kq * 8);
        }
      }
    }
#pragma unroll
    for (int u = 0; u < U; ++u) {
#pragma unroll
      for (int mt = 0; mt < MT; ++mt) {
        acc_g[mt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(ag[u], b[u][mt], acc_g[mt], 0, 0, 0);
        acc_u[mt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(au[u], b[u][mt], acc_u[mt], 0, 0, 0);
      }
    }
  }

  __shared__ float lds_g[GS_NWAVES][MT * 16][16];
  __shared__ float lds_u[GS_NWAVES][MT * 16][16];
#pragma unroll
  for (int mt = 0; mt < MT; ++mt) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      lds_g[wave][mt * 16 + row][kq * 4 + r] = acc_g[mt][r];
      lds_u[wave][mt * 16 + row][kq * 4 + r] = acc_u[mt][r];
    }
  }
  __syncthreads();

  for (int idx = tid; idx < MT * 16 * 16; idx += 256) {
    const int m = idx >> 4;
    const int n = idx & 15;
    if (m >= M) continue;
    const float g = lds_g[0][m][n] + lds_g[1][m][n] + lds_g[2][m][n] + lds_g[3][m][n];
    const float u = lds_u[0][m][n] + lds_u[1][m][n] + lds_u[2][m][n] + lds_u[3][m][n];
    const float act = g / (1.f + __expf(-g));
    y[(long)m * I + n0 + n] = __float2bfloat16(act * u);
  }
}

void launch_gemm_skinny_gated(__hip_bfloat16* y, const __hip_bfloat16* x,
                              const __hip_bfloat16* w13, int M, int I, int K,
                              hipStream_t stream) {
  dim3 grid(I / 16);
  dim3 block(256);
  if (M <= 16)
    hipLaunchKernelGGL(gemm_skinny_gated_kernel<1>, grid, block, 0, stream, y, x, w13, M, I, K);
  else if (M <= 32)
    hipLaunchKernelGGL(gemm_skinny_gated_kernel<2>, grid, block, 0, stream, y, x, w13, M, I, K);
  else
    hipLaunchKernelGGL(gemm_skinny_gated_kernel<4>, grid, block, 0, stream, y, x, w13, M, I, K);
}

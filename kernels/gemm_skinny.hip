#include <cstdlib>
// Skinny decode GEMM (M <= 64): y[M,N] = x[M,K] · W[N,K]^T, bf16 in/out.
//
// hipBLASLt's picks for the llama decode shapes run at 22-43% of the HBM
// roofline (tools/gemm_bench.py, r1).  v1 of this kernel read MFMA fragments
// straight from HBM; per-lane 16B pieces of 16 rows made every access a
// scattered line request and each workgroup re-read all of x (the W stream
// evicts x from L2), so useful bandwidth capped at ~1.8 TB/s.  v2 is a real
// tiled GEMM shaped for the streaming bound:
//   - grid (N/128, KS): each workgroup owns 128 W rows and 1/KS of K.
//     KS (pure shape function, hipGraph-safe) tops the grid up to >=512
//     workgroups; KS>1 writes f32 partials merged by a tiny second kernel.
//   - per 64-k chunk, W[128, 64] (16 KB) and x[64, 64] (8 KB) tiles are
//     staged HBM->LDS with 128B-contiguous-per-row coalescing, XOR-swizzled
//     ((row&7)<<4, guide G4) so the 16-rows-per-lane fragment reads are
//     bank-conflict-free; double-buffered so chunk c+1 streams while c
//     computes.  x is read once per WG (amortized over 128 N rows).
//   - each wave owns NT=2 16-row n-tiles; per 32-k slice it does
//     NT*MT v_mfma_f32_16x16x32_bf16 with both operands from LDS.
//
// x rows past M are staged as zeros; output rows past M are never written.

#include "common.h"

typedef __attribute__((ext_vector_type(8))) short bf16x8_t;
typedef __attribute__((ext_vector_type(4))) float f32x4_t;
typedef __attribute__((ext_vector_type(16))) float f32x16_t;

#define GS_NWAVES 4
#define GS_NT 2                      // n-tiles (16 rows) per wave
#define GS_ROWS (GS_NWAVES * GS_NT * 16)  // 128 W rows per workgroup
#define GS_KB 64                     // k elems per staged chunk

// byte offset of (row, byte_in_row) in a [rows][GS_KB] bf16 LDS tile with the
// (row&7)<<4 XOR swizzle; row stride = GS_KB*2 = 128 B
DEVINLINE int swz(int row, int byte_in_row) {
  return row * (GS_KB * 2) + (byte_in_row ^ ((row & 7) << 4));
}

template <int MT>  // m tiles of 16 rows per workgroup (up to 8 = 128 rows)
__global__ __launch_bounds__(256, 2) void gemm_skinny_kernel(
    __hip_bfloat16* __restrict__ y,  // [M, N]   (KS == 1)
    float* __restrict__ part,        // [KS, M, N] f32 (KS > 1; else null)
    const __hip_bfloat16* __restrict__ x,  // [M, K]
    const __hip_bfloat16* __restrict__ w,  // [N, K]
    const int M,
    const int N,
    const int K) {
  const int n_blk = blockIdx.x;      // which 128-row block of W
  const int ks = blockIdx.y;
  const int m0 = blockIdx.z * (MT * 16);  // this workgroup's m rows
  const int KS = gridDim.y;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int row16 = lane & 15;
  const int kq = lane >> 4;

  const int k_per_wg = K / KS;
  const int k_lo = ks * k_per_wg;
  const int nchunks = k_per_wg / GS_KB;
  const long n_base = (long)n_blk * GS_ROWS;

  constexpr int XROWS = MT * 16;
  constexpr int XPASS = (XROWS + 31) / 32;
  __shared__ __hip_bfloat16 w_lds[2][GS_ROWS * GS_KB];
  __shared__ __hip_bfloat16 x_lds[2][XROWS * GS_KB];

  // ---- staging: 256 threads, 8 per row (8 x 16 B = 128 B contiguous) -----
  const int st_row = tid / 8;        // 0..31 per pass
  const int st_byte = (tid & 7) * 16;
  const __hip_bfloat16* w_base = w + (n_base + st_row) * (long)K + k_lo + st_byte / 2;
  const __hip_bfloat16* x_base =
      x + (long)(m0 + st_row) * K + k_lo + st_byte / 2;

  bf16x8_t st_w[GS_ROWS / 32];       // 4 passes of W
  bf16x8_t st_x[XPASS];              // x rows m0 .. m0+XROWS
  auto stage_load = [&](int chunk) {
    const int koff = chunk * GS_KB;
#pragma unroll
    for (int p = 0; p < GS_ROWS / 32; ++p)
      st_w[p] = *reinterpret_cast<const bf16x8_t*>(w_base + (long)(p * 32) * K + koff);
#pragma unroll
    for (int p = 0; p < XPASS; ++p) {
      const int r = p * 32 + st_row;
      st_x[p] = bf16x8_t{};
      if (r < XROWS && m0 + r < M)
        st_x[p] = *reinterpret_cast<const bf16x8_t*>(x_base + (long)(p * 32) * K + koff);
    }
  };
  auto stage_write = [&](int buf) {
#pragma unroll
    for (int p = 0; p < GS_ROWS / 32; ++p)
      *reinterpret_cast<bf16x8_t*>(
          reinterpret_cast<char*>(w_lds[buf]) + swz(p * 32 + st_row, st_byte)) = st_w[p];
#pragma unroll
    for (int p = 0; p < XPASS; ++p) {
      const int r = p * 32 + st_row;
      if (r < XROWS)
        *reinterpret_cast<bf16x8_t*>(
            reinterpret_cast<char*>(x_lds[buf]) + swz(r, st_byte)) = st_x[p];
    }
  };

  f32x4_t acc[GS_NT][MT];
#pragma unroll
  for (int nt = 0; nt < GS_NT; ++nt)
#pragma unroll
    for (int mt = 0; mt < MT; ++mt) acc[nt][mt] = f32x4_t{};

  // T14 pipeline (guide §5): write chunk c from regs (its loads had a full
  // iteration to land), immediately re-issue loads for c+1 into the same
  // regs, barrier, compute c.  No load-latency stall inside the loop.
  stage_load(0);

  for (int c = 0; c < nchunks; ++c) {
    const int buf = c & 1;
    stage_write(buf);
    if (c + 1 < nchunks) stage_load(c + 1);
    __syncthreads();  // chunk c staged for everyone

#pragma unroll
    for (int kc = 0; kc < 2; ++kc) {
      const int byte = kc * 64 + kq * 16;
      bf16x8_t a[GS_NT], b[MT];
#pragma unroll
      for (int nt = 0; nt < GS_NT; ++nt) {
        const int r = (wave * GS_NT + nt) * 16 + row16;
        a[nt] = *reinterpret_cast<const bf16x8_t*>(
            reinterpret_cast<const char*>(w_lds[buf]) + swz(r, byte));
      }
#pragma unroll
      for (int mt = 0; mt < MT; ++mt) {
        b[mt] = *reinterpret_cast<const bf16x8_t*>(
            reinterpret_cast<const char*>(x_lds[buf]) + swz(mt * 16 + row16, byte));
      }
#pragma unroll
      for (int nt = 0; nt < GS_NT; ++nt)
#pragma unroll
        for (int mt = 0; mt < MT; ++mt)
          acc[nt][mt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a[nt], b[mt], acc[nt][mt], 0, 0, 0);
    }
  }

  // ---- epilogue: C frag D[n_local = kq*4+reg][m = row16] per (nt, mt) ----
#pragma unroll
  for (int nt = 0; nt < GS_NT; ++nt) {
    const long n0 = n_base + (wave * GS_NT + nt) * 16 + kq * 4;
#pragma unroll
    for (int mt = 0; mt < MT; ++mt) {
      const int m = m0 + mt * 16 + row16;
      if (m >= M) continue;
      if (part == nullptr) {
        // pack 4 f32 -> 4 bf16 (8 B) at y[m][n0..n0+4)
        union { unsigned u[2]; } o;
        asm volatile("v_cvt_pk_bf16_f32 %0, %1, %2"
                     : "=v"(o.u[0]) : "v"(acc[nt][mt][0]), "v"(acc[nt][mt][1]));
        asm volatile("v_cvt_pk_bf16_f32 %0, %1, %2"
                     : "=v"(o.u[1]) : "v"(acc[nt][mt][2]), "v"(acc[nt][mt][3]));
        *reinterpret_cast<unsigned*>(y + (long)m * N + n0) = o.u[0];
        *reinterpret_cast<unsigned*>(y + (long)m * N + n0 + 2) = o.u[1];
      } else {
        float* p = part + ((long)ks * M + m) * N + n0;
        *reinterpret_cast<f32x4_t*>(p) = acc[nt][mt];
      }
    }
  }
}


// ---------------------------------------------------------------------------
// Large-M tiled GEMM (M > 64): classic 128x128 macro-tile on
// v_mfma_f32_32x32x16_bf16.  The 2x2 wave grid gives each wave a 64x64 C
// tile (2x2 of 32x32 MFMA tiles, 64 f32 accum regs); x and W chunk tiles
// ([128, 64k], 16 KB each) are staged with the same XOR-swizzled T14
// double-buffer pipeline as the skinny kernel.  Replaces hipBLASLt for the
// prefill and large-decode-batch linears (measured ~34% MFU there, r1).
// ---------------------------------------------------------------------------

__global__ __launch_bounds__(256, 2) void gemm_tile_kernel(
    __hip_bfloat16* __restrict__ y,  // [M, N]   (KS == 1)
    float* __restrict__ part,        // [KS, M, N] f32 (KS > 1; else null)
    const __hip_bfloat16* __restrict__ x,  // [M, K]
    const __hip_bfloat16* __restrict__ w,  // [N, K]
    const int M,
    const int N,
    const int K) {
  const int n_blk = blockIdx.x;
  const int ks = blockIdx.y;
  const int m0 = blockIdx.z * 128;
  const int KS = gridDim.y;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int col32 = lane & 31;
  const int half = lane >> 5;
  const int mh = wave >> 1;   // wave's m half (64 rows)
  const int nh = wave & 1;    // wave's n half (64 cols)

  const int k_per_wg = K / KS;
  const int k_lo = ks * k_per_wg;
  const int nchunks = k_per_wg / GS_KB;
  const long n_base = (long)n_blk * 128;

  __shared__ __hip_bfloat16 w_lds[2][128 * GS_KB];
  __shared__ __hip_bfloat16 x_lds[2][128 * GS_KB];

  const int st_row = tid / 8;        // 0..31 per pass
  const int st_byte = (tid & 7) * 16;
  const __hip_bfloat16* w_base = w + (n_base + st_row) * (long)K + k_lo + st_byte / 2;
  const __hip_bfloat16* x_base =
      x + (long)(m0 + st_row) * K + k_lo + st_byte / 2;

  bf16x8_t st_w[4], st_x[4];
  auto stage_load = [&](int chunk) {
    const int koff = chunk * GS_KB;
#pragma unroll
    for (int p = 0; p < 4; ++p) {
      st_w[p] = *reinterpret_cast<const bf16x8_t*>(w_base + (long)(p * 32) * K + koff);
      st_x[p] = bf16x8_t{};
      if (m0 + p * 32 + st_row < M)
        st_x[p] = *reinterpret_cast<const bf16x8_t*>(x_base + (long)(p * 32) * K + koff);
    }
  };
  auto stage_write = [&](int buf) {
#pragma unroll
    for (int p = 0; p < 4; ++p) {
      *reinterpret_cast<bf16x8_t*>(
          reinterpret_cast<char*>(w_lds[buf]) + swz(p * 32 + st_row, st_byte)) = st_w[p];
      *reinterpret_cast<bf16x8_t*>(
          reinterpret_cast<char*>(x_lds[buf]) + swz(p * 32 + st_row, st_byte)) = st_x[p];
    }
  };

  f32x16_t acc[2][2];  // [mt][nt] 32x32 tiles
  acc[0][0] = f32x16_t{}; acc[0][1] = f32x16_t{};
  acc[1][0] = f32x16_t{}; acc[1][1] = f32x16_t{};

  stage_load(0);

  for (int c = 0; c < nchunks; ++c) {
    const int buf = c & 1;
    stage_write(buf);
    if (c + 1 < nchunks) stage_load(c + 1);
    __syncthreads();

#pragma unroll
    for (int kc = 0; kc < 4; ++kc) {  // 16-k slices of the 64-k chunk
      const int byte = kc * 32 + half * 16;
      bf16x8_t a[2], b[2];
#pragma unroll
      for (int mt = 0; mt < 2; ++mt)
        a[mt] = *reinterpret_cast<const bf16x8_t*>(
            reinterpret_cast<const char*>(x_lds[buf]) +
            swz(mh * 64 + mt * 32 + col32, byte));
#pragma unroll
      for (int nt = 0; nt < 2; ++nt)
        b[nt] = *reinterpret_cast<const bf16x8_t*>(
            reinterpret_cast<const char*>(w_lds[buf]) +
            swz(nh * 64 + nt * 32 + col32, byte));
#pragma unroll
      for (int mt = 0; mt < 2; ++mt)
#pragma unroll
        for (int nt = 0; nt < 2; ++nt)
          acc[mt][nt] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a[mt], b[nt], acc[mt][nt], 0, 0, 0);
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
        const int m = m0 + mh * 64 + mt * 32 + (r & 3) + 8 * (r >> 2) + 4 * half;
        if (m >= M) continue;
        if (part == nullptr)
          y[(long)m * N + n] = __float2bfloat16(acc[mt][nt][r]);
        else
          part[((long)ks * M + m) * N + n] = acc[mt][nt][r];
      }
    }
  }
}

// y[m][n] = sum_ks part[ks][m][n], bf16 out
__global__ void gemm_skinny_merge_kernel(
    __hip_bfloat16* __restrict__ y,
    const float* __restrict__ part,
    const int M,
    const int N,
    const int KS) {
  const long idx = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (idx >= (long)M * N) return;
  float s = 0.f;
  for (int ks = 0; ks < KS; ++ks) s += part[(long)ks * M * N + idx];
  y[idx] = __float2bfloat16(s);
}

// KS: pure function of shapes (hipGraph-stable): top the grid up to >=512
// workgroups while keeping each WG's K range a multiple of 64 and >=256.
int gemm_skinny_num_ksplit(int N, int K, int M) {
  // VTA_GEMM_MIN_KDEPTH tunes the fill-vs-pipeline-depth trade
  // (k elems each workgroup keeps; deeper = better ramp, fewer WGs).
  static const int min_depth = [] {
    const char* e = getenv("VTA_GEMM_MIN_KDEPTH");
    return e ? atoi(e) : 1024;
  }();
  const int cols = N / GS_ROWS;
  const int zb = M > 64 ? (M + 127) / 128 : 1;
  int ks = 1;
  while (ks < 16 && cols * zb * ks * 2 <= 512 && (K / (ks * 2)) % GS_KB == 0 &&
         K / (ks * 2) >= min_depth)
    ks *= 2;
  return ks;
}

void launch_gemm_skinny(__hip_bfloat16* y, float* part, const __hip_bfloat16* x,
                        const __hip_bfloat16* w, int M, int N, int K, int KS,
                        hipStream_t stream) {
  dim3 block(256);
  float* p = KS > 1 ? part : nullptr;
#define GS_CASE(MT, ZB)                                                      \
  hipLaunchKernelGGL(gemm_skinny_kernel<MT>, dim3(N / GS_ROWS, KS, ZB),      \
                     block, 0, stream, y, p, x, w, M, N, K)
  if (M <= 16) GS_CASE(1, 1);
  else if (M <= 32) GS_CASE(2, 1);
  else if (M <= 64) GS_CASE(4, 1);
  else
    hipLaunchKernelGGL(gemm_tile_kernel,
                       dim3(N / 128, KS, (M + 127) / 128), block, 0, stream,
                       y, p, x, w, M, N, K);
#undef GS_CASE
  if (KS > 1) {
    const long total = (long)M * N;
    hipLaunchKernelGGL(gemm_skinny_merge_kernel,
                       dim3((total + 255) / 256), dim3(256), 0, stream, y,
                       part, M, N, KS);
  }
}

// ---------------------------------------------------------------------------
// Gated variant for the SwiGLU MLP: with w13 = [gate; up] ([2I, K]),
//   y[m, i] = silu(x·Wg^T)[m, i] * (x·Wu^T)[m, i]        (y: [M, I])
// Same tiling; each workgroup streams the gate tile AND the matching up tile
// (sharing the x tile), so the [M, 2I] intermediate never exists.
// ---------------------------------------------------------------------------

template <int MT>
__global__ __launch_bounds__(256, 2) void gemm_skinny_gated_kernel(
    __hip_bfloat16* __restrict__ y,   // [M, I]  (KS == 1)
    float* __restrict__ part,         // [KS, 2, M, I] f32 (KS > 1)
    const __hip_bfloat16* __restrict__ x,    // [M, K]
    const __hip_bfloat16* __restrict__ w13,  // [2I, K]
    const int M,
    const int I,
    const int K) {
  const int n_blk = blockIdx.x;
  const int ks = blockIdx.y;
  const int m0 = blockIdx.z * (MT * 16);
  const int KS = gridDim.y;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int row16 = lane & 15;
  const int kq = lane >> 4;

  const int k_per_wg = K / KS;
  const int k_lo = ks * k_per_wg;
  const int nchunks = k_per_wg / GS_KB;
  const long n_base = (long)n_blk * GS_ROWS;

  constexpr int XROWS = MT * 16;
  constexpr int XPASS = (XROWS + 31) / 32;
  __shared__ __hip_bfloat16 wg_lds[2][GS_ROWS * GS_KB];
  __shared__ __hip_bfloat16 wu_lds[2][GS_ROWS * GS_KB];
  __shared__ __hip_bfloat16 x_lds[2][XROWS * GS_KB];

  const int st_row = tid / 8;
  const int st_byte = (tid & 7) * 16;
  const __hip_bfloat16* wg_base = w13 + (n_base + st_row) * (long)K + k_lo + st_byte / 2;
  const __hip_bfloat16* wu_base =
      w13 + ((long)I + n_base + st_row) * K + k_lo + st_byte / 2;
  const __hip_bfloat16* x_base =
      x + (long)(m0 + st_row) * K + k_lo + st_byte / 2;

  bf16x8_t st_g[GS_ROWS / 32], st_u[GS_ROWS / 32], st_x[XPASS];
  auto stage_load = [&](int chunk) {
    const int koff = chunk * GS_KB;
#pragma unroll
    for (int p = 0; p < GS_ROWS / 32; ++p) {
      st_g[p] = *reinterpret_cast<const bf16x8_t*>(wg_base + (long)(p * 32) * K + koff);
      st_u[p] = *reinterpret_cast<const bf16x8_t*>(wu_base + (long)(p * 32) * K + koff);
    }
#pragma unroll
    for (int p = 0; p < XPASS; ++p) {
      const int r = p * 32 + st_row;
      st_x[p] = bf16x8_t{};
      if (r < XROWS && m0 + r < M)
        st_x[p] = *reinterpret_cast<const bf16x8_t*>(x_base + (long)(p * 32) * K + koff);
    }
  };
  auto stage_write = [&](int buf) {
#pragma unroll
    for (int p = 0; p < GS_ROWS / 32; ++p) {
      *reinterpret_cast<bf16x8_t*>(
          reinterpret_cast<char*>(wg_lds[buf]) + swz(p * 32 + st_row, st_byte)) = st_g[p];
      *reinterpret_cast<bf16x8_t*>(
          reinterpret_cast<char*>(wu_lds[buf]) + swz(p * 32 + st_row, st_byte)) = st_u[p];
    }
#pragma unroll
    for (int p = 0; p < XPASS; ++p) {
      const int r = p * 32 + st_row;
      if (r < XROWS)
        *reinterpret_cast<bf16x8_t*>(
            reinterpret_cast<char*>(x_lds[buf]) + swz(r, st_byte)) = st_x[p];
    }
  };

  f32x4_t acc_g[GS_NT][MT], acc_u[GS_NT][MT];
#pragma unroll
  for (int nt = 0; nt < GS_NT; ++nt)
#pragma unroll
    for (int mt = 0; mt < MT; ++mt) {
      acc_g[nt][mt] = f32x4_t{};
      acc_u[nt][mt] = f32x4_t{};
    }

  stage_load(0);

  for (int c = 0; c < nchunks; ++c) {
    const int buf = c & 1;
    stage_write(buf);
    if (c + 1 < nchunks) stage_load(c + 1);
    __syncthreads();

#pragma unroll
    for (int kc = 0; kc < 2; ++kc) {
      const int byte = kc * 64 + kq * 16;
      bf16x8_t ag[GS_NT], au[GS_NT], b[MT];
#pragma unroll
      for (int nt = 0; nt < GS_NT; ++nt) {
        const int r = (wave * GS_NT + nt) * 16 + row16;
        ag[nt] = *reinterpret_cast<const bf16x8_t*>(
            reinterpret_cast<const char*>(wg_lds[buf]) + swz(r, byte));
        au[nt] = *reinterpret_cast<const bf16x8_t*>(
            reinterpret_cast<const char*>(wu_lds[buf]) + swz(r, byte));
      }
#pragma unroll
      for (int mt = 0; mt < MT; ++mt)
        b[mt] = *reinterpret_cast<const bf16x8_t*>(
            reinterpret_cast<const char*>(x_lds[buf]) + swz(mt * 16 + row16, byte));
#pragma unroll
      for (int nt = 0; nt < GS_NT; ++nt)
#pragma unroll
        for (int mt = 0; mt < MT; ++mt) {
          acc_g[nt][mt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(ag[nt], b[mt], acc_g[nt][mt], 0, 0, 0);
          acc_u[nt][mt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(au[nt], b[mt], acc_u[nt][mt], 0, 0, 0);
        }
    }
  }

#pragma unroll
  for (int nt = 0; nt < GS_NT; ++nt) {
    const long n0 = n_base + (wave * GS_NT + nt) * 16 + kq * 4;
#pragma unroll
    for (int mt = 0; mt < MT; ++mt) {
      const int m = m0 + mt * 16 + row16;
      if (m >= M) continue;
      if (part == nullptr) {
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const float g = acc_g[nt][mt][r];
          const float act = g / (1.f + __expf(-g));
          y[(long)m * I + n0 + r] = __float2bfloat16(act * acc_u[nt][mt][r]);
        }
      } else {
        float* pg = part + (((long)ks * 2 + 0) * M + m) * I + n0;
        float* pu = part + (((long)ks * 2 + 1) * M + m) * I + n0;
        *reinterpret_cast<f32x4_t*>(pg) = acc_g[nt][mt];
        *reinterpret_cast<f32x4_t*>(pu) = acc_u[nt][mt];
      }
    }
  }
}

// y[m][i] = silu(sum_ks g) * (sum_ks u)
__global__ void gemm_gated_merge_kernel(
    __hip_bfloat16* __restrict__ y,
    const float* __restrict__ part,  // [KS, 2, M, I]
    const int M,
    const int I,
    const int KS) {
  const long idx = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (idx >= (long)M * I) return;
  float g = 0.f, u = 0.f;
  for (int ks = 0; ks < KS; ++ks) {
    g += part[((long)ks * 2 + 0) * M * I + idx];
    u += part[((long)ks * 2 + 1) * M * I + idx];
  }
  const float act = g / (1.f + __expf(-g));
  y[idx] = __float2bfloat16(act * u);
}

void launch_gemm_skinny_gated(__hip_bfloat16* y, float* part,
                              const __hip_bfloat16* x,
                              const __hip_bfloat16* w13, int M, int I, int K,
                              int KS, hipStream_t stream) {
  dim3 block(256);
  float* p = KS > 1 ? part : nullptr;
#define GG_CASE(MT, ZB)                                                       \
  hipLaunchKernelGGL(gemm_skinny_gated_kernel<MT>,                            \
                     dim3(I / GS_ROWS, KS, ZB), block, 0, stream, y,          \
                     p, x, w13, M, I, K)
  if (M <= 16) GG_CASE(1, 1);
  else if (M <= 32) GG_CASE(2, 1);
  else if (M <= 64) GG_CASE(4, 1);
  else GG_CASE(8, (M + 127) / 128);
#undef GG_CASE
  if (KS > 1) {
    const long total = (long)M * I;
    hipLaunchKernelGGL(gemm_gated_merge_kernel,
                       dim3((total + 255) / 256), dim3(256), 0, stream, y,
                       part, M, I, KS);
  }
}

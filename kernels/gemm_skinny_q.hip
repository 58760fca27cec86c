// Weight-only-quantized skinny decode GEMM: y = x @ Wq^T,
// activations bf16, weights int8 (per-output-channel scale) or packed int4
// (group-128 scales, offset-binary nibbles).  Same streaming structure as
// gemm_skinny.hip — the W tile is dequantised to bf16 during the LDS stage
// write, so the MFMA inner loop is unchanged while the HBM weight stream
// shrinks 2x (int8) / 4x (int4); at M <= 64 these GEMMs are weight-stream
// bound, so that is the speedup.
//
// Reference surface: --quantize {awq,gptq,squeezellm} (reference
// tgis_utils/args.py:128-138).  Checkpoint-specific AWQ/GPTQ scale formats
// don't exist in this offline environment; the engine applies round-to-
// nearest (RTN) quantisation of the loaded weights into these layouts
// (int4 group-128 for the 4-bit surfaces, int8 for --quantize int8).

#include "common.h"

typedef __attribute__((ext_vector_type(8))) short bf16x8_t;
typedef __attribute__((ext_vector_type(4))) float f32x4_t;

#define GQ_NWAVES 4
#define GQ_NT 2
#define GQ_ROWS (GQ_NWAVES * GQ_NT * 16)
#define GQ_KB 64
#define GQ_GROUP 128  // int4 scale group along K

DEVINLINE int gq_swz(int row, int byte_in_row) {
  return row * (GQ_KB * 2) + (byte_in_row ^ ((row & 7) << 4));
}

// QBITS: 8 (int8, per-row scale) or 4 (packed nibbles, [N][K/128] scales)
template <int MT, int QBITS>
__global__ __launch_bounds__(256, 2) void gemm_skinny_q_kernel(
    __hip_bfloat16* __restrict__ y,        // [M, N]
    float* __restrict__ part,              // [KS, M, N] (KS > 1) or null
    const __hip_bfloat16* __restrict__ x,  // [M, K]
    const unsigned char* __restrict__ wq,  // int8: [N][K]; int4: [N][K/2]
    const float* __restrict__ wscale,      // int8: [N]; int4: [N][K/128]
    const int M, const int N, const int K) {
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
  const int nchunks = k_per_wg / GQ_KB;
  const long n_base = (long)n_blk * GQ_ROWS;

  constexpr int XROWS = MT * 16;
  constexpr int XPASS = (XROWS + 31) / 32;
  __shared__ __hip_bfloat16 w_lds[2][GQ_ROWS * GQ_KB];
  __shared__ __hip_bfloat16 x_lds[2][XROWS * GQ_KB];

  const int st_row = tid / 8;        // 0..31 per pass
  const int st_elem = (tid & 7) * 8;  // first of this thread's 8 k elems
  const __hip_bfloat16* x_base =
      x + (long)(m0 + st_row) * K + k_lo + st_elem;

  // quantized W source for this thread's 8 elems
  typedef __attribute__((ext_vector_type(8))) char i8x8_t;
  union QW {
    i8x8_t b8;      // int8: 8 bytes
    unsigned u4;    // int4: 4 bytes = 8 nibbles
  };
  QW st_w[GQ_ROWS / 32];
  bf16x8_t st_x[XPASS];

  auto stage_load = [&](int chunk) {
    const int koff = chunk * GQ_KB;
#pragma unroll
    for (int p = 0; p < GQ_ROWS / 32; ++p) {
      const long row = n_base + p * 32 + st_row;
      if (QBITS == 8) {
        st_w[p].b8 = *reinterpret_cast<const i8x8_t*>(
            wq + row * K + k_lo + koff + st_elem);
      } else {
        st_w[p].u4 = *reinterpret_cast<const unsigned*>(
            wq + row * (K / 2) + (k_lo + koff + st_elem) / 2);
      }
    }
#pragma unroll
    for (int p = 0; p < XPASS; ++p) {
      const int r = p * 32 + st_row;
      st_x[p] = bf16x8_t{};
      if (r < XROWS && m0 + r < M)
        st_x[p] = *reinterpret_cast<const bf16x8_t*>(
            x_base + (long)(p * 32) * K + koff);
    }
  };

  auto stage_write = [&](int buf, int chunk) {
    const int koff = chunk * GQ_KB;
#pragma unroll
    for (int p = 0; p < GQ_ROWS / 32; ++p) {
      const long row = n_base + p * 32 + st_row;
      float scale;
      if (QBITS == 8) {
        scale = wscale[row];
      } else {
        scale = wscale[row * (K / GQ_GROUP) + (k_lo + koff + st_elem) / GQ_GROUP];
      }
      bf16x8_t v;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float q;
        if (QBITS == 8) {
          q = (float)st_w[p].b8[j];
        } else {
          q = (float)(int)((st_w[p].u4 >> (4 * j)) & 0xF) - 8.f;
        }
        v[j] = __builtin_bit_cast(short, __float2bfloat16(q * scale));
      }
      *reinterpret_cast<bf16x8_t*>(
          reinterpret_cast<char*>(w_lds[buf]) +
          gq_swz(p * 32 + st_row, st_elem * 2)) = v;
    }
#pragma unroll
    for (int p = 0; p < XPASS; ++p) {
      const int r = p * 32 + st_row;
      if (r < XROWS)
        *reinterpret_cast<bf16x8_t*>(
            reinterpret_cast<char*>(x_lds[buf]) + gq_swz(r, st_elem * 2)) =
            st_x[p];
    }
  };

  f32x4_t acc[GQ_NT][MT];
#pragma unroll
  for (int nt = 0; nt < GQ_NT; ++nt)
#pragma unroll
    for (int mt = 0; mt < MT; ++mt) acc[nt][mt] = f32x4_t{};

  stage_load(0);

  for (int c = 0; c < nchunks; ++c) {
    const int buf = c & 1;
    stage_write(buf, c);
    if (c + 1 < nchunks) stage_load(c + 1);
    __syncthreads();

#pragma unroll
    for (int kc = 0; kc < 2; ++kc) {
      const int byte = kc * 64 + kq * 16;
      bf16x8_t a[GQ_NT], b[MT];
#pragma unroll
      for (int nt = 0; nt < GQ_NT; ++nt) {
        const int r = (wave * GQ_NT + nt) * 16 + row16;
        a[nt] = *reinterpret_cast<const bf16x8_t*>(
            reinterpret_cast<const char*>(w_lds[buf]) + gq_swz(r, byte));
      }
#pragma unroll
      for (int mt = 0; mt < MT; ++mt) {
        b[mt] = *reinterpret_cast<const bf16x8_t*>(
            reinterpret_cast<const char*>(x_lds[buf]) +
            gq_swz(mt * 16 + row16, byte));
      }
#pragma unroll
      for (int nt = 0; nt < GQ_NT; ++nt)
#pragma unroll
        for (int mt = 0; mt < MT; ++mt)
          acc[nt][mt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a[nt], b[mt], acc[nt][mt], 0, 0, 0);
    }
  }

#pragma unroll
  for (int nt = 0; nt < GQ_NT; ++nt) {
    const long n0 = n_base + (wave * GQ_NT + nt) * 16 + kq * 4;
#pragma unroll
    for (int mt = 0; mt < MT; ++mt) {
      const int m = m0 + mt * 16 + row16;
      if (m >= M) continue;
      if (part == nullptr) {
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

int gemm_skinny_num_ksplit(int N, int K, int M);  // gemm_skinny.hip
__global__ void gemm_skinny_merge_kernel(__hip_bfloat16*, const float*, int,
                                         int, int);

template <int QBITS>
static void launch_q(__hip_bfloat16* y, float* part, const __hip_bfloat16* x,
                     const unsigned char* wq, const float* ws, int M, int N,
                     int K, int KS, hipStream_t stream) {
  const int mt = (min(M, 64) + 15) / 16;
  dim3 grid(N / GQ_ROWS, KS, (M + mt * 16 - 1) / (mt * 16));
  dim3 block(256);
  switch (mt) {
    case 1:
      hipLaunchKernelGGL((gemm_skinny_q_kernel<1, QBITS>), grid, block, 0,
                         stream, y, part, x, wq, ws, M, N, K);
      break;
    case 2:
      hipLaunchKernelGGL((gemm_skinny_q_kernel<2, QBITS>), grid, block, 0,
                         stream, y, part, x, wq, ws, M, N, K);
      break;
    case 3:
      hipLaunchKernelGGL((gemm_skinny_q_kernel<3, QBITS>), grid, block, 0,
                         stream, y, part, x, wq, ws, M, N, K);
      break;
    default:
      hipLaunchKernelGGL((gemm_skinny_q_kernel<4, QBITS>), grid, block, 0,
                         stream, y, part, x, wq, ws, M, N, K);
      break;
  }
}

void launch_gemm_skinny_q(__hip_bfloat16* y, float* part,
                          const __hip_bfloat16* x, const unsigned char* wq,
                          const float* ws, int qbits, int M, int N, int K,
                          int KS, hipStream_t stream) {
  if (qbits == 8)
    launch_q<8>(y, KS > 1 ? part : nullptr, x, wq, ws, M, N, K, KS, stream);
  else
    launch_q<4>(y, KS > 1 ? part : nullptr, x, wq, ws, M, N, K, KS, stream);
  if (KS > 1) {
    const long total = (long)M * N;
    hipLaunchKernelGGL(gemm_skinny_merge_kernel, dim3((total + 255) / 256),
                       dim3(256), 0, stream, y, part, M, N, KS);
  }
}

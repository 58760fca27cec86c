// MFMA fragment-layout probes.  Computes D = A·B with the ASSUMED per-lane
// operand layouts for the gfx950 bf16 MFMA shapes; the GPU test compares
// against a torch matmul so a wrong layout assumption fails loudly before it
// can hide inside the attention kernel (guide §5.4 rule 16).
//
// Assumed layouts (v_mfma_f32_32x32x16_bf16, one wave):
//   A[32m x 16k]: lane l holds A[l&31][(l>>5)*8 + j], j = 0..7
//   B[16k x 32n]: lane l holds B[(l>>5)*8 + j][l&31]
//   D[32m x 32n]: lane l holds D[(reg&3) + 8*(reg>>2) + 4*(l>>5)][l&31], reg 0..15
// And for v_mfma_f32_16x16x32_bf16:
//   A[16m x 32k]: lane l holds A[l&15][(l>>4)*8 + j]
//   B[32k x 16n]: lane l holds B[(l>>4)*8 + j][l&15]
//   D[16m x 16n]: lane l holds D[(l>>4)*4 + reg][l&15], reg 0..3

#include "common.h"

typedef __attribute__((ext_vector_type(8))) short bf16x8_t;
typedef __attribute__((ext_vector_type(16))) float f32x16_t;
typedef __attribute__((ext_vector_type(4))) float f32x4_t;

__global__ void mfma_probe_32x32x16(const unsigned short* __restrict__ a,  // [32,16]
                                    const unsigned short* __restrict__ b,  // [16,32]
                                    float* __restrict__ d) {               // [32,32]
  const int lane = threadIdx.x & 63;
  bf16x8_t av, bv;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    av[j] = (short)a[(lane & 31) * 16 + (lane >> 5) * 8 + j];
    bv[j] = (short)b[((lane >> 5) * 8 + j) * 32 + (lane & 31)];
  }
  f32x16_t acc = {};
  acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(av, bv, acc, 0, 0, 0);
#pragma unroll
  for (int reg = 0; reg < 16; ++reg) {
    const int row = (reg & 3) + 8 * (reg >> 2) + 4 * (lane >> 5);
    d[row * 32 + (lane & 31)] = acc[reg];
  }
}

__global__ void mfma_probe_16x16x32(const unsigned short* __restrict__ a,  // [16,32]
                                    const unsigned short* __restrict__ b,  // [32,16]
                                    float* __restrict__ d) {               // [16,16]
  const int lane = threadIdx.x & 63;
  bf16x8_t av, bv;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    av[j] = (short)a[(lane & 15) * 32 + (lane >> 4) * 8 + j];
    bv[j] = (short)b[((lane >> 4) * 8 + j) * 16 + (lane & 15)];
  }
  f32x4_t acc = {};
  acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(av, bv, acc, 0, 0, 0);
#pragma unroll
  for (int reg = 0; reg < 4; ++reg) {
    const int row = (lane >> 4) * 4 + reg;
    d[row * 16 + (lane & 15)] = acc[reg];
  }
}

void launch_mfma_probe_32(const unsigned short* a, const unsigned short* b,
                          float* d, hipStream_t s) {
  hipLaunchKernelGGL(mfma_probe_32x32x16, dim3(1), dim3(64), 0, s, a, b, d);
}

void launch_mfma_probe_16(const unsigned short* a, const unsigned short* b,
                          float* d, hipStream_t s) {
  hipLaunchKernelGGL(mfma_probe_16x16x32, dim3(1), dim3(64), 0, s, a, b, d);
}

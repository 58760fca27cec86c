// MFMA paged decode attention — replaces the VALU decode
// kernel for bf16 head_dim 64/128, GQA group <= 8.
//
// The VALU kernel spent its time in per-slot cross-lane reduction chains
// (rocprofv3 r1: 155 us/call vs a ~17 us HBM roofline at batch 64, ctx 512).
// Here Q·K^T runs on v_mfma_f32_32x32x16_bf16 tiles with the swapped-operand
// trick from the prefill kernel (S^T = K·Q^T, so softmax stats are per-lane
// column with ONE shfl per 32-slot tile), P is repacked with
// v_cvt_pk_bf16_f32 + v_permlane32_swap into the P·V A-operand, and V tiles
// are staged per-wave through LDS so the transposed B-fragment gather stays
// on-chip.
//
// Flash-decoding split: grid (nseq, kvh, npart); the context is cut into
// npart contiguous tile ranges and each workgroup's 4 waves take tiles
// round-robin inside their range.  Every wave writes an independent partial
// (m, l, acc_f32) — no cross-wave merge in this kernel — and a small merge
// kernel reduces the npart*4 partials per (seq, head).  npart is a pure
// function of tensor shapes, so the launch is hipGraph-stable while the
// in-kernel split adapts to the runtime seq_len.

#include "common.h"
#include <float.h>

typedef __attribute__((ext_vector_type(8))) short bf16x8_t;
typedef __attribute__((ext_vector_type(16))) float f32x16_t;

// 8 consecutive KV elements as bf16: direct for bf16 caches, hardware
// cvt_pk_f32_fp8 for e4m3 caches (fp8 KV halves the decode HBM stream)
template <typename CT>
DEVINLINE bf16x8_t load_kv8(const CT* p);
template <>
DEVINLINE bf16x8_t load_kv8<__hip_bfloat16>(const __hip_bfloat16* p) {
  return *reinterpret_cast<const bf16x8_t*>(p);
}
template <>
DEVINLINE bf16x8_t load_kv8<unsigned char>(const unsigned char* p) {
  return __builtin_bit_cast(
      bf16x8_t, fp8x8_to_bf16x8(*reinterpret_cast<const u32x2_vec_t*>(p)));
}

#define KVT 32     // kv slots per mfma tile
#define NWAVES 4   // waves per workgroup

template <int HEAD_DIM, typename CT>
__global__ __launch_bounds__(256, 2) void paged_decode_mfma_kernel(
    float* __restrict__ part_acc,   // [nseq, nheads, P, HEAD_DIM] f32
    float* __restrict__ part_ml,    // [nseq, nheads, P, 2] f32 (m, l)
    const __hip_bfloat16* __restrict__ q,        // [nseq, nheads, HD]
    const CT* __restrict__ k_cache,  // [nb, bs, kvh, HD] bf16 | e4m3
    const CT* __restrict__ v_cache,
    const int* __restrict__ block_tables,  // [nseq, max_blocks]
    const int* __restrict__ seq_lens,      // [nseq]
    const float scale,
    const int nheads,
    const int kvh,
    const int group,       // nheads / kvh, <= 8
    const int block_size,
    const int max_blocks) {
  constexpr int KCH = HEAD_DIM / 16;  // k-chunks for the QK^T mfma
  constexpr int DT = HEAD_DIM / 32;   // 32-col output tiles

  const int seq = blockIdx.x;
  const int kv_head = blockIdx.y;
  const int part = blockIdx.z;
  const int npart = gridDim.z;
  const int seq_len = seq_lens[seq];

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int col = lane & 31;   // kv row of the A frag AND q col of the B frag
  const int half = lane >> 5;

  // this partition's tile range; waves stride by NWAVES inside it
  const int tiles_total = (seq_len + KVT - 1) / KVT;
  const int tpp = (tiles_total + npart - 1) / npart;
  const int tile_lo = part * tpp;
  const int tile_hi = min(tile_lo + tpp, tiles_total);

  const int head0 = kv_head * group;
  const int* btable = block_tables + (long)seq * max_blocks;
  const long kv_row_stride = (long)kvh * HEAD_DIM;

  // per-wave V staging tile (linear; the B gather reads rows across lanes)
  __shared__ __hip_bfloat16 v_lds_all[NWAVES * KVT * HEAD_DIM];
  __hip_bfloat16* v_lds = v_lds_all + wave * KVT * HEAD_DIM;

  // ---- Q as B fragments: lane holds Q[q=col][ks*16 + half*8 + j] ---------
  bf16x8_t qb[KCH];
  if (col < group) {
    const __hip_bfloat16* q_row = q + ((long)seq * nheads + head0 + col) * HEAD_DIM;
#pragma unroll
    for (int ks = 0; ks < KCH; ++ks)
      qb[ks] = *reinterpret_cast<const bf16x8_t*>(q_row + ks * 16 + half * 8);
  } else {
#pragma unroll
    for (int ks = 0; ks < KCH; ++ks) qb[ks] = bf16x8_t{};
  }

  float m_state = -FLT_MAX;  // per q col (duplicated on lane and lane^32)
  float l_state = 0.f;
  f32x16_t acc_o[DT];
#pragma unroll
  for (int dt = 0; dt < DT; ++dt) acc_o[dt] = f32x16_t{};

  constexpr int VPASS = KVT * HEAD_DIM / 8 / 64;  // V staging passes per wave

  for (int t = tile_lo + wave; t < tile_hi; t += NWAVES) {
    const int kv_base = t * KVT;

    // ---- issue ALL of this tile's HBM loads first (K frags + V rows) -----
    // K is needed first (QK^T mfma), so its 8 loads go out before V's; the
    // mfma chain then only waits on vmcnt covering K while V stays in
    // flight, and the V->LDS writes drain afterwards.  Keeping every load
    // of the tile outstanding together (instead of load->ds_write per
    // pass) collapses ~9 HBM round trips per tile into ~1.
    bf16x8_t ka[KCH];
    {
      const int pos = kv_base + col;
      if (pos < seq_len) {
        const int block = btable[pos / block_size];
        const CT* k_row =
            k_cache + ((long)block * block_size + pos % block_size) * kv_row_stride +
            (long)kv_head * HEAD_DIM;
#pragma unroll
        for (int ks = 0; ks < KCH; ++ks)
          ka[ks] = load_kv8<CT>(k_row + ks * 16 + half * 8);
      } else {
#pragma unroll
        for (int ks = 0; ks < KCH; ++ks) ka[ks] = bf16x8_t{};
      }
    }
    bf16x8_t vv[VPASS];
    {
      constexpr int LPR = HEAD_DIM / 8;        // lanes per row
      constexpr int ROWS_PER_PASS = 64 / LPR;  // rows per pass per wave
      const int r_in_pass = lane / LPR;
      const int d8 = (lane % LPR) * 8;
#pragma unroll
      for (int pass = 0; pass < VPASS; ++pass) {
        const int s = pass * ROWS_PER_PASS + r_in_pass;
        const int pos = kv_base + s;
        vv[pass] = bf16x8_t{};
        if (pos < seq_len) {
          const int block = btable[pos / block_size];
          const long row = ((long)block * block_size + pos % block_size) *
                               kv_row_stride +
                           (long)kv_head * HEAD_DIM + d8;
          vv[pass] = load_kv8<CT>(v_cache + row);
        }
      }
#pragma unroll
      for (int pass = 0; pass < VPASS; ++pass) {
        const int s = pass * ROWS_PER_PASS + r_in_pass;
        *reinterpret_cast<bf16x8_t*>(v_lds + s * HEAD_DIM + d8) = vv[pass];
      }
    }

    // ---- S^T = K · Q^T ---------------------------------------------------
    f32x16_t acc_s{};
#pragma unroll
    for (int ks = 0; ks < KCH; ++ks)
      acc_s = __builtin_amdgcn_mfma_f32_32x32x16_bf16(ka[ks], qb[ks], acc_s, 0, 0, 0);

    // ---- online softmax (stats per q col = this lane's col) --------------
    float s_val[16];
    float local_max = -FLT_MAX;
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int kv_pos = kv_base + (r & 3) + 8 * (r >> 2) + 4 * half;
      float sv = acc_s[r] * scale;
      if (kv_pos >= seq_len) sv = -FLT_MAX;
      s_val[r] = sv;
      local_max = fmaxf(local_max, sv);
    }
    const float tile_max = fmaxf(local_max, __shfl_xor(local_max, 32, 64));
    const float m_new = fmaxf(m_state, tile_max);
    const float rescale = __expf(m_state - m_new);
    m_state = m_new;

    float p[16];
    float local_sum = 0.f;
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      p[r] = s_val[r] == -FLT_MAX ? 0.f : __expf(s_val[r] - m_new);
      local_sum += p[r];
    }
    l_state = l_state * rescale + local_sum + __shfl_xor(local_sum, 32, 64);

    // rescale factors redistributed to the PV accumulator's q rows
    float f_reg[16];
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int qrow = (r & 3) + 8 * (r >> 2) + 4 * half;
      f_reg[r] = __shfl(rescale, qrow, 64);
    }
#pragma unroll
    for (int dt = 0; dt < DT; ++dt) {
#pragma unroll
      for (int r = 0; r < 16; ++r) acc_o[dt][r] *= f_reg[r];
    }

    // ---- pack P to bf16 A-fragments (cvt_pk + permlane32_swap) -----------
    unsigned int pk[8];
#pragma unroll
    for (int i = 0; i < 8; ++i) {
      asm volatile("v_cvt_pk_bf16_f32 %0, %1, %2"
                   : "=v"(pk[i]) : "v"(p[2 * i]), "v"(p[2 * i + 1]));
    }
    bf16x8_t pa[2];
#pragma unroll
    for (int c = 0; c < 2; ++c) {
      unsigned x0 = pk[4 * c + 0], x1 = pk[4 * c + 1];
      unsigned y0 = pk[4 * c + 2], y1 = pk[4 * c + 3];
      {
        auto r0 = __builtin_amdgcn_permlane32_swap(x0, y0, false, false);
        x0 = r0[0]; y0 = r0[1];
        auto r1 = __builtin_amdgcn_permlane32_swap(x1, y1, false, false);
        x1 = r1[0]; y1 = r1[1];
      }
      union { unsigned u[4]; bf16x8_t v; } u;
      u.u[0] = x0; u.u[1] = x1; u.u[2] = y0; u.u[3] = y1;
      pa[c] = u.v;
    }

    // ---- PV: out[q][d] += P[q][kv] V[kv][d] ------------------------------
#pragma unroll
    for (int dt = 0; dt < DT; ++dt) {
#pragma unroll
      for (int c = 0; c < 2; ++c) {
        bf16x8_t b;
#pragma unroll
        for (int j = 0; j < 8; ++j)
          b[j] = *reinterpret_cast<const short*>(
              v_lds + (c * 16 + half * 8 + j) * HEAD_DIM + dt * 32 + col);
        acc_o[dt] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(pa[c], b, acc_o[dt], 0, 0, 0);
      }
    }
  }

  // ---- per-wave partial store: (m, l) per q col + f32 accumulator --------
  const int P = npart * NWAVES;
  const int pw = part * NWAVES + wave;
  if (lane < group) {
    float* ml = part_ml + (((long)seq * nheads + head0 + lane) * P + pw) * 2;
    ml[0] = m_state;
    ml[1] = l_state;
  }
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    const int qrow = (r & 3) + 8 * (r >> 2) + 4 * half;
    if (qrow >= group) continue;
    float* acc_row =
        part_acc + (((long)seq * nheads + head0 + qrow) * P + pw) * HEAD_DIM;
#pragma unroll
    for (int dt = 0; dt < DT; ++dt) acc_row[dt * 32 + col] = acc_o[dt][r];
  }
}

// Reduce the P partials of each (seq, head) row: out = sum_p f_p acc_p / sum_p f_p l_p
template <int HEAD_DIM>
__global__ void decode_merge_kernel(
    __hip_bfloat16* __restrict__ out,      // [nseq, nheads, HD]
    const float* __restrict__ part_acc,    // [nseq, nheads, P, HD]
    const float* __restrict__ part_ml,     // [nseq, nheads, P, 2]
    const int nheads,
    const int P) {
  const int seq = blockIdx.x;
  const int head = blockIdx.y;
  const int d = threadIdx.x;
  const long row = (long)seq * nheads + head;
  const float* ml = part_ml + row * P * 2;
  const float* acc = part_acc + row * P * HEAD_DIM + d;

  float M = -FLT_MAX;
  for (int p = 0; p < P; ++p) M = fmaxf(M, ml[2 * p]);
  float L = 0.f, A = 0.f;
  for (int p = 0; p < P; ++p) {
    const float f = __expf(ml[2 * p] - M);
    L += f * ml[2 * p + 1];
    A += f * acc[(long)p * HEAD_DIM];
  }
  const float inv = L > 0.f ? 1.f / L : 0.f;
  out[row * HEAD_DIM + d] = __float2bfloat16(A * inv);
}

int decode_mfma_num_partitions(int nseq, int kvh, int max_context) {
  // pure function of shapes (hipGraph-stable): fill >=512 workgroups, but
  // keep >=2 tiles per partition at the allocated max context
  int npart = 512 / (nseq * kvh > 0 ? nseq * kvh : 1);
  if (npart < 1) npart = 1;
  if (npart > 32) npart = 32;
  const int max_tiles = (max_context + KVT - 1) / KVT;
  const int cap = max_tiles / 2 > 0 ? max_tiles / 2 : 1;
  if (npart > cap) npart = cap;
  return npart;
}

template <typename CT>
static void launch_decode_ct(__hip_bfloat16* out, float* part_acc,
                             float* part_ml, const __hip_bfloat16* q,
                             const CT* kc, const CT* vc, const int* bt,
                             const int* sl, float scale, int nseq, int nheads,
                             int kvh, int head_dim, int block_size,
                             int max_blocks, int npart, hipStream_t stream) {
  const int group = nheads / kvh;
  dim3 grid(nseq, kvh, npart);
  dim3 block(256);
  dim3 mgrid(nseq, nheads);
  const int P = npart * NWAVES;
  switch (head_dim) {
    case 64:
      hipLaunchKernelGGL((paged_decode_mfma_kernel<64, CT>), grid, block, 0,
                         stream, part_acc, part_ml, q, kc, vc, bt, sl, scale,
                         nheads, kvh, group, block_size, max_blocks);
      hipLaunchKernelGGL(decode_merge_kernel<64>, mgrid, dim3(64), 0, stream,
                         out, part_acc, part_ml, nheads, P);
      break;
    case 128:
      hipLaunchKernelGGL((paged_decode_mfma_kernel<128, CT>), grid, block, 0,
                         stream, part_acc, part_ml, q, kc, vc, bt, sl, scale,
                         nheads, kvh, group, block_size, max_blocks);
      hipLaunchKernelGGL(decode_merge_kernel<128>, mgrid, dim3(128), 0, stream,
                         out, part_acc, part_ml, nheads, P);
      break;
    default:
      abort();
  }
}

void launch_paged_decode_mfma(__hip_bfloat16* out, float* part_acc,
                              float* part_ml, const __hip_bfloat16* q,
                              const __hip_bfloat16* kc, const __hip_bfloat16* vc,
                              const int* bt, const int* sl, float scale,
                              int nseq, int nheads, int kvh, int head_dim,
                              int block_size, int max_blocks, int npart,
                              hipStream_t stream) {
  launch_decode_ct<__hip_bfloat16>(out, part_acc, part_ml, q, kc, vc, bt, sl,
                                   scale, nseq, nheads, kvh, head_dim,
                                   block_size, max_blocks, npart, stream);
}

void launch_paged_decode_mfma_fp8(
    __hip_bfloat16* out, float* part_acc, float* part_ml,
    const __hip_bfloat16* q, const unsigned char* kc, const unsigned char* vc,
    const int* bt, const int* sl, float scale, int nseq, int nheads, int kvh,
    int head_dim, int block_size, int max_blocks, int npart,
    hipStream_t stream) {
  launch_decode_ct<unsigned char>(out, part_acc, part_ml, q, kc, vc, bt, sl,
                                  scale, nseq, nheads, kvh, head_dim,
                                  block_size, max_blocks, npart, stream);
}

// MFMA prefill attention: flash-style causal attention over
// the paged KV cache on v_mfma_f32_32x32x16_bf16 tiles.
//
// Geometry: one workgroup (4 waves) per (seq, head, 128-row query tile);
// each wave owns a 32-row query sub-tile.  K and V tiles (32 slots x
// HEAD_DIM) are staged cooperatively into LDS once per workgroup and read by
// all 4 waves.
//
// Swapped-operand trick (guide §B attn): the score MFMA computes
// S^T = K · Q^T, so each lane's C column is one query row and the row-wise
// softmax statistics (m, l) reduce with one local max + one shfl_xor(32).
// P is packed to bf16 with v_cvt_pk_bf16_f32 and redistributed with
// v_permlane32_swap so it feeds the P·V MFMA's A operand directly (T12/T21).
//
// LDS: the K tile is XOR-swizzled (byte ^= (row&7)<<4, guide G4) because its
// A-fragment reads walk 32 distinct rows at one column slice; the V tile
// stays linear because its B-fragment gather reads one row across lanes.

#include "common.h"
#include <float.h>

typedef __attribute__((ext_vector_type(8))) short bf16x8_t;
typedef __attribute__((ext_vector_type(16))) float f32x16_t;

#define KVT 32  // kv slots per tile

template <int HEAD_DIM>
__global__ __launch_bounds__(256, 2) void paged_prefill_mfma_kernel(
    __hip_bfloat16* __restrict__ out,            // [total_q, nheads, HD]
    const __hip_bfloat16* __restrict__ q,        // [total_q, nheads, HD]
    const __hip_bfloat16* __restrict__ k_cache,  // [nb, bs, kvh, HD]
    const __hip_bfloat16* __restrict__ v_cache,
    const int* __restrict__ block_tables,     // [nseq, max_blocks]
    const int* __restrict__ query_start_loc,  // [nseq+1]
    const int* __restrict__ seq_lens,         // [nseq]
    const float scale,
    const int nheads,
    const int kvh,
    const int block_size,
    const int max_blocks) {
  constexpr int KCH = HEAD_DIM / 16;  // k-chunks for the QK^T mfma
  constexpr int DT = HEAD_DIM / 32;   // 32-col output tiles

  const int seq = blockIdx.x;
  const int head = blockIdx.y;
  const int q_tile = blockIdx.z;
  const int kv_head = head / (nheads / kvh);

  const int q_start = query_start_loc[seq];
  const int q_len = query_start_loc[seq + 1] - q_start;
  const int seq_len = seq_lens[seq];
  const int tile_base = q_tile * 128;
  if (tile_base >= q_len) return;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int col = lane & 31;   // q column / d column of MFMA fragments
  const int half = lane >> 5;  // 0 or 1

  const int* btable = block_tables + (long)seq * max_blocks;
  const long kv_row_stride = (long)kvh * HEAD_DIM;

  // ---- LDS tiles, double-buffered: tile t+1 streams in while the mfma
  // phase reads tile t, so the per-tile HBM round trip is overlapped and
  // only one __syncthreads per tile remains.
  // K rows padded to HEAD_DIM+8 elems (272 B at hd=128): row base banks then
  // stride 4 per row (gcd 16 with 64 banks) -> A-fragment reads are 2-way
  // conflicted instead of the 4-way XOR classes, and staging writes stay
  // contiguous b128 (PMC r1: SQ_LDS_BANK_CONFLICT ~34% of LDS cycles).
  __shared__ __hip_bfloat16 k_lds2[2][KVT * (HEAD_DIM + 8)];
  __shared__ __hip_bfloat16 v_lds2[2][KVT * HEAD_DIM];  // linear

  // ---- load this wave's Q sub-tile as B fragments ----------------------
  // B[k][q]: lane holds Q[q=col][ks*16 + half*8 + j]
  const int q_row_local = tile_base + wave * 32 + col;
  const bool q_valid = q_row_local < q_len;
  const int q_pos = seq_len - q_len + q_row_local;  // kv position of the row
  bf16x8_t qb[KCH];
  if (q_valid) {
    const __hip_bfloat16* q_row =
        q + ((long)(q_start + q_row_local) * nheads + head) * HEAD_DIM;
#pragma unroll
    for (int ks = 0; ks < KCH; ++ks)
      qb[ks] = *reinterpret_cast<const bf16x8_t*>(q_row + ks * 16 + half * 8);
  } else {
#pragma unroll
    for (int ks = 0; ks < KCH; ++ks) qb[ks] = bf16x8_t{};
  }

  float m_state = -FLT_MAX;  // per q row (duplicated on lane and lane+32)
  float l_state = 0.f;
  f32x16_t acc_o[DT];
#pragma unroll
  for (int dt = 0; dt < DT; ++dt) acc_o[dt] = f32x16_t{};

  // row index of C/D register r (q row for PV output, kv row for scores)
  // row(r) = (r&3) + 8*(r>>2) + 4*half

  // causal bound for the whole workgroup
  const int wg_rows = min(128, q_len - tile_base);
  const int wg_max_pos = seq_len - q_len + tile_base + wg_rows - 1;
  const int kv_limit = min(seq_len, wg_max_pos + 1);

  constexpr int LPR = HEAD_DIM / 8;       // staging lanes per row (16B each)
  constexpr int ROWS_PER_PASS = 256 / LPR;
  constexpr int NPASS = KVT / ROWS_PER_PASS;
  const int r_in_pass = tid / LPR;
  const int d8 = (tid % LPR) * 8;

  // stage tile (kv_base) into LDS buffer `buf`: phase 0 = issue loads into
  // regs, phase 1 = LDS writes (so all of a tile's HBM loads are in flight
  // together)
  bf16x8_t st_k[NPASS], st_v[NPASS];
  auto stage_load = [&](int kv_base) {
#pragma unroll
    for (int pass = 0; pass < NPASS; ++pass) {
      const int s = pass * ROWS_PER_PASS + r_in_pass;
      const int pos = kv_base + s;
      st_k[pass] = bf16x8_t{};
      st_v[pass] = bf16x8_t{};
      if (pos < kv_limit) {
        const int block = btable[pos / block_size];
        const long row = ((long)block * block_size + pos % block_size) *
                             kv_row_stride +
                         (long)kv_head * HEAD_DIM + d8;
        st_k[pass] = *reinterpret_cast<const bf16x8_t*>(k_cache + row);
        st_v[pass] = *reinterpret_cast<const bf16x8_t*>(v_cache + row);
      }
    }
  };
  auto stage_write = [&](int buf) {
#pragma unroll
    for (int pass = 0; pass < NPASS; ++pass) {
      const int s = pass * ROWS_PER_PASS + r_in_pass;
      const int k_byte = (s * (HEAD_DIM + 8) + d8) * 2;
      *reinterpret_cast<bf16x8_t*>(
          reinterpret_cast<char*>(k_lds2[buf]) + k_byte) = st_k[pass];
      *reinterpret_cast<bf16x8_t*>(v_lds2[buf] + s * HEAD_DIM + d8) = st_v[pass];
    }
  };

  // T14 pipeline: write tile t from regs (loaded a full iteration ago),
  // re-issue loads for t+1, barrier, compute t.
  stage_load(0);

  for (int kv_base = 0; kv_base < kv_limit; kv_base += KVT) {
    const int buf = (kv_base / KVT) & 1;
    const __hip_bfloat16* k_lds = k_lds2[buf];
    const __hip_bfloat16* v_lds = v_lds2[buf];
    stage_write(buf);
    const int next_base = kv_base + KVT;
    if (next_base < kv_limit) stage_load(next_base);
    __syncthreads();  // tile (kv_base) fully staged for all waves

    // ---- S^T = K · Q^T -------------------------------------------------
    f32x16_t acc_s{};
#pragma unroll
    for (int ks = 0; ks < KCH; ++ks) {
      // A[kv=col][k = ks*16 + half*8 + j] from the padded K tile
      const int byte = (col * (HEAD_DIM + 8) + ks * 16 + half * 8) * 2;
      bf16x8_t a = *reinterpret_cast<const bf16x8_t*>(
          reinterpret_cast<const char*>(k_lds) + byte);
      acc_s = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, qb[ks], acc_s, 0, 0, 0);
    }

    // ---- online softmax (stats per q row = this lane's col) --------------
    // Fast path for fully-visible tiles (strictly below the causal diagonal
    // of every q row of this wave AND inside seq_len — the common case for
    // long prompts): no per-element mask math.  PMC r1: the masked loops
    // were ~1/3 of the kernel's VALU issue.
    const int wave_q_pos_min = seq_len - q_len + tile_base + wave * 32;
    // wave-uniform: the q_len term implies every lane's q row is valid
    const bool full_tile =
        (kv_base + KVT - 1 <= wave_q_pos_min) &&
        (kv_base + KVT <= seq_len) && (tile_base + wave * 32 + 31 < q_len);
    float s_val[16];
    float local_max = -FLT_MAX;
    if (full_tile) {
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        s_val[r] = acc_s[r] * scale;
        local_max = fmaxf(local_max, s_val[r]);
      }
    } else {
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int kv_local = (r & 3) + 8 * (r >> 2) + 4 * half;
        const int kv_pos = kv_base + kv_local;
        float sv = acc_s[r] * scale;
        if (!q_valid || kv_pos > q_pos || kv_pos >= seq_len) sv = -FLT_MAX;
        s_val[r] = sv;
        local_max = fmaxf(local_max, sv);
      }
    }
    const float tile_max = fmaxf(local_max, __shfl_xor(local_max, 32, 64));
    const float m_new = fmaxf(m_state, tile_max);
    const float rescale = __expf(m_state - m_new);
    m_state = m_new;

    float p[16];
    float local_sum = 0.f;
    if (full_tile) {
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        p[r] = __expf(s_val[r] - m_new);
        local_sum += p[r];
      }
    } else {
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        p[r] = s_val[r] == -FLT_MAX ? 0.f : __expf(s_val[r] - m_new);
        local_sum += p[r];
      }
    }
    l_state = l_state * rescale + local_sum + __shfl_xor(local_sum, 32, 64);

    // rescale factors redistributed to the PV accumulator's q rows: the
    // factor for q row qr lives identically on lanes qr and qr+32
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

    // ---- pack P to bf16 A-fragments (cvt_pk + permlane32_swap) ---------
    // reg r -> kv_local (r&3)+8*(r>>2)+4*half.  Chunk ks2=0 covers kv 0..15,
    // ks2=1 covers 16..31.
    unsigned int pk[8];
#pragma unroll
    for (int i = 0; i < 8; ++i) {
      asm volatile("v_cvt_pk_bf16_f32 %0, %1, %2"
                   : "=v"(pk[i]) : "v"(p[2 * i]), "v"(p[2 * i + 1]));
    }
    // pk[0..1] = kv(0..3)+4h, pk[2..3] = kv(8..11)+4h,
    // pk[4..5] = kv(16..19)+4h, pk[6..7] = kv(24..27)+4h
    bf16x8_t pa[2];
#pragma unroll
    for (int c = 0; c < 2; ++c) {  // c=0 -> kv 0..15, c=1 -> kv 16..31
      unsigned x0 = pk[4 * c + 0], x1 = pk[4 * c + 1];
      unsigned y0 = pk[4 * c + 2], y1 = pk[4 * c + 3];
      {
        auto r0 = __builtin_amdgcn_permlane32_swap(x0, y0, false, false);
        x0 = r0[0]; y0 = r0[1];
        auto r1 = __builtin_amdgcn_permlane32_swap(x1, y1, false, false);
        x1 = r1[0]; y1 = r1[1];
      }
      // lanes<32: [x0 x1 | y0 y1] = kv c*16 + (0..7)
      // lanes>=32: [x0 x1 | y0 y1] = kv c*16 + (8..15)
      union { unsigned u[4]; bf16x8_t v; } u;
      u.u[0] = x0; u.u[1] = x1; u.u[2] = y0; u.u[3] = y1;
      pa[c] = u.v;
    }

    // ---- PV: out[q][d] += P[q][kv] V[kv][d] ----------------------------
#pragma unroll
    for (int dt = 0; dt < DT; ++dt) {
#pragma unroll
      for (int c = 0; c < 2; ++c) {
        // B[k=kv][d=col]: lane gathers V[c*16 + half*8 + j][dt*32 + col]
        bf16x8_t b;
#pragma unroll
        for (int j = 0; j < 8; ++j)
          b[j] = *reinterpret_cast<const short*>(
              v_lds + (c * 16 + half * 8 + j) * HEAD_DIM + dt * 32 + col);
        acc_o[dt] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(pa[c], b, acc_o[dt], 0, 0, 0);
      }
    }

  }

  // ---- epilogue: normalize by l (per output q row) and store -----------
  const float l_inv_own = l_state > 0.f ? 1.f / l_state : 0.f;
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    const int qrow = (r & 3) + 8 * (r >> 2) + 4 * half;
    const float l_inv = __shfl(l_inv_own, qrow, 64);
    const int q_local = tile_base + wave * 32 + qrow;
    if (q_local >= q_len) continue;
    __hip_bfloat16* out_row =
        out + ((long)(q_start + q_local) * nheads + head) * HEAD_DIM;
#pragma unroll
    for (int dt = 0; dt < DT; ++dt)
      out_row[dt * 32 + col] = __float2bfloat16(acc_o[dt][r] * l_inv);
  }
}

void launch_paged_prefill_mfma(
    __hip_bfloat16* out, const __hip_bfloat16* q, const __hip_bfloat16* kc,
    const __hip_bfloat16* vc, const int* bt, const int* qsl, const int* sl,
    float scale, int nseq, int nheads, int kvh, int head_dim, int block_size,
    int max_blocks, int max_query_len, hipStream_t stream) {
  const int qtiles = (max_query_len + 127) / 128;
  dim3 grid(nseq, nheads, qtiles);
  dim3 block(256);
  switch (head_dim) {
    case 64:
      hipLaunchKernelGGL(paged_prefill_mfma_kernel<64>, grid, block, 0, stream,
                         out, q, kc, vc, bt, qsl, sl, scale, nheads, kvh,
                         block_size, max_blocks);
      break;
    case 128:
      hipLaunchKernelGGL(paged_prefill_mfma_kernel<128>, grid, block, 0,
                         stream, out, q, kc, vc, bt, qsl, sl, scale, nheads,
                         kvh, block_size, max_blocks);
      break;
    default:
      abort();
  }
}

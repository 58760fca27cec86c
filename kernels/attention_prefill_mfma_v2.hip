// MFMA prefill attention v2: flash-style causal attention over
// the paged KV cache, rebuilt on the CDNA4 ladder that lifts the r1 kernel's
// 11-16% MFU: KVBLK=64 tiles, a zero-conflict XOR-16 K image, hardware
// transpose reads (ds_read_b64_tr_b16) for the V fragments, async staging
// (global loads issued under softmax+PV), defer-max rescaling, and exp2 with
// the softmax scale folded into the logits.
//
// Geometry: one workgroup = 8 waves = one 256-row query tile per (seq, head);
// each wave owns 32 q rows.  K/V tiles of 64 kv slots x HEAD_DIM are staged
// cooperatively, double-buffered, one barrier per tile.
//
// LDS images (per buffer, HD=128):
//   K: [64][128] bf16, byte ^= ((row & 15) << 4)  -> ds_read_b128 A-fragments
//      hit 16 distinct banks per lane group (zero conflict).
//   V: 16 subtiles [32 kv][16 d] row-major, stride 1152 B (1024 + 128 pad so
//      d0-adjacent subtiles sit 32 dword-banks apart) -> each 16-lane quarter
//      tr-reads a [4 kv][16 d] window, conflict-free, B-fragments at b64 rate
//      (the r1 kernel's per-element V gathers were 64 ds_read_u16 per tile).
//
// Swapped-operand score MFMA (S^T = K.Q^T) keeps each lane's P row local:
// row stats need one fmax chain + one shfl_xor(32); P packs to bf16 with
// v_cvt_pk_bf16_f32 + permlane32_swap straight into the PV A-fragments.

#include "common.h"
#include <float.h>

typedef __attribute__((ext_vector_type(8))) short bf16x8_t;
typedef __attribute__((ext_vector_type(16))) float f32x16_t;

template <typename CT>
DEVINLINE bf16x8_t load_kv8_v2(const CT* p);
template <>
DEVINLINE bf16x8_t load_kv8_v2<__hip_bfloat16>(const __hip_bfloat16* p) {
  return *reinterpret_cast<const bf16x8_t*>(p);
}
template <>
DEVINLINE bf16x8_t load_kv8_v2<unsigned char>(const unsigned char* p) {
  return __builtin_bit_cast(
      bf16x8_t, fp8x8_to_bf16x8(*reinterpret_cast<const u32x2_vec_t*>(p)));
}
typedef __attribute__((ext_vector_type(4))) __bf16 bf16x4_b;
typedef __attribute__((address_space(3))) bf16x4_b* lds_tr_ptr;

#define KVT2 64         // kv slots per tile
#define QTILE2 256      // q rows per workgroup (8 waves x 32)
#define VSUB_STRIDE 1152  // bytes per [32][16] V subtile incl. 128-B pad
#define LOG2E 1.44269504f
#define DEFER_MAX_THR 11.5f  // log2-domain threshold (= 8 nats, T13)

// cross-half (lane <-> lane+32) reductions.  permlane32_swap is a LANE-LOCAL
// register-half exchange (tools/probe_swap.hip), not a cross-lane shuffle,
// so the partner value must come from a real shuffle.
__device__ inline float cross_half_max(float v) {
  return fmaxf(v, __shfl_xor(v, 32, 64));
}

__device__ inline float cross_half_sum(float v) {
  return v + __shfl_xor(v, 32, 64);
}

template <int HEAD_DIM, typename CT, int NW>
__global__ __launch_bounds__(NW * 64, 512 / (NW * 64)) void paged_prefill_mfma_v2_kernel(
    __hip_bfloat16* __restrict__ out,            // [total_q, nheads, HD]
    const __hip_bfloat16* __restrict__ q,        // [total_q, nheads, HD]
    const CT* __restrict__ k_cache,  // [nb, bs, kvh, HD] bf16 | e4m3
    const CT* __restrict__ v_cache,
    const int* __restrict__ block_tables,     // [nseq, max_blocks]
    const int* __restrict__ query_start_loc,  // [nseq+1]
    const int* __restrict__ seq_lens,         // [nseq]
    const float scale,
    const int nheads,
    const int kvh,
    const int block_size,
    const int max_blocks) {
  static_assert(HEAD_DIM == 128, "v2 is specialised for HD=128");
  constexpr int KCH = HEAD_DIM / 16;  // QK^T k-chunks (8)
  constexpr int DT = HEAD_DIM / 32;   // 32-col output tiles (4)
  constexpr int QT = NW * 32;         // q rows per workgroup
  constexpr int NTHREADS = NW * 64;
  // staging: each thread covers (s, d8) slots; pass count scales with size
  constexpr int ST_PASS = (KVT2 * (HEAD_DIM / 8)) / NTHREADS;

  const int seq = blockIdx.x;
  const int head = blockIdx.y;
  const int q_tile = blockIdx.z;
  const int kv_head = head / (nheads / kvh);

  const int q_start = query_start_loc[seq];
  const int q_len = query_start_loc[seq + 1] - q_start;
  const int seq_len = seq_lens[seq];
  const int tile_base = q_tile * QT;
  if (tile_base >= q_len) return;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int col = lane & 31;   // q column of the score fragments
  const int half = lane >> 5;  // 0 or 1

  const int* btable = block_tables + (long)seq * max_blocks;
  const long kv_row_stride = (long)kvh * HEAD_DIM;

  // K: [64][256B] XOR-swizzled; V: 16 x 1152B subtiles
  __shared__ __hip_bfloat16 k_lds2[2][KVT2 * HEAD_DIM];
  __shared__ char v_lds2[2][16 * VSUB_STRIDE];

  // ---- per-wave Q sub-tile as B fragments ------------------------------
  const int q_row_local = tile_base + wave * 32 + col;
  const bool q_valid = q_row_local < q_len;
  const int q_pos = seq_len - q_len + q_row_local;
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

  float m2_state = -FLT_MAX;  // running max, log2 domain
  float l_state = 0.f;
  f32x16_t acc_o[DT];
#pragma unroll
  for (int dt = 0; dt < DT; ++dt) acc_o[dt] = f32x16_t{};

  const int wg_rows = min(QT, q_len - tile_base);
  const int wg_max_pos = seq_len - q_len + tile_base + wg_rows - 1;
  const int kv_limit = min(seq_len, wg_max_pos + 1);
  // last kv position any of this wave's rows may attend to
  const int wave_rows = min(32, q_len - tile_base - wave * 32);
  const int wave_max_pos =
      seq_len - q_len + tile_base + wave * 32 + max(wave_rows - 1, 0);
  const int wave_q_pos_min = seq_len - q_len + tile_base + wave * 32;

  // ---- staging map: thread t covers slot (t % ROWS_PER_PASS, t / RPP * 8)
  // with ROWS_PER_PASS = NTHREADS/16 rows per pass (16 8-elem slots per row)
  constexpr int RPP = NTHREADS / 16;
  const int st_s = tid % RPP;
  const int st_d8 = (tid / RPP) * 8;
  bf16x8_t st_k[ST_PASS], st_v[ST_PASS];
  auto stage_load = [&](int kv_base) {
#pragma unroll
    for (int pass = 0; pass < ST_PASS; ++pass) {
      const int s = st_s + pass * RPP;
      const int pos = kv_base + s;
      st_k[pass] = bf16x8_t{};
      st_v[pass] = bf16x8_t{};
      if (pos < kv_limit) {
        const int block = btable[pos / block_size];
        const long row = ((long)block * block_size + pos % block_size) *
                             kv_row_stride +
                         (long)kv_head * HEAD_DIM + st_d8;
        st_k[pass] = load_kv8_v2<CT>(k_cache + row);
        st_v[pass] = load_kv8_v2<CT>(v_cache + row);
      }
    }
  };
  auto stage_write = [&](int buf) {
#pragma unroll
    for (int pass = 0; pass < ST_PASS; ++pass) {
      const int s = st_s + pass * RPP;
      const int k_byte = (s * (HEAD_DIM * 2) + st_d8 * 2) ^ ((s & 15) << 4);
      *reinterpret_cast<bf16x8_t*>(
          reinterpret_cast<char*>(k_lds2[buf]) + k_byte) = st_k[pass];
      const int sub = ((s >> 5) << 3) + (st_d8 >> 4);
      const int v_byte = sub * VSUB_STRIDE + (s & 31) * 32 + (st_d8 & 15) * 2;
      *reinterpret_cast<bf16x8_t*>(v_lds2[buf] + v_byte) = st_v[pass];
    }
  };

  stage_load(0);
  stage_write(0);

  const float sc2 = scale * LOG2E;  // fold softmax scale into exp2 domain

  for (int kv_base = 0; kv_base < kv_limit; kv_base += KVT2) {
    const int buf = (kv_base / KVT2) & 1;
    __syncthreads();  // publish tile (kv_base) to all waves

    // wave-uniform: MFMAs must run with a full EXEC mask — a per-lane guard
    // here drops the masked lanes' K fragments from the score MFMA
    const bool wave_active =
        (tile_base + wave * 32 < q_len) && (kv_base <= wave_max_pos);
    const char* k_lds = reinterpret_cast<const char*>(k_lds2[buf]);
    const char* v_lds = v_lds2[buf];

    f32x16_t acc_s[2];
    if (wave_active) {
      // ---- S^T = K . Q^T per 32-kv block -------------------------------
#pragma unroll
      for (int kb = 0; kb < 2; ++kb) {
        acc_s[kb] = f32x16_t{};
        const int krow = kb * 32 + col;
#pragma unroll
        for (int ks = 0; ks < KCH; ++ks) {
          const int byte =
              (krow * (HEAD_DIM * 2) + ks * 32 + half * 16) ^ ((krow & 15) << 4);
          bf16x8_t a = *reinterpret_cast<const bf16x8_t*>(k_lds + byte);
          acc_s[kb] =
              __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, qb[ks], acc_s[kb], 0, 0, 0);
        }
      }
    }

    // ---- async stage: issue next tile's HBM loads now ------------------
    const int next_base = kv_base + KVT2;
    if (next_base < kv_limit) stage_load(next_base);

    if (wave_active) {
      // ---- online softmax; the running max m2_state lives in the RAW
      // score domain (sc2 > 0 commutes with max), so the P pass is one
      // fma + exp2 per element: p = exp2(acc*sc2 - m*sc2).
      // fast path: tile strictly below every row's causal diagonal
      const bool full_tile = (kv_base + KVT2 - 1 <= wave_q_pos_min) &&
                             (kv_base + KVT2 <= seq_len) &&
                             (tile_base + wave * 32 + 31 < q_len);
      float local_max = -FLT_MAX;
      if (full_tile) {
#pragma unroll
        for (int kb = 0; kb < 2; ++kb)
#pragma unroll
          for (int r = 0; r < 16; ++r)
            local_max = fmaxf(local_max, acc_s[kb][r]);
      } else {
#pragma unroll
        for (int kb = 0; kb < 2; ++kb)
#pragma unroll
          for (int r = 0; r < 16; ++r) {
            const int kv_pos =
                kv_base + kb * 32 + (r & 3) + 8 * (r >> 2) + 4 * half;
            float v = acc_s[kb][r];
            if (!q_valid || kv_pos > q_pos || kv_pos >= seq_len) v = -FLT_MAX;
            acc_s[kb][r] = v;
            local_max = fmaxf(local_max, v);
          }
      }
      const float tile_max = cross_half_max(local_max);

      // defer-max (T13): skip the O/l rescale while the running max holds.
      // The decision is wave-uniform and taken BEFORE any of this tile's P
      // enters O or l, so everything at the old scale rescales exactly once.
      const float thr = DEFER_MAX_THR / sc2;
      const bool need_rescale = !__all(
          m2_state != -FLT_MAX && tile_max <= m2_state + thr);
      float m2_new = m2_state;
      if (need_rescale) {
        m2_new = fmaxf(m2_state, tile_max);
        const float rescale = exp2f((m2_state - m2_new) * sc2);
        l_state *= rescale;
        float f_reg[16];
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          const int qrow = (r & 3) + 8 * (r >> 2) + 4 * half;
          f_reg[r] = __shfl(rescale, qrow, 64);
        }
#pragma unroll
        for (int dt = 0; dt < DT; ++dt)
#pragma unroll
          for (int r = 0; r < 16; ++r) acc_o[dt][r] *= f_reg[r];
        m2_state = m2_new;
      }

      // p = exp2(fma(acc, sc2, bias)); masked rows hold -FLT_MAX whose fma
      // lands at -inf -> exp2 gives 0, no select needed
      const float bias = (m2_new == -FLT_MAX ? 0.f : -m2_new) * sc2;
      float p[32];
      float local_sum = 0.f;
#pragma unroll
      for (int kb = 0; kb < 2; ++kb)
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          const float v = __builtin_amdgcn_exp2f(
              __builtin_fmaf(acc_s[kb][r], sc2, bias));
          p[kb * 16 + r] = v;
          local_sum += v;
        }
      l_state += cross_half_sum(local_sum);

      // ---- P -> bf16 A-fragments (cvt_pk + permlane32_swap, T12) -------
      bf16x8_t pa[4];
#pragma unroll
      for (int kb = 0; kb < 2; ++kb) {
        unsigned int pk[8];
#pragma unroll
        for (int i = 0; i < 8; ++i) {
          asm volatile("v_cvt_pk_bf16_f32 %0, %1, %2"
                       : "=v"(pk[i])
                       : "v"(p[kb * 16 + 2 * i]), "v"(p[kb * 16 + 2 * i + 1]));
        }
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
          pa[kb * 2 + c] = u.v;
        }
      }

      // ---- PV via hardware transpose reads, per d0 block ---------------
      // quarter q of the wave covers (d0 parity = q&1, kv +8 per q>>1)
      const int quart = lane >> 4;
#pragma unroll
      for (int dt = 0; dt < DT; ++dt) {
        const int d0 = dt * 2 + (quart & 1);
        bf16x4_b ra[4], rb[4];
#pragma unroll
        for (int c = 0; c < 4; ++c) {
          const int sub = ((c >> 1) << 3) + d0;
          const int r0 = (c & 1) * 16 + (quart >> 1) * 8;
          const char* w =
              v_lds + sub * VSUB_STRIDE + r0 * 32 + (lane & 15) * 8;
          ra[c] = __builtin_amdgcn_ds_read_tr16_b64_v4bf16((lds_tr_ptr)w);
          rb[c] = __builtin_amdgcn_ds_read_tr16_b64_v4bf16((lds_tr_ptr)(w + 128));
        }
#pragma unroll
        for (int c = 0; c < 4; ++c) {
          union { bf16x4_b h[2]; bf16x8_t v; } u;
          u.h[0] = ra[c];
          u.h[1] = rb[c];
          acc_o[dt] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              pa[c], u.v, acc_o[dt], 0, 0, 0);
        }
      }
    }

    // ---- write next tile into the other buffer -------------------------
    if (next_base < kv_limit) stage_write(buf ^ 1);
  }

  // ---- epilogue: normalize and store -----------------------------------
  // NOTE: every lane stores here — in the PV output the lane indexes a d
  // column and carries ALL 32 q rows for it; q-row validity is per row
  // (q_local check below), not per lane.
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

#include <cstdlib>

template <typename CT>
static void launch_prefill_v2_ct(
    __hip_bfloat16* out, const __hip_bfloat16* q, const CT* kc, const CT* vc,
    const int* bt, const int* qsl, const int* sl, float scale, int nseq,
    int nheads, int kvh, int head_dim, int block_size, int max_blocks,
    int max_query_len, hipStream_t stream) {
  if (head_dim != 128) abort();
  static const int nw = [] {
    const char* e = getenv("VTA_PREFILL_WAVES");
    // measured: 8-wave/256-row wins (317/560 TF vs 302/498 for 4-wave —
    // the doubled per-CU staging traffic outweighs the causal-skew saving)
    return (e && e[0] == '4') ? 4 : 8;
  }();
  if (nw == 8) {
    const int qtiles = (max_query_len + 255) / 256;
    hipLaunchKernelGGL((paged_prefill_mfma_v2_kernel<128, CT, 8>),
                       dim3(nseq, nheads, qtiles), dim3(512), 0, stream, out,
                       q, kc, vc, bt, qsl, sl, scale, nheads, kvh, block_size,
                       max_blocks);
  } else {
    const int qtiles = (max_query_len + 127) / 128;
    hipLaunchKernelGGL((paged_prefill_mfma_v2_kernel<128, CT, 4>),
                       dim3(nseq, nheads, qtiles), dim3(256), 0, stream, out,
                       q, kc, vc, bt, qsl, sl, scale, nheads, kvh, block_size,
                       max_blocks);
  }
}

void launch_paged_prefill_mfma_v2(
    __hip_bfloat16* out, const __hip_bfloat16* q, const __hip_bfloat16* kc,
    const __hip_bfloat16* vc, const int* bt, const int* qsl, const int* sl,
    float scale, int nseq, int nheads, int kvh, int head_dim, int block_size,
    int max_blocks, int max_query_len, hipStream_t stream) {
  launch_prefill_v2_ct<__hip_bfloat16>(out, q, kc, vc, bt, qsl, sl, scale,
                                       nseq, nheads, kvh, head_dim,
                                       block_size, max_blocks, max_query_len,
                                       stream);
}

void launch_paged_prefill_mfma_v2_fp8(
    __hip_bfloat16* out, const __hip_bfloat16* q, const unsigned char* kc,
    const unsigned char* vc, const int* bt, const int* qsl, const int* sl,
    float scale, int nseq, int nheads, int kvh, int head_dim, int block_size,
    int max_blocks, int max_query_len, hipStream_t stream) {
  launch_prefill_v2_ct<unsigned char>(out, q, kc, vc, bt, qsl, sl, scale,
                                      nseq, nheads, kvh, head_dim, block_size,
                                      max_blocks, max_query_len, stream);
}

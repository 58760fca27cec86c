// Paged-KV attention kernels.
//
// Cache layout: [num_blocks, block_size, kv_heads, head_dim]; a (slot, head)
// row is head_dim contiguous elements (256 B at hd=128 bf16).
//
// Decode (one query token per sequence) is HBM-bound: the whole KV context is
// streamed once.  Geometry: one workgroup of 4 waves per (seq, kv_head); all
// GROUP query heads of the kv head are computed together so K/V are read
// once per GROUP heads.  Within a wave, lanes are split into slot-groups of
// LPS = head_dim/8 lanes; each lane reads 16 B (8 elements) of a K/V row, so
// a wave streams 64*16 B = 1 KiB per instruction fully coalesced.  Online
// softmax state (m, l) is wave-uniform; partial accumulators merge across
// slot-groups by shfl and across waves through LDS.
//
// Prefill (chunked, causal, queries are the tail of the sequence) reuses the
// same slot-group structure with a 4-wave workgroup per (seq, head, 16-row
// query tile); each wave owns 4 query rows.  This version is VALU dot-product
// based — correctness-first; the MFMA tile version replaces it (tracked in
// kernels/README).

#include "common.h"
#include <float.h>

// 8-element per-lane fragment: one 16-B load for 16-bit dtypes, two for f32
template <typename T>
struct alignas(16) Elems8 {
  T data[8];
};

template <typename T>
DEVINLINE Elems8<T> load8(const T* p) {
  return *reinterpret_cast<const Elems8<T>*>(p);
}

template <typename T>
DEVINLINE void store8(T* p, const Elems8<T>& v) {
  *reinterpret_cast<Elems8<T>*>(p) = v;
}

// dot of 8-element slices held per lane
template <typename T>
DEVINLINE float dot8(const float* qf, const Elems8<T>& kv) {
  float acc = 0.f;
#pragma unroll
  for (int j = 0; j < 8; ++j) acc += qf[j] * ScalarOps<T>::to_f32(kv.data[j]);
  return acc;
}

// ---------------------------------------------------------------------------
// decode
// ---------------------------------------------------------------------------

template <typename T, int HEAD_DIM, int GROUP>
__global__ __launch_bounds__(256) void paged_decode_kernel(
    T* __restrict__ out,            // [nseq, nheads, HEAD_DIM]
    const T* __restrict__ q,        // [nseq, nheads, HEAD_DIM]
    const T* __restrict__ k_cache,  // [nblocks, bs, kvh, HEAD_DIM]
    const T* __restrict__ v_cache,
    const int* __restrict__ block_tables,  // [nseq, max_blocks]
    const int* __restrict__ seq_lens,      // [nseq]
    const float scale,
    const int kvh,
    const int block_size,
    const int max_blocks) {
  constexpr int LPS = HEAD_DIM / 8;   // lanes per slot
  constexpr int SPW = WAVE_SIZE / LPS;  // slots per wave per iteration
  constexpr int NWAVES = 4;

  const int seq = blockIdx.x;
  const int kv_head = blockIdx.y;
  const int seq_len = seq_lens[seq];
  const int lane = threadIdx.x & (WAVE_SIZE - 1);
  const int wave = threadIdx.x / WAVE_SIZE;
  const int slot_in_wave = lane / LPS;
  const int dim_off = (lane % LPS) * 8;
  const int nheads = kvh * GROUP;
  const int* btable = block_tables + (long)seq * max_blocks;

  // q fragments: 8 f32 per lane per grouped head
  float qf[GROUP][8];
#pragma unroll
  for (int g = 0; g < GROUP; ++g) {
    const T* q_row = q + ((long)seq * nheads + kv_head * GROUP + g) * HEAD_DIM;
    Elems8<T> v = load8(q_row + dim_off);
#pragma unroll
    for (int j = 0; j < 8; ++j) qf[g][j] = ScalarOps<T>::to_f32(v.data[j]);
  }

  float m[GROUP], l[GROUP], acc[GROUP][8];
#pragma unroll
  for (int g = 0; g < GROUP; ++g) {
    m[g] = -FLT_MAX;
    l[g] = 0.f;
#pragma unroll
    for (int j = 0; j < 8; ++j) acc[g][j] = 0.f;
  }

  const int slots_per_iter = NWAVES * SPW;
  const long row_stride = (long)kvh * HEAD_DIM;

  for (int base = 0; base < seq_len; base += slots_per_iter) {
    const int pos = base + wave * SPW + slot_in_wave;
    const bool valid = pos < seq_len;
    Elems8<T> k8, v8;
    long row = 0;
    if (valid) {
      const int block = btable[pos / block_size];
      row = ((long)block * block_size + pos % block_size) * row_stride +
            (long)kv_head * HEAD_DIM + dim_off;
      k8 = load8(k_cache + row);
      v8 = load8(v_cache + row);
    }
    float p[GROUP];
#pragma unroll
    for (int g = 0; g < GROUP; ++g) {
      float qk = valid ? dot8(qf[g], k8) : 0.f;
      qk = group_reduce_sum<LPS>(qk) * scale;
      if (!valid) qk = -FLT_MAX;
      // wave-wide max (qk uniform within each slot-group)
      const float m_cand = wave_reduce_max(qk);
      const float m_new = fmaxf(m[g], m_cand);
      const float rescale = __expf(m[g] - m_new);
      p[g] = valid ? __expf(qk - m_new) : 0.f;
      // sum p over the wave's distinct slots (count each group once)
      const float contrib = (lane % LPS == 0) ? p[g] : 0.f;
      l[g] = l[g] * rescale + wave_reduce_sum(contrib);
      m[g] = m_new;
#pragma unroll
      for (int j = 0; j < 8; ++j) acc[g][j] = acc[g][j] * rescale;
    }
#pragma unroll
    for (int g = 0; g < GROUP; ++g) {
      if (valid) {
#pragma unroll
        for (int j = 0; j < 8; ++j)
          acc[g][j] += p[g] * ScalarOps<T>::to_f32(v8.data[j]);
      }
    }
  }

  // fold the wave's slot-group accumulator copies together (they share m/l)
#pragma unroll
  for (int g = 0; g < GROUP; ++g) {
#pragma unroll
    for (int j = 0; j < 8; ++j) {
#pragma unroll
      for (int off = 32; off >= LPS; off >>= 1)
        acc[g][j] += __shfl_xor(acc[g][j], off, WAVE_SIZE);
    }
  }

  // cross-wave merge through LDS
  __shared__ float lds_acc[NWAVES][GROUP][HEAD_DIM];
  __shared__ float lds_ml[NWAVES][GROUP][2];
  if (lane < LPS) {
#pragma unroll
    for (int g = 0; g < GROUP; ++g) {
#pragma unroll
      for (int j = 0; j < 8; ++j) lds_acc[wave][g][dim_off + j] = acc[g][j];
    }
  }
  if (lane == 0) {
#pragma unroll
    for (int g = 0; g < GROUP; ++g) {
      lds_ml[wave][g][0] = m[g];
      lds_ml[wave][g][1] = l[g];
    }
  }
  __syncthreads();

  // all 256 threads cooperate on the final merge over (g, d)
  for (int idx = threadIdx.x; idx < GROUP * HEAD_DIM; idx += blockDim.x) {
    const int g = idx / HEAD_DIM;
    const int d = idx % HEAD_DIM;
    float M = -FLT_MAX;
#pragma unroll
    for (int w = 0; w < NWAVES; ++w) M = fmaxf(M, lds_ml[w][g][0]);
    float L = 0.f, A = 0.f;
#pragma unroll
    for (int w = 0; w < NWAVES; ++w) {
      const float f = __expf(lds_ml[w][g][0] - M);
      L += lds_ml[w][g][1] * f;
      A += lds_acc[w][g][d] * f;
    }
    const float inv_l = L > 0.f ? 1.f / L : 0.f;
    out[((long)seq * nheads + kv_head * GROUP + g) * HEAD_DIM + d] =
        ScalarOps<T>::from_f32(A * inv_l);
  }
}

// ---------------------------------------------------------------------------
// prefill (causal, queries = tail of the sequence)
// ---------------------------------------------------------------------------

template <typename T, int HEAD_DIM>
__global__ __launch_bounds__(256) void paged_prefill_kernel(
    T* __restrict__ out,            // [total_q, nheads, HEAD_DIM]
    const T* __restrict__ q,        // [total_q, nheads, HEAD_DIM]
    const T* __restrict__ k_cache,  // [nblocks, bs, kvh, HEAD_DIM]
    const T* __restrict__ v_cache,
    const int* __restrict__ block_tables,     // [nseq, max_blocks]
    const int* __restrict__ query_start_loc,  // [nseq+1]
    const int* __restrict__ seq_lens,         // [nseq]
    const float scale,
    const int nheads,
    const int kvh,
    const int block_size,
    const int max_blocks) {
  constexpr int LPS = HEAD_DIM / 8;
  constexpr int SPW = WAVE_SIZE / LPS;
  constexpr int NWAVES = 4;
  constexpr int QR = 4;  // query rows per wave
  constexpr int QTILE = NWAVES * QR;  // per workgroup

  const int seq = blockIdx.x;
  const int head = blockIdx.y;
  const int q_tile = blockIdx.z;
  const int kv_head = head / (nheads / kvh);

  const int q_start = query_start_loc[seq];
  const int q_len = query_start_loc[seq + 1] - q_start;
  const int seq_len = seq_lens[seq];
  const int tile_base = q_tile * QTILE;
  if (tile_base >= q_len) return;

  const int lane = threadIdx.x & (WAVE_SIZE - 1);
  const int wave = threadIdx.x / WAVE_SIZE;
  const int slot_in_wave = lane / LPS;
  const int dim_off = (lane % LPS) * 8;
  const int* btable = block_tables + (long)seq * max_blocks;
  const long row_stride = (long)kvh * HEAD_DIM;

  // this wave's query rows (within the tile)
  float qf[QR][8];
  int qpos[QR];  // global kv position of each query row
  bool qvalid[QR];
#pragma unroll
  for (int r = 0; r < QR; ++r) {
    const int qi = tile_base + wave * QR + r;
    qvalid[r] = qi < q_len;
    qpos[r] = seq_len - q_len + qi;
    if (qvalid[r]) {
      const T* q_row = q + ((long)(q_start + qi) * nheads + head) * HEAD_DIM;
      Elems8<T> v = load8(q_row + dim_off);
#pragma unroll
      for (int j = 0; j < 8; ++j) qf[r][j] = ScalarOps<T>::to_f32(v.data[j]);
    } else {
#pragma unroll
      for (int j = 0; j < 8; ++j) qf[r][j] = 0.f;
    }
  }

  float m[QR], l[QR], acc[QR][8];
#pragma unroll
  for (int r = 0; r < QR; ++r) {
    m[r] = -FLT_MAX;
    l[r] = 0.f;
#pragma unroll
    for (int j = 0; j < 8; ++j) acc[r][j] = 0.f;
  }

  // causal bound for this wave: the furthest key any of its rows can see
  const int max_pos = qpos[QR - 1];

  for (int base = 0; base + 0 <= max_pos; base += SPW) {
    const int pos = base + slot_in_wave;
    const bool valid = pos <= max_pos && pos < seq_len;
    Elems8<T> k8, v8;
    if (valid) {
      const int block = btable[pos / block_size];
      const long row = ((long)block * block_size + pos % block_size) * row_stride +
                       (long)kv_head * HEAD_DIM + dim_off;
      k8 = load8(k_cache + row);
      v8 = load8(v_cache + row);
    }
    float p[QR];
#pragma unroll
    for (int r = 0; r < QR; ++r) {
      const bool see = valid && pos <= qpos[r];
      float qk = see ? dot8(qf[r], k8) : 0.f;
      qk = group_reduce_sum<LPS>(qk) * scale;
      if (!see) qk = -FLT_MAX;
      const float m_cand = wave_reduce_max(qk);
      const float m_new = fmaxf(m[r], m_cand);
      const float rescale = __expf(m[r] - m_new);
      p[r] = see ? __expf(qk - m_new) : 0.f;
      const float contrib = (lane % LPS == 0) ? p[r] : 0.f;
      l[r] = l[r] * rescale + wave_reduce_sum(contrib);
      m[r] = m_new;
#pragma unroll
      for (int j = 0; j < 8; ++j) acc[r][j] *= rescale;
      if (see) {
#pragma unroll
        for (int j = 0; j < 8; ++j)
          acc[r][j] += p[r] * ScalarOps<T>::to_f32(v8.data[j]);
      }
    }
  }

  // fold slot-group copies (same m/l frame)
#pragma unroll
  for (int r = 0; r < QR; ++r) {
#pragma unroll
    for (int j = 0; j < 8; ++j) {
#pragma unroll
      for (int off = 32; off >= LPS; off >>= 1)
        acc[r][j] += __shfl_xor(acc[r][j], off, WAVE_SIZE);
    }
  }

  // no cross-wave state: each wave owns its own query rows; lanes < LPS store
#pragma unroll
  for (int r = 0; r < QR; ++r) {
    if (!qvalid[r]) continue;
    const int qi = tile_base + wave * QR + r;
    const float inv_l = l[r] > 0.f ? 1.f / l[r] : 0.f;
    if (lane < LPS) {
      T* out_row = out + ((long)(q_start + qi) * nheads + head) * HEAD_DIM;
      Elems8<T> o;
#pragma unroll
      for (int j = 0; j < 8; ++j)
        o.data[j] = ScalarOps<T>::from_f32(acc[r][j] * inv_l);
      store8(out_row + dim_off, o);
    }
  }
}

// ---------------------------------------------------------------------------
// launchers with head-dim / group dispatch
// ---------------------------------------------------------------------------

template <typename T>
void launch_paged_decode(T* out, const T* q, const T* kc, const T* vc,
                         const int* bt, const int* sl, float scale, int nseq,
                         int nheads, int kvh, int head_dim, int block_size,
                         int max_blocks, hipStream_t stream) {
  const int group = nheads / kvh;
  dim3 grid(nseq, kvh);
  dim3 block(256);

#define DECODE_CASE(HD, G)                                                    \
  hipLaunchKernelGGL((paged_decode_kernel<T, HD, G>), grid, block, 0, stream, \
                     out, q, kc, vc, bt, sl, scale, kvh, block_size,          \
                     max_blocks)

#define DECODE_HD(HD)                                                         \
  switch (group) {                                                            \
    case 1: DECODE_CASE(HD, 1); break;                                        \
    case 2: DECODE_CASE(HD, 2); break;                                        \
    case 4: DECODE_CASE(HD, 4); break;                                        \
    case 5: DECODE_CASE(HD, 5); break;                                        \
    case 6: DECODE_CASE(HD, 6); break;                                        \
    case 7: DECODE_CASE(HD, 7); break;                                        \
    case 8: DECODE_CASE(HD, 8); break;                                        \
    case 3: DECODE_CASE(HD, 3); break;                                        \
    default: abort();                                                         \
  }

  switch (head_dim) {
    case 16: DECODE_HD(16); break;
    case 32: DECODE_HD(32); break;
    case 64: DECODE_HD(64); break;
    case 128: DECODE_HD(128); break;
    default: abort();
  }
#undef DECODE_HD
#undef DECODE_CASE
}

template <typename T>
void launch_paged_prefill(T* out, const T* q, const T* kc, const T* vc,
                          const int* bt, const int* qsl, const int* sl,
                          float scale, int nseq, int nheads, int kvh,
                          int head_dim, int block_size, int max_blocks,
                          int max_query_len, hipStream_t stream) {
  const int qtiles = (max_query_len + 15) / 16;
  dim3 grid(nseq, nheads, qtiles);
  dim3 block(256);

#define PREFILL_CASE(HD)                                                      \
  hipLaunchKernelGGL((paged_prefill_kernel<T, HD>), grid, block, 0, stream,   \
                     out, q, kc, vc, bt, qsl, sl, scale, nheads, kvh,         \
                     block_size, max_blocks)

  switch (head_dim) {
    case 16: PREFILL_CASE(16); break;
    case 32: PREFILL_CASE(32); break;
    case 64: PREFILL_CASE(64); break;
    case 128: PREFILL_CASE(128); break;
    default: abort();
  }
#undef PREFILL_CASE
}

#define INSTANTIATE(T)                                                        \
  template void launch_paged_decode<T>(T*, const T*, const T*, const T*,      \
                                       const int*, const int*, float, int,    \
                                       int, int, int, int, int, hipStream_t); \
  template void launch_paged_prefill<T>(T*, const T*, const T*, const T*,     \
                                        const int*, const int*, const int*,   \
                                        float, int, int, int, int, int, int,  \
                                        int, hipStream_t);

INSTANTIATE(float)
INSTANTIATE(__hip_bfloat16)
INSTANTIATE(__half)

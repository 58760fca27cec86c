// Torch extension bindings for the CDNA4 kernels.
// Built in-tree as vllm_tgis_adapter_amd/_C.so by kernels/build.py
// (hipcc --offload-arch=gfx950).

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include <hip/hip_bf16.h>
#include <hip/hip_fp16.h>

// launchers (defined in the .hip translation units)
template <typename T>
void launch_rms_norm(T*, const T*, const T*, float, int, int, hipStream_t);
template <typename T>
void launch_fused_add_rms_norm(T*, T*, const T*, float, int, int, hipStream_t);
template <typename T>
void launch_silu_and_mul(T*, const T*, int, int, hipStream_t);
template <typename T>
void launch_rotary_embedding(const long*, T*, T*, const float*, int, int, int,
                             int, hipStream_t);
template <typename T>
void launch_reshape_and_cache(const T*, const T*, T*, T*, const long*, int,
                              int, hipStream_t);
template <typename T>
void launch_paged_decode(T*, const T*, const T*, const T*, const int*,
                         const int*, float, int, int, int, int, int, int,
                         hipStream_t);
template <typename T>
void launch_paged_prefill(T*, const T*, const T*, const T*, const int*,
                          const int*, const int*, float, int, int, int, int,
                          int, int, int, hipStream_t);
void launch_mfma_probe_32(const unsigned short*, const unsigned short*, float*,
                          hipStream_t);
void launch_mfma_probe_16(const unsigned short*, const unsigned short*, float*,
                          hipStream_t);
void launch_paged_prefill_mfma_v2(__hip_bfloat16*, const __hip_bfloat16*,
                                  const __hip_bfloat16*, const __hip_bfloat16*,
                                  const int*, const int*, const int*, float,
                                  int, int, int, int, int, int, int,
                                  hipStream_t);
void launch_paged_prefill_mfma(__hip_bfloat16*, const __hip_bfloat16*,
                               const __hip_bfloat16*, const __hip_bfloat16*,
                               const int*, const int*, const int*, float, int,
                               int, int, int, int, int, int, hipStream_t);
int decode_mfma_num_partitions(int nseq, int kvh, int max_context);
int gemm_skinny_num_ksplit(int N, int K, int M);
void launch_logsoftmax_topk(float*, int*, float*, int*,
                            const __hip_bfloat16*, const long*, int, int,
                            int, hipStream_t);
void launch_gemm_tile(__hip_bfloat16*, const __hip_bfloat16*,
                      const __hip_bfloat16*, int, int, int, hipStream_t);
void launch_xgmi_allreduce(const unsigned long long*, __hip_bfloat16*,
                           const __hip_bfloat16*, long, long, int, int,
                           hipStream_t);
void launch_paged_decode_mfma_fp8(__hip_bfloat16*, float*, float*,
                                  const __hip_bfloat16*, const unsigned char*,
                                  const unsigned char*, const int*,
                                  const int*, float, int, int, int, int, int,
                                  int, int, hipStream_t);
void launch_paged_prefill_mfma_v2_fp8(__hip_bfloat16*, const __hip_bfloat16*,
                                      const unsigned char*,
                                      const unsigned char*, const int*,
                                      const int*, const int*, float, int, int,
                                      int, int, int, int, int, hipStream_t);
void launch_reshape_and_cache_fp8(const __hip_bfloat16*,
                                  const __hip_bfloat16*, unsigned char*,
                                  unsigned char*, const long*, int, int,
                                  hipStream_t);
void launch_gemm_skinny_q(__hip_bfloat16*, float*, const __hip_bfloat16*,
                          const unsigned char*, const float*, int, int, int,
                          int, int, hipStream_t);
void launch_gemm_skinny(__hip_bfloat16*, float*, const __hip_bfloat16*,
                        const __hip_bfloat16*, int, int, int, int,
                        hipStream_t);
void launch_gemm_skinny_gated(__hip_bfloat16*, float*, const __hip_bfloat16*,
                              const __hip_bfloat16*, int, int, int, int,
                              hipStream_t);
void launch_paged_decode_mfma(__hip_bfloat16*, float*, float*,
                              const __hip_bfloat16*, const __hip_bfloat16*,
                              const __hip_bfloat16*, const int*, const int*,
                              float, int, int, int, int, int, int, int,
                              hipStream_t);
template <typename T>
void launch_lora_shrink(float*, const T*, const T*, const int*, int, int, int,
                        hipStream_t);
template <typename T>
void launch_lora_expand(T*, const float*, const T*, const int*, const float*,
                        int, int, int, int, int, hipStream_t);
template <typename T>
void launch_sample_argmax(long*, const T*, const float*, const float*, int,
                          int, hipStream_t);
void launch_moe_gemm(__hip_bfloat16*, const __hip_bfloat16*,
                     const __hip_bfloat16*, const int*, int, int, int, int,
                     bool, hipStream_t);

namespace {

hipStream_t current_stream() {
  return at::hip::getCurrentHIPStream().stream();
}

template <typename T>
T* ptr(torch::Tensor& t) {
  return reinterpret_cast<T*>(t.data_ptr());
}
template <typename T>
const T* cptr(const torch::Tensor& t) {
  return reinterpret_cast<const T*>(t.data_ptr());
}

#define DISPATCH_FLOATING(DTYPE, BODY)                                   \
  switch (DTYPE) {                                                       \
    case at::ScalarType::BFloat16: {                                     \
      using scalar_t = __hip_bfloat16;                                   \
      BODY;                                                              \
      break;                                                             \
    }                                                                    \
    case at::ScalarType::Half: {                                         \
      using scalar_t = __half;                                           \
      BODY;                                                              \
      break;                                                             \
    }                                                                    \
    case at::ScalarType::Float: {                                        \
      using scalar_t = float;                                            \
      BODY;                                                              \
      break;                                                             \
    }                                                                    \
    default:                                                             \
      TORCH_CHECK(false, "unsupported dtype");                           \
  }

void rms_norm(torch::Tensor out, torch::Tensor input, torch::Tensor weight,
              double eps) {
  const int hidden = input.size(-1);
  const int rows = input.numel() / hidden;
  TORCH_CHECK(input.is_contiguous() && out.is_contiguous());
  TORCH_CHECK(hidden % 8 == 0, "hidden must be a multiple of 8");
  DISPATCH_FLOATING(input.scalar_type(), {
    launch_rms_norm<scalar_t>(ptr<scalar_t>(out), cptr<scalar_t>(input),
                              cptr<scalar_t>(weight), (float)eps, rows, hidden,
                              current_stream());
  });
}

void fused_add_rms_norm(torch::Tensor x, torch::Tensor residual,
                        torch::Tensor weight, double eps) {
  const int hidden = x.size(-1);
  const int rows = x.numel() / hidden;
  TORCH_CHECK(x.is_contiguous() && residual.is_contiguous());
  DISPATCH_FLOATING(x.scalar_type(), {
    launch_fused_add_rms_norm<scalar_t>(
        ptr<scalar_t>(x), ptr<scalar_t>(residual), cptr<scalar_t>(weight),
        (float)eps, rows, hidden, current_stream());
  });
}

void silu_and_mul(torch::Tensor out, torch::Tensor x) {
  const int d = out.size(-1);
  const int rows = out.numel() / d;
  TORCH_CHECK(x.size(-1) == 2 * d);
  TORCH_CHECK(d % 8 == 0, "intermediate size must be a multiple of 8");
  DISPATCH_FLOATING(x.scalar_type(), {
    launch_silu_and_mul<scalar_t>(ptr<scalar_t>(out), cptr<scalar_t>(x), rows,
                                  d, current_stream());
  });
}

void rotary_embedding(torch::Tensor positions, torch::Tensor q,
                      torch::Tensor k, int64_t head_dim,
                      torch::Tensor cos_sin_cache) {
  const int tokens = positions.size(0);
  const int nq = q.size(-1) / head_dim;
  const int nk = k.size(-1) / head_dim;
  TORCH_CHECK(positions.scalar_type() == at::ScalarType::Long);
  TORCH_CHECK(cos_sin_cache.scalar_type() == at::ScalarType::Float);
  DISPATCH_FLOATING(q.scalar_type(), {
    launch_rotary_embedding<scalar_t>(
        positions.data_ptr<long>(), ptr<scalar_t>(q), ptr<scalar_t>(k),
        cos_sin_cache.data_ptr<float>(), tokens, nq, nk, (int)head_dim,
        current_stream());
  });
}

void reshape_and_cache(torch::Tensor k, torch::Tensor v, torch::Tensor k_cache,
                       torch::Tensor v_cache, torch::Tensor slot_mapping) {
  const int tokens = k.size(0);
  const int row_elems = k_cache.size(2) * k_cache.size(3);
  TORCH_CHECK(slot_mapping.scalar_type() == at::ScalarType::Long);
  if (k_cache.scalar_type() == at::ScalarType::Byte) {
    TORCH_CHECK(k.scalar_type() == at::ScalarType::BFloat16,
                "fp8 KV cache takes bf16 K/V inputs");
    TORCH_CHECK(row_elems % 8 == 0);
    launch_reshape_and_cache_fp8(
        reinterpret_cast<const __hip_bfloat16*>(k.data_ptr()),
        reinterpret_cast<const __hip_bfloat16*>(v.data_ptr()),
        reinterpret_cast<unsigned char*>(k_cache.data_ptr()),
        reinterpret_cast<unsigned char*>(v_cache.data_ptr()),
        slot_mapping.data_ptr<long>(), tokens, row_elems, current_stream());
    return;
  }
  DISPATCH_FLOATING(k.scalar_type(), {
    launch_reshape_and_cache<scalar_t>(
        cptr<scalar_t>(k), cptr<scalar_t>(v), ptr<scalar_t>(k_cache),
        ptr<scalar_t>(v_cache), slot_mapping.data_ptr<long>(), tokens,
        row_elems, current_stream());
  });
}

void paged_attention_decode(torch::Tensor out, torch::Tensor q,
                            torch::Tensor k_cache, torch::Tensor v_cache,
                            torch::Tensor block_tables, torch::Tensor seq_lens,
                            double scale) {
  const int nseq = q.size(0);
  const int nheads = q.size(1);
  const int head_dim = q.size(2);
  const int kvh = k_cache.size(2);
  const int block_size = k_cache.size(1);
  const int max_blocks = block_tables.size(1);
  TORCH_CHECK(block_tables.scalar_type() == at::ScalarType::Int);
  TORCH_CHECK(seq_lens.scalar_type() == at::ScalarType::Int);
  const int group = nheads / kvh;
  static const bool force_valu = [] {
    const char* e = getenv("VTA_DECODE_VALU");
    return e && e[0] == '1';
  }();
  const bool kv_fp8 = k_cache.scalar_type() == at::ScalarType::Byte;
  if (kv_fp8) {
    TORCH_CHECK(q.scalar_type() == at::ScalarType::BFloat16 &&
                    (head_dim == 64 || head_dim == 128) && group >= 1 &&
                    group <= 8,
                "fp8 KV cache requires the bf16 MFMA decode path");
    const int npart = decode_mfma_num_partitions(nseq, kvh, max_blocks * block_size);
    const int P = npart * 4;
    auto opts = q.options().dtype(at::ScalarType::Float);
    auto part_acc = at::empty({nseq, nheads, P, head_dim}, opts);
    auto part_ml = at::empty({nseq, nheads, P, 2}, opts);
    launch_paged_decode_mfma_fp8(
        reinterpret_cast<__hip_bfloat16*>(out.data_ptr()),
        part_acc.data_ptr<float>(), part_ml.data_ptr<float>(),
        reinterpret_cast<const __hip_bfloat16*>(q.data_ptr()),
        reinterpret_cast<const unsigned char*>(k_cache.data_ptr()),
        reinterpret_cast<const unsigned char*>(v_cache.data_ptr()),
        block_tables.data_ptr<int>(), seq_lens.data_ptr<int>(), (float)scale,
        nseq, nheads, kvh, head_dim, block_size, max_blocks, npart,
        current_stream());
    return;
  }
  if (!force_valu && q.scalar_type() == at::ScalarType::BFloat16 &&
      (head_dim == 64 || head_dim == 128) && group >= 1 && group <= 8 &&
      nheads == kvh * group && nseq > 0) {
    const int npart = decode_mfma_num_partitions(nseq, kvh, max_blocks * block_size);
    const int P = npart * 4;
    auto opts = q.options().dtype(at::ScalarType::Float);
    auto part_acc = at::empty({nseq, nheads, P, head_dim}, opts);
    auto part_ml = at::empty({nseq, nheads, P, 2}, opts);
    launch_paged_decode_mfma(
        reinterpret_cast<__hip_bfloat16*>(out.data_ptr()),
        part_acc.data_ptr<float>(), part_ml.data_ptr<float>(),
        reinterpret_cast<const __hip_bfloat16*>(q.data_ptr()),
        reinterpret_cast<const __hip_bfloat16*>(k_cache.data_ptr()),
        reinterpret_cast<const __hip_bfloat16*>(v_cache.data_ptr()),
        block_tables.data_ptr<int>(), seq_lens.data_ptr<int>(), (float)scale,
        nseq, nheads, kvh, head_dim, block_size, max_blocks, npart,
        current_stream());
    return;
  }
  DISPATCH_FLOATING(q.scalar_type(), {
    launch_paged_decode<scalar_t>(
        ptr<scalar_t>(out), cptr<scalar_t>(q), cptr<scalar_t>(k_cache),
        cptr<scalar_t>(v_cache), block_tables.data_ptr<int>(),
        seq_lens.data_ptr<int>(), (float)scale, nseq, nheads, kvh, head_dim,
        block_size, max_blocks, current_stream());
  });
}

void paged_attention_prefill(torch::Tensor out, torch::Tensor q,
                             torch::Tensor k_cache, torch::Tensor v_cache,
                             torch::Tensor block_tables,
                             torch::Tensor query_start_loc,
                             torch::Tensor seq_lens, double scale,
                             int64_t max_query_len) {
  const int nseq = seq_lens.size(0);
  const int nheads = q.size(1);
  const int head_dim = q.size(2);
  const int kvh = k_cache.size(2);
  const int block_size = k_cache.size(1);
  const int max_blocks = block_tables.size(1);
  static const bool force_valu = [] {
    const char* e = getenv("VTA_PREFILL_VALU");
    return e && e[0] == '1';
  }();
  static const bool force_v1 = [] {
    const char* e = getenv("VTA_PREFILL_V1");
    return e && e[0] == '1';
  }();
  if (k_cache.scalar_type() == at::ScalarType::Byte) {
    TORCH_CHECK(q.scalar_type() == at::ScalarType::BFloat16 && head_dim == 128,
                "fp8 KV cache requires the bf16 v2 prefill path (hd=128)");
    launch_paged_prefill_mfma_v2_fp8(
        reinterpret_cast<__hip_bfloat16*>(out.data_ptr()),
        reinterpret_cast<const __hip_bfloat16*>(q.data_ptr()),
        reinterpret_cast<const unsigned char*>(k_cache.data_ptr()),
        reinterpret_cast<const unsigned char*>(v_cache.data_ptr()),
        block_tables.data_ptr<int>(), query_start_loc.data_ptr<int>(),
        seq_lens.data_ptr<int>(), (float)scale, nseq, nheads, kvh, head_dim,
        block_size, max_blocks, (int)max_query_len, current_stream());
    return;
  }
  if (!force_valu && !force_v1 &&
      q.scalar_type() == at::ScalarType::BFloat16 && head_dim == 128) {
    launch_paged_prefill_mfma_v2(
        reinterpret_cast<__hip_bfloat16*>(out.data_ptr()),
        reinterpret_cast<const __hip_bfloat16*>(q.data_ptr()),
        reinterpret_cast<const __hip_bfloat16*>(k_cache.data_ptr()),
        reinterpret_cast<const __hip_bfloat16*>(v_cache.data_ptr()),
        block_tables.data_ptr<int>(), query_start_loc.data_ptr<int>(),
        seq_lens.data_ptr<int>(), (float)scale, nseq, nheads, kvh, head_dim,
        block_size, max_blocks, (int)max_query_len, current_stream());
    return;
  }
  if (!force_valu && q.scalar_type() == at::ScalarType::BFloat16 &&
      (head_dim == 64 || head_dim == 128)) {
    launch_paged_prefill_mfma(
        reinterpret_cast<__hip_bfloat16*>(out.data_ptr()),
        reinterpret_cast<const __hip_bfloat16*>(q.data_ptr()),
        reinterpret_cast<const __hip_bfloat16*>(k_cache.data_ptr()),
        reinterpret_cast<const __hip_bfloat16*>(v_cache.data_ptr()),
        block_tables.data_ptr<int>(), query_start_loc.data_ptr<int>(),
        seq_lens.data_ptr<int>(), (float)scale, nseq, nheads, kvh, head_dim,
        block_size, max_blocks, (int)max_query_len, current_stream());
    return;
  }
  DISPATCH_FLOATING(q.scalar_type(), {
    launch_paged_prefill<scalar_t>(
        ptr<scalar_t>(out), cptr<scalar_t>(q), cptr<scalar_t>(k_cache),
        cptr<scalar_t>(v_cache), block_tables.data_ptr<int>(),
        query_start_loc.data_ptr<int>(), seq_lens.data_ptr<int>(),
        (float)scale, nseq, nheads, kvh, head_dim, block_size, max_blocks,
        (int)max_query_len, current_stream());
  });
}

// ---- direct-xGMI all-reduce plumbing (E15) --------------------------------
// Raw hipMalloc (NOT the torch caching allocator: hipIpcGetMemHandle needs
// the allocation base).  Returns (device_ptr, ipc_handle_bytes).
std::tuple<int64_t, py::bytes> xar_alloc(int64_t nbytes) {
  void* p = nullptr;
  TORCH_CHECK(hipMalloc(&p, nbytes) == hipSuccess, "xar_alloc failed");
  TORCH_CHECK(hipMemset(p, 0, nbytes) == hipSuccess);
  TORCH_CHECK(hipDeviceSynchronize() == hipSuccess);
  hipIpcMemHandle_t h;
  TORCH_CHECK(hipIpcGetMemHandle(&h, p) == hipSuccess,
              "hipIpcGetMemHandle failed");
  return {reinterpret_cast<int64_t>(p),
          py::bytes(reinterpret_cast<const char*>(&h), sizeof(h))};
}

int64_t xar_open(const std::string& handle_bytes) {
  TORCH_CHECK(handle_bytes.size() == sizeof(hipIpcMemHandle_t));
  hipIpcMemHandle_t h;
  memcpy(&h, handle_bytes.data(), sizeof(h));
  void* p = nullptr;
  TORCH_CHECK(hipIpcOpenMemHandle(&p, h, hipIpcMemLazyEnablePeerAccess) ==
                  hipSuccess,
              "hipIpcOpenMemHandle failed");
  return reinterpret_cast<int64_t>(p);
}

void xgmi_allreduce(torch::Tensor tensor, torch::Tensor bufs, int64_t cap,
                    int64_t rank, int64_t world) {
  TORCH_CHECK(tensor.scalar_type() == at::ScalarType::BFloat16);
  TORCH_CHECK(tensor.is_contiguous() && tensor.is_cuda());
  TORCH_CHECK(bufs.scalar_type() == at::ScalarType::Long && bufs.is_cuda());
  TORCH_CHECK(tensor.numel() * 2 <= cap, "tensor exceeds staging capacity");
  launch_xgmi_allreduce(
      reinterpret_cast<const unsigned long long*>(bufs.data_ptr()),
      reinterpret_cast<__hip_bfloat16*>(tensor.data_ptr()),
      reinterpret_cast<const __hip_bfloat16*>(tensor.data_ptr()),
      tensor.numel(), cap, (int)rank, (int)world, current_stream());
}

void gemm_tile(torch::Tensor y, torch::Tensor x, torch::Tensor w) {
  const int M = x.size(0);
  const int K = x.size(1);
  const int N = w.size(0);
  TORCH_CHECK(x.scalar_type() == at::ScalarType::BFloat16 &&
              w.scalar_type() == at::ScalarType::BFloat16);
  TORCH_CHECK(x.is_contiguous() && w.is_contiguous() && y.is_contiguous());
  TORCH_CHECK(M >= 1 && N % 128 == 0 && K % 64 == 0);
  TORCH_CHECK(w.size(1) == K && y.size(0) == M && y.size(1) == N);
  launch_gemm_tile(reinterpret_cast<__hip_bfloat16*>(y.data_ptr()),
                   reinterpret_cast<const __hip_bfloat16*>(x.data_ptr()),
                   reinterpret_cast<const __hip_bfloat16*>(w.data_ptr()), M, N,
                   K, current_stream());
}

void gemm_skinny(torch::Tensor y, torch::Tensor x, torch::Tensor w) {
  const int M = x.size(0);
  const int K = x.size(1);
  const int N = w.size(0);
  TORCH_CHECK(x.scalar_type() == at::ScalarType::BFloat16 &&
              w.scalar_type() == at::ScalarType::BFloat16);
  TORCH_CHECK(x.is_contiguous() && w.is_contiguous() && y.is_contiguous());
  TORCH_CHECK(M >= 1 && N % 128 == 0 && K % 64 == 0);
  TORCH_CHECK(w.size(1) == K && y.size(0) == M && y.size(1) == N);
  const int KS = gemm_skinny_num_ksplit(N, K, M);
  torch::Tensor part;
  float* pp = nullptr;
  if (KS > 1) {
    part = at::empty({KS, M, N}, x.options().dtype(at::ScalarType::Float));
    pp = part.data_ptr<float>();
  }
  launch_gemm_skinny(reinterpret_cast<__hip_bfloat16*>(y.data_ptr()), pp,
                     reinterpret_cast<const __hip_bfloat16*>(x.data_ptr()),
                     reinterpret_cast<const __hip_bfloat16*>(w.data_ptr()), M,
                     N, K, KS, current_stream());
}

void gemm_skinny_q(torch::Tensor y, torch::Tensor x, torch::Tensor wq,
                   torch::Tensor wscale, int64_t qbits) {
  const int M = x.size(0);
  const int K = x.size(1);
  const int N = wq.size(0);
  TORCH_CHECK(x.scalar_type() == at::ScalarType::BFloat16);
  TORCH_CHECK(wq.scalar_type() == at::ScalarType::Byte ||
              wq.scalar_type() == at::ScalarType::Char);
  TORCH_CHECK(wscale.scalar_type() == at::ScalarType::Float);
  TORCH_CHECK(x.is_contiguous() && wq.is_contiguous() &&
              wscale.is_contiguous() && y.is_contiguous());
  TORCH_CHECK(M >= 1 && M <= 64 && N % 128 == 0 && K % 128 == 0);
  TORCH_CHECK(qbits == 8 || qbits == 4);
  TORCH_CHECK(wq.size(1) == (qbits == 8 ? K : K / 2));
  const int KS = gemm_skinny_num_ksplit(N, K, M);
  torch::Tensor part;
  float* pp = nullptr;
  if (KS > 1) {
    part = at::empty({KS, M, N}, x.options().dtype(at::ScalarType::Float));
    pp = part.data_ptr<float>();
  }
  launch_gemm_skinny_q(
      reinterpret_cast<__hip_bfloat16*>(y.data_ptr()), pp,
      reinterpret_cast<const __hip_bfloat16*>(x.data_ptr()),
      reinterpret_cast<const unsigned char*>(wq.data_ptr()),
      wscale.data_ptr<float>(), (int)qbits, M, N, K, KS, current_stream());
}

void gemm_skinny_gated(torch::Tensor y, torch::Tensor x, torch::Tensor w13) {
  const int M = x.size(0);
  const int K = x.size(1);
  const int I = w13.size(0) / 2;
  TORCH_CHECK(x.scalar_type() == at::ScalarType::BFloat16 &&
              w13.scalar_type() == at::ScalarType::BFloat16);
  TORCH_CHECK(x.is_contiguous() && w13.is_contiguous() && y.is_contiguous());
  TORCH_CHECK(M >= 1 && I % 128 == 0 && K % 64 == 0);
  TORCH_CHECK(w13.size(0) % 2 == 0 && w13.size(1) == K);
  TORCH_CHECK(y.size(0) == M && y.size(1) == I);
  const int KS = gemm_skinny_num_ksplit(I, K, M);
  torch::Tensor part;
  float* pp = nullptr;
  if (KS > 1) {
    part = at::empty({KS, 2, M, I}, x.options().dtype(at::ScalarType::Float));
    pp = part.data_ptr<float>();
  }
  launch_gemm_skinny_gated(
      reinterpret_cast<__hip_bfloat16*>(y.data_ptr()), pp,
      reinterpret_cast<const __hip_bfloat16*>(x.data_ptr()),
      reinterpret_cast<const __hip_bfloat16*>(w13.data_ptr()), M, I, K, KS,
      current_stream());
}

void lora_bgmv(torch::Tensor out, torch::Tensor x, torch::Tensor a_stack,
               torch::Tensor b_stack, torch::Tensor tmp, torch::Tensor slots,
               torch::Tensor scales, int64_t off) {
  const int Tn = x.size(0);
  const int K = x.size(1);
  const int R = a_stack.size(1);
  const int N = b_stack.size(1);
  const int out_w = out.size(1);
  TORCH_CHECK(out.size(0) == Tn && tmp.size(0) == Tn && tmp.size(1) == R);
  TORCH_CHECK(a_stack.size(2) == K && b_stack.size(2) == R);
  TORCH_CHECK(R <= 64 && K % 8 == 0);
  TORCH_CHECK(off + N <= out_w);
  TORCH_CHECK(slots.scalar_type() == at::ScalarType::Int);
  TORCH_CHECK(tmp.scalar_type() == at::ScalarType::Float &&
              scales.scalar_type() == at::ScalarType::Float);
  TORCH_CHECK(x.is_contiguous() && out.is_contiguous() &&
              a_stack.is_contiguous() && b_stack.is_contiguous());
  DISPATCH_FLOATING(x.scalar_type(), {
    launch_lora_shrink<scalar_t>(tmp.data_ptr<float>(), cptr<scalar_t>(x),
                                 cptr<scalar_t>(a_stack),
                                 slots.data_ptr<int>(), Tn, R, K,
                                 current_stream());
    launch_lora_expand<scalar_t>(ptr<scalar_t>(out), tmp.data_ptr<float>(),
                                 cptr<scalar_t>(b_stack),
                                 slots.data_ptr<int>(),
                                 scales.data_ptr<float>(), Tn, R, N, out_w,
                                 (int)off, current_stream());
  });
}

void moe_gemm(torch::Tensor y, torch::Tensor x, torch::Tensor w,
              torch::Tensor seg_off, bool gated) {
  const int T = x.size(0);
  const int K = x.size(1);
  const int E = w.size(0);
  const long wn = w.size(1);
  const int N = gated ? (int)(wn / 2) : (int)wn;
  TORCH_CHECK(x.scalar_type() == at::ScalarType::BFloat16 &&
              w.scalar_type() == at::ScalarType::BFloat16 &&
              y.scalar_type() == at::ScalarType::BFloat16);
  TORCH_CHECK(seg_off.scalar_type() == at::ScalarType::Int &&
              seg_off.numel() == E + 1);
  TORCH_CHECK(N % 128 == 0 && K % 64 == 0 && w.size(2) == K);
  TORCH_CHECK(y.size(0) == T && y.size(1) == N);
  TORCH_CHECK(x.is_contiguous() && w.is_contiguous() && y.is_contiguous());
  if (T == 0) return;
  launch_moe_gemm(reinterpret_cast<__hip_bfloat16*>(y.data_ptr()),
                  reinterpret_cast<const __hip_bfloat16*>(x.data_ptr()),
                  reinterpret_cast<const __hip_bfloat16*>(w.data_ptr()),
                  seg_off.data_ptr<int>(), E, N, K, T, gated,
                  current_stream());
}

void sample_argmax(torch::Tensor out, torch::Tensor logits,
                   torch::Tensor temps, c10::optional<torch::Tensor> noise) {
  const int N = logits.size(0);
  const int V = logits.size(1);
  TORCH_CHECK(out.scalar_type() == at::ScalarType::Long && out.size(0) == N);
  TORCH_CHECK(temps.scalar_type() == at::ScalarType::Float && temps.size(0) == N);
  TORCH_CHECK(logits.is_contiguous());
  const float* np_ = nullptr;
  if (noise.has_value()) {
    TORCH_CHECK(noise->scalar_type() == at::ScalarType::Float &&
                noise->is_contiguous() && noise->size(0) == N &&
                noise->size(1) == V);
    np_ = noise->data_ptr<float>();
  }
  DISPATCH_FLOATING(logits.scalar_type(), {
    launch_sample_argmax<scalar_t>(out.data_ptr<long>(),
                                   cptr<scalar_t>(logits),
                                   temps.data_ptr<float>(), np_, N, V,
                                   current_stream());
  });
}

void logsoftmax_topk(torch::Tensor topv, torch::Tensor topi,
                     torch::Tensor chosen_lp, torch::Tensor ranks,
                     torch::Tensor logits, torch::Tensor chosen) {
  const int N = logits.size(0);
  const int V = logits.size(1);
  const int K = topv.size(1);
  TORCH_CHECK(logits.scalar_type() == at::ScalarType::BFloat16 &&
              logits.is_contiguous());
  TORCH_CHECK(V % 8 == 0, "vocab must be a multiple of 8");
  TORCH_CHECK(K <= 16, "K <= 16");
  TORCH_CHECK(topv.scalar_type() == at::ScalarType::Float &&
              topi.scalar_type() == at::ScalarType::Int &&
              topv.is_contiguous() && topi.is_contiguous() &&
              topi.size(1) == K && topv.size(0) == N && topi.size(0) == N);
  TORCH_CHECK(chosen_lp.scalar_type() == at::ScalarType::Float &&
              chosen_lp.size(0) == N);
  TORCH_CHECK(ranks.scalar_type() == at::ScalarType::Int && ranks.size(0) == N);
  TORCH_CHECK(chosen.scalar_type() == at::ScalarType::Long &&
              chosen.is_contiguous() && chosen.size(0) == N);
  launch_logsoftmax_topk(
      topv.data_ptr<float>(), topi.data_ptr<int>(),
      chosen_lp.data_ptr<float>(), ranks.data_ptr<int>(),
      reinterpret_cast<const __hip_bfloat16*>(logits.data_ptr()),
      chosen.data_ptr<long>(), N, V, K, current_stream());
}

void mfma_probe(torch::Tensor a, torch::Tensor b, torch::Tensor d, int64_t shape) {
  TORCH_CHECK(a.scalar_type() == at::ScalarType::BFloat16);
  auto* ap = reinterpret_cast<const unsigned short*>(a.data_ptr());
  auto* bp = reinterpret_cast<const unsigned short*>(b.data_ptr());
  auto* dp = d.data_ptr<float>();
  if (shape == 32)
    launch_mfma_probe_32(ap, bp, dp, current_stream());
  else
    launch_mfma_probe_16(ap, bp, dp, current_stream());
}

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("mfma_probe", &mfma_probe, "MFMA fragment layout probe");
  m.def("rms_norm", &rms_norm, "fused RMSNorm (CDNA4)");
  m.def("fused_add_rms_norm", &fused_add_rms_norm,
        "in-place residual-add + RMSNorm (CDNA4)");
  m.def("silu_and_mul", &silu_and_mul, "SwiGLU activation (CDNA4)");
  m.def("rotary_embedding", &rotary_embedding, "NeoX rotary, in place (CDNA4)");
  m.def("reshape_and_cache", &reshape_and_cache, "paged KV cache write (CDNA4)");
  m.def("paged_attention_decode", &paged_attention_decode,
        "paged decode attention (CDNA4)");
  m.def("paged_attention_prefill", &paged_attention_prefill,
        "paged causal prefill attention (CDNA4)");
  m.def("xar_alloc", &xar_alloc,
        "raw hipMalloc + IPC handle for the xGMI all-reduce staging buffer");
  m.def("xar_open", &xar_open, "open a peer's IPC handle -> device pointer");
  m.def("xgmi_allreduce", &xgmi_allreduce,
        "in-place one-shot all-reduce over peer-mapped xGMI buffers");
  m.def("gemm_tile", &gemm_tile,
        "128x128-tile glds-staged bf16 GEMM for decode batches (CDNA4)");
  m.def("gemm_skinny_q", &gemm_skinny_q,
        "weight-only-quantized (int8 / int4-g128) skinny decode GEMM");
  m.def("gemm_skinny", &gemm_skinny,
        "skinny decode GEMM y = x @ w^T, M <= 64 (CDNA4 MFMA)");
  m.def("gemm_skinny_gated", &gemm_skinny_gated,
        "fused gate/up skinny GEMM + SiLU-mul, M <= 64 (CDNA4 MFMA)");
  m.def("moe_gemm", &moe_gemm,
        "grouped MoE GEMM over expert segments (CDNA4 MFMA)");
  m.def("logsoftmax_topk", &logsoftmax_topk,
        "fused log-softmax + top-K + sampled-token rank (one HBM pass)");
  m.def("sample_argmax", &sample_argmax,
        "fused temperature/gumbel-race sampling + greedy argmax (CDNA4)");
  m.def("lora_bgmv", &lora_bgmv,
        "batched multi-LoRA shrink+expand for mixed-adapter batches (CDNA4)");
}

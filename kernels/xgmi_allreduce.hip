// Direct-xGMI one-shot all-reduce.
//
// MI355X xGMI is point-to-point (7 links x ~153 GB/s per GPU), so a ring
// all-reduce pays 2*S*(N-1)/N bytes per link; the one-shot form — every rank
// reads every peer's staged shard directly over xGMI and reduces locally —
// moves only S bytes per link and skips the ring's 2(N-1) latency steps.
// Used for the TP decode tensors (<= a few MB); larger prefill tensors stay
// on RCCL (caller's policy, parallel/__init__.py).
//
// Protocol per op (two kernels, stream-ordered, hipGraph-capturable —
// sequencing state lives on DEVICE so capture/replay works):
//   stage:  gate on peers having drained op seq-1 (local "done" mailbox),
//           copy input -> own staging buffer, system-fence, last block
//           writes seq into every peer's "ready" mailbox over xGMI.
//   reduce: every block polls the LOCAL ready mailbox (peers push — no
//           remote polling), acquire-fence, sums all ranks' staging shards
//           (f32 accumulate), stores bf16; last block advances the device
//           seq and pushes "done" to every peer.
// Mailbox values are monotonically increasing sequence numbers (no ABA).
// Spins carry a clock64 timeout that poisons the error flag instead of
// hanging the GPU.

#include "common.h"

#define XAR_MAXW 8

typedef __attribute__((ext_vector_type(8))) short bf16x8_t;

struct XarMeta {
  unsigned ready[XAR_MAXW];
  unsigned done[XAR_MAXW];
  unsigned counter[2];
  unsigned long long seq;
  unsigned error;
};

__device__ inline XarMeta* xar_meta(unsigned long long base, long cap) {
  return reinterpret_cast<XarMeta*>(base + cap);
}

__device__ inline unsigned xar_load_sys(const unsigned* p) {
  return __hip_atomic_load(p, __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_SYSTEM);
}

__device__ inline void xar_store_sys(unsigned* p, unsigned v) {
  __hip_atomic_store(p, v, __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_SYSTEM);
}

// spin until *p >= want; ~4e9 cycles timeout -> error flag
__device__ inline bool xar_wait(const unsigned* p, unsigned want,
                                unsigned* err) {
  long long t0 = clock64();
  while (xar_load_sys(p) < want) {
    if (clock64() - t0 > 4000000000LL) {
      xar_store_sys(err, 1u);
      return false;
    }
  }
  return true;
}

__global__ void xgmi_ar_stage_kernel(
    const unsigned long long* __restrict__ bufs,  // [world] peer base ptrs
    const __hip_bfloat16* __restrict__ inp,
    const long nelems, const long cap, const int rank, const int world) {
  XarMeta* my = xar_meta(bufs[rank], cap);
  const unsigned seq = (unsigned)my->seq + 1;

  // gate: all peers must have drained op seq-1 from our staging buffer
  for (int p = 0; p < world; ++p)
    if (!xar_wait(&my->done[p], seq - 1, &my->error)) return;

  __hip_bfloat16* staging = reinterpret_cast<__hip_bfloat16*>(bufs[rank]);
  const long vecs = nelems / 8;
  const long idx0 = (long)blockIdx.x * blockDim.x + threadIdx.x;
  const long stride = (long)gridDim.x * blockDim.x;
  for (long i = idx0; i < vecs; i += stride)
    reinterpret_cast<bf16x8_t*>(staging)[i] =
        reinterpret_cast<const bf16x8_t*>(inp)[i];
  for (long i = vecs * 8 + idx0; i < nelems; i += stride)
    staging[i] = inp[i];

  __threadfence_system();
  __syncthreads();
  if (threadIdx.x == 0) {
    const unsigned old = atomicAdd(&my->counter[0], 1u);
    if (old == gridDim.x - 1) {
      my->counter[0] = 0;
      __threadfence_system();
      for (int p = 0; p < world; ++p)
        xar_store_sys(&xar_meta(bufs[p], cap)->ready[rank], seq);
    }
  }
}

__global__ void xgmi_ar_reduce_kernel(
    const unsigned long long* __restrict__ bufs,
    __hip_bfloat16* __restrict__ out,
    const long nelems, const long cap, const int rank, const int world) {
  XarMeta* my = xar_meta(bufs[rank], cap);
  const unsigned seq = (unsigned)my->seq + 1;

  for (int p = 0; p < world; ++p)
    if (!xar_wait(&my->ready[p], seq, &my->error)) return;
  __threadfence_system();  // acquire: peers' staged data now visible

  const __hip_bfloat16* src[XAR_MAXW];
  for (int p = 0; p < world; ++p)
    src[p] = reinterpret_cast<const __hip_bfloat16*>(bufs[p]);

  const long vecs = nelems / 8;
  const long idx0 = (long)blockIdx.x * blockDim.x + threadIdx.x;
  const long stride = (long)gridDim.x * blockDim.x;
  for (long i = idx0; i < vecs; i += stride) {
    float acc[8] = {};
    for (int p = 0; p < world; ++p) {
      const bf16x8_t v = reinterpret_cast<const bf16x8_t*>(src[p])[i];
#pragma unroll
      for (int j = 0; j < 8; ++j)
        acc[j] += __bfloat162float(
            __builtin_bit_cast(__hip_bfloat16, (short)v[j]));
    }
    bf16x8_t o;
#pragma unroll
    for (int j = 0; j < 8; ++j)
      o[j] = __builtin_bit_cast(short, __float2bfloat16(acc[j]));
    reinterpret_cast<bf16x8_t*>(out)[i] = o;
  }
  for (long i = vecs * 8 + idx0; i < nelems; i += stride) {
    float acc = 0.f;
    for (int p = 0; p < world; ++p) acc += __bfloat162float(src[p][i]);
    out[i] = __float2bfloat16(acc);
  }

  __syncthreads();
  if (threadIdx.x == 0) {
    const unsigned old = atomicAdd(&my->counter[1], 1u);
    if (old == gridDim.x - 1) {
      my->counter[1] = 0;
      my->seq = seq;  // device-side sequencing: capture/replay-safe
      __threadfence_system();
      for (int p = 0; p < world; ++p)
        xar_store_sys(&xar_meta(bufs[p], cap)->done[rank], seq);
    }
  }
}

void launch_xgmi_allreduce(const unsigned long long* bufs_dev,
                           __hip_bfloat16* out, const __hip_bfloat16* inp,
                           long nelems, long cap, int rank, int world,
                           hipStream_t stream) {
  const int threads = 256;
  long vecs = (nelems + 7) / 8;
  int blocks = (int)((vecs + threads - 1) / threads);
  if (blocks > 512) blocks = 512;
  if (blocks < 1) blocks = 1;
  hipLaunchKernelGGL(xgmi_ar_stage_kernel, dim3(blocks), dim3(threads), 0,
                     stream, bufs_dev, inp, nelems, cap, rank, world);
  hipLaunchKernelGGL(xgmi_ar_reduce_kernel, dim3(blocks), dim3(threads), 0,
                     stream, bufs_dev, out, nelems, cap, rank, world);
}

#include <hip/hip_runtime.h>
#include <cstdio>
typedef __attribute__((ext_vector_type(2))) unsigned int u32x2;
__global__ void probe(short* out, short* out2) {
  __shared__ short lds[2048];
  for (int i = threadIdx.x; i < 2048; i += blockDim.x) lds[i] = (short)i;
  __syncthreads();
  // opaque consumer: keeps the LDS array + its writes alive (the asm below
  // references LDS only through raw integer offsets)
  if ((int)out[0] == -32768) out[0] = lds[threadIdx.x];
  // uniform base address experiment
  u32x2 v;
  unsigned addr = 0;  // byte offset into LDS
  asm volatile("ds_read_b64_tr_b16 %0, %1\n\ts_waitcnt lgkmcnt(0)"
               : "=v"(v) : "v"(addr) : "memory");
  union { u32x2 u; short s[4]; } c; c.u = v;
  for (int j = 0; j < 4; ++j) out[threadIdx.x * 4 + j] = c.s[j];
  // per-lane address experiment: lane provides addr = lane*8 bytes
  unsigned addr2 = threadIdx.x * 8;
  asm volatile("ds_read_b64_tr_b16 %0, %1\n\ts_waitcnt lgkmcnt(0)"
               : "=v"(v) : "v"(addr2) : "memory");
  c.u = v;
  for (int j = 0; j < 4; ++j) out2[threadIdx.x * 4 + j] = c.s[j];
}
int main() {
  short *o1, *o2;
  hipMalloc(&o1, 64*4*2); hipMalloc(&o2, 64*4*2);
  hipLaunchKernelGGL(probe, dim3(1), dim3(64), 0, 0, o1, o2);
  hipDeviceSynchronize();
  short h1[256], h2[256];
  hipMemcpy(h1, o1, sizeof(h1), hipMemcpyDeviceToHost);
  hipMemcpy(h2, o2, sizeof(h2), hipMemcpyDeviceToHost);
  printf("uniform base=0: lane: elems\n");
  for (int l = 0; l < 64; ++l) {
    printf("%2d: %4d %4d %4d %4d%s", l, h1[l*4], h1[l*4+1], h1[l*4+2], h1[l*4+3], (l%4==3)?"\n":"   ");
  }
  printf("\nper-lane addr=lane*8:\n");
  for (int l = 0; l < 64; ++l) {
    printf("%2d: %4d %4d %4d %4d%s", l, h2[l*4], h2[l*4+1], h2[l*4+2], h2[l*4+3], (l%4==3)?"\n":"   ");
  }
  return 0;
}

#include <hip/hip_runtime.h>
#include <cstdio>
typedef unsigned short u16;
typedef u16 u16x4 __attribute__((ext_vector_type(4)));

// each lane passes addr = base + per-lane offset by SCHEME; dumps 4 outputs
extern "C" __global__ void probe(u16* out, int scheme) {
  __shared__ u16 lds[4096];
  for (int i = threadIdx.x; i < 4096; i += blockDim.x) lds[i] = (u16)i;
  __syncthreads();
  const int lane = threadIdx.x & 63;
  int off_elems = 0;
  if (scheme == 1) off_elems = (lane & 15) * 4;     // 8B per lane within group
  if (scheme == 2) off_elems = lane * 4;            // 8B per lane across wave
  if (scheme == 3) off_elems = (lane >> 4) * 64;    // group base only
  if (scheme == 4) off_elems = (lane & 15) * 4 + (lane >> 4) * 64;
  unsigned addr = off_elems * 2;  // bytes
  u16x4 r;
  asm volatile("ds_read_b64_tr_b16 %0, %1\n\ts_waitcnt lgkmcnt(0)"
               : "=v"(r) : "v"(addr) : "memory");
  __builtin_amdgcn_sched_barrier(0);
  if (threadIdx.x < 64) {
    for (int j = 0; j < 4; ++j) out[lane * 4 + j] = r[j];
  }
  // keep the LDS array alive: a data-dependent read the compiler cannot
  // prove dead (out[] contents are unknown at entry)
  if (out[0] == 0xFFFFu) out[1] = lds[out[2] & 4095];
}
int main() {
  u16* d;
  hipMalloc(&d, 64 * 4 * sizeof(u16));
  u16 h[256];
  for (int s = 1; s <= 4; ++s) {
    hipLaunchKernelGGL(probe, dim3(1), dim3(64), 0, 0, d, s);
    hipMemcpy(h, d, sizeof(h), hipMemcpyDeviceToHost);
    printf("scheme %d:\n", s);
    for (int l = 0; l < 32; ++l) {
      printf("  lane %2d: %4d %4d %4d %4d\n", l, h[l*4], h[l*4+1], h[l*4+2], h[l*4+3]);
      if (l == 7) l = 14;  // skip middle lanes for brevity
    }
  }
  hipFree(d);
  return 0;
}

// Common device helpers for ant_ray_amd CDNA4 (gfx950) kernels.
//
// Conventions (per the CDNA4 playbook):
//   * wavefront = 64 lanes (never 32); block size multiples of 64
//   * bf16 traffic vectorized as ushort8 (16 B/lane) — scalar bf16 loads are
//     2-2.5x slower on this chip
//   * reductions: __shfl_down over 64 lanes, then LDS across waves
//   * memory-bound kernels aim at the ~6.3 TB/s achievable HBM ceiling
#pragma once
#include <hip/hip_runtime.h>

#include <cstdint>

#define WAVE 64

typedef unsigned short ushort_t;
typedef ushort_t ushortx8 __attribute__((ext_vector_type(8)));
typedef ushort_t ushortx4 __attribute__((ext_vector_type(4)));
typedef float floatx8 __attribute__((ext_vector_type(8)));
typedef float floatx4 __attribute__((ext_vector_type(4)));

// bf16 (stored as ushort) <-> f32. bf16->f32 is an exact shift; f32->bf16 is
// round-to-nearest-even with NaN guard.
__device__ __forceinline__ float bf2f(ushort_t u) {
  return __builtin_bit_cast(float, (uint32_t)u << 16);
}

__device__ __forceinline__ ushort_t f2bf(float f) {
  uint32_t x = __builtin_bit_cast(uint32_t, f);
  if ((x & 0x7fffffffu) > 0x7f800000u) return (ushort_t)((x >> 16) | 0x40);  // NaN
  uint32_t r = (x + 0x7fffu + ((x >> 16) & 1u)) >> 16;
  return (ushort_t)r;
}

__device__ __forceinline__ floatx8 bf8_to_f32x8(ushortx8 v) {
  floatx8 o;
#pragma unroll
  for (int i = 0; i < 8; ++i) o[i] = bf2f(v[i]);
  return o;
}

__device__ __forceinline__ ushortx8 f32x8_to_bf8(floatx8 v) {
  ushortx8 o;
#pragma unroll
  for (int i = 0; i < 8; ++i) o[i] = f2bf(v[i]);
  return o;
}

// ---- reductions --------------------------------------------------------

__device__ __forceinline__ float wave_reduce_sum(float v) {
#pragma unroll
  for (int off = WAVE / 2; off > 0; off >>= 1) v += __shfl_down(v, off);
  return v;
}

__device__ __forceinline__ float wave_reduce_max(float v) {
#pragma unroll
  for (int off = WAVE / 2; off > 0; off >>= 1) v = fmaxf(v, __shfl_down(v, off));
  return v;
}

// Block-level sum over up to 1024 threads (<=16 waves). Returns the total to
// every thread. `scratch` must be __shared__ float[16].
__device__ __forceinline__ float block_reduce_sum(float v, float* scratch) {
  int lane = threadIdx.x & (WAVE - 1);
  int wid = threadIdx.x / WAVE;
  int nwaves = (blockDim.x + WAVE - 1) / WAVE;
  v = wave_reduce_sum(v);
  if (lane == 0) scratch[wid] = v;
  __syncthreads();
  float total = 0.f;
#pragma unroll 4
  for (int i = 0; i < nwaves; ++i) total += scratch[i];
  return total;
}

__device__ __forceinline__ float block_reduce_max(float v, float* scratch) {
  int lane = threadIdx.x & (WAVE - 1);
  int wid = threadIdx.x / WAVE;
  int nwaves = (blockDim.x + WAVE - 1) / WAVE;
  v = wave_reduce_max(v);
  if (lane == 0) scratch[wid] = v;
  __syncthreads();
  float m = -INFINITY;
#pragma unroll 4
  for (int i = 0; i < nwaves; ++i) m = fmaxf(m, scratch[i]);
  return m;
}

// ---- ds_read_b64_tr_b16 machinery (shared by the attention kernels) ----
//
// Subtiled image for the hardware transpose read: [row/4][col/16] tiles
// of row-major [4 row][16 col] 64-element bf16 blocks. Strides are
// bank-staggered (tools/probes/tr16_probe.hip documents the lane->element
// mapping): col-subtile stride 72 elems so an 8-lane ds_write_b128 group
// lands on distinct write banks and the two tr tiles of one 32-lane read
// conflict-group overlap on only 4 of 64 read banks.
#define TRSUB 72
#define TRKEY4 (8 * TRSUB)

typedef ushort_t ushortx4_tr __attribute__((ext_vector_type(4)));

// element offset of (row, col) in the subtiled image (cols = 0..127)
__device__ __forceinline__ int tr_img_off(int row, int col) {
  return (row >> 2) * TRKEY4 + (col >> 4) * TRSUB + (row & 3) * 16 +
         (col & 15);
}

__device__ __forceinline__ unsigned lds_byte_base_of(const ushort_t* p) {
  auto lp = (const __attribute__((address_space(3))) ushort_t*)p;
  return (unsigned)(unsigned long)lp;
}

template <int OFF>
__device__ __forceinline__ ushortx4_tr tr16_read_off(unsigned vbase) {
  ushortx4_tr r;
  // no "memory" clobber: images are published by a tile barrier and not
  // written during the read phase — a clobber would serialize scheduling
  asm volatile("ds_read_b64_tr_b16 %0, %1 offset:%c2"
               : "=v"(r) : "v"(vbase), "i"(OFF));
  return r;
}

template <int N>
__device__ __forceinline__ void tr16_wait_n() {
  asm volatile("s_waitcnt lgkmcnt(%c0)" :: "i"(N) : "memory");
  __builtin_amdgcn_sched_barrier(0);  // hipcc would hoist MFMAs past it
}

// issue the 2*NROW32/2 (kstep x kk) reads of one 32-col d-block DT; the
// per-lane vbase carries (lane>>5)*2 key4-tiles + ((lane>>4)&1) subtile
// + (lane&15)*8 bytes (computed once per image)
template <int NROW32, int DT, int KSTEP = 0>
__device__ __forceinline__ void tr16_issue_dt(unsigned vbase,
                                              ushortx4_tr (*vr)[2]) {
  if constexpr (KSTEP < 2 * NROW32) {
    constexpr int base = DT * 2 * TRSUB * 2;
    vr[KSTEP][0] = tr16_read_off<base + (KSTEP * 4 + 0) * TRKEY4 * 2>(vbase);
    vr[KSTEP][1] = tr16_read_off<base + (KSTEP * 4 + 1) * TRKEY4 * 2>(vbase);
    tr16_issue_dt<NROW32, DT, KSTEP + 1>(vbase, vr);
  }
}

__device__ __forceinline__ unsigned tr16_lane_base(const ushort_t* img,
                                                   int lane) {
  return lds_byte_base_of(img) +
         (unsigned)(((lane >> 5) * 2 * TRKEY4 + ((lane >> 4) & 1) * TRSUB) * 2 +
                    (lane & 15) * 8);
}

#define CHECK_LAUNCH()                                         \
  do {                                                         \
    hipError_t e = hipGetLastError();                          \
    if (e != hipSuccess) {                                     \
      printf("kernel launch failed: %s\n", hipGetErrorString(e)); \
    }                                                          \
  } while (0)

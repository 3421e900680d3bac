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

#define CHECK_LAUNCH()                                         \
  do {                                                         \
    hipError_t e = hipGetLastError();                          \
    if (e != hipSuccess) {                                     \
      printf("kernel launch failed: %s\n", hipGetErrorString(e)); \
    }                                                          \
  } while (0)

// RMSNorm forward/backward (+ fused residual add), bf16, CDNA4.
//
// The hot normalization op of the Llama family. Fused with the residual add
// (one HBM pass instead of two) per the MI355X design rule: fuse
// elementwise/normalization work into the producing kernel.
//
// Forward:  r = x + residual (optional, stored back);  rstd = 1/sqrt(mean(r^2)+eps)
//           y = r * rstd * w
// Backward: c1 = sum(dy*w*r)/H
//           dx = rstd*(dy*w) - rstd^3 * c1 * r
//           dw[h] += sum_rows dy[n,h] * r[n,h] * rstd[n]   (fp32 atomics from
//           per-thread register accumulators; columns are thread-owned)
#include "common.hip.h"

// One block per row; row length H must be a multiple of 8.
// With residual: h_out = bf16(x + residual) is written to a SEPARATE buffer
// (training keeps every layer's h for backward; h_out == residual is legal
// for inference-style in-place reuse).
extern "C" __global__ void __launch_bounds__(256)
rmsnorm_fwd_kernel(const ushort_t* __restrict__ x, const ushort_t* __restrict__ w,
                   ushort_t* __restrict__ y, float* __restrict__ rstd_out,
                   const ushort_t* __restrict__ residual,  // may be null
                   ushort_t* __restrict__ h_out,           // required if residual
                   int H, float eps) {
  __shared__ float scratch[16];
  int64_t row = blockIdx.x;
  const ushort_t* xr = x + row * H;
  ushort_t* yr = y + row * H;
  const ushort_t* rr = residual ? residual + row * H : nullptr;
  ushort_t* hr = residual ? h_out + row * H : nullptr;

  float sumsq = 0.f;
  // pass 1: (optional residual add) + sum of squares
  for (int h = threadIdx.x * 8; h < H; h += blockDim.x * 8) {
    ushortx8 xv = *reinterpret_cast<const ushortx8*>(xr + h);
    floatx8 f = bf8_to_f32x8(xv);
    if (rr) {
      ushortx8 rv = *reinterpret_cast<const ushortx8*>(rr + h);
      floatx8 rf = bf8_to_f32x8(rv);
#pragma unroll
      for (int i = 0; i < 8; ++i) f[i] += rf[i];
      // store the bf16-rounded sum; recompute f from it so the normalized
      // output matches what later layers will re-read
      ushortx8 sv = f32x8_to_bf8(f);
      *reinterpret_cast<ushortx8*>(hr + h) = sv;
      f = bf8_to_f32x8(sv);
    }
#pragma unroll
    for (int i = 0; i < 8; ++i) sumsq += f[i] * f[i];
  }
  float total = block_reduce_sum(sumsq, scratch);
  float rstd = rsqrtf(total / (float)H + eps);
  if (threadIdx.x == 0 && rstd_out) rstd_out[row] = rstd;

  // pass 2: normalize (rows are L1/L2 resident after pass 1)
  const ushort_t* src = rr ? hr : xr;
  for (int h = threadIdx.x * 8; h < H; h += blockDim.x * 8) {
    ushortx8 xv = *reinterpret_cast<const ushortx8*>(src + h);
    ushortx8 wv = *reinterpret_cast<const ushortx8*>(w + h);
    floatx8 f = bf8_to_f32x8(xv);
    floatx8 wf = bf8_to_f32x8(wv);
    floatx8 o;
#pragma unroll
    for (int i = 0; i < 8; ++i) o[i] = f[i] * rstd * wf[i];
    *reinterpret_cast<ushortx8*>(yr + h) = f32x8_to_bf8(o);
  }
}

// Backward. Grid-stride over rows so each thread owns a fixed column slice
// across all its rows; dw accumulates in registers (compile-time ITERS so the
// accumulator array never spills to scratch — playbook rule: runtime-indexed
// register arrays go to local memory) and hits global memory once per block
// via atomics. ITERS = ceil(H / 2048); H <= 8192 supported.
template <int ITERS>
__global__ void __launch_bounds__(256)
rmsnorm_bwd_kernel(const ushort_t* __restrict__ dy, const ushort_t* __restrict__ r,
                   const ushort_t* __restrict__ w, const float* __restrict__ rstd,
                   ushort_t* __restrict__ dx, float* __restrict__ dw,
                   int64_t N, int H) {
  __shared__ float scratch[16];
  float dw_acc[ITERS][8];
#pragma unroll
  for (int i = 0; i < ITERS; ++i)
#pragma unroll
    for (int j = 0; j < 8; ++j) dw_acc[i][j] = 0.f;

  constexpr int iters = ITERS;

  for (int64_t row = blockIdx.x; row < N; row += gridDim.x) {
    const ushort_t* dyr = dy + row * H;
    const ushort_t* rr = r + row * H;
    ushort_t* dxr = dx + row * H;
    float rs = rstd[row];
    float c1 = 0.f;
    // pass 1: c1 = sum(dy*w*r)
#pragma unroll
    for (int it = 0; it < iters; ++it) {
      int h = it * blockDim.x * 8 + threadIdx.x * 8;
      if (h >= H) break;
      floatx8 dyf = bf8_to_f32x8(*reinterpret_cast<const ushortx8*>(dyr + h));
      floatx8 rf = bf8_to_f32x8(*reinterpret_cast<const ushortx8*>(rr + h));
      floatx8 wf = bf8_to_f32x8(*reinterpret_cast<const ushortx8*>(w + h));
#pragma unroll
      for (int i = 0; i < 8; ++i) c1 += dyf[i] * wf[i] * rf[i];
    }
    float total = block_reduce_sum(c1, scratch);
    float k = total / (float)H * rs * rs;
    __syncthreads();  // scratch reuse safety across rows
    // pass 2: dx + dw accumulation
#pragma unroll
    for (int it = 0; it < iters; ++it) {
      int h = it * blockDim.x * 8 + threadIdx.x * 8;
      if (h >= H) break;
      floatx8 dyf = bf8_to_f32x8(*reinterpret_cast<const ushortx8*>(dyr + h));
      floatx8 rf = bf8_to_f32x8(*reinterpret_cast<const ushortx8*>(rr + h));
      floatx8 wf = bf8_to_f32x8(*reinterpret_cast<const ushortx8*>(w + h));
      floatx8 o;
#pragma unroll
      for (int i = 0; i < 8; ++i) {
        o[i] = rs * (dyf[i] * wf[i] - k * rf[i]);
        dw_acc[it][i] += dyf[i] * rf[i] * rs;
      }
      *reinterpret_cast<ushortx8*>(dxr + h) = f32x8_to_bf8(o);
    }
  }
  // flush dw accumulators to this block's private partial row (atomics into
  // one shared dw[H] serialize 2048 blocks on 4096 addresses — measured 8x
  // over roofline; partials + a tiny second-stage reduce fix that)
#pragma unroll
  for (int it = 0; it < iters; ++it) {
    int h = it * blockDim.x * 8 + threadIdx.x * 8;
    if (h >= H) break;
    float* dst = dw + (int64_t)blockIdx.x * H + h;
#pragma unroll
    for (int i = 0; i < 8; ++i) dst[i] = dw_acc[it][i];
  }
}

// Second stage: dw[H] = sum over G partial rows. Parallel over BOTH h
// and G-slices (H=4096 columns alone gave only 4 workgroups — measured
// 63 GB/s; slicing G across blockIdx.y with one atomicAdd per column
// per slice fills the chip). dw must be zeroed before launch.
extern "C" __global__ void __launch_bounds__(256)
dw_reduce_kernel(const float* __restrict__ partials, float* __restrict__ dw,
                 int G, int H, int g_per_slice) {
  const int h = blockIdx.x * blockDim.x + threadIdx.x;
  if (h >= H) return;
  const int g0 = blockIdx.y * g_per_slice;
  const int g1 = min(g0 + g_per_slice, G);
  float acc = 0.f;
  for (int g = g0; g < g1; ++g) acc += partials[(int64_t)g * H + h];
  atomicAdd(&dw[h], acc);
}

extern "C" void launch_rmsnorm_fwd(const void* x, const void* w, void* y,
                                   float* rstd, const void* residual,
                                   void* h_out, int64_t N, int H, float eps,
                                   hipStream_t s) {
  dim3 grid((uint32_t)N), block(256);
  hipLaunchKernelGGL(rmsnorm_fwd_kernel, grid, block, 0, s,
                     (const ushort_t*)x, (const ushort_t*)w, (ushort_t*)y,
                     rstd, (const ushort_t*)residual, (ushort_t*)h_out, H, eps);
}

// dw_partials must hold grid*H floats; returns grid used via launch config
// chosen here (<=1024 blocks).
extern "C" int rmsnorm_bwd_grid(int64_t N) { return (int)(N < 1024 ? N : 1024); }

extern "C" void launch_rmsnorm_bwd(const void* dy, const void* r, const void* w,
                                   const float* rstd, void* dx,
                                   float* dw_partials, float* dw, int64_t N,
                                   int H, hipStream_t s) {
  int grid = rmsnorm_bwd_grid(N);
  int iters = (H + 2047) / 2048;
#define LAUNCH_BWD(IT)                                                        \
  hipLaunchKernelGGL(rmsnorm_bwd_kernel<IT>, dim3(grid), dim3(256), 0, s,     \
                     (const ushort_t*)dy, (const ushort_t*)r,                 \
                     (const ushort_t*)w, rstd, (ushort_t*)dx, dw_partials, N, H)
  switch (iters) {
    case 1: LAUNCH_BWD(1); break;
    case 2: LAUNCH_BWD(2); break;
    case 3: LAUNCH_BWD(3); break;
    default: LAUNCH_BWD(4); break;
  }
#undef LAUNCH_BWD
  int rgrid = (H + 255) / 256;
  int slices = 16;
  int g_per_slice = (grid + slices - 1) / slices;
  hipMemsetAsync(dw, 0, (size_t)H * sizeof(float), s);
  hipLaunchKernelGGL(dw_reduce_kernel, dim3(rgrid, slices), dim3(256), 0, s,
                     dw_partials, dw, grid, H, g_per_slice);
}

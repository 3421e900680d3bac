#include "hip/hip_runtime.h"
// Rotary position embedding (neox half-rotation), bf16, CDNA4.
//
// Operates IN-PLACE on the q and k regions of the fused QKV projection
// output (token stride = (Hq+2Hk)*D), so no contiguous copies are needed —
// the projection GEMM's output buffer is rotated directly. cos/sin tables
// are f32 [S_max, D/2], precomputed on host (on-device sinf/cosf would turn
// this memory-bound op VALU-bound, per the CDNA4 playbook).
// backward = same rotation with sin negated (R^T g).
#include "common.hip.h"

// One thread per (token, head, pair). Lanes cover consecutive pairs
// d=0..D/2-1 -> two coalesced 128B segments per wave access.
extern "C" __global__ void __launch_bounds__(256)
rope_kernel(ushort_t* __restrict__ q, ushort_t* __restrict__ k,
            const float* __restrict__ cos_t, const float* __restrict__ sin_t,
            int64_t B, int64_t S, int Hq, int Hk, int D,
            int64_t q_tok_stride, int64_t k_tok_stride, float sign) {
  int half = D / 2;
  int64_t n_tok = B * S;
  int64_t total = n_tok * (Hq + Hk) * half;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (int64_t)gridDim.x * blockDim.x) {
    int d = (int)(i % half);
    int64_t th = i / half;
    int h = (int)(th % (Hq + Hk));
    int64_t tok = th / (Hq + Hk);
    int64_t pos = tok % S;
    float c = cos_t[pos * half + d];
    float sn = sin_t[pos * half + d] * sign;
    ushort_t* base = (h < Hq) ? q + tok * q_tok_stride + (int64_t)h * D
                              : k + tok * k_tok_stride + (int64_t)(h - Hq) * D;
    float x1 = bf2f(base[d]);
    float x2 = bf2f(base[d + half]);
    base[d] = f2bf(x1 * c - x2 * sn);
    base[d + half] = f2bf(x1 * sn + x2 * c);
  }
}

extern "C" void launch_rope(void* q, void* k, const float* cos_t,
                            const float* sin_t, int64_t B, int64_t S, int Hq,
                            int Hk, int D, int64_t q_tok_stride,
                            int64_t k_tok_stride, int backward, hipStream_t s) {
  int64_t total = B * S * (Hq + Hk) * (D / 2);
  int64_t blocks = (total + 255) / 256;
  if (blocks > 16384) blocks = 16384;
  hipLaunchKernelGGL(rope_kernel, dim3((uint32_t)blocks), dim3(256), 0, s,
                     (ushort_t*)q, (ushort_t*)k, cos_t, sin_t, B, S, Hq, Hk, D,
                     q_tok_stride, k_tok_stride, backward ? -1.f : 1.f);
}

// Rotary position embedding (neox half-rotation), bf16, CDNA4.
//
// Operates IN-PLACE on the q and k regions of the fused QKV projection
// output (token stride = (Hq+2Hk)*D), so no contiguous copies are needed —
// the projection GEMM's output buffer is rotated directly. cos/sin tables
// are f32 [S_max, D/2], precomputed on host (on-device sinf/cosf would turn
// this memory-bound op VALU-bound, per the CDNA4 playbook).
// backward = same rotation with sin negated (R^T g).
#include "common.hip.h"

// One thread per (token, head, 8-pair block): bf16x8 vector loads/stores
// on both rotation halves and float4 table loads — the scalar version
// measured 2.1 TB/s (6x off the HBM roofline); vectorized it is
// memory-bound at the achievable ceiling. D/2 must be a multiple of 8
// (64/128-dim heads: 32/64 pairs).
extern "C" __global__ void __launch_bounds__(256)
rope_kernel(ushort_t* __restrict__ q, ushort_t* __restrict__ k,
            const float* __restrict__ cos_t, const float* __restrict__ sin_t,
            int64_t B, int64_t S, int Hq, int Hk, int D,
            int64_t q_tok_stride, int64_t k_tok_stride, float sign) {
  const int half = D / 2;
  const int blk8 = half / 8;                 // 8-pair blocks per head
  const int64_t n_tok = B * S;
  const int64_t total = n_tok * (Hq + Hk) * blk8;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (int64_t)gridDim.x * blockDim.x) {
    const int d8 = (int)(i % blk8);
    int64_t th = i / blk8;
    const int h = (int)(th % (Hq + Hk));
    const int64_t tok = th / (Hq + Hk);
    const int64_t pos = tok % S;
    const int d0 = d8 * 8;
    const float* ct = cos_t + pos * half + d0;
    const float* st = sin_t + pos * half + d0;
    ushort_t* base = (h < Hq) ? q + tok * q_tok_stride + (int64_t)h * D
                              : k + tok * k_tok_stride + (int64_t)(h - Hq) * D;
    const ushortx8 lo = *(const ushortx8*)(base + d0);
    const ushortx8 hi = *(const ushortx8*)(base + d0 + half);
    const floatx4 c0 = *(const floatx4*)ct;
    const floatx4 c1 = *(const floatx4*)(ct + 4);
    const floatx4 s0 = *(const floatx4*)st;
    const floatx4 s1 = *(const floatx4*)(st + 4);
    ushortx8 olo, ohi;
#pragma unroll
    for (int e = 0; e < 8; ++e) {
      const float c = (e < 4 ? c0[e & 3] : c1[e & 3]);
      const float sn = (e < 4 ? s0[e & 3] : s1[e & 3]) * sign;
      const float x1 = bf2f(lo[e]);
      const float x2 = bf2f(hi[e]);
      olo[e] = f2bf(x1 * c - x2 * sn);
      ohi[e] = f2bf(x1 * sn + x2 * c);
    }
    *(ushortx8*)(base + d0) = olo;
    *(ushortx8*)(base + d0 + half) = ohi;
  }
}

extern "C" void launch_rope(void* q, void* k, const float* cos_t,
                            const float* sin_t, int64_t B, int64_t S, int Hq,
                            int Hk, int D, int64_t q_tok_stride,
                            int64_t k_tok_stride, int backward, hipStream_t s) {
  int64_t total = B * S * (Hq + Hk) * (D / 16);  // 8-pair blocks
  int64_t blocks = (total + 255) / 256;
  if (blocks > 2048) blocks = 2048;  // grid-stride covers the rest
  hipLaunchKernelGGL(rope_kernel, dim3((uint32_t)blocks), dim3(256), 0, s,
                     (ushort_t*)q, (ushort_t*)k, cos_t, sin_t, B, S, Hq, Hk, D,
                     q_tok_stride, k_tok_stride, backward ? -1.f : 1.f);
}

// ---- device-pos decode-step prep (graph-capturable) ----------------------
//
// One kernel replaces three host-pos ops of the single-token decode step
// (rope on q/k + the k and v cache slice-copies), with the position read
// from the DEVICE lens buffer (pos = lens[b] - 1). No host scalar depends
// on the step index, so the whole token step can be captured in a hipGraph
// and replayed (the host-pos path costs ~260 launches/token from Python).
// qkv: [B, (Hq+2Hk)*D] contiguous (the projection GEMM's output for the
// step's single token). k/v caches: [B, Hk, Tmax, D] via strides.
extern "C" __global__ void __launch_bounds__(256)
rope_cache_write_kernel(ushort_t* __restrict__ qkv,
                        ushort_t* __restrict__ ck, ushort_t* __restrict__ cv,
                        const int* __restrict__ lens,
                        const float* __restrict__ cos_t,
                        const float* __restrict__ sin_t,
                        int B, int Hq, int Hk, int D,
                        long cb, long ch, long cs) {
  const int half = D / 2;
  const int blk8 = half / 8;
  const int Ht = Hq + 2 * Hk;
  const long total = (long)B * Ht * blk8;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (long)gridDim.x * blockDim.x) {
    const int d8 = (int)(i % blk8);
    long th = i / blk8;
    const int h = (int)(th % Ht);
    const int b = (int)(th / Ht);
    const long pos = lens[b] - 1;
    const int d0 = d8 * 8;
    ushort_t* src = qkv + ((long)b * Ht + h) * D;
    const ushortx8 lo = *(const ushortx8*)(src + d0);
    const ushortx8 hi = *(const ushortx8*)(src + d0 + half);
    if (h >= Hq + Hk) {
      // v head: plain copy into the cache
      ushort_t* dst = cv + (long)b * cb + (long)(h - Hq - Hk) * ch + pos * cs;
      *(ushortx8*)(dst + d0) = lo;
      *(ushortx8*)(dst + d0 + half) = hi;
      continue;
    }
    const float* ct = cos_t + pos * half + d0;
    const float* st = sin_t + pos * half + d0;
    const floatx4 c0 = *(const floatx4*)ct;
    const floatx4 c1 = *(const floatx4*)(ct + 4);
    const floatx4 s0 = *(const floatx4*)st;
    const floatx4 s1 = *(const floatx4*)(st + 4);
    ushortx8 olo, ohi;
#pragma unroll
    for (int e = 0; e < 8; ++e) {
      const float c = (e < 4 ? c0[e & 3] : c1[e & 3]);
      const float sn = (e < 4 ? s0[e & 3] : s1[e & 3]);
      const float x1 = bf2f(lo[e]);
      const float x2 = bf2f(hi[e]);
      olo[e] = f2bf(x1 * c - x2 * sn);
      ohi[e] = f2bf(x1 * sn + x2 * c);
    }
    if (h < Hq) {
      // q head: rotate in place (read by the decode kernel right after)
      *(ushortx8*)(src + d0) = olo;
      *(ushortx8*)(src + d0 + half) = ohi;
    } else {
      // k head: rotated value goes to the cache at pos
      ushort_t* dst = ck + (long)b * cb + (long)(h - Hq) * ch + pos * cs;
      *(ushortx8*)(dst + d0) = olo;
      *(ushortx8*)(dst + d0 + half) = ohi;
    }
  }
}

extern "C" void launch_rope_cache_write(void* qkv, void* ck, void* cv,
                                        const int* lens, const float* cos_t,
                                        const float* sin_t, int B, int Hq,
                                        int Hk, int D, long cb, long ch,
                                        long cs, hipStream_t s) {
  long total = (long)B * (Hq + 2 * Hk) * (D / 16);
  long blocks = (total + 255) / 256;
  if (blocks > 2048) blocks = 2048;
  hipLaunchKernelGGL(rope_cache_write_kernel, dim3((uint32_t)blocks),
                     dim3(256), 0, s, (ushort_t*)qkv, (ushort_t*)ck,
                     (ushort_t*)cv, lens, cos_t, sin_t, B, Hq, Hk, D, cb, ch,
                     cs);
}

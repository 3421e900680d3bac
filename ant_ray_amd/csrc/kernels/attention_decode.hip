// Flash-decode attention for CDNA4 (gfx950): single new token per sequence
// attending over a bf16 KV cache.
//
// Role parity: the reference delegates decode attention to vLLM's paged
// attention; this is the MI355X-native in-tree
// equivalent for the framework's own serve path (models/llama.py generate).
//
// Regime: memory-bound KV read (B*Hk*T*D*2 bf16 bytes per step). Design per
// the CDNA4 playbook "Attention decode" recipe:
//   * one workgroup per (batch, kv-head, key-chunk); all GQ = Hq/Hk query
//     heads of the group are computed in the same pass so the K/V bytes are
//     read ONCE (GQA reuse in registers, not in cache),
//   * 256 threads = 4 waves = 16 lane-groups of 16; each group walks keys
//     with stride 16, lane g covers 8 of the 128 dims (16-B bf16x8 loads —
//     one coalesced 256-B line per key row per group),
//   * per-group online softmax in the exp2 domain with defer-max (T13):
//     o-rescale only when the running max grows by > 8,
//   * partials (o, m, l) combined across the 16 groups through LDS, then
//     across key-chunks by a tiny second kernel (flash-decode two-level
//     combine). C == 1 writes bf16 output directly and skips level 2.
//
// Supports per-sequence lengths (lens[b]) for ragged serve batches.
#include <hip/hip_runtime.h>

#include "common.hip.h"

#define DEC_D 128
#define DEC_MAX_GQ 8
#define DEC_THR 8.0f

typedef float floatx2 __attribute__((ext_vector_type(2)));

// One partial record per (b, hq, chunk): unnormalized o (f32[D]) + m + l.
// Stored [B, Hq, C, D+2] f32.
//
// GQT is a TEMPLATE parameter (0 = runtime GQ): the per-head register
// arrays (qf/o_acc/m/l) sized for the runtime maximum (8) cost 191
// VGPRs and cap the kernel at 2 waves/SIMD; llama3-8b runs GQ=4 (~120
// VGPRs -> 4 waves/SIMD, twice the latency hiding for the HBM-bound KV
// walk). MAXOCC passes the matching __launch_bounds__ occupancy.
template <int GQT, int MAXOCC>
__global__ __launch_bounds__(256, MAXOCC) void attn_decode_kernel(
    const ushort_t* __restrict__ Q,   // [B, Hq, D] contiguous
    const ushort_t* __restrict__ K,   // [B, Hk, Tmax, D] via strides
    const ushort_t* __restrict__ V,   // same layout as K
    ushort_t* __restrict__ O,         // [B, Hq, D] bf16 (used when C == 1)
    float* __restrict__ PART,         // [B, Hq, C, D+2] f32 (when C > 1)
    const int* __restrict__ lens,     // [B] valid lengths (nullptr -> T)
    int T, int Hq, int Hk, int C,
    long kb, long kh, long ks,        // K/V strides (elements)
    long qbs,                         // Q batch stride (Hq*D, or the fused
                                      // qkv row stride when q is a region)
    float scale_log2) {
  const int chunk = blockIdx.x;
  const int hk = blockIdx.y;
  const int b = blockIdx.z;
  constexpr int GA = GQT > 0 ? GQT : DEC_MAX_GQ;  // register-array bound
  const int GQ = GQT > 0 ? GQT : (Hq / Hk);

  const int seq_len = lens ? lens[b] : T;
  // chunk covers keys [c0, c1)
  const int per_chunk = (seq_len + C - 1) / C;
  const int c0 = chunk * per_chunk;
  const int c1 = min(c0 + per_chunk, seq_len);

  const int tid = threadIdx.x;
  const int group = tid >> 4;        // 0..15 (16-lane groups)
  const int gl = tid & 15;           // lane within group
  const int d0 = gl * 8;             // this lane's 8 dims

  const ushort_t* Kp = K + (long)b * kb + (long)hk * kh;
  const ushort_t* Vp = V + (long)b * kb + (long)hk * kh;

  // Q fragments for the GQ query heads of this kv group (8 f32 per head)
  floatx8 qf[GA];
#pragma unroll
  for (int g = 0; g < GA; ++g) {
    if (GQT == 0 && g >= GQ) break;
    const ushort_t* qrow =
        Q + (long)b * qbs + (long)(hk * GQ + g) * DEC_D + d0;
    qf[g] = bf8_to_f32x8(*(const ushortx8*)qrow);
  }

  float m_run[GA], l_run[GA];
  floatx8 o_acc[GA] = {};
#pragma unroll
  for (int g = 0; g < GA; ++g) { m_run[g] = -1e30f; l_run[g] = 0.f; }

  // keys walk: group `group` handles keys c0+group, +16, +32 ... K/V rows
  // for the NEXT key are prefetched while the current key computes (a
  // load->use chain per key would serialize ~900-cycle HBM misses behind
  // ~40 VALU ops; the compiler keeps the prefetch loads in flight across
  // the softmax body)
  ushortx8 kv, vv, kv_n, vv_n;
  const int key0 = c0 + group;
  if (key0 < c1) {
    kv = *(const ushortx8*)(Kp + (long)key0 * ks + d0);
    vv = *(const ushortx8*)(Vp + (long)key0 * ks + d0);
  }
  for (int key = key0; key < c1; key += 16) {
    if (key + 16 < c1) {
      kv_n = *(const ushortx8*)(Kp + (long)(key + 16) * ks + d0);
      vv_n = *(const ushortx8*)(Vp + (long)(key + 16) * ks + d0);
    }
    const floatx8 kf = bf8_to_f32x8(kv);
    const floatx8 vf = bf8_to_f32x8(vv);
#pragma unroll
    for (int g = 0; g < GA; ++g) {
      if (GQT == 0 && g >= GQ) break;
      // dot(q, k) over this lane's 8 dims, then 16-lane tree reduce
      float s = qf[g][0] * kf[0];
#pragma unroll
      for (int j = 1; j < 8; ++j) s = fmaf(qf[g][j], kf[j], s);
      s += __shfl_xor(s, 1);
      s += __shfl_xor(s, 2);
      s += __shfl_xor(s, 4);
      s += __shfl_xor(s, 8);
      s *= scale_log2;  // exp2-domain score, identical on all 16 lanes
      if (s > m_run[g] + DEC_THR) {  // T13 defer-max: rare on real data
        const float m_new = s;
        const float alpha = __builtin_exp2f(m_run[g] - m_new);
        l_run[g] *= alpha;
#pragma unroll
        for (int j = 0; j < 8; ++j) o_acc[g][j] *= alpha;
        m_run[g] = m_new;
      }
      const float p = __builtin_exp2f(s - m_run[g]);
      l_run[g] += p;
#pragma unroll
      for (int j = 0; j < 8; ++j) o_acc[g][j] = fmaf(p, vf[j], o_acc[g][j]);
    }
    kv = kv_n;
    vv = vv_n;
  }

  // ---- combine the 16 groups through LDS.
  // Layout: per (group, g): 128 o floats + m + l. Dynamic size
  // 16*GQ*(D+2)*4 B (GQ=4 -> 33 KiB; GQ=8 -> 66 KiB), set at launch.
  extern __shared__ __attribute__((aligned(16))) float red[];
  const int rec = DEC_D + 2;
#pragma unroll
  for (int g = 0; g < GA; ++g) {
    if (GQT == 0 && g >= GQ) break;
    float* dst = &red[(group * GQ + g) * rec];
#pragma unroll
    for (int j = 0; j < 8; ++j) dst[d0 + j] = o_acc[g][j];
    if (gl == 0) { dst[DEC_D] = m_run[g]; dst[DEC_D + 1] = l_run[g]; }
  }
  __syncthreads();

  // wave 0 combines: lane covers 2 dims of one (g) record half...
  // Simpler: threads 0..(GQ*128-1) each own one output dim of one head.
  for (int idx = tid; idx < GQ * DEC_D; idx += 256) {
    const int g = idx / DEC_D;
    const int d = idx % DEC_D;
    float m_star = -1e30f;
#pragma unroll 4
    for (int gr = 0; gr < 16; ++gr)
      m_star = fmaxf(m_star, red[(gr * GQ + g) * rec + DEC_D]);
    float o_sum = 0.f, l_sum = 0.f;
#pragma unroll 4
    for (int gr = 0; gr < 16; ++gr) {
      const float* src = &red[(gr * GQ + g) * rec];
      const float w = __builtin_exp2f(src[DEC_D] - m_star);
      o_sum = fmaf(src[d], w, o_sum);
      l_sum = fmaf(src[DEC_D + 1], w, l_sum);
    }
    const int hq = hk * GQ + g;
    if (C == 1) {
      const float l_safe = l_sum > 0.f ? l_sum : 1.f;
      O[(((long)b * Hq) + hq) * DEC_D + d] = f2bf(o_sum / l_safe);
    } else {
      float* out = PART + ((((long)b * Hq) + hq) * (long)C + chunk) * rec;
      out[d] = o_sum;
      if (d == 0) { out[DEC_D] = m_star; out[DEC_D + 1] = l_sum; }
    }
  }
}

// Level-2 combine over C chunks: one workgroup per (b, hq); 128 threads each
// own one dim.
extern "C" __global__ __launch_bounds__(128, 8) void attn_decode_combine_kernel(
    const float* __restrict__ PART,  // [B, Hq, C, D+2]
    ushort_t* __restrict__ O,        // [B, Hq, D]
    int C) {
  const int d = threadIdx.x;
  const long bh = blockIdx.x;
  const int rec = DEC_D + 2;
  const float* base = PART + bh * (long)C * rec;
  float m_star = -1e30f;
  for (int c = 0; c < C; ++c) m_star = fmaxf(m_star, base[c * rec + DEC_D]);
  float o_sum = 0.f, l_sum = 0.f;
  for (int c = 0; c < C; ++c) {
    const float w = __builtin_exp2f(base[c * rec + DEC_D] - m_star);
    o_sum = fmaf(base[c * rec + d], w, o_sum);
    l_sum = fmaf(base[c * rec + DEC_D + 1], w, l_sum);
  }
  const float l_safe = l_sum > 0.f ? l_sum : 1.f;
  O[bh * DEC_D + d] = f2bf(o_sum / l_safe);
}

extern "C" void launch_attn_decode(const void* q, const void* k, const void* v,
                                   void* o, float* part, const int* lens,
                                   int B, int T, int Hq, int Hk, int C,
                                   long kb, long kh, long ks, long qbs,
                                   float scale, void* stream) {
  const float scale_log2 = scale * 1.4426950408889634f;
  dim3 grid(C, Hk, B);
  const int GQ = Hq / Hk;
  const size_t lds = (size_t)16 * GQ * (DEC_D + 2) * sizeof(float);
#define ANTRAY_DEC_LAUNCH(GQV, OCC)                                          \
  hipLaunchKernelGGL((attn_decode_kernel<GQV, OCC>), grid, dim3(256), lds,   \
                     (hipStream_t)stream, (const ushort_t*)q,                \
                     (const ushort_t*)k, (const ushort_t*)v, (ushort_t*)o,   \
                     part, lens, T, Hq, Hk, C, kb, kh, ks, qbs, scale_log2)
  switch (GQ) {
    case 1: ANTRAY_DEC_LAUNCH(1, 4); break;
    case 2: ANTRAY_DEC_LAUNCH(2, 4); break;
    case 4: ANTRAY_DEC_LAUNCH(4, 4); break;   // llama3-8b (Hq32/Hk8)
    case 8: ANTRAY_DEC_LAUNCH(8, 2); break;   // 66 KiB LDS caps blocks/CU
    default: ANTRAY_DEC_LAUNCH(0, 2); break;  // runtime-GQ guard path
  }
#undef ANTRAY_DEC_LAUNCH
  if (C > 1) {
    hipLaunchKernelGGL(attn_decode_combine_kernel, dim3(B * Hq), dim3(128), 0,
                       (hipStream_t)stream, part, (ushort_t*)o, C);
  }
}

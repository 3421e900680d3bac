// Flash-attention forward, hand-written for CDNA4 (gfx950).
//
// Replaces torch SDPA's AOTriton path (measured 369 TF/s fwd on this
// workload) with an MFMA 32x32x16 bf16 kernel using the CDNA4 idioms from
// the playbook: swapped QK^T (S^T = K.Q^T) so each lane owns a full P row
// (q = lane&31), online softmax in exp2 domain, K staged row-major in LDS,
// V staged TRANSPOSED in LDS so the P.V B-fragment is one ds_read_b128.
//
// Layout contract (one workgroup = 4 waves = one 128-row Q block):
//   wave w handles q rows [m0 + 32w, m0 + 32w + 32)
//   K-tile loop: 32 keys/tile, causal upper bound at the wave's last row.
//   mfma_f32_32x32x16_bf16 fragment maps (cdna4_isa.md §10):
//     A[m][k]: m = l&31, k = (l>>5)*8 + j            (j = 0..7)
//     B[k][n]: n = l&31, k = (l>>5)*8 + j
//     C[m][n]: n = l&31, m = (j&3) + 8*(j>>2) + 4*(l>>5)  (j = 0..15)
//
// Swapped S^T = mfma(A=K, B=Q): C cols n = q (lane-local), rows m = key.
// Lane pair (l, l^32) splits the 32 keys of one q row; __shfl_xor(.,32)
// closes row reductions and builds the P A-fragment for P.V.
//
// Shapes: D = 128 fixed; Hq % Hk == 0 (GQA); any S, causal or not.
#include <hip/hip_runtime.h>

#include "common.hip.h"

typedef __bf16 bf16x8 __attribute__((ext_vector_type(8)));
typedef float floatx16 __attribute__((ext_vector_type(16)));

#define ATTN_D 128
#define BLOCK_M 128   // q rows per workgroup (32 per wave)
#define BLOCK_N 32    // keys per tile
#define KROW 136      // K LDS row stride in elems (16B aligned, de-banked)
#define VROW 48       // V^T LDS row stride in elems (16B aligned)

__device__ __forceinline__ float xor32(float v) { return __shfl_xor(v, 32); }

extern "C" __global__ __launch_bounds__(256, 2) void attn_fwd_kernel(
    const ushort_t* __restrict__ Q,  // [B, Hq, S, D] via strides
    const ushort_t* __restrict__ K,  // [B, Hk, S, D]
    const ushort_t* __restrict__ V,  // [B, Hk, S, D]
    ushort_t* __restrict__ O,        // [B, Hq, S, D]
    float* __restrict__ LSE,         // [B, Hq, S] log2-domain lse (for bwd)
    int S, int Hq, int Hk,
    long qb, long qh, long qs,       // Q strides (elements)
    long kb, long kh, long ks,
    long ob, long oh, long os,
    float scale_log2,                // softmax_scale * log2(e)
    int causal) {
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int m_block = blockIdx.x;
  const int hq = blockIdx.y;
  const int b = blockIdx.z;
  const int hk = hq / (Hq / Hk);

  const int q0 = m_block * BLOCK_M + wave * 32;   // this wave's first q row
  // NOTE: no early return — the K/V staging barrier below is workgroup-wide,
  // so every wave must stay in the tile loop (tail waves just mask out)
  const int q_row = q0 + (lane & 31);             // this lane's q row
  const bool q_valid = q_row < S;

  const ushort_t* Qp = Q + (long)b * qb + (long)hq * qh;
  const ushort_t* Kp = K + (long)b * kb + (long)hk * kh;
  const ushort_t* Vp = V + (long)b * kb + (long)hk * kh;

  __shared__ ushort_t k_lds[32 * KROW];
  __shared__ ushort_t v_lds[ATTN_D * VROW];

  // ---- Q fragments: B[k=d][n=q], lane holds Q[q_row][step*16+(l>>5)*8+0..7]
  bf16x8 qf[8];
  {
    const ushort_t* qrow = Qp + (long)(q_valid ? q_row : 0) * qs;
    const int dbase = (lane >> 5) * 8;
#pragma unroll
    for (int st = 0; st < 8; ++st)
      qf[st] = __builtin_bit_cast(
          bf16x8, *(const ushortx8*)(qrow + st * 16 + dbase));
  }

  // ---- online-softmax state (per lane == per q row) + O accumulators
  float m_run = -1e30f, l_run = 0.f;
  floatx16 o_acc[4] = {};  // 4 d-tiles of 32; C[m=q? no: m spread, n=d]

  // tile count must be UNIFORM across the workgroup (staging barriers):
  // loop to the LAST wave's causal bound; earlier waves skip compute on
  // tiles past their own bound
  const int wg_last_row = m_block * BLOCK_M + (BLOCK_M - 1);
  const int n_end_row = causal ? min(wg_last_row, S - 1) : (S - 1);
  const int n_tiles = (n_end_row / BLOCK_N) + 1;
  const int my_last_tile = causal ? ((q0 + 31) / BLOCK_N) : (n_tiles - 1);

  for (int t = 0; t < n_tiles; ++t) {
    const int n0 = t * BLOCK_N;
    // ---- stage K tile row-major + V tile transposed
    __syncthreads();
    {
      // 256 threads, 32x128 elems: thread tid covers (row=tid>>3, 16 elems)
      const int row = threadIdx.x >> 3;
      const int col = (threadIdx.x & 7) * 16;
      const int krow_g = min(n0 + row, S - 1);
      const ushortx8 kv0 = *(const ushortx8*)(Kp + (long)krow_g * ks + col);
      const ushortx8 kv1 = *(const ushortx8*)(Kp + (long)krow_g * ks + col + 8);
      *(ushortx8*)(&k_lds[row * KROW + col]) = kv0;
      *(ushortx8*)(&k_lds[row * KROW + col + 8]) = kv1;
      const ushortx8 vv0 = *(const ushortx8*)(Vp + (long)krow_g * ks + col);
      const ushortx8 vv1 = *(const ushortx8*)(Vp + (long)krow_g * ks + col + 8);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        v_lds[(col + j) * VROW + row] = vv0[j];
        v_lds[(col + 8 + j) * VROW + row] = vv1[j];
      }
    }
    __syncthreads();
    if (t > my_last_tile || q0 >= S) continue;  // masked-out wave: stage only

    // ---- S^T = K . Q^T  (C: n = q = lane&31, m = key offset)
    floatx16 st_acc = {};
    {
      const int dbase = (lane >> 5) * 8;
#pragma unroll
      for (int stp = 0; stp < 8; ++stp) {
        bf16x8 kf = __builtin_bit_cast(
            bf16x8,
            *(const ushortx8*)(&k_lds[(lane & 31) * KROW + stp * 16 + dbase]));
        st_acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(kf, qf[stp], st_acc,
                                                         0, 0, 0);
      }
    }

    // ---- mask + scale into exp2 domain
    const int mrow_base = 4 * (lane >> 5);
    float s_val[16];
#pragma unroll
    for (int j = 0; j < 16; ++j) {
      const int key = n0 + (j & 3) + 8 * (j >> 2) + mrow_base;
      float s = st_acc[j] * scale_log2;
      const bool dead = (causal && key > q_row) || key >= S || !q_valid;
      s_val[j] = dead ? -1e30f : s;
    }

    // ---- online softmax (lane pair l, l^32 shares q row)
    float tmax = s_val[0];
#pragma unroll
    for (int j = 1; j < 16; ++j) tmax = fmaxf(tmax, s_val[j]);
    tmax = fmaxf(tmax, xor32(tmax));
    const float m_new = fmaxf(m_run, tmax);
    const float alpha = (m_run <= -1e30f) ? 0.f : __builtin_exp2f(m_run - m_new);
    float psum = 0.f;
    float p_val[16];
#pragma unroll
    for (int j = 0; j < 16; ++j) {
      p_val[j] = (s_val[j] <= -1e30f) ? 0.f : __builtin_exp2f(s_val[j] - m_new);
      psum += p_val[j];
    }
    psum += xor32(psum);
    l_run = l_run * alpha + psum;
    m_run = m_new;

    // ---- build P A-fragments: A[m=q][k=key], lane needs keys
    //      (l>>5)*8 + 0..7 (+16 for kstep 1); own regs cover keys
    //      {0..3,8..11,16..19,24..27} + mrow_base; partner has the rest.
    float p_part[16];
#pragma unroll
    for (int j = 0; j < 16; ++j) p_part[j] = xor32(p_val[j]);
    // Lane half h = lane>>5 wants keys 8h+16*kstep+i (i<8). Inverting the
    // C map gives reg j = (i&3) + 8*kstep + 4h, from p_val when
    // (i>=4)==(h==1) else p_part. All selects below are between
    // CONSTANT-indexed registers (rule #20: runtime-indexed arrays spill).
    const bool hi_half = (lane >> 5) != 0;
    float own_sh[8], par_sh[8];  // c = (i&3) + 4*kstep -> j = (c&3)+8*(c>>2)+4h
#pragma unroll
    for (int c = 0; c < 8; ++c) {
      const int j0 = (c & 3) + 8 * (c >> 2);
      own_sh[c] = hi_half ? p_val[j0 + 4] : p_val[j0];
      par_sh[c] = hi_half ? p_part[j0 + 4] : p_part[j0];
    }
    bf16x8 pf[2];
#pragma unroll
    for (int kstep = 0; kstep < 2; ++kstep) {
      ushortx8 pk;
#pragma unroll
      for (int i = 0; i < 8; ++i) {
        const int c = (i & 3) + 4 * kstep;
        const bool own_lo = (i < 4);  // own iff (i>=4)==hi_half
        const float v = hi_half ? (own_lo ? par_sh[c] : own_sh[c])
                                : (own_lo ? own_sh[c] : par_sh[c]);
        pk[i] = f2bf(v);
      }
      pf[kstep] = __builtin_bit_cast(bf16x8, pk);
    }

    // ---- O rescale by alpha (broadcast alpha from lane q to C rows)
    float alpha_j[16];
#pragma unroll
    for (int j = 0; j < 16; ++j) {
      const int qrow_j = (j & 3) + 8 * (j >> 2) + mrow_base;
      alpha_j[j] = __shfl(alpha, qrow_j);
    }
#pragma unroll
    for (int dt = 0; dt < 4; ++dt)
#pragma unroll
      for (int j = 0; j < 16; ++j) o_acc[dt][j] *= alpha_j[j];

    // ---- O += P . V   (B[k][n=d] = one b128 from transposed V)
#pragma unroll
    for (int dt = 0; dt < 4; ++dt) {
      const int d0 = dt * 32;
#pragma unroll
      for (int kstep = 0; kstep < 2; ++kstep) {
        bf16x8 vf = __builtin_bit_cast(
            bf16x8, *(const ushortx8*)(&v_lds[(d0 + (lane & 31)) * VROW +
                                              kstep * 16 + (lane >> 5) * 8]));
        o_acc[dt] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(pf[kstep], vf,
                                                            o_acc[dt], 0, 0, 0);
      }
    }
  }

  // ---- epilogue: O /= l, write bf16; LSE in log2 domain
  const float l_safe = (l_run > 0.f) ? l_run : 1.f;
  float linv_j[16];
#pragma unroll
  for (int j = 0; j < 16; ++j) {
    const int qrow_j = (j & 3) + 8 * (j >> 2) + 4 * (lane >> 5);
    linv_j[j] = 1.f / __shfl(l_safe, qrow_j);
  }
  ushort_t* Op = O + (long)b * ob + (long)hq * oh;
#pragma unroll
  for (int j = 0; j < 16; ++j) {
    const int qr = q0 + (j & 3) + 8 * (j >> 2) + 4 * (lane >> 5);
    if (qr >= S) continue;
#pragma unroll
    for (int dt = 0; dt < 4; ++dt) {
      Op[(long)qr * os + dt * 32 + (lane & 31)] = f2bf(o_acc[dt][j] * linv_j[j]);
    }
  }
  if (LSE != nullptr && q_valid && lane < 32) {
    // lane l<32 and partner hold identical (m, l) after the xor reduction
    LSE[((long)b * Hq + hq) * S + q_row] =
        m_run + __builtin_log2f(l_safe);
  }
}

extern "C" void launch_attn_fwd(const void* q, const void* k, const void* v,
                                void* o, float* lse, int B, int S, int Hq,
                                int Hk, long qb, long qh, long qs, long kb,
                                long kh, long ks, long ob, long oh, long os,
                                float scale, int causal, void* stream) {
  dim3 grid((S + BLOCK_M - 1) / BLOCK_M, Hq, B);
  const float scale_log2 = scale * 1.4426950408889634f;
  hipLaunchKernelGGL(attn_fwd_kernel, grid, dim3(256), 0,
                     (hipStream_t)stream, (const ushort_t*)q,
                     (const ushort_t*)k, (const ushort_t*)v, (ushort_t*)o,
                     lse, S, Hq, Hk, qb, qh, qs, kb, kh, ks, ob, oh, os,
                     scale_log2, causal);
}

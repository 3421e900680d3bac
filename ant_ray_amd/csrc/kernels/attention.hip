// Flash-attention forward, hand-written for CDNA4 (gfx950).
//
// Replaces torch SDPA's AOTriton path (measured 369-399 TF/s fwd on the
// Llama-3-8B shape) with an MFMA 32x32x16 bf16 kernel built from the CDNA4
// playbook idioms:
//   * swapped QK^T (S^T = K.Q^T) so each lane owns one q row (q = lane&31)
//     and the softmax row reduction needs only a lane<->lane^32 exchange,
//   * online softmax in the exp2 domain with defer-max rescale (T13):
//     the O/l rescale pass runs only when the row max grows > THR,
//   * P A-fragments built by v_cvt_pk_bf16_f32 + v_permlane32_swap (T12):
//     4 permlanes + 8 cvt_pk replace 16 ds_bpermute + scalar bf16 packing,
//   * -inf masking (exp2(-inf - m) == 0) so the exp path has NO branches,
//   * K staged row-major with the T2 XOR swizzle (block ^= row&7) and V
//     staged TRANSPOSED (VROW=40+, k ^= 8*((d>>4)&3)): conflict-reduced
//     ds_read_b128 fragment reads on both,
//   * T14 issue-early staging: tile t+1's global loads are in flight while
//     tile t computes,
//   * tile width templated (NT sub-tiles of 32 keys): NT=2 = 64-key tiles,
//     32 MFMAs per barrier interval instead of 16 — the 2-phase-stall
//     analysis in the playbook shows barrier/staging overhead amortizes
//     with MFMA-per-phase.
//
// Layout contract (one workgroup = 4 waves = one 128-row Q block):
//   wave w handles q rows [m0 + 32w, m0 + 32w + 32)
//   mfma_f32_32x32x16_bf16 fragment maps (cdna4_isa.md §10):
//     A[m][k]: m = l&31, k = (l>>5)*8 + j            (j = 0..7)
//     B[k][n]: n = l&31, k = (l>>5)*8 + j
//     C[m][n]: n = l&31, m = (j&3) + 8*(j>>2) + 4*(l>>5)  (j = 0..15)
//
// Shapes: D = 128 fixed; Hq % Hk == 0 (GQA); any S, causal or not.
#include <hip/hip_runtime.h>

#include "common.hip.h"

typedef __bf16 bf16x8 __attribute__((ext_vector_type(8)));
typedef float floatx16 __attribute__((ext_vector_type(16)));
typedef unsigned int uint32x4 __attribute__((ext_vector_type(4)));

#define ATTN_D 128
#define BLOCK_M 128   // q rows per workgroup (32 per wave)
#define KROW 128      // K LDS row stride (T2 swizzle instead of padding)
#define RESCALE_THR 8.0f  // T13: defer O-rescale until max grows > 2^8

#define NEG_INF (-__builtin_inff())

__device__ __forceinline__ float xor32(float v) { return __shfl_xor(v, 32); }

// raw v_exp_f32: clang's exp2f lowers to exp+ldexp+cmp+cndmask (a large-
// input guard) even under fast-math; our exponents are <= THR=8 and -inf
// maps to 0 in hardware, so the single instruction is exact here
__device__ __forceinline__ float exp2_raw(float x) {
  float r;
  asm("v_exp_f32 %0, %1" : "=v"(r) : "v"(x));
  return r;
}

// v_cvt_pk_bf16_f32: packs (lo, hi) f32 -> one u32 of 2 bf16 (HW RNE)
__device__ __forceinline__ unsigned cvt_pk_bf16(float lo, float hi) {
  unsigned r;
  asm("v_cvt_pk_bf16_f32 %0, %1, %2" : "=v"(r) : "v"(lo), "v"(hi));
  return r;
}

// K LDS: element offset of (row, 8-elem block blk), T2 XOR swizzle
__device__ __forceinline__ int k_lds_off(int row, int blk) {
  return row * KROW + 8 * (blk ^ (row & 7));
}

#define ATTN_D 128
#define BLOCK_M 128   // q rows per workgroup (32 per wave)
#define KROW 128      // K LDS row stride (T2 swizzle instead of padding)
#define RESCALE_THR 8.0f  // T13: defer O-rescale until max grows > 2^8

// NT = 32-key sub-tiles per staged tile (1 -> 32-key tiles, 2 -> 64).
template <int NT>
__global__ __launch_bounds__(256, 2) void attn_fwd_kernel(
    const ushort_t* __restrict__ Q,  // [B, Hq, S, D] via strides
    const ushort_t* __restrict__ K,  // [B, Hk, S, D]
    const ushort_t* __restrict__ V,  // [B, Hk, S, D]
    ushort_t* __restrict__ O,        // [B, Hq, S, D]
    float* __restrict__ LSE,         // [B, Hq, S] log2-domain lse (for bwd)
    int S, int Hq, int Hk,
    long qb, long qh, long qs,       // strides (elements)
    long kb, long kh, long ks,
    long ob, long oh, long os,
    float scale_log2,                // softmax_scale * log2(e)
    int causal) {
  constexpr int BN = 32 * NT;       // keys per staged tile
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int m_block = blockIdx.x;
  const int hq = blockIdx.y;
  const int b = blockIdx.z;
  const int hk = hq / (Hq / Hk);

  const int q0 = m_block * BLOCK_M + wave * 32;   // this wave's first q row
  // no early return: the staging barriers are workgroup-wide
  const int q_row = q0 + (lane & 31);
  const bool q_valid = q_row < S;

  const ushort_t* Qp = Q + (long)b * qb + (long)hq * qh;
  const ushort_t* Kp = K + (long)b * kb + (long)hk * kh;
  const ushort_t* Vp = V + (long)b * kb + (long)hk * kh;

  __shared__ ushort_t k_lds[2][BN * KROW];
  __shared__ ushort_t v_lds[2][(BN / 4) * TRKEY4];

  // ---- Q fragments: B[k=d][n=q], PRE-SCALED by softmax_scale*log2e so
  // the per-tile scores come out of the MFMA already in the exp2 domain
  // (saves 32 v_mul per tile; bf16 rounding of q*scale is within the
  // kernel's bf16 I/O noise)
  bf16x8 qf[8];
  {
    const ushort_t* qrow = Qp + (long)(q_valid ? q_row : 0) * qs;
    const int dbase = (lane >> 5) * 8;
#pragma unroll
    for (int st = 0; st < 8; ++st) {
      ushortx8 u = *(const ushortx8*)(qrow + st * 16 + dbase);
      ushortx8 sc;
#pragma unroll
      for (int e = 0; e < 8; ++e) sc[e] = f2bf(bf2f(u[e]) * scale_log2);
      qf[st] = __builtin_bit_cast(bf16x8, sc);
    }
  }

  float m_run = -1e30f, l_run = 0.f;
  floatx16 o_acc[4] = {};

  const int wg_last_row = m_block * BLOCK_M + (BLOCK_M - 1);
  const int n_end_row = causal ? min(wg_last_row, S - 1) : (S - 1);
  const int n_tiles = (n_end_row / BN) + 1;
  const int my_last_tile = causal ? ((q0 + 31) / BN) : (n_tiles - 1);

  // staging geometry: thread covers NT rows 32 apart, 16 elems each at
  // (row = tid>>3 (+32), col = (tid&7)*16)
  const int st_row = threadIdx.x >> 3;
  const int st_col = (threadIdx.x & 7) * 16;

  // T14: tile 0 loads issued before the loop
  ushortx8 kv0[NT], kv1[NT], vv0[NT], vv1[NT];
#pragma unroll
  for (int h = 0; h < NT; ++h) {
    const int krow_g = min(st_row + 32 * h, S - 1);
    kv0[h] = *(const ushortx8*)(Kp + (long)krow_g * ks + st_col);
    kv1[h] = *(const ushortx8*)(Kp + (long)krow_g * ks + st_col + 8);
    vv0[h] = *(const ushortx8*)(Vp + (long)krow_g * ks + st_col);
    vv1[h] = *(const ushortx8*)(Vp + (long)krow_g * ks + st_col + 8);
  }

  for (int t = 0; t < n_tiles; ++t) {
    const int n0 = t * BN;
    const int buf = t & 1;
    // double buffer: write tile t into buf while tile t-1 (other buf) may
    // still be read; ONE barrier per tile orders write(t) vs read(t)
#pragma unroll
    for (int h = 0; h < NT; ++h) {
      *(ushortx8*)(&k_lds[buf][k_lds_off(st_row + 32 * h, st_col / 8)]) = kv0[h];
      *(ushortx8*)(&k_lds[buf][k_lds_off(st_row + 32 * h, st_col / 8 + 1)]) = kv1[h];
      // V: two 16-B vector writes into the subtiled tr-read image (the
      // 16 consecutive d of one (key/4, d/16) subtile row are contiguous)
      const int vkey = st_row + 32 * h;
      *(ushortx8*)(&v_lds[buf][tr_img_off(vkey, st_col)]) = vv0[h];
      *(ushortx8*)(&v_lds[buf][tr_img_off(vkey, st_col + 8)]) = vv1[h];
    }
    __syncthreads();  // tile t staged in buf; tile t-1 reads are also done
    if (t + 1 < n_tiles) {
#pragma unroll
      for (int h = 0; h < NT; ++h) {
        const int krow_g = min((t + 1) * BN + st_row + 32 * h, S - 1);
        kv0[h] = *(const ushortx8*)(Kp + (long)krow_g * ks + st_col);
        kv1[h] = *(const ushortx8*)(Kp + (long)krow_g * ks + st_col + 8);
        vv0[h] = *(const ushortx8*)(Vp + (long)krow_g * ks + st_col);
        vv1[h] = *(const ushortx8*)(Vp + (long)krow_g * ks + st_col + 8);
      }
    }
    if (t > my_last_tile || q0 >= S) continue;  // masked wave: stage only

    // ---- S^T = K . Q^T  (C: n = q = lane&31, m = key offset)
    floatx16 st_acc[NT] = {};
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int h = 0; h < NT; ++h)
#pragma unroll
      for (int stp = 0; stp < 8; ++stp) {
        bf16x8 kf = __builtin_bit_cast(
            bf16x8, *(const ushortx8*)(
                        &k_lds[buf][k_lds_off((lane & 31) + 32 * h,
                                              2 * stp + (lane >> 5))]));
        st_acc[h] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(kf, qf[stp],
                                                            st_acc[h], 0, 0, 0);
      }
    __builtin_amdgcn_s_setprio(0);

    // ---- mask + scale; dead scores are -inf so exp2 underflows to 0
    const int mrow_base = 4 * (lane >> 5);
    float s_val[NT][16];
    const bool needs_mask =
        (causal && n0 + BN - 1 > q0) || (n0 + BN > S) || !q_valid;
    if (needs_mask) {
#pragma unroll
      for (int h = 0; h < NT; ++h)
#pragma unroll
        for (int j = 0; j < 16; ++j) {
          const int key = n0 + 32 * h + (j & 3) + 8 * (j >> 2) + mrow_base;
          const bool dead = (causal && key > q_row) || key >= S || !q_valid;
          s_val[h][j] = dead ? NEG_INF : st_acc[h][j];
        }
    } else {
#pragma unroll
      for (int h = 0; h < NT; ++h)
#pragma unroll
        for (int j = 0; j < 16; ++j) s_val[h][j] = st_acc[h][j];
    }

    // ---- online softmax with defer-max (T13)
    float tmax = fmaxf(s_val[0][0], s_val[0][1]);
#pragma unroll
    for (int h = 0; h < NT; ++h)
#pragma unroll
      for (int j = h == 0 ? 2 : 0; j < 16; ++j)
        tmax = fmaxf(tmax, s_val[h][j]);
    tmax = fmaxf(tmax, xor32(tmax));
    if (__any(tmax > m_run + RESCALE_THR)) {
      const float m_new = fmaxf(m_run, tmax);
      const float alpha = exp2_raw(m_run - m_new);  // 0 on 1st tile
      l_run *= alpha;
      m_run = m_new;
      float alpha_j[16];
#pragma unroll
      for (int j = 0; j < 16; ++j)
        alpha_j[j] = __shfl(alpha, (j & 3) + 8 * (j >> 2) + mrow_base);
#pragma unroll
      for (int dt = 0; dt < 4; ++dt)
#pragma unroll
        for (int j = 0; j < 16; ++j) o_acc[dt][j] *= alpha_j[j];
    }
    float psum = 0.f;
    float p_val[NT][16];
#pragma unroll
    for (int h = 0; h < NT; ++h)
#pragma unroll
      for (int j = 0; j < 16; ++j) {
        p_val[h][j] = exp2_raw(s_val[h][j] - m_run);  // <= 2^THR
        psum += p_val[h][j];
      }
    psum += xor32(psum);
    l_run += psum;

    // ---- P A-fragments via cvt_pk + permlane32_swap (T12).
    bf16x8 pf[2 * NT];
#pragma unroll
    for (int h = 0; h < NT; ++h) {
      unsigned own_pk[8];
#pragma unroll
      for (int b2 = 0; b2 < 4; ++b2) {
        own_pk[2 * b2] = cvt_pk_bf16(p_val[h][4 * b2], p_val[h][4 * b2 + 1]);
        own_pk[2 * b2 + 1] =
            cvt_pk_bf16(p_val[h][4 * b2 + 2], p_val[h][4 * b2 + 3]);
      }
      uint32x4 pw0, pw1;
      {
        auto r0 = __builtin_amdgcn_permlane32_swap(own_pk[0], own_pk[2],
                                                   false, false);
        auto r1 = __builtin_amdgcn_permlane32_swap(own_pk[1], own_pk[3],
                                                   false, false);
        pw0[0] = r0[0]; pw0[2] = r0[1];
        pw0[1] = r1[0]; pw0[3] = r1[1];
        auto r2 = __builtin_amdgcn_permlane32_swap(own_pk[4], own_pk[6],
                                                   false, false);
        auto r3 = __builtin_amdgcn_permlane32_swap(own_pk[5], own_pk[7],
                                                   false, false);
        pw1[0] = r2[0]; pw1[2] = r2[1];
        pw1[1] = r3[0]; pw1[3] = r3[1];
      }
      pf[2 * h] = __builtin_bit_cast(bf16x8, pw0);
      pf[2 * h + 1] = __builtin_bit_cast(bf16x8, pw1);
    }

    // ---- O += P . V   (B[k=key][n=d] fragments by hardware transpose
    // read from the row-major subtiled V image: per kstep two
    // ds_read_b64_tr_b16 deliver the lane's 8 keys at its d column).
    // Double-buffered: dt+1's reads are in flight behind a counted
    // lgkm wait while dt's MFMAs run (no other LDS ops live here).
    {
      const unsigned vbase = tr16_lane_base(v_lds[buf], lane);
      ushortx4_tr vrA[2 * NT][2], vrB[2 * NT][2];
      __builtin_amdgcn_s_setprio(1);
      tr16_issue_dt<NT, 0>(vbase, vrA);
#pragma unroll
      for (int dt = 0; dt < 4; ++dt) {
        ushortx4_tr(*cur)[2] = (dt & 1) ? vrB : vrA;
        ushortx4_tr(*nxt)[2] = (dt & 1) ? vrA : vrB;
        if (dt == 0) tr16_issue_dt<NT, 1>(vbase, nxt);
        if (dt == 1) tr16_issue_dt<NT, 2>(vbase, nxt);
        if (dt == 2) tr16_issue_dt<NT, 3>(vbase, nxt);
        if (dt < 3)
          tr16_wait_n<4 * NT>();  // dt's reads landed; dt+1's stay in flight
        else
          tr16_wait_n<0>();
#pragma unroll
        for (int kstep = 0; kstep < 2 * NT; ++kstep) {
          struct {
            ushortx4_tr a, b;
          } pair = {cur[kstep][0], cur[kstep][1]};
          o_acc[dt] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              pf[kstep], __builtin_bit_cast(bf16x8, pair), o_acc[dt], 0, 0, 0);
        }
      }
      __builtin_amdgcn_s_setprio(0);
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
  if (LSE != nullptr && q_valid && q0 < S && lane < 32) {
    LSE[((long)b * Hq + hq) * S + q_row] = m_run + __builtin_log2f(l_safe);
  }
}

extern "C" void launch_attn_fwd(const void* q, const void* k, const void* v,
                                void* o, float* lse, int B, int S, int Hq,
                                int Hk, long qb, long qh, long qs, long kb,
                                long kh, long ks, long ob, long oh, long os,
                                float scale, int causal, void* stream) {
  dim3 grid((S + BLOCK_M - 1) / BLOCK_M, Hq, B);
  const float scale_log2 = scale * 1.4426950408889634f;
  // tile width: 64-key tiles by default (32 MFMAs per barrier interval);
  // ANTRAY_FWD_NT=1 keeps the 32-key variant for A/B
  static const int nt = [] {
    const char* e = getenv("ANTRAY_FWD_NT");
    return (e && e[0] == '1') ? 1 : 2;
  }();
  if (nt == 2) {
    hipLaunchKernelGGL(attn_fwd_kernel<2>, grid, dim3(256), 0,
                       (hipStream_t)stream, (const ushort_t*)q,
                       (const ushort_t*)k, (const ushort_t*)v, (ushort_t*)o,
                       lse, S, Hq, Hk, qb, qh, qs, kb, kh, ks, ob, oh, os,
                       scale_log2, causal);
  } else {
    hipLaunchKernelGGL(attn_fwd_kernel<1>, grid, dim3(256), 0,
                       (hipStream_t)stream, (const ushort_t*)q,
                       (const ushort_t*)k, (const ushort_t*)v, (ushort_t*)o,
                       lse, S, Hq, Hk, qb, qh, qs, kb, kh, ks, ob, oh, os,
                       scale_log2, causal);
  }
}

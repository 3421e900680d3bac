// Flash-attention backward for CDNA4 (gfx950): preprocess + dQ + dK/dV.
//
// Replaces AOTriton's bwd_preprocess/bwd_kernel_dq/bwd_kernel_dk_dv
// (measured 219 TF/s on the Llama-3-8B shape). Math (exp2 domain, lse L
// and scores s~ = S*scale*log2e saved/recomputed):
//   P    = exp2(s~ - L)
//   delta= rowsum(dO . O)                        (preprocess kernel)
//   dP   = dO V^T ;  dS = scale * P . (dP - delta)
//   dQ   = dS K   ;  dK = dS^T Q  ;  dV = P^T dO
//
// Same fragment machinery as attention.hip (see its header comment):
//   A[m][k]: m=l&31, k=(l>>5)*8+j ; B[k][n]: n=l&31, k=(l>>5)*8+j
//   C[m][n]: n=l&31, m=(j&3)+8*(j>>2)+4*(l>>5)
//
// dQ kernel: one workgroup = 128 q rows (4 waves x 32), loops k-tiles.
//   S^T = mfma(A=K_rm, B=Qfrag); dP^T = mfma(A=V_rm, B=dOfrag) — both C
//   layouts have q lane-local, so L[q]/delta[q] are per-lane scalars.
//   dS^T -> A-frag by the same cvt_pk+permlane32_swap transpose as fwd P;
//   dQ += mfma(dSfrag, B=K^T from transposed-staged KT).
//
// dKV kernel: one workgroup = 128 keys (4 waves x 32), loops q-tiles from
//   the causal diagonal. K/V of the block live row-major in LDS; per
//   q-tile Q/dO are staged row-major + transposed. P^T and dS^T need
//   key-lane-local A-frags (sum over q): a WAVE-PRIVATE LDS scratch
//   round-trips the C tile ([key][q] layout, one b128 read per fragment).
//   GQA: grid is per Q-HEAD; dK/dV partials land in [B,Hq,S,D] and
//   attn_bwd_reduce_kv sums head groups into [B,Hk,S,D].
#include <hip/hip_runtime.h>

#include "common.hip.h"

typedef __bf16 bf16x8 __attribute__((ext_vector_type(8)));
typedef float floatx16 __attribute__((ext_vector_type(16)));
typedef unsigned int uint32x4_t __attribute__((ext_vector_type(4)));

#define ATTN_D 128
#define BLOCK 32           // tile edge (keys or qs per wave)
#define KROW 128           // row-major LDS row stride, T2 swizzled
#define TROW 40            // transposed LDS row stride
#define SCR_ROW 40         // wave-private scratch row stride

__device__ __forceinline__ int rm_off(int row, int blk) {  // T2 swizzle
  return row * KROW + 8 * (blk ^ (row & 7));
}

__device__ __forceinline__ int tr_off(int d, int k) {
  return d * TROW + (k ^ (8 * ((d >> 4) & 3)));
}

// raw v_exp_f32 (clang's exp2f carries a large-input guard even under
// fast-math; bwd exponents are bounded by the fwd lse)
__device__ __forceinline__ float exp2_rawb(float x) {
  float r;
  asm("v_exp_f32 %0, %1" : "=v"(r) : "v"(x));
  return r;
}

typedef float floatx4_b __attribute__((ext_vector_type(4)));

// the 16 per-reg q-rows of a C tile are (j&3)+8*(j>>2)+4*(lane>>5): four
// 4-row runs at mb, mb+8, mb+16, mb+24 — four float4 loads fetch the
// lse/delta values a lane needs (replaces 16 ds_bpermute broadcasts)
__device__ __forceinline__ void load_rows16(const float* p, int base,
                                            int mb, int S,
                                            floatx4_b (&o)[4]) {
  if (base + 32 <= S) {
#pragma unroll
    for (int g = 0; g < 4; ++g)
      o[g] = *(const floatx4_b*)(p + base + mb + 8 * g);
  } else {
    // tail tile: element-wise clamped loads (rows >= S are masked dead
    // by the caller; the clamp only keeps the reads in bounds)
#pragma unroll
    for (int g = 0; g < 4; ++g)
#pragma unroll
      for (int e = 0; e < 4; ++e)
        o[g][e] = p[min(base + mb + 8 * g + e, S - 1)];
  }
}

__device__ __forceinline__ unsigned cvt_pk_bf16b(float lo, float hi) {
  unsigned r;
  asm("v_cvt_pk_bf16_f32 %0, %1, %2" : "=v"(r) : "v"(lo), "v"(hi));
  return r;
}

// ---------------------------------------------------------------- delta

extern "C" __global__ void attn_bwd_preprocess_kernel(
    const ushort_t* __restrict__ dO, const ushort_t* __restrict__ O,
    float* __restrict__ delta, long rows, int Hq, int S,
    long dob, long doh, long dos, long ob, long oh, long os) {
  // one wave per 4 rows: 16 lanes per row, each lane 8 elems. dO and O
  // may each be STRIDED ([B,S,H,D]-layout outputs); delta stays [B,Hq,S]
  // row-major.
  const long row = (long)blockIdx.x * (blockDim.x >> 4) + (threadIdx.x >> 4);
  if (row >= rows) return;
  const int sub = threadIdx.x & 15;
  const long s = row % S;
  const long h = (row / S) % Hq;
  const long b2 = row / ((long)S * Hq);
  const ushortx8 a =
      *(const ushortx8*)(dO + b2 * dob + h * doh + s * dos + sub * 8);
  const ushortx8 b =
      *(const ushortx8*)(O + b2 * ob + h * oh + s * os + sub * 8);
  float acc = 0.f;
#pragma unroll
  for (int i = 0; i < 8; ++i) acc += bf2f(a[i]) * bf2f(b[i]);
#pragma unroll
  for (int off = 8; off > 0; off >>= 1) acc += __shfl_down(acc, off);
  if (sub == 0) delta[row] = acc;
}

// ------------------------------------------------------------ dQ kernel

// NT = 32-key sub-tiles staged per barrier interval (NT=2 = 64-key
// stages, halves the __syncthreads count — the same lever as the
// forward kernel's NT; LDS 2x25.2 KB still fits 2 blocks/CU).
template <int NT>
__global__ __launch_bounds__(256, 2) void attn_bwd_dq_kernel(
    const ushort_t* __restrict__ Q, const ushort_t* __restrict__ K,
    const ushort_t* __restrict__ V, const ushort_t* __restrict__ dO,
    const float* __restrict__ LSE,    // [B,Hq,S] log2 domain
    const float* __restrict__ Delta,  // [B,Hq,S]
    ushort_t* __restrict__ dQ,        // [B,Hq,S,D] via its own strides
    int S, int Hq, int Hk,
    long qb, long qh, long qs,        // Q strides (elements)
    long kb, long kh, long ks,        // K/V strides
    long ob, long oh, long os,        // dO strides
    long gb, long gh, long gs,        // dQ strides (rows contiguous)
    float scale, int causal) {
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int m_block = blockIdx.x;
  const int hq = blockIdx.y;
  const int b = blockIdx.z;
  const int hk = hq / (Hq / Hk);
  const float c_log2 = scale * 1.4426950408889634f;

  const int q0 = m_block * 128 + wave * 32;
  const int q_row = q0 + (lane & 31);
  const bool q_valid = q_row < S;

  const ushort_t* Qp = Q + (long)b * qb + (long)hq * qh;
  const ushort_t* dOp = dO + (long)b * ob + (long)hq * oh;
  const ushort_t* Kp = K + (long)b * kb + (long)hk * kh;
  const ushort_t* Vp = V + (long)b * kb + (long)hk * kh;
  const float* Lp = LSE + ((long)b * Hq + hq) * S;
  const float* Dp = Delta + ((long)b * Hq + hq) * S;

  __shared__ ushort_t k_rm[NT * 32 * KROW];
  __shared__ ushort_t v_rm[NT * 32 * KROW];
  __shared__ ushort_t k_img[NT * 8 * TRKEY4];  // tr16 image: K^T B-frags

  // Q and dO fragments (B: lane n = q, kdim = d slices)
  bf16x8 qf[8], dof[8];
  {
    const long r = (long)(q_valid ? q_row : 0) * qs;
    const long ro = (long)(q_valid ? q_row : 0) * os;
    const int dbase = (lane >> 5) * 8;
#pragma unroll
    for (int st = 0; st < 8; ++st) {
      qf[st] = __builtin_bit_cast(bf16x8,
                                  *(const ushortx8*)(Qp + r + st * 16 + dbase));
      dof[st] = __builtin_bit_cast(
          bf16x8, *(const ushortx8*)(dOp + ro + st * 16 + dbase));
    }
  }
  const float L_q = q_valid ? Lp[q_row] : 0.f;
  const float D_q = q_valid ? Dp[q_row] : 0.f;

  floatx16 dq_acc[4] = {};

  const int wg_last_row = m_block * 128 + 127;
  const int n_end_row = causal ? min(wg_last_row, S - 1) : (S - 1);
  const int n_tiles = (n_end_row / BLOCK) + 1;
  const int my_last_tile = causal ? ((q0 + 31) / BLOCK) : (n_tiles - 1);

  const int st_row = threadIdx.x >> 3;
  const int st_col = (threadIdx.x & 7) * 16;

  // T14 double-buffered prefetch (same pattern as the dv/dk kernels): tile
  // t+1's global loads are issued right after tile t is staged, so the
  // inter-barrier critical path never waits on HBM latency.
  ushortx8 k0[NT], k1[NT], v0[NT], v1[NT];
#pragma unroll
  for (int hh = 0; hh < NT; ++hh) {
    const int krow_g = min(32 * hh + st_row, S - 1);
    k0[hh] = *(const ushortx8*)(Kp + (long)krow_g * ks + st_col);
    k1[hh] = *(const ushortx8*)(Kp + (long)krow_g * ks + st_col + 8);
    v0[hh] = *(const ushortx8*)(Vp + (long)krow_g * ks + st_col);
    v1[hh] = *(const ushortx8*)(Vp + (long)krow_g * ks + st_col + 8);
  }
  const int n_tiles_st = (n_tiles + NT - 1) / NT;   // staged (NT-wide) tiles
  for (int ts = 0; ts < n_tiles_st; ++ts) {
    __syncthreads();
#pragma unroll
    for (int hh = 0; hh < NT; ++hh) {
      *(ushortx8*)(&k_rm[rm_off(32 * hh + st_row, st_col / 8)]) = k0[hh];
      *(ushortx8*)(&k_rm[rm_off(32 * hh + st_row, st_col / 8 + 1)]) = k1[hh];
      *(ushortx8*)(&v_rm[rm_off(32 * hh + st_row, st_col / 8)]) = v0[hh];
      *(ushortx8*)(&v_rm[rm_off(32 * hh + st_row, st_col / 8 + 1)]) = v1[hh];
      *(ushortx8*)(&k_img[hh * 8 * TRKEY4 + tr_img_off(st_row, st_col)]) =
          k0[hh];
      *(ushortx8*)(&k_img[hh * 8 * TRKEY4 +
                          tr_img_off(st_row, st_col + 8)]) = k1[hh];
    }
    __syncthreads();
    if (ts + 1 < n_tiles_st) {
#pragma unroll
      for (int hh = 0; hh < NT; ++hh) {
        const int krow_g = min((ts + 1) * NT * BLOCK + 32 * hh + st_row,
                               S - 1);
        k0[hh] = *(const ushortx8*)(Kp + (long)krow_g * ks + st_col);
        k1[hh] = *(const ushortx8*)(Kp + (long)krow_g * ks + st_col + 8);
        v0[hh] = *(const ushortx8*)(Vp + (long)krow_g * ks + st_col);
        v1[hh] = *(const ushortx8*)(Vp + (long)krow_g * ks + st_col + 8);
      }
    }

#pragma unroll
    for (int h = 0; h < NT; ++h) {
    const int t = ts * NT + h;
    if (t >= n_tiles || t > my_last_tile || q0 >= S) continue;
    const int n0 = t * BLOCK;

    // S^T and dP^T (C: n = q lane-local, m = key)
    floatx16 st_acc = {}, dp_acc = {};
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int stp = 0; stp < 8; ++stp) {
      const int blk = 2 * stp + (lane >> 5);
      bf16x8 kf = __builtin_bit_cast(
          bf16x8,
          *(const ushortx8*)(&k_rm[rm_off(32 * h + (lane & 31), blk)]));
      bf16x8 vf = __builtin_bit_cast(
          bf16x8,
          *(const ushortx8*)(&v_rm[rm_off(32 * h + (lane & 31), blk)]));
      st_acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(kf, qf[stp], st_acc,
                                                       0, 0, 0);
      dp_acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(vf, dof[stp], dp_acc,
                                                       0, 0, 0);
    }
    __builtin_amdgcn_s_setprio(0);

    // dS^T = scale * P^T (dP^T - delta); dead entries 0 via P=exp2(-inf)=0
    const int mrow_base = 4 * (lane >> 5);
    float ds_val[16];
#pragma unroll
    for (int j = 0; j < 16; ++j) {
      const int key = n0 + (j & 3) + 8 * (j >> 2) + mrow_base;
      const bool dead = (causal && key > q_row) || key >= S || !q_valid;
      const float p = dead ? 0.f
                           : exp2_rawb(st_acc[j] * c_log2 - L_q);
      ds_val[j] = scale * p * (dp_acc[j] - D_q);
    }

    // transpose dS^T C -> A-frag (m=q, kdim=key) via cvt_pk + permlane
    unsigned own_pk[8];
#pragma unroll
    for (int b2 = 0; b2 < 4; ++b2) {
      own_pk[2 * b2] = cvt_pk_bf16b(ds_val[4 * b2], ds_val[4 * b2 + 1]);
      own_pk[2 * b2 + 1] = cvt_pk_bf16b(ds_val[4 * b2 + 2], ds_val[4 * b2 + 3]);
    }
    bf16x8 dsf[2];
    {
      auto r0 = __builtin_amdgcn_permlane32_swap(own_pk[0], own_pk[2], false, false);
      auto r1 = __builtin_amdgcn_permlane32_swap(own_pk[1], own_pk[3], false, false);
      auto r2 = __builtin_amdgcn_permlane32_swap(own_pk[4], own_pk[6], false, false);
      auto r3 = __builtin_amdgcn_permlane32_swap(own_pk[5], own_pk[7], false, false);
      unsigned w0[4] = {r0[0], r1[0], r0[1], r1[1]};
      unsigned w1[4] = {r2[0], r3[0], r2[1], r3[1]};
      dsf[0] = __builtin_bit_cast(bf16x8, *(uint32x4_t*)w0);
      dsf[1] = __builtin_bit_cast(bf16x8, *(uint32x4_t*)w1);
    }

    // dQ += dS . K  (B[k=key][n=d] fragments via ds_read_b64_tr_b16
    // from the row-major subtiled K image; dt+1's reads pipeline behind
    // dt's MFMAs — only our tr ops are on lgkm in this phase)
    {
      const unsigned kbase =
          tr16_lane_base(k_img + h * 8 * TRKEY4, lane);
      ushortx4_tr vrA[2][2], vrB[2][2];
      __builtin_amdgcn_s_setprio(1);
      tr16_issue_dt<1, 0>(kbase, vrA);
#pragma unroll
      for (int dt = 0; dt < 4; ++dt) {
        ushortx4_tr(*cur)[2] = (dt & 1) ? vrB : vrA;
        ushortx4_tr(*nxt)[2] = (dt & 1) ? vrA : vrB;
        if (dt == 0) tr16_issue_dt<1, 1>(kbase, nxt);
        if (dt == 1) tr16_issue_dt<1, 2>(kbase, nxt);
        if (dt == 2) tr16_issue_dt<1, 3>(kbase, nxt);
        if (dt < 3)
          tr16_wait_n<4>();
        else
          tr16_wait_n<0>();
#pragma unroll
        for (int kstep = 0; kstep < 2; ++kstep) {
          struct {
            ushortx4_tr a, b;
          } pair = {cur[kstep][0], cur[kstep][1]};
          dq_acc[dt] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              dsf[kstep], __builtin_bit_cast(bf16x8, pair), dq_acc[dt],
              0, 0, 0);
        }
      }
      __builtin_amdgcn_s_setprio(0);
    }
    }  // NT sub-tile loop
  }

  // epilogue: write dQ (C: m = q reg-spread, n = d lane)
  ushort_t* dQp = dQ + (long)b * gb + (long)hq * gh;
#pragma unroll
  for (int j = 0; j < 16; ++j) {
    const int qr = q0 + (j & 3) + 8 * (j >> 2) + 4 * (lane >> 5);
    if (qr >= S) continue;
#pragma unroll
    for (int dt = 0; dt < 4; ++dt)
      dQp[(long)qr * gs + dt * 32 + (lane & 31)] = f2bf(dq_acc[dt][j]);
  }
}

// ------------------------------------------------------ dV / dK kernels
//
// Split (instead of one fused dKV kernel) so each stays under the 256-VGPR
// occupancy-2 line: the fused version needed 128 accumulator VGPRs + two
// C tiles and spilled. dV re-derives P from (S^T, lse); dK re-derives S^T
// and dP^T. +25% MFMA vs fused, but 2 waves/SIMD instead of 1 hides the
// LDS/HBM latency that dominated the fused kernel (PMC: 37% WAIT_ANY).

// NQ = 32-row q sub-tiles staged per barrier interval (NQ=2 halves the
// __syncthreads count per q row, same win as the forward kernel's NT).
template <int NQ, int OCC = 2, bool KLDS = false>
__global__ __launch_bounds__(256, OCC) void attn_bwd_dv_kernel(
    const ushort_t* __restrict__ Q, const ushort_t* __restrict__ K,
    const ushort_t* __restrict__ dO,
    const float* __restrict__ LSE,
    ushort_t* __restrict__ dV_out,   // [B,Hk,S,D] via its own strides
    int S, int Hq, int Hk,
    long qb, long qh, long qs,
    long kb, long kh, long ks,
    long ob, long oh, long os,       // dO strides
    long gb, long gh, long gs,       // dV strides (rows contiguous)
    float scale, int causal) {
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int k_block = blockIdx.x;
  const int hk = blockIdx.y;        // KV head: the q-head group is looped
  const int b = blockIdx.z;
  const int group = Hq / Hk;
  const float c_log2 = scale * 1.4426950408889634f;

  const int key0 = k_block * 128 + wave * 32;
  const int key_row = key0 + (lane & 31);

  const ushort_t* Kp = K + (long)b * kb + (long)hk * kh;

  __shared__ ushort_t q_rm[NQ * 32 * KROW];
  __shared__ ushort_t do_img[NQ * 8 * TRKEY4];  // tr16 image: dO^T B-frags
  // KLDS: the block's 128 K rows live in LDS instead of 64 VGPRs of
  // per-lane registers — the register diet (190 -> ~130) buys a third
  // wave per SIMD, which matters because PMC shows dv 46% wait-bound
  // (gpurun_out/r2l). Reads come back as one ds_read_b128 per MFMA.
  __shared__ ushort_t k_lds[KLDS ? 128 * KROW : 8];

  // own K rows — the B operand of S = Q.K^T (B[k=d][n=key]: n =
  // lane-local key, k-slices = d — the same per-lane bytes an
  // A-fragment holds, so the load is unchanged)
  bf16x8 kfr[KLDS ? 1 : 8];
  if (!KLDS) {
    const long kg = (long)min(key_row, S - 1) * ks;
    const int dbase = (lane >> 5) * 8;
#pragma unroll
    for (int st = 0; st < 8; ++st)
      kfr[KLDS ? 0 : st] = __builtin_bit_cast(
          bf16x8, *(const ushortx8*)(Kp + kg + st * 16 + dbase));
  } else {
    // stage all 128 rows once: thread covers rows (tid>>3) + 32*r, 16
    // dims at (tid&7)*16 — same T2-swizzled row-major image as q_rm
    const int sr = threadIdx.x >> 3;
    const int sc = (threadIdx.x & 7) * 16;
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int krow = k_block * 128 + 32 * r + sr;
      const long kg = (long)min(krow, S - 1) * ks;
      *(ushortx8*)(&k_lds[rm_off(32 * r + sr, sc / 8)]) =
          *(const ushortx8*)(Kp + kg + sc);
      *(ushortx8*)(&k_lds[rm_off(32 * r + sr, sc / 8 + 1)]) =
          *(const ushortx8*)(Kp + kg + sc + 8);
    }
    // the first q-tile barrier below makes these writes visible
  }

  floatx16 dv_acc[4] = {};
  constexpr int QB = 32 * NQ;       // q rows per staged tile
  // 128 is a multiple of QB for NQ in {1,2}, so the causal start tile is
  // exact (no dead half-tiles at the diagonal)
  const int q_start_tile = causal ? (k_block * 128) / QB : 0;
  const int n_q_tiles = (S + QB - 1) / QB;
  const int st_row = threadIdx.x >> 3;
  const int st_col = (threadIdx.x & 7) * 16;

  for (int g = 0; g < group; ++g) {
  const int hq = hk * group + g;
  const ushort_t* Qp = Q + (long)b * qb + (long)hq * qh;
  const ushort_t* dOp = dO + (long)b * ob + (long)hq * oh;
  const float* Lp = LSE + ((long)b * Hq + hq) * S;
  ushortx8 qa[NQ], qa2[NQ], da[NQ], da2[NQ];
#pragma unroll
  for (int h = 0; h < NQ; ++h) {
    const int qg = min(q_start_tile * QB + 32 * h + st_row, S - 1);
    qa[h] = *(const ushortx8*)(Qp + (long)qg * qs + st_col);
    qa2[h] = *(const ushortx8*)(Qp + (long)qg * qs + st_col + 8);
    da[h] = *(const ushortx8*)(dOp + (long)qg * os + st_col);
    da2[h] = *(const ushortx8*)(dOp + (long)qg * os + st_col + 8);
  }

  for (int t = q_start_tile; t < n_q_tiles; ++t) {
    __syncthreads();
#pragma unroll
    for (int h = 0; h < NQ; ++h) {
      *(ushortx8*)(&q_rm[rm_off(32 * h + st_row, st_col / 8)]) = qa[h];
      *(ushortx8*)(&q_rm[rm_off(32 * h + st_row, st_col / 8 + 1)]) = qa2[h];
      *(ushortx8*)(&do_img[h * 8 * TRKEY4 + tr_img_off(st_row, st_col)]) =
          da[h];
      *(ushortx8*)(&do_img[h * 8 * TRKEY4 +
                           tr_img_off(st_row, st_col + 8)]) = da2[h];
    }
    __syncthreads();
    if (t + 1 < n_q_tiles) {
#pragma unroll
      for (int h = 0; h < NQ; ++h) {
        const int qg = min((t + 1) * QB + 32 * h + st_row, S - 1);
        qa[h] = *(const ushortx8*)(Qp + (long)qg * qs + st_col);
        qa2[h] = *(const ushortx8*)(Qp + (long)qg * qs + st_col + 8);
        da[h] = *(const ushortx8*)(dOp + (long)qg * os + st_col);
        da2[h] = *(const ushortx8*)(dOp + (long)qg * os + st_col + 8);
      }
    }

#pragma unroll
    for (int h = 0; h < NQ; ++h) {
    const int tq0 = t * QB + 32 * h;
    // S = Q.K^T with q REG-SPREAD, key LANE-LOCAL (operand roles swapped
    // vs the old scratch version): C[m=q][n=key] — P^T's A-fragment
    // (A[m=key][k=q]) then comes from the SAME in-register
    // cvt_pk+permlane transpose the forward kernel uses for P (T12),
    // removing the wave-scratch LDS round trip entirely.
    floatx16 st_acc = {};
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int stp = 0; stp < 8; ++stp) {
      bf16x8 qfr = __builtin_bit_cast(
          bf16x8,
          *(const ushortx8*)(&q_rm[rm_off(32 * h + (lane & 31),
                                          2 * stp + (lane >> 5))]));
      const bf16x8 kop =
          KLDS ? __builtin_bit_cast(
                     bf16x8,
                     *(const ushortx8*)(&k_lds[rm_off(
                         wave * 32 + (lane & 31), 2 * stp + (lane >> 5))]))
               : kfr[KLDS ? 0 : stp];
      st_acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(qfr, kop, st_acc,
                                                       0, 0, 0);
    }
    __builtin_amdgcn_s_setprio(0);

    // per-q-row lse fetched once per lane, broadcast per reg row by
    // shfl (LDS-local ~50cy; an r2m experiment with global float4 loads
    // instead put 200+cy L2 latency on the critical path and lost 3%)
    const float L_lane = Lp[min(tq0 + (lane & 31), S - 1)];
    const int mrow_base = 4 * (lane >> 5);
    const int key_here = key0 + (lane & 31);
    float p_val[16];
#pragma unroll
    for (int j = 0; j < 16; ++j) {
      const int qrow_j = (j & 3) + 8 * (j >> 2) + mrow_base;
      const int q_abs = tq0 + qrow_j;
      const float L_q = __shfl(L_lane, qrow_j);
      const bool dead = (causal && key_here > q_abs) || q_abs >= S ||
                        key_here >= S;
      p_val[j] = dead ? 0.f
                      : exp2_rawb(st_acc[j] * c_log2 - L_q);
    }
    // C[q][key] -> A[m=key][k=q] fragments in-register (fwd T12 pattern)
    unsigned own_pk[8];
#pragma unroll
    for (int b2 = 0; b2 < 4; ++b2) {
      own_pk[2 * b2] = cvt_pk_bf16b(p_val[4 * b2], p_val[4 * b2 + 1]);
      own_pk[2 * b2 + 1] = cvt_pk_bf16b(p_val[4 * b2 + 2], p_val[4 * b2 + 3]);
    }
    bf16x8 pf[2];
    {
      auto r0 = __builtin_amdgcn_permlane32_swap(own_pk[0], own_pk[2], false, false);
      auto r1 = __builtin_amdgcn_permlane32_swap(own_pk[1], own_pk[3], false, false);
      auto r2 = __builtin_amdgcn_permlane32_swap(own_pk[4], own_pk[6], false, false);
      auto r3 = __builtin_amdgcn_permlane32_swap(own_pk[5], own_pk[7], false, false);
      unsigned w0[4] = {r0[0], r1[0], r0[1], r1[1]};
      unsigned w1[4] = {r2[0], r3[0], r2[1], r3[1]};
      pf[0] = __builtin_bit_cast(bf16x8, *(uint32x4_t*)w0);
      pf[1] = __builtin_bit_cast(bf16x8, *(uint32x4_t*)w1);
    }
    {
      const unsigned dbase =
          tr16_lane_base(do_img + h * 8 * TRKEY4, lane);
      ushortx4_tr vrA[2][2], vrB[2][2];
      __builtin_amdgcn_s_setprio(1);
      tr16_issue_dt<1, 0>(dbase, vrA);
#pragma unroll
      for (int dt = 0; dt < 4; ++dt) {
        ushortx4_tr(*cur)[2] = (dt & 1) ? vrB : vrA;
        ushortx4_tr(*nxt)[2] = (dt & 1) ? vrA : vrB;
        if (dt == 0) tr16_issue_dt<1, 1>(dbase, nxt);
        if (dt == 1) tr16_issue_dt<1, 2>(dbase, nxt);
        if (dt == 2) tr16_issue_dt<1, 3>(dbase, nxt);
        if (dt < 3)
          tr16_wait_n<4>();
        else
          tr16_wait_n<0>();
#pragma unroll
        for (int kstep = 0; kstep < 2; ++kstep) {
          struct {
            ushortx4_tr a, b;
          } pair = {cur[kstep][0], cur[kstep][1]};
          dv_acc[dt] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              pf[kstep], __builtin_bit_cast(bf16x8, pair), dv_acc[dt],
              0, 0, 0);
        }
      }
      __builtin_amdgcn_s_setprio(0);
    }
    }  // NQ sub-tile loop
  }
  }  // q-head group loop

  ushort_t* dVh = dV_out + (long)b * gb + (long)hk * gh;
#pragma unroll
  for (int j = 0; j < 16; ++j) {
    const int kr = key0 + (j & 3) + 8 * (j >> 2) + 4 * (lane >> 5);
    if (kr >= S) continue;
#pragma unroll
    for (int dt = 0; dt < 4; ++dt)
      dVh[(long)kr * gs + dt * 32 + (lane & 31)] = f2bf(dv_acc[dt][j]);
  }
}

template <int NQ>
__global__ __launch_bounds__(256, 2) void attn_bwd_dk_kernel(
    const ushort_t* __restrict__ Q, const ushort_t* __restrict__ K,
    const ushort_t* __restrict__ V, const ushort_t* __restrict__ dO,
    const float* __restrict__ LSE, const float* __restrict__ Delta,
    ushort_t* __restrict__ dK_out,   // [B,Hk,S,D] via its own strides
    int S, int Hq, int Hk,
    long qb, long qh, long qs,
    long kb, long kh, long ks,
    long ob, long oh, long os,       // dO strides
    long gb, long gh, long gs,       // dK strides (rows contiguous)
    float scale, int causal) {
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int k_block = blockIdx.x;
  const int hk = blockIdx.y;        // KV head
  const int b = blockIdx.z;
  const int group = Hq / Hk;
  const float c_log2 = scale * 1.4426950408889634f;

  const int key0 = k_block * 128 + wave * 32;
  const int key_row = key0 + (lane & 31);

  const ushort_t* Kp = K + (long)b * kb + (long)hk * kh;
  const ushort_t* Vp = V + (long)b * kb + (long)hk * kh;

  __shared__ ushort_t q_rm[NQ * 32 * KROW];
  __shared__ ushort_t do_rm[NQ * 32 * KROW];
  __shared__ ushort_t q_img[NQ * 8 * TRKEY4];  // tr16 image: Q^T B-frags

  // own K/V rows in registers — B operands of S = Q.K^T and dP = dO.V^T
  // (key lane-local; the same per-lane bytes an A-fragment holds)
  bf16x8 kfr[8], vfr[8];
  {
    const long kg = (long)min(key_row, S - 1) * ks;
    const int dbase = (lane >> 5) * 8;
#pragma unroll
    for (int st = 0; st < 8; ++st) {
      kfr[st] = __builtin_bit_cast(
          bf16x8, *(const ushortx8*)(Kp + kg + st * 16 + dbase));
      vfr[st] = __builtin_bit_cast(
          bf16x8, *(const ushortx8*)(Vp + kg + st * 16 + dbase));
    }
  }

  floatx16 dk_acc[4] = {};
  constexpr int QB = 32 * NQ;       // q rows per staged tile
  const int q_start_tile = causal ? (k_block * 128) / QB : 0;
  const int n_q_tiles = (S + QB - 1) / QB;
  const int st_row = threadIdx.x >> 3;
  const int st_col = (threadIdx.x & 7) * 16;

  for (int g = 0; g < group; ++g) {
  const int hq = hk * group + g;
  const ushort_t* Qp = Q + (long)b * qb + (long)hq * qh;
  const ushort_t* dOp = dO + (long)b * ob + (long)hq * oh;
  const float* Lp = LSE + ((long)b * Hq + hq) * S;
  const float* Dp = Delta + ((long)b * Hq + hq) * S;
  // double-buffered q/do tile prefetch (same pattern as attn_bwd_dv):
  // next tile's global loads issue OUTSIDE the LDS critical section so
  // the barrier never waits on HBM latency
  ushortx8 qa[NQ], qa2[NQ], da[NQ], da2[NQ];
#pragma unroll
  for (int h = 0; h < NQ; ++h) {
    const int qg = min(q_start_tile * QB + 32 * h + st_row, S - 1);
    qa[h] = *(const ushortx8*)(Qp + (long)qg * qs + st_col);
    qa2[h] = *(const ushortx8*)(Qp + (long)qg * qs + st_col + 8);
    da[h] = *(const ushortx8*)(dOp + (long)qg * os + st_col);
    da2[h] = *(const ushortx8*)(dOp + (long)qg * os + st_col + 8);
  }
  for (int t = q_start_tile; t < n_q_tiles; ++t) {
    __syncthreads();
#pragma unroll
    for (int h = 0; h < NQ; ++h) {
      *(ushortx8*)(&q_rm[rm_off(32 * h + st_row, st_col / 8)]) = qa[h];
      *(ushortx8*)(&q_rm[rm_off(32 * h + st_row, st_col / 8 + 1)]) = qa2[h];
      *(ushortx8*)(&do_rm[rm_off(32 * h + st_row, st_col / 8)]) = da[h];
      *(ushortx8*)(&do_rm[rm_off(32 * h + st_row, st_col / 8 + 1)]) = da2[h];
      *(ushortx8*)(&q_img[h * 8 * TRKEY4 + tr_img_off(st_row, st_col)]) =
          qa[h];
      *(ushortx8*)(&q_img[h * 8 * TRKEY4 +
                          tr_img_off(st_row, st_col + 8)]) = qa2[h];
    }
    __syncthreads();
    if (t + 1 < n_q_tiles) {
#pragma unroll
      for (int h = 0; h < NQ; ++h) {
        const int qg = min((t + 1) * QB + 32 * h + st_row, S - 1);
        qa[h] = *(const ushortx8*)(Qp + (long)qg * qs + st_col);
        qa2[h] = *(const ushortx8*)(Qp + (long)qg * qs + st_col + 8);
        da[h] = *(const ushortx8*)(dOp + (long)qg * os + st_col);
        da2[h] = *(const ushortx8*)(dOp + (long)qg * os + st_col + 8);
      }
    }

#pragma unroll
    for (int h = 0; h < NQ; ++h) {
    const int tq0 = t * QB + 32 * h;
    // S and dP with q REG-SPREAD, key LANE-LOCAL (roles swapped, as in
    // attn_bwd_dv): dS^T's A-fragment comes from the in-register T12
    // transpose — no wave-scratch LDS round trip.
    floatx16 st_acc = {}, dp_acc = {};
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int stp = 0; stp < 8; ++stp) {
      const int blk = 2 * stp + (lane >> 5);
      bf16x8 qfr = __builtin_bit_cast(
          bf16x8,
          *(const ushortx8*)(&q_rm[rm_off(32 * h + (lane & 31), blk)]));
      st_acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(qfr, kfr[stp], st_acc,
                                                       0, 0, 0);
      bf16x8 dofr = __builtin_bit_cast(
          bf16x8,
          *(const ushortx8*)(&do_rm[rm_off(32 * h + (lane & 31), blk)]));
      dp_acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(dofr, vfr[stp], dp_acc,
                                                       0, 0, 0);
    }
    __builtin_amdgcn_s_setprio(0);

    // per-q-row lse/delta via float4 loads (see attn_bwd_dv)
    const int q_lane = min(tq0 + (lane & 31), S - 1);
    const float L_lane = Lp[q_lane];
    const float D_lane = Dp[q_lane];
    const int mrow_base = 4 * (lane >> 5);
    const int key_here = key0 + (lane & 31);
    float ds_val[16];
#pragma unroll
    for (int j = 0; j < 16; ++j) {
      const int qrow_j = (j & 3) + 8 * (j >> 2) + mrow_base;
      const int q_abs = tq0 + qrow_j;
      const float L_q = __shfl(L_lane, qrow_j);
      const float D_q = __shfl(D_lane, qrow_j);
      const bool dead = (causal && key_here > q_abs) || q_abs >= S ||
                        key_here >= S;
      const float p = dead ? 0.f
                           : exp2_rawb(st_acc[j] * c_log2 - L_q);
      ds_val[j] = scale * p * (dp_acc[j] - D_q);
    }
    unsigned own_pk[8];
#pragma unroll
    for (int b2 = 0; b2 < 4; ++b2) {
      own_pk[2 * b2] = cvt_pk_bf16b(ds_val[4 * b2], ds_val[4 * b2 + 1]);
      own_pk[2 * b2 + 1] = cvt_pk_bf16b(ds_val[4 * b2 + 2], ds_val[4 * b2 + 3]);
    }
    bf16x8 dsf[2];
    {
      auto r0 = __builtin_amdgcn_permlane32_swap(own_pk[0], own_pk[2], false, false);
      auto r1 = __builtin_amdgcn_permlane32_swap(own_pk[1], own_pk[3], false, false);
      auto r2 = __builtin_amdgcn_permlane32_swap(own_pk[4], own_pk[6], false, false);
      auto r3 = __builtin_amdgcn_permlane32_swap(own_pk[5], own_pk[7], false, false);
      unsigned w0[4] = {r0[0], r1[0], r0[1], r1[1]};
      unsigned w1[4] = {r2[0], r3[0], r2[1], r3[1]};
      dsf[0] = __builtin_bit_cast(bf16x8, *(uint32x4_t*)w0);
      dsf[1] = __builtin_bit_cast(bf16x8, *(uint32x4_t*)w1);
    }
    {
      const unsigned qb2 =
          tr16_lane_base(q_img + h * 8 * TRKEY4, lane);
      ushortx4_tr vrA[2][2], vrB[2][2];
      __builtin_amdgcn_s_setprio(1);
      tr16_issue_dt<1, 0>(qb2, vrA);
#pragma unroll
      for (int dt = 0; dt < 4; ++dt) {
        ushortx4_tr(*cur)[2] = (dt & 1) ? vrB : vrA;
        ushortx4_tr(*nxt)[2] = (dt & 1) ? vrA : vrB;
        if (dt == 0) tr16_issue_dt<1, 1>(qb2, nxt);
        if (dt == 1) tr16_issue_dt<1, 2>(qb2, nxt);
        if (dt == 2) tr16_issue_dt<1, 3>(qb2, nxt);
        if (dt < 3)
          tr16_wait_n<4>();
        else
          tr16_wait_n<0>();
#pragma unroll
        for (int kstep = 0; kstep < 2; ++kstep) {
          struct {
            ushortx4_tr a, b;
          } pair = {cur[kstep][0], cur[kstep][1]};
          dk_acc[dt] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              dsf[kstep], __builtin_bit_cast(bf16x8, pair), dk_acc[dt],
              0, 0, 0);
        }
      }
      __builtin_amdgcn_s_setprio(0);
    }
    }  // NQ sub-tile loop
  }
  }  // q-head group loop

  ushort_t* dKh = dK_out + (long)b * gb + (long)hk * gh;
#pragma unroll
  for (int j = 0; j < 16; ++j) {
    const int kr = key0 + (j & 3) + 8 * (j >> 2) + 4 * (lane >> 5);
    if (kr >= S) continue;
#pragma unroll
    for (int dt = 0; dt < 4; ++dt)
      dKh[(long)kr * gs + dt * 32 + (lane & 31)] = f2bf(dk_acc[dt][j]);
  }
}

// sum Q-head partial dK/dV into the KV heads: [B,Hq,S,D] -> [B,Hk,S,D]
extern "C" __global__ void attn_bwd_reduce_kv_kernel(
    const ushort_t* __restrict__ dKp, const ushort_t* __restrict__ dVp,
    ushort_t* __restrict__ dK, ushort_t* __restrict__ dV,
    int Hq, int Hk, long SD, long total) {
  const long i = (long)blockIdx.x * blockDim.x + threadIdx.x;  // over Hk*S*D/8
  if (i * 8 >= total) return;
  const int group = Hq / Hk;
  const long hk = (i * 8) / SD % Hk;
  const long b = (i * 8) / (SD * Hk);
  const long inner = (i * 8) % SD;
  floatx8 ka = {}, va = {};
#pragma unroll 4
  for (int g = 0; g < group; ++g) {
    const long src = ((b * Hq) + hk * group + g) * SD + inner;
    const floatx8 kx = bf8_to_f32x8(*(const ushortx8*)(dKp + src));
    const floatx8 vx = bf8_to_f32x8(*(const ushortx8*)(dVp + src));
#pragma unroll
    for (int e = 0; e < 8; ++e) {
      ka[e] += kx[e];
      va[e] += vx[e];
    }
  }
  const long dst = (b * Hk + hk) * SD + inner;
  *(ushortx8*)(dK + dst) = f32x8_to_bf8(ka);
  *(ushortx8*)(dV + dst) = f32x8_to_bf8(va);
}

// ---------------------------------------------------------------- hosts

extern "C" void launch_attn_bwd_preprocess(const void* dO, const void* o,
                                           float* delta, long rows, int Hq,
                                           int S, long dob, long doh,
                                           long dos, long ob, long oh,
                                           long os, void* stream) {
  const int waves_per_block = 4;
  const long rows_per_block = waves_per_block * 4;
  dim3 grid((rows + rows_per_block - 1) / rows_per_block);
  hipLaunchKernelGGL(attn_bwd_preprocess_kernel, grid, dim3(256), 0,
                     (hipStream_t)stream, (const ushort_t*)dO,
                     (const ushort_t*)o, delta, rows, Hq, S, dob, doh, dos,
                     ob, oh, os);
}

extern "C" void launch_attn_bwd_dq(const void* q, const void* k, const void* v,
                                   const void* dO, const float* lse,
                                   const float* delta, void* dq, int B, int S,
                                   int Hq, int Hk, long qb, long qh, long qs,
                                   long kb, long kh, long ks, long ob,
                                   long oh, long os, long gb, long gh,
                                   long gs, float scale,
                                   int causal, void* stream) {
  dim3 grid((S + 127) / 128, Hq, B);
  static const int dq_nt = [] {
    const char* e = getenv("ANTRAY_BWD_DQ_NT");
    return e ? atoi(e) : 1;
  }();
  if (dq_nt >= 2) {
    hipLaunchKernelGGL(attn_bwd_dq_kernel<2>, grid, dim3(256), 0,
                       (hipStream_t)stream, (const ushort_t*)q,
                       (const ushort_t*)k, (const ushort_t*)v,
                       (const ushort_t*)dO, lse, delta, (ushort_t*)dq, S, Hq,
                       Hk, qb, qh, qs, kb, kh, ks, ob, oh, os, gb, gh, gs,
                       scale, causal);
  } else {
    hipLaunchKernelGGL(attn_bwd_dq_kernel<1>, grid, dim3(256), 0,
                       (hipStream_t)stream, (const ushort_t*)q,
                       (const ushort_t*)k, (const ushort_t*)v,
                       (const ushort_t*)dO, lse, delta, (ushort_t*)dq, S, Hq,
                       Hk, qb, qh, qs, kb, kh, ks, ob, oh, os, gb, gh, gs,
                       scale, causal);
  }
}

extern "C" void launch_attn_bwd_dkv(const void* q, const void* k,
                                    const void* v, const void* dO,
                                    const float* lse, const float* delta,
                                    void* dkp, void* dvp, int B, int S,
                                    int Hq, int Hk, long qb, long qh, long qs,
                                    long kb, long kh, long ks, long ob,
                                    long oh, long os, long gb, long gh,
                                    long gs, float scale,
                                    int causal, void* stream,
                                    void* stream2) {
  dim3 grid((S + 127) / 128, Hk, B);
  // staged-q-tile width. Measured A/B (gpurun_out/r2g/ab.log, B6 S4096):
  // dv<2> (64-row stages, 214 VGPR no spill) is NEUTRAL vs dv<1> (8.96
  // vs 8.91 ms f+b — barrier count is not the dv bottleneck) and dk<2>
  // needs ~280 VGPR -> spills 104 B/lane and LOSES (9.72 ms). Default
  // stays 1; ANTRAY_BWD_NQ=2 -> dv 64-row, =3 -> both 64-row, for
  // re-measuring on future silicon/compilers.
  static const int nq = [] {
    const char* e = getenv("ANTRAY_BWD_NQ");
    return e ? atoi(e) : 1;
  }();
  // dv variant: default 4 = K rows in LDS, 160 VGPR, 3 waves/SIMD —
  // measured 8.70 vs 8.92 ms f+b against the 2-wave register-resident
  // variant (PMC r2l showed dv 46% wait-bound; the third wave hides
  // it). 3 = force-capped 3-wave variant (spills 80 B/lane, LOSES at
  // 9.32 ms — kept for re-measurement); 2 = original.
  static const int dv_occ = [] {
    const char* e = getenv("ANTRAY_BWD_DV_OCC");
    return e ? atoi(e) : 4;
  }();
  if (nq >= 2) {
    hipLaunchKernelGGL(attn_bwd_dv_kernel<2>, grid, dim3(256), 0,
                       (hipStream_t)stream, (const ushort_t*)q,
                       (const ushort_t*)k, (const ushort_t*)dO, lse,
                       (ushort_t*)dvp, S, Hq, Hk, qb, qh, qs, kb, kh, ks,
                       ob, oh, os, gb, gh, gs, scale, causal);
  } else if (dv_occ >= 4) {
    hipLaunchKernelGGL((attn_bwd_dv_kernel<1, 3, true>), grid, dim3(256), 0,
                       (hipStream_t)stream, (const ushort_t*)q,
                       (const ushort_t*)k, (const ushort_t*)dO, lse,
                       (ushort_t*)dvp, S, Hq, Hk, qb, qh, qs, kb, kh, ks,
                       ob, oh, os, gb, gh, gs, scale, causal);
  } else if (dv_occ >= 3) {
    hipLaunchKernelGGL((attn_bwd_dv_kernel<1, 3>), grid, dim3(256), 0,
                       (hipStream_t)stream, (const ushort_t*)q,
                       (const ushort_t*)k, (const ushort_t*)dO, lse,
                       (ushort_t*)dvp, S, Hq, Hk, qb, qh, qs, kb, kh, ks,
                       ob, oh, os, gb, gh, gs, scale, causal);
  } else {
    hipLaunchKernelGGL(attn_bwd_dv_kernel<1>, grid, dim3(256), 0,
                       (hipStream_t)stream, (const ushort_t*)q,
                       (const ushort_t*)k, (const ushort_t*)dO, lse,
                       (ushort_t*)dvp, S, Hq, Hk, qb, qh, qs, kb, kh, ks,
                       ob, oh, os, gb, gh, gs, scale, causal);
  }
  if (nq >= 3) {
    hipLaunchKernelGGL(attn_bwd_dk_kernel<2>, grid, dim3(256), 0,
                       (hipStream_t)(stream2 ? stream2 : stream),
                       (const ushort_t*)q,
                       (const ushort_t*)k, (const ushort_t*)v,
                       (const ushort_t*)dO, lse, delta, (ushort_t*)dkp, S, Hq,
                       Hk, qb, qh, qs, kb, kh, ks, ob, oh, os, gb, gh, gs,
                       scale, causal);
  } else {
    hipLaunchKernelGGL(attn_bwd_dk_kernel<1>, grid, dim3(256), 0,
                       (hipStream_t)(stream2 ? stream2 : stream),
                       (const ushort_t*)q,
                       (const ushort_t*)k, (const ushort_t*)v,
                       (const ushort_t*)dO, lse, delta, (ushort_t*)dkp, S, Hq,
                       Hk, qb, qh, qs, kb, kh, ks, ob, oh, os, gb, gh, gs,
                       scale, causal);
  }
}

extern "C" void launch_attn_bwd_reduce_kv(const void* dkp, const void* dvp,
                                          void* dk, void* dv, int B, int Hq,
                                          int Hk, int S, void* stream) {
  const long SD = (long)S * ATTN_D;
  const long total = (long)B * Hk * SD;
  const long n_vec = total / 8;
  dim3 grid((n_vec + 255) / 256);
  hipLaunchKernelGGL(attn_bwd_reduce_kv_kernel, grid, dim3(256), 0,
                     (hipStream_t)stream, (const ushort_t*)dkp,
                     (const ushort_t*)dvp, (ushort_t*)dk, (ushort_t*)dv, Hq,
                     Hk, SD, total);
}

// Torch bindings for the CDNA4 kernels (host-only C++; device code lives in
// kernels/*.hip, linked as hipcc-compiled objects — see csrc/build.py).
#include <torch/extension.h>

#include <c10/hip/HIPStream.h>
#include <hip/hip_runtime.h>

extern "C" {
void launch_rmsnorm_fwd(const void*, const void*, void*, float*, const void*,
                        void*, int64_t, int, float, hipStream_t);
void launch_rmsnorm_bwd(const void*, const void*, const void*, const float*,
                        void*, float*, float*, int64_t, int, hipStream_t);
int rmsnorm_bwd_grid(int64_t);
void launch_rope(void*, void*, const float*, const float*, int64_t, int64_t,
                 int, int, int, int64_t, int64_t, int, hipStream_t);
void launch_swiglu_fwd(const void*, void*, int64_t, int64_t, hipStream_t);
void launch_swiglu_bwd(const void*, const void*, void*, int64_t, int64_t,
                       hipStream_t);
void launch_cross_entropy(void*, const int32_t*, float*, int64_t, int64_t,
                          float, int32_t, int, hipStream_t);
void launch_adamw(float*, void*, const void*, float*, float*, int64_t, float,
                  float, float, float, float, float, float, float, hipStream_t);
void launch_bf16_scale(void*, float, int64_t, hipStream_t);
void launch_attn_fwd(const void*, const void*, const void*, void*, float*,
                     int, int, int, int, long, long, long, long, long, long,
                     long, long, long, float, int, void*);
void launch_attn_bwd_preprocess(const void*, const void*, float*, long, int,
                                int, long, long, long, long, long, long,
                                void*);
void launch_attn_bwd_dq(const void*, const void*, const void*, const void*,
                        const float*, const float*, void*, int, int, int, int,
                        long, long, long, long, long, long, long, long, long,
                        long, long, long, float, int, void*);
void launch_attn_bwd_dkv(const void*, const void*, const void*, const void*,
                         const float*, const float*, void*, void*, int, int,
                         int, int, long, long, long, long, long, long, long,
                         long, long, long, long, long, float, int, void*,
                         void*);
void launch_attn_bwd_reduce_kv(const void*, const void*, void*, void*, int,
                               int, int, int, void*);
void launch_bf16_to_f32(const void*, float*, int64_t, hipStream_t);
void launch_f32_to_bf16(const float*, void*, int64_t, hipStream_t);
void launch_attn_decode(const void*, const void*, const void*, void*, float*,
                        const int*, int, int, int, int, int, long, long, long,
                        long, float, void*);
void launch_rope_cache_write(void*, void*, void*, const int*, const float*,
                             const float*, int, int, int, int, long, long,
                             long, hipStream_t);
void launch_cast_affine_u8(const void*, void*, const float*, const float*,
                           long, int, int, int, void*);
void launch_cast_affine_f32(const void*, void*, const float*, const float*,
                            long, int, int, int, void*);
void launch_nhwc_to_nchw_u8(const void*, void*, const float*, const float*,
                            long, long, int, int, void*);
}

namespace {

hipStream_t cur_stream() { return c10::hip::getCurrentHIPStream().stream(); }

void check_bf16(const torch::Tensor& t, const char* name) {
  TORCH_CHECK(t.is_cuda(), name, " must be on GPU");
  TORCH_CHECK(t.scalar_type() == torch::kBFloat16, name, " must be bf16");
  TORCH_CHECK(t.is_contiguous(), name, " must be contiguous");
}

// Returns (y, rstd, h) — h is the fresh residual-sum buffer when `residual`
// is given, else an alias of x.
std::tuple<torch::Tensor, torch::Tensor, torch::Tensor> rmsnorm_fwd(
    torch::Tensor x, torch::Tensor w,
    c10::optional<torch::Tensor> residual, double eps) {
  check_bf16(x, "x");
  check_bf16(w, "w");
  int64_t H = x.size(-1);
  int64_t N = x.numel() / H;
  TORCH_CHECK(H % 8 == 0, "H must be a multiple of 8");
  auto y = torch::empty_like(x);
  auto rstd = torch::empty({N}, x.options().dtype(torch::kFloat32));
  const void* res_ptr = nullptr;
  void* h_ptr = nullptr;
  torch::Tensor h = x;
  if (residual.has_value()) {
    check_bf16(*residual, "residual");
    res_ptr = residual->data_ptr();
    h = torch::empty_like(x);
    h_ptr = h.data_ptr();
  }
  launch_rmsnorm_fwd(x.data_ptr(), w.data_ptr(), y.data_ptr(),
                     rstd.data_ptr<float>(), res_ptr, h_ptr, N, (int)H,
                     (float)eps, cur_stream());
  return {y, rstd, h};
}

std::tuple<torch::Tensor, torch::Tensor> rmsnorm_bwd(torch::Tensor dy,
                                                     torch::Tensor r,
                                                     torch::Tensor w,
                                                     torch::Tensor rstd) {
  check_bf16(dy, "dy");
  check_bf16(r, "r");
  check_bf16(w, "w");
  int64_t H = dy.size(-1);
  int64_t N = dy.numel() / H;
  TORCH_CHECK(H <= 8192, "rmsnorm_bwd supports H<=8192");
  auto dx = torch::empty_like(dy);
  auto dw = torch::empty({H}, dy.options().dtype(torch::kFloat32));
  int grid = rmsnorm_bwd_grid(N);
  auto partials = torch::empty({grid, H}, dy.options().dtype(torch::kFloat32));
  launch_rmsnorm_bwd(dy.data_ptr(), r.data_ptr(), w.data_ptr(),
                     rstd.data_ptr<float>(), dx.data_ptr(),
                     partials.data_ptr<float>(), dw.data_ptr<float>(), N,
                     (int)H, cur_stream());
  return {dx, dw};
}

// q/k are [B,S,H,D] views (possibly strided slices of a fused qkv buffer);
// requires the innermost [H,D] block per token to be contiguous.
void rope_(torch::Tensor q, torch::Tensor k, torch::Tensor cos_t,
           torch::Tensor sin_t, bool backward) {
  TORCH_CHECK(q.is_cuda() && q.scalar_type() == torch::kBFloat16);
  TORCH_CHECK(k.is_cuda() && k.scalar_type() == torch::kBFloat16);
  TORCH_CHECK(q.dim() == 4 && k.dim() == 4, "q/k must be [B,S,H,D]");
  TORCH_CHECK(q.stride(3) == 1 && q.stride(2) == q.size(3), "q [H,D] must be contiguous");
  TORCH_CHECK(k.stride(3) == 1 && k.stride(2) == k.size(3), "k [H,D] must be contiguous");
  TORCH_CHECK(cos_t.scalar_type() == torch::kFloat32 && cos_t.is_contiguous());
  int64_t B = q.size(0), S = q.size(1);
  int Hq = (int)q.size(2), Hk = (int)k.size(2), D = (int)q.size(3);
  // the kernel walks a FLAT token index (B*S tokens at one uniform
  // stride), so dim0 must fold into dim1. A size-1 S dim (single-token
  // decode step) reports a synthetic stride — use the batch stride as the
  // token stride there (tok == b).
  int64_t q_tok = q.stride(1), k_tok = k.stride(1);
  if (S == 1) {
    q_tok = B > 1 ? q.stride(0) : q_tok;
    k_tok = B > 1 ? k.stride(0) : k_tok;
  } else {
    TORCH_CHECK(B == 1 || q.stride(0) == S * q.stride(1),
                "q batch stride must fold");
    TORCH_CHECK(B == 1 || k.stride(0) == S * k.stride(1),
                "k batch stride must fold");
  }
  TORCH_CHECK(cos_t.size(0) >= S && cos_t.size(1) == D / 2, "cos table too small");
  TORCH_CHECK(D % 16 == 0, "rope: head_dim must be a multiple of 16");
  launch_rope(q.data_ptr(), k.data_ptr(), cos_t.data_ptr<float>(),
              sin_t.data_ptr<float>(), B, S, Hq, Hk, D, q_tok, k_tok,
              backward ? 1 : 0, cur_stream());
}

torch::Tensor swiglu_fwd(torch::Tensor gu) {
  check_bf16(gu, "gate_up");
  int64_t I2 = gu.size(-1);
  int64_t N = gu.numel() / I2;
  int64_t I = I2 / 2;
  TORCH_CHECK(I % 8 == 0, "intermediate size must be a multiple of 8");
  auto sizes = gu.sizes().vec();
  sizes.back() = I;
  auto out = torch::empty(sizes, gu.options());
  launch_swiglu_fwd(gu.data_ptr(), out.data_ptr(), N, I, cur_stream());
  return out;
}

torch::Tensor swiglu_bwd(torch::Tensor dout, torch::Tensor gu) {
  check_bf16(dout, "dout");
  check_bf16(gu, "gate_up");
  int64_t I2 = gu.size(-1);
  int64_t N = gu.numel() / I2;
  auto dgu = torch::empty_like(gu);
  launch_swiglu_bwd(dout.data_ptr(), gu.data_ptr(), dgu.data_ptr(), N, I2 / 2,
                    cur_stream());
  return dgu;
}

torch::Tensor cross_entropy_fwd_bwd(torch::Tensor logits, torch::Tensor targets,
                                    double grad_scale, int64_t ignore_index,
                                    bool write_dlogits) {
  check_bf16(logits, "logits");
  TORCH_CHECK(targets.scalar_type() == torch::kInt32 && targets.is_contiguous());
  int64_t V = logits.size(-1);
  int64_t N = logits.numel() / V;
  TORCH_CHECK(targets.numel() == N, "targets size mismatch");
  TORCH_CHECK(V % 8 == 0, "V must be a multiple of 8");
  auto loss = torch::empty({N}, logits.options().dtype(torch::kFloat32));
  launch_cross_entropy(logits.data_ptr(), targets.data_ptr<int32_t>(),
                       loss.data_ptr<float>(), N, V, (float)grad_scale,
                       (int32_t)ignore_index, write_dlogits ? 1 : 0,
                       cur_stream());
  return loss;
}

void adamw_(torch::Tensor p, torch::Tensor p_bf16, torch::Tensor g,
            torch::Tensor m, torch::Tensor v, double lr, double b1, double b2,
            double eps, double wd, int64_t step, double grad_scale) {
  TORCH_CHECK(p.scalar_type() == torch::kFloat32 && p.is_contiguous());
  check_bf16(p_bf16, "p_bf16");
  check_bf16(g, "grad");
  int64_t n = p.numel();
  TORCH_CHECK(n % 4 == 0, "flat param count must be a multiple of 4");
  TORCH_CHECK(p_bf16.numel() == n && g.numel() == n && m.numel() == n &&
              v.numel() == n);
  float bc1 = 1.f / (1.f - powf((float)b1, (float)step));
  float bc2 = 1.f / (1.f - powf((float)b2, (float)step));
  launch_adamw(p.data_ptr<float>(), p_bf16.data_ptr(), g.data_ptr(),
               m.data_ptr<float>(), v.data_ptr<float>(), n, (float)lr,
               (float)b1, (float)b2, (float)eps, (float)wd, bc1, bc2,
               (float)grad_scale, cur_stream());
}

void bf16_scale_(torch::Tensor x, double scale) {
  check_bf16(x, "x");
  TORCH_CHECK(x.numel() % 8 == 0);
  launch_bf16_scale(x.data_ptr(), (float)scale, x.numel(), cur_stream());
}

torch::Tensor bf16_to_f32(torch::Tensor x) {
  check_bf16(x, "x");
  TORCH_CHECK(x.numel() % 8 == 0);
  auto y = torch::empty(x.sizes(), x.options().dtype(torch::kFloat32));
  launch_bf16_to_f32(x.data_ptr(), y.data_ptr<float>(), x.numel(), cur_stream());
  return y;
}

void f32_to_bf16_(torch::Tensor x, torch::Tensor y) {
  TORCH_CHECK(x.scalar_type() == torch::kFloat32 && x.is_contiguous());
  check_bf16(y, "y");
  TORCH_CHECK(x.numel() == y.numel() && x.numel() % 8 == 0);
  launch_f32_to_bf16(x.data_ptr<float>(), y.data_ptr(), x.numel(), cur_stream());
}

}  // namespace


// Flash attention forward. q: [B, Hq, S, D], k/v: [B, Hk, S, D] (strided ok
// along the head dim so fused-qkv views work; the S,D inner block must be
// row-contiguous). Returns (o, lse) — lse in log2 domain, empty unless
// need_lse.
std::tuple<torch::Tensor, torch::Tensor> attn_fwd(torch::Tensor q,
                                                  torch::Tensor k,
                                                  torch::Tensor v,
                                                  double scale, bool causal,
                                                  bool need_lse) {
  TORCH_CHECK(q.is_cuda() && q.scalar_type() == torch::kBFloat16, "q bf16 gpu");
  TORCH_CHECK(q.dim() == 4 && k.dim() == 4 && v.dim() == 4, "q/k/v 4-D");
  int B = q.size(0), Hq = q.size(1), S = q.size(2), D = q.size(3);
  int Hk = k.size(1);
  TORCH_CHECK(D == 128, "attn_fwd: D must be 128");
  TORCH_CHECK(Hq % Hk == 0, "attn_fwd: Hq % Hk != 0");
  TORCH_CHECK(q.stride(3) == 1 && k.stride(3) == 1 && v.stride(3) == 1,
              "innermost dim must be contiguous");
  TORCH_CHECK(k.strides() == v.strides() && k.sizes() == v.sizes(),
              "k and v must share layout");
  // o in [B, S, Hq, D] memory layout (presented as a [B,Hq,S,D] strided
  // view): the model consumes o as [B, S, Hq*D] without a transpose copy
  auto o = torch::empty({B, S, Hq, D}, q.options())
               .permute({0, 2, 1, 3});
  torch::Tensor lse;
  float* lse_ptr = nullptr;
  if (need_lse) {
    lse = torch::empty({B, Hq, S}, q.options().dtype(torch::kFloat32));
    lse_ptr = lse.data_ptr<float>();
  } else {
    lse = torch::empty({0}, q.options().dtype(torch::kFloat32));
  }
  launch_attn_fwd(q.data_ptr(), k.data_ptr(), v.data_ptr(), o.data_ptr(),
                  lse_ptr, B, S, Hq, Hk, q.stride(0), q.stride(1), q.stride(2),
                  k.stride(0), k.stride(1), k.stride(2), o.stride(0),
                  o.stride(1), o.stride(2), (float)scale, causal ? 1 : 0,
                  (void*)cur_stream());
  return {o, lse};
}


// Flash-attention backward. Inputs as attn_fwd plus o (fwd output), lse
// (log2 domain from attn_fwd(need_lse=True)) and grad dO. Returns
// (dq, dk, dv) in [B,Hq,S,D]/[B,Hk,S,D] contiguous bf16.
std::tuple<torch::Tensor, torch::Tensor, torch::Tensor> attn_bwd(
    torch::Tensor dout, torch::Tensor q, torch::Tensor k, torch::Tensor v,
    torch::Tensor o, torch::Tensor lse, double scale, bool causal,
    c10::optional<torch::Tensor> dq_out, c10::optional<torch::Tensor> dk_out,
    c10::optional<torch::Tensor> dv_out) {
  TORCH_CHECK(q.dim() == 4 && q.size(3) == 128, "attn_bwd: q [B,Hq,S,128]");
  int B = q.size(0), Hq = q.size(1), S = q.size(2);
  int Hk = k.size(1);
  TORCH_CHECK(q.stride(3) == 1 && k.stride(3) == 1 && v.stride(3) == 1,
              "innermost dim must be contiguous");
  TORCH_CHECK(k.strides() == v.strides(), "k/v must share layout");
  // dout/o may be [B,S,H,D]-layout strided views (independently); the
  // kernels read rows through strides — no contiguous copies
  auto dc = dout.stride(3) == 1 ? dout : dout.contiguous();
  TORCH_CHECK(o.stride(3) == 1, "o rows must be contiguous");
  auto stream = cur_stream();
  long rows = (long)B * Hq * S;
  auto delta = torch::empty({B, Hq, S}, q.options().dtype(torch::kFloat32));
  launch_attn_bwd_preprocess(dc.data_ptr(), o.data_ptr(),
                             delta.data_ptr<float>(), rows, Hq, S,
                             dc.stride(0), dc.stride(1), dc.stride(2),
                             o.stride(0), o.stride(1), o.stride(2),
                             (void*)stream);
  // gradients may land in caller-provided STRIDED tensors (e.g. views of
  // one fused dqkv buffer — the rope+attention fused backward skips the
  // autograd slice-scatter entirely); rows must be contiguous
  auto pick = [&](c10::optional<torch::Tensor>& t, int H) {
    if (t.has_value()) {
      TORCH_CHECK(t->dim() == 4 && t->size(1) == H && t->stride(3) == 1,
                  "grad out must be [B,H,S,128] with contiguous rows");
      return *t;
    }
    return torch::empty({B, H, S, 128}, q.options());
  };
  auto dq = pick(dq_out, Hq);
  auto dk = pick(dk_out, Hk);
  auto dv = pick(dv_out, Hk);
  TORCH_CHECK(dk.strides() == dv.strides(), "dk/dv must share layout");
  // the three gradient kernels are independent and each leaves the MFMA
  // pipe ~65% idle at 2 waves/SIMD: run them CONCURRENTLY on two side
  // streams so their workgroups interleave on the CUs
  static hipStream_t side1 = nullptr, side2 = nullptr;
  static hipEvent_t ev_pre = nullptr, ev_s1 = nullptr, ev_s2 = nullptr;
  if (side1 == nullptr) {
    hipStreamCreateWithFlags(&side1, hipStreamNonBlocking);
    hipStreamCreateWithFlags(&side2, hipStreamNonBlocking);
    hipEventCreateWithFlags(&ev_pre, hipEventDisableTiming);
    hipEventCreateWithFlags(&ev_s1, hipEventDisableTiming);
    hipEventCreateWithFlags(&ev_s2, hipEventDisableTiming);
  }
  hipEventRecord(ev_pre, stream);       // delta/lse ready
  hipStreamWaitEvent(side1, ev_pre, 0);
  hipStreamWaitEvent(side2, ev_pre, 0);
  launch_attn_bwd_dq(q.data_ptr(), k.data_ptr(), v.data_ptr(),
                     dc.data_ptr(), lse.data_ptr<float>(),
                     delta.data_ptr<float>(), dq.data_ptr(), B, S, Hq, Hk,
                     q.stride(0), q.stride(1), q.stride(2), k.stride(0),
                     k.stride(1), k.stride(2), dc.stride(0), dc.stride(1),
                     dc.stride(2), dq.stride(0), dq.stride(1), dq.stride(2),
                     (float)scale, causal ? 1 : 0,
                     (void*)stream);
  launch_attn_bwd_dkv(q.data_ptr(), k.data_ptr(), v.data_ptr(),
                      dc.data_ptr(), lse.data_ptr<float>(),
                      delta.data_ptr<float>(), dk.data_ptr(), dv.data_ptr(),
                      B, S, Hq, Hk, q.stride(0), q.stride(1), q.stride(2),
                      k.stride(0), k.stride(1), k.stride(2), dc.stride(0),
                      dc.stride(1), dc.stride(2), dk.stride(0), dk.stride(1),
                      dk.stride(2), (float)scale,
                      causal ? 1 : 0, (void*)side1, (void*)side2);
  hipEventRecord(ev_s1, side1);
  hipEventRecord(ev_s2, side2);
  hipStreamWaitEvent(stream, ev_s1, 0);
  hipStreamWaitEvent(stream, ev_s2, 0);
    return {dq, dk, dv};
}

// Flash-decode: one new token per sequence over a bf16 KV cache.
// q: [B, Hq, D] contiguous; k/v: [B, Hk, Tmax, D] (strided along b/h ok,
// rows contiguous); T = valid cache length (uniform), or per-sequence
// lengths in `lens` (int32 [B] on device; then T = max). Returns o
// [B, Hq, D] bf16.
torch::Tensor attn_decode(torch::Tensor q, torch::Tensor k, torch::Tensor v,
                          int64_t T, double scale,
                          c10::optional<torch::Tensor> lens) {
  TORCH_CHECK(q.is_cuda() && q.scalar_type() == torch::kBFloat16, "q bf16 gpu");
  TORCH_CHECK(q.dim() == 3 && q.is_contiguous(), "q must be [B,Hq,D] contig");
  TORCH_CHECK(k.dim() == 4 && v.dim() == 4, "k/v must be [B,Hk,T,D]");
  int B = q.size(0), Hq = q.size(1), D = q.size(2);
  int Hk = k.size(1);
  TORCH_CHECK(D == 128, "attn_decode: D must be 128");
  TORCH_CHECK(Hq % Hk == 0 && Hq / Hk <= 8, "attn_decode: Hq/Hk must be <=8");
  TORCH_CHECK(k.stride(3) == 1 && v.stride(3) == 1, "rows must be contiguous");
  TORCH_CHECK(k.strides() == v.strides(), "k/v must share layout");
  TORCH_CHECK(T >= 1 && T <= k.size(2), "bad cache length");
  const int* lens_ptr = nullptr;
  if (lens.has_value()) {
    TORCH_CHECK(lens->is_cuda() && lens->scalar_type() == torch::kInt32 &&
                lens->numel() == B, "lens must be int32 [B] on device");
    lens_ptr = lens->data_ptr<int>();
  }
  // chunk count: fill the chip (~1024 workgroups) without splitting
  // below 128 keys per chunk. At the kernel's original 2 waves/SIMD a
  // 1024-WG target lost 15% (r2e A/B); after the GQ-templated register
  // diet raised occupancy to 4 waves/SIMD it WINS (r2k sweep: B8 T4096
  // 2.86 -> 3.41 TB/s, B32 T2048 3.05 -> 4.43; 2048+ regress again).
  // ANTRAY_DEC_WGS overrides the target for experiments.
  static const long wg_target = [] {
    const char* e = getenv("ANTRAY_DEC_WGS");
    return e ? atol(e) : 1024L;
  }();
  int C = (int)std::min<long>(
      std::max<long>(1, wg_target / std::max(1, B * Hk)),
      std::max<long>(1, ((long)T + 127) / 128));
  auto o = torch::empty({B, Hq, D}, q.options());
  torch::Tensor part;
  float* part_ptr = nullptr;
  if (C > 1) {
    part = torch::empty({(long)B * Hq * C * (D + 2)},
                        q.options().dtype(torch::kFloat32));
    part_ptr = part.data_ptr<float>();
  }
  launch_attn_decode(q.data_ptr(), k.data_ptr(), v.data_ptr(), o.data_ptr(),
                     part_ptr, lens_ptr, B, (int)T, Hq, Hk, C, k.stride(0),
                     k.stride(1), k.stride(2), (long)Hq * D, (float)scale,
                     (void*)cur_stream());
  return o;
}

// Device-pos single-token decode step (graph-capturable): rope q/k +
// cache writes + flash-decode in one call, with the position read from
// the DEVICE lens buffer (pos = lens[b]-1) — no host scalar depends on
// the step index, so generate() captures the whole token step in a
// hipGraph and replays it (the host-pos path pays ~260 Python launches
// per token). qkv: [B, 1, (Hq+2Hk)*D] or [B, (Hq+2Hk)*D] contiguous.
torch::Tensor decode_step_attn(torch::Tensor qkv, torch::Tensor ck,
                               torch::Tensor cv, torch::Tensor lens,
                               torch::Tensor cos_t, torch::Tensor sin_t,
                               int64_t Hq, int64_t Hk, double scale) {
  TORCH_CHECK(qkv.is_cuda() && qkv.scalar_type() == torch::kBFloat16 &&
              qkv.is_contiguous(), "qkv must be contiguous bf16 gpu");
  TORCH_CHECK(ck.dim() == 4 && cv.dim() == 4 && ck.strides() == cv.strides(),
              "caches must be [B,Hk,T,D] with a shared layout");
  TORCH_CHECK(ck.stride(3) == 1, "cache rows must be contiguous");
  TORCH_CHECK(lens.is_cuda() && lens.scalar_type() == torch::kInt32,
              "lens must be int32 on device");
  int B = ck.size(0), D = ck.size(3);
  int Tmax = ck.size(2);
  TORCH_CHECK(D == 128, "decode_step: D must be 128");
  TORCH_CHECK(qkv.numel() == (long)B * (Hq + 2 * Hk) * D, "qkv shape");
  TORCH_CHECK(lens.numel() == B, "lens must be [B]");
  TORCH_CHECK(cos_t.scalar_type() == torch::kFloat32 &&
              cos_t.is_contiguous() && cos_t.size(0) >= Tmax,
              "cos table must cover the cache");
  launch_rope_cache_write(qkv.data_ptr(), ck.data_ptr(), cv.data_ptr(),
                          lens.data_ptr<int>(), cos_t.data_ptr<float>(),
                          sin_t.data_ptr<float>(), B, (int)Hq, (int)Hk, D,
                          ck.stride(0), ck.stride(1), ck.stride(2),
                          cur_stream());
  static const long wg_target2 = [] {
    const char* e = getenv("ANTRAY_DEC_WGS");
    return e ? atol(e) : 1024L;
  }();
  // C is sized from the STATIC Tmax (graph shapes cannot depend on the
  // step); chunks past lens[b] compute empty ranges and combine to 0
  int C = (int)std::min<long>(
      std::max<long>(1, wg_target2 / std::max(1, B * (int)Hk)),
      std::max<long>(1, ((long)Tmax + 127) / 128));
  auto o = torch::empty({B, Hq, D}, qkv.options());
  torch::Tensor part;
  float* part_ptr = nullptr;
  if (C > 1) {
    part = torch::empty({(long)B * Hq * C * (D + 2)},
                        qkv.options().dtype(torch::kFloat32));
    part_ptr = part.data_ptr<float>();
  }
  launch_attn_decode(qkv.data_ptr(), ck.data_ptr(), cv.data_ptr(),
                     o.data_ptr(), part_ptr, lens.data_ptr<int>(), B, Tmax,
                     (int)Hq, (int)Hk, C, ck.stride(0), ck.stride(1),
                     ck.stride(2), (long)(Hq + 2 * Hk) * D, (float)scale,
                     (void*)cur_stream());
  return o;
}

// Fused cast + per-channel affine: y = (x - shift) * scale. x u8 or f32;
// out bf16 or f32. scale/shift: f32 tensors of numel 1 (scalar) or C
// (= x's innermost dim). Data-plane collate hot path.
torch::Tensor cast_affine(torch::Tensor x, torch::Tensor scale,
                          torch::Tensor shift, torch::ScalarType out_dtype) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous(), "x must be contiguous GPU");
  TORCH_CHECK(scale.is_cuda() && scale.scalar_type() == torch::kFloat32 &&
              shift.is_cuda() && shift.scalar_type() == torch::kFloat32,
              "scale/shift must be f32 on GPU");
  TORCH_CHECK(out_dtype == torch::kBFloat16 || out_dtype == torch::kFloat32,
              "out dtype must be bf16 or f32");
  long n = x.numel();
  int C = (int)x.size(-1);
  int per_channel = scale.numel() > 1;
  if (per_channel) {
    TORCH_CHECK(scale.numel() == C && shift.numel() == C,
                "per-channel scale/shift must match innermost dim");
  }
  auto y = torch::empty(x.sizes(), x.options().dtype(out_dtype));
  int out_f32 = out_dtype == torch::kFloat32;
  if (x.scalar_type() == torch::kUInt8) {
    launch_cast_affine_u8(x.data_ptr(), y.data_ptr(),
                          scale.data_ptr<float>(), shift.data_ptr<float>(), n,
                          C, per_channel, out_f32, (void*)cur_stream());
  } else if (x.scalar_type() == torch::kFloat32) {
    launch_cast_affine_f32(x.data_ptr(), y.data_ptr(),
                           scale.data_ptr<float>(), shift.data_ptr<float>(),
                           n, C, per_channel, out_f32, (void*)cur_stream());
  } else {
    TORCH_CHECK(false, "cast_affine supports u8 or f32 input");
  }
  return y;
}

// Fused NHWC u8 -> NCHW normalize: x [N,H,W,C] u8 -> [N,C,H,W] bf16/f32.
torch::Tensor nhwc_to_nchw(torch::Tensor x, torch::Tensor scale,
                           torch::Tensor shift, torch::ScalarType out_dtype) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() && x.dim() == 4 &&
              x.scalar_type() == torch::kUInt8, "x must be [N,H,W,C] u8 GPU");
  long N = x.size(0), H = x.size(1), W = x.size(2);
  int C = (int)x.size(3);
  TORCH_CHECK(scale.numel() == C && shift.numel() == C &&
              scale.is_cuda() && shift.is_cuda(), "scale/shift must be [C]");
  auto y = torch::empty({N, C, H, W}, x.options().dtype(out_dtype));
  launch_nhwc_to_nchw_u8(x.data_ptr(), y.data_ptr(), scale.data_ptr<float>(),
                         shift.data_ptr<float>(), N * H * W, H * W, C,
                         out_dtype == torch::kFloat32,
                         (void*)cur_stream());
  return y;
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("rmsnorm_fwd", &rmsnorm_fwd, "fused (add+)rmsnorm forward",
        py::arg("x"), py::arg("w"), py::arg("residual") = py::none(),
        py::arg("eps") = 1e-5);
  m.def("rmsnorm_bwd", &rmsnorm_bwd);
  m.def("rope_", &rope_, py::arg("q"), py::arg("k"), py::arg("cos"),
        py::arg("sin"), py::arg("backward") = false);
  m.def("swiglu_fwd", &swiglu_fwd);
  m.def("swiglu_bwd", &swiglu_bwd);
  m.def("cross_entropy_fwd_bwd", &cross_entropy_fwd_bwd, py::arg("logits"),
        py::arg("targets"), py::arg("grad_scale") = 1.0,
        py::arg("ignore_index") = -100, py::arg("write_dlogits") = true);
  m.def("adamw_", &adamw_);
  m.def("attn_fwd", &attn_fwd, py::arg("q"), py::arg("k"), py::arg("v"),
        py::arg("scale"), py::arg("causal") = true,
        py::arg("need_lse") = false);
  m.def("attn_bwd", &attn_bwd, py::arg("dout"), py::arg("q"), py::arg("k"),
        py::arg("v"), py::arg("o"), py::arg("lse"), py::arg("scale"),
        py::arg("causal") = true, py::arg("dq_out") = py::none(),
        py::arg("dk_out") = py::none(), py::arg("dv_out") = py::none());
  m.def("decode_step_attn", &decode_step_attn, py::arg("qkv"), py::arg("ck"),
        py::arg("cv"), py::arg("lens"), py::arg("cos"), py::arg("sin"),
        py::arg("hq"), py::arg("hk"), py::arg("scale"));
  m.def("attn_decode", &attn_decode, py::arg("q"), py::arg("k"), py::arg("v"),
        py::arg("T"), py::arg("scale"), py::arg("lens") = py::none());
  m.def("cast_affine", &cast_affine, py::arg("x"), py::arg("scale"),
        py::arg("shift"), py::arg("out_dtype"));
  m.def("nhwc_to_nchw", &nhwc_to_nchw, py::arg("x"), py::arg("scale"),
        py::arg("shift"), py::arg("out_dtype"));
  m.def("bf16_scale_", &bf16_scale_);
  m.def("bf16_to_f32", &bf16_to_f32);
  m.def("f32_to_bf16_", &f32_to_bf16_);
}

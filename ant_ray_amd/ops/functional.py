"""Autograd wrappers over the CDNA4 kernels (GPU) / torch references (CPU).

These replace the library-call sites the reference delegates to PyTorch
(the reference ships zero CUDA kernels) with hand-written
HIP kernels on the MI355X path.
"""
from __future__ import annotations

import torch

from ant_ray_amd.ops import reference as ref


def _hip():
    from ant_ray_amd.ops import hip_ops

    return hip_ops()


def _use_hip(t: torch.Tensor) -> bool:
    return t.is_cuda


# ---------------------------------------------------------------- rmsnorm


class _RMSNormFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, w, eps):
        y, rstd, _ = _hip().rmsnorm_fwd(x, w, None, eps)
        ctx.save_for_backward(x, w, rstd)
        return y

    @staticmethod
    def backward(ctx, dy):
        x, w, rstd = ctx.saved_tensors
        dx, dw = _hip().rmsnorm_bwd(dy.contiguous(), x, w, rstd)
        return dx, dw.to(w.dtype), None


def rmsnorm(x: torch.Tensor, w: torch.Tensor, eps: float = 1e-5) -> torch.Tensor:
    if _use_hip(x):
        return _RMSNormFn.apply(x.contiguous(), w, eps)
    # CPU reference with autograd
    xf = x.float()
    rstd = torch.rsqrt(xf.pow(2).mean(-1, keepdim=True) + eps)
    return (xf * rstd * w.float()).to(x.dtype)


class _FusedAddRMSNormFn(torch.autograd.Function):
    """(x, residual) -> (y, h) where h = bf16(x+residual) is a FRESH buffer
    (training keeps each layer's h for backward), y = rmsnorm(h)*w — one
    fused kernel pass."""

    @staticmethod
    def forward(ctx, x, residual, w, eps):
        y, rstd, h = _hip().rmsnorm_fwd(x, w, residual, eps)
        ctx.save_for_backward(h, w, rstd)
        return y, h

    @staticmethod
    def backward(ctx, dy, dh_out):
        h, w, rstd = ctx.saved_tensors
        dx, dw = _hip().rmsnorm_bwd(dy.contiguous(), h, w, rstd)
        if dh_out is not None:
            dx = dx + dh_out
        return dx, dx, dw.to(w.dtype), None


def fused_add_rmsnorm(x, residual, w, eps: float = 1e-5):
    """Returns (normalized, h=x+residual). h is a fresh tensor."""
    if _use_hip(x):
        return _FusedAddRMSNormFn.apply(x.contiguous(), residual.contiguous(), w, eps)
    h = (x.float() + residual.float()).to(x.dtype)
    return rmsnorm(h, w, eps), h


# ------------------------------------------------------------------- rope


class _RopeQKVFn(torch.autograd.Function):
    """Rotates the q,k regions of the fused qkv buffer in place."""

    @staticmethod
    def forward(ctx, qkv, cos, sin, Hq, Hk, D):
        B, S, _ = qkv.shape
        q = qkv[:, :, : Hq * D].view(B, S, Hq, D)
        k = qkv[:, :, Hq * D : (Hq + Hk) * D].view(B, S, Hk, D)
        _hip().rope_(q, k, cos, sin, False)
        ctx.meta = (Hq, Hk, D)
        ctx.tables = (cos, sin)
        ctx.mark_dirty(qkv)
        return qkv

    @staticmethod
    def backward(ctx, dqkv):
        Hq, Hk, D = ctx.meta
        cos, sin = ctx.tables
        B, S, _ = dqkv.shape
        # clone: the incoming gradient may be used elsewhere by autograd and
        # the inverse rotation below is in-place
        dqkv = dqkv.contiguous().clone()
        dq = dqkv[:, :, : Hq * D].view(B, S, Hq, D)
        dk = dqkv[:, :, Hq * D : (Hq + Hk) * D].view(B, S, Hk, D)
        _hip().rope_(dq, dk, cos, sin, True)
        return dqkv, None, None, None, None, None


def rope_qkv(qkv: torch.Tensor, cos: torch.Tensor, sin: torch.Tensor,
             Hq: int, Hk: int, D: int) -> torch.Tensor:
    """qkv: [B, S, (Hq+2*Hk)*D] fused projection output; rotates q,k in place
    (GPU) and returns the buffer."""
    if _use_hip(qkv):
        return _RopeQKVFn.apply(qkv, cos, sin, Hq, Hk, D)
    B, S, _ = qkv.shape
    q = qkv[:, :, : Hq * D].view(B, S, Hq, D)
    k = qkv[:, :, Hq * D : (Hq + Hk) * D].view(B, S, Hk, D)
    v = qkv[:, :, (Hq + Hk) * D :]
    qr = ref.rope_apply(q, cos, sin).reshape(B, S, Hq * D)
    kr = ref.rope_apply(k, cos, sin).reshape(B, S, Hk * D)
    return torch.cat([qr, kr, v], dim=-1)


# ----------------------------------------------------------------- swiglu


class _SwiGLUFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, gu):
        out = _hip().swiglu_fwd(gu)
        ctx.save_for_backward(gu)
        return out

    @staticmethod
    def backward(ctx, dout):
        (gu,) = ctx.saved_tensors
        return _hip().swiglu_bwd(dout.contiguous(), gu)


def swiglu(gate_up: torch.Tensor) -> torch.Tensor:
    if _use_hip(gate_up):
        return _SwiGLUFn.apply(gate_up.contiguous())
    I = gate_up.shape[-1] // 2
    g = gate_up[..., :I].float()
    u = gate_up[..., I:].float()
    return (torch.nn.functional.silu(g) * u).to(gate_up.dtype)


# -------------------------------------------------------------- attention


class _AttentionFn(torch.autograd.Function):
    """Training path: custom flash fwd saving (q,k,v,o,lse); bwd runs the
    hand-written dq/dkv kernels (csrc/kernels/attention_bwd.hip)."""

    @staticmethod
    def forward(ctx, q, k, v, causal, scale):
        o, lse = _hip().attn_fwd(q, k, v, float(scale), causal, True)
        ctx.save_for_backward(q, k, v, o, lse)
        ctx.causal = causal
        ctx.scale = float(scale)
        return o

    @staticmethod
    def backward(ctx, dout):
        q, k, v, o, lse = ctx.saved_tensors
        dq, dk, dv = _hip().attn_bwd(dout, q, k, v, o, lse, ctx.scale,
                                     ctx.causal)
        return dq, dk, dv, None, None


def attention(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
              causal: bool = True, scale=None,
              need_lse: bool = False):
    """Flash attention forward on the hand-written CDNA4 MFMA kernel
    (csrc/kernels/attention.hip). q: [B,Hq,S,D], k/v: [B,Hk,S,D] (GQA by
    head-count ratio), D=128, bf16. Returns o (and lse in log2 domain when
    need_lse). CPU fallback: exact fp32 reference."""
    if scale is None:
        scale = q.shape[-1] ** -0.5
    if _use_hip(q):
        if torch.is_grad_enabled() and (q.requires_grad or k.requires_grad
                                        or v.requires_grad):
            if need_lse:
                raise ValueError("need_lse not supported on the grad path")
            return _AttentionFn.apply(q, k, v, causal, float(scale))
        o, lse = _hip().attn_fwd(q, k, v, float(scale), causal, need_lse)
        return (o, lse) if need_lse else o
    o = ref.attention_ref(q, k, v, causal, float(scale))
    if need_lse:
        raise NotImplementedError("need_lse is GPU-only")
    return o


class _RopeAttentionFn(torch.autograd.Function):
    """Fused RoPE + flash attention over the packed qkv projection.

    Forward: rotate q/k IN PLACE on the qkv buffer, run the flash kernel
    on the strided head views. Backward: dq/dk/dv are written by the bwd
    kernels DIRECTLY into the strided slices of ONE dqkv buffer (the
    kernels take output strides), then the inverse rotation runs in place
    — this removes autograd's slice-backward zero-fill + three scatter
    copies and the rope-backward clone (~40 GB/step of glue traffic on
    the llama3-8b bench shape)."""

    @staticmethod
    def forward(ctx, qkv, cos, sin, Hq, Hk, D, causal, scale):
        B, S, _ = qkv.shape
        q4 = qkv[..., : Hq * D].view(B, S, Hq, D)
        k4 = qkv[..., Hq * D : (Hq + Hk) * D].view(B, S, Hk, D)
        _hip().rope_(q4, k4, cos, sin, False)
        q = q4.transpose(1, 2)
        k = k4.transpose(1, 2)
        v = qkv[..., (Hq + Hk) * D :].view(B, S, Hk, D).transpose(1, 2)
        o, lse = _hip().attn_fwd(q, k, v, float(scale), causal, True)
        ctx.save_for_backward(qkv, o, lse, cos, sin)
        ctx.meta = (Hq, Hk, D, causal, float(scale))
        # the rotated qkv must be an OUTPUT to be marked dirty (autograd
        # contract for in-place); callers ignore it
        ctx.mark_dirty(qkv)
        ctx.set_materialize_grads(False)
        return o, qkv

    @staticmethod
    def backward(ctx, dout, dqkv_passthrough):
        qkv, o, lse, cos, sin = ctx.saved_tensors
        if dout is None:
            return dqkv_passthrough, None, None, None, None, None, None, None
        Hq, Hk, D, causal, scale = ctx.meta
        B, S, _ = qkv.shape
        q = qkv[..., : Hq * D].view(B, S, Hq, D).transpose(1, 2)
        k = qkv[..., Hq * D : (Hq + Hk) * D].view(B, S, Hk, D).transpose(1, 2)
        v = qkv[..., (Hq + Hk) * D :].view(B, S, Hk, D).transpose(1, 2)
        dqkv = torch.empty_like(qkv)
        dq = dqkv[..., : Hq * D].view(B, S, Hq, D).transpose(1, 2)
        dk = dqkv[..., Hq * D : (Hq + Hk) * D].view(B, S, Hk, D).transpose(1, 2)
        dv = dqkv[..., (Hq + Hk) * D :].view(B, S, Hk, D).transpose(1, 2)
        _hip().attn_bwd(dout, q, k, v, o, lse, scale, causal,
                        dq_out=dq, dk_out=dk, dv_out=dv)
        dq4 = dqkv[..., : Hq * D].view(B, S, Hq, D)
        dk4 = dqkv[..., Hq * D : (Hq + Hk) * D].view(B, S, Hk, D)
        _hip().rope_(dq4, dk4, cos, sin, True)  # inverse rotation, in place
        if dqkv_passthrough is not None:
            dqkv = dqkv + dqkv_passthrough
        return dqkv, None, None, None, None, None, None, None


def rope_attention(qkv: torch.Tensor, cos, sin, Hq: int, Hk: int, D: int,
                   causal: bool = True, scale=None) -> torch.Tensor:
    """Fused RoPE(qkv in place) + flash attention; returns o [B,Hq,S,D].
    GPU-only fast path for the training hot loop (models/llama.py)."""
    if scale is None:
        scale = D ** -0.5
    o, _ = _RopeAttentionFn.apply(qkv, cos, sin, Hq, Hk, D, causal,
                                  float(scale))
    return o


def attention_decode(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
                     seq_len: int, scale=None, lens=None) -> torch.Tensor:
    """Flash-decode: one new token per sequence over a bf16 KV cache
    (csrc/kernels/attention_decode.hip — flash-decode two-level combine,
    GQA heads share one K/V read). q: [B,Hq,D]; k/v: [B,Hk,Tmax,D] with
    seq_len valid rows (or per-sequence int32 `lens` on device). Inference
    only (no autograd). CPU fallback: exact fp32 reference."""
    if scale is None:
        scale = q.shape[-1] ** -0.5
    if _use_hip(q):
        return _hip().attn_decode(q.contiguous(), k, v, int(seq_len),
                                  float(scale), lens)
    kc = k[:, :, :seq_len]
    vc = v[:, :, :seq_len]
    return ref.attention_decode_ref(q, kc, vc, lens=lens, scale=float(scale))


def chunked_prefill_attention(q: torch.Tensor, k_new: torch.Tensor,
                              v_new: torch.Tensor, ck: torch.Tensor,
                              cv: torch.Tensor, pos: int,
                              scale=None) -> torch.Tensor:
    """Attention for CHUNKED PREFILL (prefix-cache hit): suffix queries
    attend over `pos` cached prefix positions (unmasked) plus the suffix
    itself (causal). GPU: square causal flash on the suffix (need_lse) +
    a bf16-GEMM / fp32-logsumexp pass over the prefix, combined by LSE
    weights — the flash-decode two-part combine, at prefill granularity.
    (A masked SDPA here falls back to the math backend and forfeits the
    prefix-cache win; this path keeps the GEMMs on hipBLASLt.)

    q: [B,Hq,S,D]; k_new/v_new: [B,Hk,S,D] rope'd suffix keys/values
    (also already written to the cache at [pos, pos+S)); ck/cv:
    [B,Hk,Tmax,D]. Returns o [B,Hq,S,D]. Inference only.
    """
    B, Hq, S, D = q.shape
    Hk = ck.shape[1]
    G = Hq // Hk
    if scale is None:
        scale = D ** -0.5
    if not _use_hip(q):
        # exact fallback: masked SDPA over prefix+suffix
        import torch.nn.functional as F

        kf = ck[:, :, : pos + S]
        vf = cv[:, :, : pos + S]
        mask = torch.ones(S, pos + S, dtype=torch.bool,
                          device=q.device).tril_(diagonal=pos)
        return F.scaled_dot_product_attention(q, kf, vf, attn_mask=mask,
                                              enable_gqa=True)
    oB, lseB = _hip().attn_fwd(q, k_new, v_new, float(scale), True, True)
    kp = ck[:, :, :pos]
    vp = cv[:, :, :pos]
    # prefix part: grouped bf16 GEMMs + fp32 running-softmax stats
    qg = q.reshape(B, Hk, G * S, D)  # q heads are kv-head-major
    sp = torch.matmul(qg, kp.transpose(-1, -2)).float() * float(scale)
    mA = sp.amax(-1, keepdim=True)                    # [B,Hk,G*S,1]
    eA = torch.exp(sp - mA)
    sA = eA.sum(-1)                                   # [B,Hk,G*S]
    oA = torch.matmul(eA.to(q.dtype), vp).float() / sA.unsqueeze(-1)
    lA = mA.squeeze(-1) + torch.log(sA)               # natural-log LSE
    # suffix flash LSE is log2-domain
    lB = lseB.view(B, Hk, G * S) * 0.6931471805599453
    m = torch.maximum(lA, lB)
    wA = torch.exp(lA - m).unsqueeze(-1)
    wB = torch.exp(lB - m).unsqueeze(-1)
    ob = oB.reshape(B, Hk, G * S, D).float()
    o = (oA * wA + ob * wB) / (wA + wB)
    return o.view(B, Hq, S, D).to(q.dtype)


def decode_step_attn(qkv: torch.Tensor, ck: torch.Tensor, cv: torch.Tensor,
                     lens: torch.Tensor, cos: torch.Tensor, sin: torch.Tensor,
                     hq: int, hk: int, scale=None) -> torch.Tensor:
    """Device-pos single-token decode step: rope q/k + cache writes +
    flash-decode in ONE call, with the token position read from the
    DEVICE `lens` buffer (pos = lens[b]-1). No host scalar depends on
    the step index, so models/llama.py captures the whole token step in
    a hipGraph and replays it. qkv: [B,1,(Hq+2Hk)*D] from the fused
    projection; caches [B,Hk,Tmax,D]; lens int32 [B] (valid length
    INCLUDING the new token). CPU fallback: exact fp32 reference (so the
    ragged/continuous decode path is testable without a GPU)."""
    D = ck.shape[-1]
    if scale is None:
        scale = D ** -0.5
    if _use_hip(qkv):
        return _hip().decode_step_attn(qkv, ck, cv, lens, cos, sin, hq, hk,
                                       float(scale))
    B = qkv.shape[0]
    flat = qkv.reshape(B, -1)
    q = flat[:, : hq * D].view(B, hq, D)
    k = flat[:, hq * D : (hq + hk) * D].view(B, hk, D)
    v = flat[:, (hq + hk) * D :].view(B, hk, D)
    pos = (lens.long() - 1).clamp(min=0)              # per-row position
    half = D // 2
    c = cos[pos].view(B, 1, half).float()
    s = sin[pos].view(B, 1, half).float()

    def rot(x):
        xf = x.float()
        x1, x2 = xf[..., :half], xf[..., half:]
        return torch.cat([x1 * c - x2 * s, x1 * s + x2 * c], -1).to(x.dtype)

    q, k = rot(q), rot(k)
    rows = torch.arange(B, device=qkv.device)
    ck[rows, :, pos] = k
    cv[rows, :, pos] = v
    return ref.attention_decode_ref(q, ck, cv, lens=lens,
                                    scale=float(scale))


# ------------------------------------------------------- data transforms


def cast_affine(x: torch.Tensor, scale=1.0, shift=0.0,
                out_dtype=torch.bfloat16) -> torch.Tensor:
    """Fused y = (cast(x) - shift) * scale for the Data collate path
    (csrc/kernels/data_transform.hip). x: u8 or f32; scale/shift scalars
    or per-channel (innermost dim) sequences. Memory-bound one-pass."""
    def as_t(v):
        if isinstance(v, torch.Tensor):
            return v.to(device=x.device, dtype=torch.float32)
        if isinstance(v, (list, tuple)):
            return torch.tensor(v, device=x.device, dtype=torch.float32)
        return torch.tensor([float(v)], device=x.device, dtype=torch.float32)

    s, h = as_t(scale), as_t(shift)
    if _use_hip(x):
        return _hip().cast_affine(x.contiguous(), s, h, out_dtype)
    v = x.float()
    if s.numel() > 1:
        v = (v - h) * s
    else:
        v = (v - h.item()) * s.item()
    return v.to(out_dtype)


def nhwc_to_nchw(x: torch.Tensor, mean, std,
                 out_dtype=torch.bfloat16) -> torch.Tensor:
    """Fused image collate: [N,H,W,C] u8 -> [N,C,H,W] normalized
    ((x - mean) / std), one HBM pass. mean/std per-channel sequences."""
    def as_t(v):
        if isinstance(v, torch.Tensor):
            return v.to(device=x.device, dtype=torch.float32)
        return torch.tensor(v, device=x.device, dtype=torch.float32)

    m, sd = as_t(mean), as_t(std)
    if _use_hip(x):
        return _hip().nhwc_to_nchw(x.contiguous(), 1.0 / sd, m, out_dtype)
    v = (x.float() - m.view(1, 1, 1, -1)) / sd.view(1, 1, 1, -1)
    return v.permute(0, 3, 1, 2).contiguous().to(out_dtype)


# ------------------------------------------------------------------ adamw


def adamw_step(p32, p_bf16, g, m, v, *, lr, b1=0.9, b2=0.95, eps=1e-8,
               wd=0.0, step, grad_scale=1.0):
    """Fused AdamW over flat buffers (fp32 master + bf16 working copy)."""
    if p32.is_cuda:
        _hip().adamw_(p32, p_bf16, g, m, v, lr, b1, b2, eps, wd, step, grad_scale)
    else:
        ref.adamw_step(p32, p_bf16, g, m, v, lr, b1, b2, eps, wd, step, grad_scale)

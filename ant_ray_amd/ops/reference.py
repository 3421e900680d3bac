"""Pure-PyTorch fp32 reference implementations of the HIP kernels.

Used (a) on CPU (tests run without a GPU), (b) as the comparison baseline for
GPU numerics tests (tests/test_ops_gpu.py compares each HIP kernel against
these at fp32).
"""
from __future__ import annotations

import torch


def rmsnorm(x: torch.Tensor, w: torch.Tensor, eps: float = 1e-5):
    xf = x.float()
    rstd = torch.rsqrt(xf.pow(2).mean(-1, keepdim=True) + eps)
    return (xf * rstd * w.float()).to(x.dtype)


def add_rmsnorm(x: torch.Tensor, residual: torch.Tensor, w: torch.Tensor, eps: float = 1e-5):
    h = (x.float() + residual.float()).to(x.dtype)
    return rmsnorm(h, w, eps), h


def swiglu(gate_up: torch.Tensor):
    I = gate_up.shape[-1] // 2
    g = gate_up[..., :I].float()
    u = gate_up[..., I:].float()
    return (torch.nn.functional.silu(g) * u).to(gate_up.dtype)


def rope_tables(dim: int, max_seq: int, theta: float = 500000.0, device="cpu"):
    inv = 1.0 / (theta ** (torch.arange(0, dim, 2, dtype=torch.float32, device=device) / dim))
    t = torch.arange(max_seq, dtype=torch.float32, device=device)
    freqs = torch.outer(t, inv)  # [S, dim/2]
    return freqs.cos().contiguous(), freqs.sin().contiguous()


def rope_apply(x: torch.Tensor, cos: torch.Tensor, sin: torch.Tensor, backward=False):
    """x: [B,S,H,D]; neox half-rotation."""
    B, S, H, D = x.shape
    half = D // 2
    xf = x.float()
    x1, x2 = xf[..., :half], xf[..., half:]
    c = cos[:S].view(1, S, 1, half)
    s = sin[:S].view(1, S, 1, half)
    if backward:
        s = -s
    o1 = x1 * c - x2 * s
    o2 = x1 * s + x2 * c
    return torch.cat([o1, o2], dim=-1).to(x.dtype)


def cross_entropy(logits: torch.Tensor, targets: torch.Tensor, ignore_index=-100):
    """Returns (per-row loss f32, dlogits wrt sum-loss)."""
    lf = logits.float()
    loss = torch.nn.functional.cross_entropy(
        lf, targets.long(), ignore_index=ignore_index, reduction="none"
    )
    p = torch.softmax(lf, dim=-1)
    valid = (targets != ignore_index).unsqueeze(-1)
    onehot = torch.zeros_like(p)
    t = targets.long().clamp(min=0)
    onehot.scatter_(1, t.unsqueeze(1), 1.0)
    dlogits = (p - onehot) * valid
    return loss, dlogits


def adamw_step(p32, p_bf16, g, m, v, lr, b1, b2, eps, wd, step, grad_scale):
    gf = g.float() * grad_scale
    m.mul_(b1).add_(gf, alpha=1 - b1)
    v.mul_(b2).addcmul_(gf, gf, value=1 - b2)
    bc1 = 1.0 / (1.0 - b1**step)
    bc2 = 1.0 / (1.0 - b2**step)
    update = (m * bc1) / ((v * bc2).sqrt() + eps) + wd * p32
    p32.add_(update, alpha=-lr)
    p_bf16.copy_(p32.to(torch.bfloat16))


def attention_decode_ref(q, k, v, lens=None, scale=None):
    """fp32 reference decode: q [B,Hq,D] (one new token), k/v [B,Hk,T,D]
    caches; lens optional per-sequence valid lengths."""
    B, Hq, D = q.shape
    Hk, T = k.shape[1], k.shape[2]
    if scale is None:
        scale = D ** -0.5
    rep = Hq // Hk
    kf = k.float().repeat_interleave(rep, dim=1)   # [B,Hq,T,D]
    vf = v.float().repeat_interleave(rep, dim=1)
    s = torch.einsum("bhd,bhtd->bht", q.float(), kf) * scale
    if lens is not None:
        pos = torch.arange(T, device=q.device).view(1, 1, T)
        mask = pos >= lens.view(B, 1, 1)
        s = s.masked_fill(mask, float("-inf"))
    p = torch.softmax(s, dim=-1)
    return torch.einsum("bht,bhtd->bhd", p, vf).to(q.dtype)


def attention_ref(q, k, v, causal=True, scale=None):
    """fp32 reference attention with GQA (q:[B,Hq,S,D], k/v:[B,Hk,S,D])."""
    import torch

    B, Hq, S, D = q.shape
    Hk = k.shape[1]
    if scale is None:
        scale = D ** -0.5
    rep = Hq // Hk
    kf = k.float().repeat_interleave(rep, dim=1)
    vf = v.float().repeat_interleave(rep, dim=1)
    s = torch.einsum("bhid,bhjd->bhij", q.float(), kf) * scale
    if causal:
        mask = torch.ones(S, S, dtype=torch.bool, device=q.device).tril()
        s = s.masked_fill(~mask, float("-inf"))
    p = torch.softmax(s, dim=-1)
    return torch.einsum("bhij,bhjd->bhid", p, vf).to(q.dtype)

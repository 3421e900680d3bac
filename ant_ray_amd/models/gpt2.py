"""GPT-2, MI355X-first implementation (BASELINE config 2: GPT-2-small DDP
bf16 on 1 MI355X). Vocab padded to a multiple of 8 (50304) for the fused CE
kernel; weights tied (wte = lm head) as in GPT-2.
"""
from __future__ import annotations

import math
from dataclasses import dataclass

import torch
import torch.nn as nn
import torch.nn.functional as F

from ant_ray_amd import ops


@dataclass
class GPT2Config:
    hidden: int = 768
    n_layers: int = 12
    n_heads: int = 12
    vocab: int = 50304  # padded 50257 -> %8
    max_seq: int = 1024
    eps: float = 1e-5

    @classmethod
    def small(cls, max_seq=1024):
        return cls(max_seq=max_seq)


class GPT2Block(nn.Module):
    def __init__(self, cfg: GPT2Config):
        super().__init__()
        self.cfg = cfg
        self.ln1 = nn.LayerNorm(cfg.hidden, eps=cfg.eps)
        self.ln2 = nn.LayerNorm(cfg.hidden, eps=cfg.eps)
        self.attn_qkv = nn.Linear(cfg.hidden, 3 * cfg.hidden)
        self.attn_out = nn.Linear(cfg.hidden, cfg.hidden)
        self.mlp_fc = nn.Linear(cfg.hidden, 4 * cfg.hidden)
        self.mlp_proj = nn.Linear(4 * cfg.hidden, cfg.hidden)

    def forward(self, x):
        cfg = self.cfg
        B, S, H = x.shape
        D = H // cfg.n_heads
        qkv = self.attn_qkv(self.ln1(x))
        q, k, v = qkv.split(H, dim=-1)
        q = q.view(B, S, cfg.n_heads, D).transpose(1, 2)
        k = k.view(B, S, cfg.n_heads, D).transpose(1, 2)
        v = v.view(B, S, cfg.n_heads, D).transpose(1, 2)
        o = F.scaled_dot_product_attention(q, k, v, is_causal=True)
        x = x + self.attn_out(o.transpose(1, 2).reshape(B, S, H))
        x = x + self.mlp_proj(F.gelu(self.mlp_fc(self.ln2(x)), approximate="tanh"))
        return x


class GPT2LMHeadModel(nn.Module):
    def __init__(self, cfg: GPT2Config, device=None):
        super().__init__()
        self.cfg = cfg
        with torch.device(device or "cpu"):
            self.wte = nn.Embedding(cfg.vocab, cfg.hidden)
            self.wpe = nn.Embedding(cfg.max_seq, cfg.hidden)
            self.blocks = nn.ModuleList(GPT2Block(cfg) for _ in range(cfg.n_layers))
            self.ln_f = nn.LayerNorm(cfg.hidden, eps=cfg.eps)
        self.to(dtype=torch.bfloat16)
        if device is not None:
            self.to(device)
        self._init_weights()

    def _init_weights(self):
        std = 0.02
        out_std = std / math.sqrt(2 * self.cfg.n_layers)
        for name, p in self.named_parameters():
            if p.dim() >= 2:
                s = out_std if ("out" in name or "proj" in name) else std
                nn.init.normal_(p, mean=0.0, std=s)

    def forward(self, tokens, targets=None):
        B, S = tokens.shape
        pos = torch.arange(S, device=tokens.device)
        x = self.wte(tokens) + self.wpe(pos)[None]
        for blk in self.blocks:
            x = blk(x)
        x = self.ln_f(x)
        flat = x.reshape(B * S, self.cfg.hidden)
        if targets is not None:
            return ops.linear_cross_entropy(flat, self.wte.weight, targets.reshape(-1))
        return (flat @ self.wte.weight.t()).view(B, S, -1)

    def num_params(self):
        return sum(p.numel() for p in self.parameters())

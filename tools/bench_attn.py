"""Micro-benchmark: custom CDNA4 flash-attn fwd vs torch SDPA (AOTriton).

Run on the GPU box:  python tools/bench_attn.py
"""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

import ant_ray_amd.ops as ops


def bench(fn, iters=20, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.time()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.time() - t0) / iters


def main():
    B, Hq, Hk, S, D = 6, 32, 8, 4096, 128
    scale = D ** -0.5
    q = torch.randn(B, Hq, S, D, device="cuda", dtype=torch.bfloat16)
    k = torch.randn(B, Hk, S, D, device="cuda", dtype=torch.bfloat16)
    v = torch.randn(B, Hk, S, D, device="cuda", dtype=torch.bfloat16)

    flops = 4 * B * Hq * S * S * D * 0.5  # causal

    t_ours = bench(lambda: ops.attention(q, k, v, causal=True))
    print(f"ours      : {t_ours*1e3:8.2f} ms  {flops/t_ours/1e12:7.1f} TF/s")

    t_sdpa = bench(lambda: torch.nn.functional.scaled_dot_product_attention(
        q, k, v, is_causal=True, enable_gqa=True))
    print(f"torch sdpa: {t_sdpa*1e3:8.2f} ms  {flops/t_sdpa/1e12:7.1f} TF/s")
    print(f"speedup: {t_sdpa/t_ours:.2f}x")


if __name__ == "__main__":
    main()

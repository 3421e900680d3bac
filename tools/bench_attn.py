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
    print(f"fwd speedup: {t_sdpa/t_ours:.2f}x")

    # ---- fwd+bwd
    bwd_flops = flops * 3.5  # S recomputed twice + 4 bwd gemms + fwd
    qg = q.clone().requires_grad_()
    kg = k.clone().requires_grad_()
    vg = v.clone().requires_grad_()
    dout = torch.randn_like(q)

    def ours_fb():
        out = ops.attention(qg, kg, vg, causal=True)
        out.backward(dout)
        qg.grad = kg.grad = vg.grad = None

    t_ours_fb = bench(ours_fb, iters=10)
    print(f"ours f+b  : {t_ours_fb*1e3:8.2f} ms  {bwd_flops/t_ours_fb/1e12:7.1f} TF/s")

    def sdpa_fb():
        out = torch.nn.functional.scaled_dot_product_attention(
            qg, kg, vg, is_causal=True, enable_gqa=True)
        out.backward(dout)
        qg.grad = kg.grad = vg.grad = None

    t_sdpa_fb = bench(sdpa_fb, iters=10)
    print(f"sdpa f+b  : {t_sdpa_fb*1e3:8.2f} ms  {bwd_flops/t_sdpa_fb/1e12:7.1f} TF/s")
    print(f"f+b speedup: {t_sdpa_fb/t_ours_fb:.2f}x")


if __name__ == "__main__":
    main()

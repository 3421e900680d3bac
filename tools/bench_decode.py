"""Decode-path benchmarks on the GPU box.

1. attn_decode kernel: ms + achieved GB/s of KV-cache read at llama3-8b
   decode shapes (the kernel is memory-bound; ceiling ~6.3 TB/s).
2. end-to-end generate() tokens/s for llama3-8b bf16 at a few batch sizes
   (weights 16 GB re-read per step -> ~2.5 ms/step floor at the HBM
   ceiling, plus KV bytes).

Run: python tools/bench_decode.py [--model llama3-8b]
"""
import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch

import ant_ray_amd.ops as ops


def bench(fn, iters=50, warmup=10):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.time()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.time() - t0) / iters


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--model", default="llama3-8b")
    ap.add_argument("--gen-batches", default="1,8,32")
    ap.add_argument("--prompt", type=int, default=512)
    ap.add_argument("--new-tokens", type=int, default=64)
    args = ap.parse_args()

    print("== attn_decode kernel (llama3-8b shapes) ==")
    Hq, Hk, D = 32, 8, 128
    for B, T in [(1, 1024), (1, 4096), (8, 1024), (8, 4096), (32, 2048)]:
        q = torch.randn(B, Hq, D, device="cuda", dtype=torch.bfloat16)
        k = torch.randn(B, Hk, T, D, device="cuda", dtype=torch.bfloat16)
        v = torch.randn(B, Hk, T, D, device="cuda", dtype=torch.bfloat16)
        t = bench(lambda: ops.attention_decode(q, k, v, seq_len=T))
        bytes_rd = B * Hk * T * D * 2 * 2
        print(f"B={B:3d} T={T:5d}: {t*1e6:8.1f} us  {bytes_rd/t/1e12:6.2f} TB/s")

    print(f"== generate() {args.model} ==")
    from ant_ray_amd.models import build_model, setup_tunableop

    setup_tunableop()

    m = build_model(args.model, device="cuda",
                    seq_len=args.prompt + args.new_tokens + 8)
    m.eval()
    vocab = m.cfg.vocab
    for B in [int(x) for x in args.gen_batches.split(",")]:
        toks = torch.randint(0, vocab, (B, args.prompt), device="cuda")
        # warm — 12 new tokens crosses the graph-capture threshold so the
        # process's expensive FIRST hipGraph instantiation (~0.7 s) lands
        # here, not in the timed region
        m.generate(toks[:, :32], max_new_tokens=12)
        torch.cuda.synchronize()
        t0 = time.time()
        out = m.generate(toks, max_new_tokens=args.new_tokens)
        torch.cuda.synchronize()
        wall = time.time() - t0
        ntok = out.shape[1] - args.prompt
        print(f"B={B:3d}: prefill+{ntok} new in {wall*1e3:8.1f} ms  "
              f"decode {B*ntok/wall:8.1f} tok/s  "
              f"({wall/ntok*1e3:6.2f} ms/step incl prefill amortized)")


if __name__ == "__main__":
    main()

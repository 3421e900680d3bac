"""A/B the native engine's prefix caching on the llama3-8b serve shape.

Scenario: a long shared system prompt (--prefix tokens) + per-request
unique suffix (--suffix), --new generated tokens. With the cache, the
shared prefix's prefill FLOPs are skipped (KV blocks copied D2D instead).

Run (GPU): python tools/bench_prefix_cache.py > gpurun_out/prefix_ab.json
"""
import argparse
import json
import os
import random
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def run(engine, prompts, new_toks, iters):
    ts = []
    for i in range(iters):
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        engine.generate_tokens([prompts[i % len(prompts)]], new_toks)
        torch.cuda.synchronize()
        ts.append(time.perf_counter() - t0)
    return ts


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--model", default="llama3-8b")
    ap.add_argument("--prefix", type=int, default=3072)
    ap.add_argument("--suffix", type=int, default=64)
    ap.add_argument("--new", type=int, default=32)
    ap.add_argument("--iters", type=int, default=6)
    args = ap.parse_args()

    os.environ["ANTRAY_PREFIX_CACHE_MB"] = "2048"
    from ant_ray_amd.llm.native_engine import NativeLLMEngine

    random.seed(0)
    sysp = [random.randrange(0, 128000) for _ in range(args.prefix)]
    prompts = [sysp + [random.randrange(0, 128000)
                       for _ in range(args.suffix)]
               for _ in range(args.iters)]

    eng = NativeLLMEngine(args.model, max_seq=4096, device="cuda")
    assert eng.prefix_cache is not None
    warm = run(eng, prompts[:1], args.new, 1)  # miss + capture + insert
    run(eng, prompts[1:2], args.new, 1)  # warm the HIT path's kernels
    hits = run(eng, prompts[2:], args.new, args.iters - 2)
    stats = eng.prefix_cache.stats()

    eng.prefix_cache = None  # disable: every call prefills everything
    run(eng, prompts[1:2], args.new, 1)  # warm (same-shape full prefill)
    cold = run(eng, prompts[2:], args.new, args.iters - 2)

    out = {
        "model": args.model, "prefix_tokens": args.prefix,
        "suffix_tokens": args.suffix, "new_tokens": args.new,
        "first_call_s": round(warm[0], 4),
        "hit_mean_s": round(sum(hits) / len(hits), 4),
        "nocache_mean_s": round(sum(cold) / len(cold), 4),
        "hit_times_s": [round(t, 4) for t in hits],
        "nocache_times_s": [round(t, 4) for t in cold],
        "speedup_vs_nocache": round(
            (sum(cold) / len(cold)) / (sum(hits) / len(hits)), 3),
        "cache_stats": {k: v for k, v in stats.items() if k != "bytes"},
        "cache_bytes_mb": round(stats["bytes"] / (1 << 20), 1),
    }
    print(json.dumps(out))


if __name__ == "__main__":
    main()

"""Serve LLM benchmark (BASELINE config 4, measured at the replicas a
1-GPU lease allows): Llama-3-8B bf16 behind a Serve deployment with the
native MI355X engine, dynamic batching, measured req/s + latency
percentiles. Writes profiles/serve_llama3_8b_<n>gpu_r02.json via --out.

Run on the GPU box:
  python tools/bench_serve_llm.py [--model llama3-8b] [--replicas 1]
      [--requests 64] [--concurrency 16] [--prompt 128] [--new-tokens 32]
"""
import argparse
import json
import os
import random
import sys
import threading
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--model", default="llama3-8b")
    ap.add_argument("--replicas", type=int, default=1)
    ap.add_argument("--requests", type=int, default=64)
    ap.add_argument("--concurrency", type=int, default=16)
    ap.add_argument("--prompt", type=int, default=128)
    ap.add_argument("--new-tokens", type=int, default=32)
    ap.add_argument("--out", default="")
    ap.add_argument("--continuous", action="store_true",
                    help="token-level continuous batching engine")
    ap.add_argument("--ragged", action="store_true",
                    help="sample prompt lengths in [16, --prompt] instead "
                         "of fixed (exposes batching-policy differences)")
    args = ap.parse_args()

    import ant_ray_amd as ray
    from ant_ray_amd import serve
    from ant_ray_amd.llm import LLMConfig, build_llm_deployment

    ray.init(num_cpus=8, num_gpus=args.replicas)
    app = build_llm_deployment(LLMConfig(
        model_loading_config={"model_id": args.model},
        engine_kwargs={"max_model_len": args.prompt + args.new_tokens + 8,
                       "max_num_seqs": args.concurrency,
                       "continuous_batching": args.continuous},
        deployment_config={"num_replicas": args.replicas},
    ))
    h = serve.run(app, name="llm", route_prefix="/llm")

    vocab = 128256 if "8b" in args.model else 1024
    rng = random.Random(0)

    def req_payload():
        plen = rng.randrange(16, args.prompt + 1) if args.ragged \
            else args.prompt
        return {"prompt_ids": [rng.randrange(vocab) for _ in range(plen)],
                "max_new_tokens": args.new_tokens}

    # warm (model build + first kernels)
    r = h.remote(req_payload()).result(timeout_s=600)
    assert len(r["token_ids"]) == args.new_tokens, r

    lat = []
    lat_lock = threading.Lock()
    sem = threading.Semaphore(args.concurrency)
    done = threading.Event()
    remaining = [args.requests]

    def fire():
        t0 = time.time()
        resp = h.remote(req_payload())

        def wait():
            try:
                resp.result(timeout_s=600)
                with lat_lock:
                    lat.append(time.time() - t0)
            finally:
                sem.release()
                with lat_lock:
                    remaining[0] -= 1
                    if remaining[0] == 0:
                        done.set()

        threading.Thread(target=wait, daemon=True).start()

    t_start = time.time()
    for _ in range(args.requests):
        sem.acquire()
        fire()
    done.wait(timeout=900)
    wall = time.time() - t_start

    lat.sort()
    n = len(lat)
    result = {
        "metric": "Serve Llama-3-8B bf16 req/s (native engine)",
        "model": args.model,
        "replicas": args.replicas,
        "requests": args.requests,
        "concurrency": args.concurrency,
        "prompt_tokens": args.prompt,
        "new_tokens": args.new_tokens,
        "req_per_s": round(n / wall, 3),
        "gen_tok_per_s": round(n * args.new_tokens / wall, 1),
        "p50_s": round(lat[n // 2], 3) if n else None,
        "p95_s": round(lat[int(n * 0.95)] if n > 1 else lat[0], 3) if n else None,
        "wall_s": round(wall, 2),
        "completed": n,
        "continuous_batching": args.continuous,
        "ragged_prompts": args.ragged,
    }
    print(json.dumps(result), flush=True)
    if args.out:
        with open(args.out, "w") as f:
            json.dump(result, f, indent=1)
    serve.shutdown()
    ray.shutdown()


if __name__ == "__main__":
    main()

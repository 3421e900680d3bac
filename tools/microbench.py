"""Core-runtime microbenchmark — mirrors the reference's ray_perf.py
(python/ray/_private/ray_perf.py:95) metric definitions so results compare
1:1 against BASELINE.md's published numbers.

Usage: python tools/microbench.py [--quick]
Prints one JSON line per metric: {"name", "value", "unit"}.
"""
import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np

import ant_ray_amd as ray


def timeit(name, fn, multiplier=1, duration=2.0):
    # warmup
    fn()
    start = time.time()
    count = 0
    while time.time() - start < duration:
        fn()
        count += 1
    elapsed = time.time() - start
    rate = count * multiplier / elapsed
    print(json.dumps({"name": name, "value": round(rate, 1),
                      "unit": "ops/s"}), flush=True)
    return rate


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--quick", action="store_true")
    args = ap.parse_args()
    dur = 1.0 if args.quick else 2.0

    ray.init(num_cpus=os.cpu_count())

    # ---- object store
    small = np.zeros(8, dtype=np.uint8)
    arr_1mb = np.zeros(1024 * 1024, dtype=np.uint8)

    ref = ray.put(small)
    timeit("single client get calls (Plasma Store)",
           lambda: [ray.get(ref) for _ in range(100)], 100, dur)
    timeit("single client put calls (Plasma Store)",
           lambda: [ray.put(small) for _ in range(100)], 100, dur)

    def put_gb():
        for _ in range(8):
            ray.put(arr_1mb)

    n = [0]
    start = time.time()
    put_gb()
    while time.time() - start < dur:
        put_gb()
        n[0] += 1
    gbps = (n[0] + 1) * 8 / 1024 / (time.time() - start)
    print(json.dumps({"name": "single client put gigabytes",
                      "value": round(gbps, 2), "unit": "GB/s"}), flush=True)

    # ---- tasks
    @ray.remote
    def tiny():
        return b"ok"

    timeit("single client tasks sync",
           lambda: [ray.get(tiny.remote()) for _ in range(100)], 100, dur)
    timeit("single client tasks async",
           lambda: ray.get([tiny.remote() for _ in range(1000)]), 1000, dur)

    # ---- actors
    @ray.remote
    class Actor:
        def ping(self):
            return b"ok"

    a = Actor.remote()
    ray.get(a.ping.remote())
    timeit("1:1 actor calls sync",
           lambda: [ray.get(a.ping.remote()) for _ in range(100)], 100, dur)
    timeit("1:1 actor calls async",
           lambda: ray.get([a.ping.remote() for _ in range(1000)]), 1000, dur)
    conc = Actor.options(max_concurrency=16).remote()
    ray.get(conc.ping.remote())
    timeit("1:1 actor calls concurrent",
           lambda: ray.get([conc.ping.remote() for _ in range(1000)]), 1000, dur)

    ray.kill(a)
    ray.kill(conc)
    n_cpu = min(os.cpu_count() or 4, 8)
    actors = [Actor.remote() for _ in range(n_cpu)]
    ray.get([x.ping.remote() for x in actors])
    timeit("n:n actor calls async",
           lambda: ray.get([x.ping.remote() for x in actors
                            for _ in range(200)]), 200 * n_cpu, dur)

    # ---- async actors
    @ray.remote
    class AsyncActor:
        async def ping(self):
            return b"ok"

    for x in actors:
        ray.kill(x)
    aa = AsyncActor.remote()
    ray.get(aa.ping.remote())
    timeit("1:1 async-actor calls sync",
           lambda: [ray.get(aa.ping.remote()) for _ in range(100)], 100, dur)
    timeit("1:1 async-actor calls async",
           lambda: ray.get([aa.ping.remote() for _ in range(1000)]), 1000, dur)

    # ---- placement groups
    ray.kill(aa)
    import time as _t
    _t.sleep(1.0)  # let killed actors release CPUs
    from ant_ray_amd.util.placement_group import (
        placement_group,
        remove_placement_group,
    )

    def pg_cycle():
        pgs = [placement_group([{"CPU": 0.01}]) for _ in range(10)]
        for pg in pgs:
            pg.wait(timeout_seconds=10)
        for pg in pgs:
            remove_placement_group(pg)

    timeit("placement group create/removal", pg_cycle, 10, dur)

    ray.shutdown()


if __name__ == "__main__":
    main()

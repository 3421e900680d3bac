"""Core-runtime microbenchmark — mirrors the reference's ray_perf.py
(python/ray/_private/ray_perf.py:95) metric definitions so results compare
1:1 against BASELINE.md's published numbers.

Usage: python tools/microbench.py [--quick]
Prints one JSON line per metric: {"name", "value", "unit"}.
"""
import argparse
import json
import os
import subprocess
import sys
import textwrap
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

    # multi client put: 10 worker processes each doing 100 puts
    # (parity ray_perf.py:113 do_put_small / put_multi_small)
    @ray.remote
    def do_put_small():
        for _ in range(100):
            ray.put(0)

    timeit("multi client put calls (Plasma Store)",
           lambda: ray.get([do_put_small.remote() for _ in range(10)]),
           1000, dur)

    # single client put gigabytes: ONE 800 MB array per put, exactly the
    # reference's shape (ray_perf.py:120 arr = np.zeros(100Mi, int64);
    # put_large puts it whole — sustained large-copy bandwidth, not
    # per-put overhead amortization)
    arr_800mb = np.zeros(100 * 1024 * 1024, dtype=np.int64)

    n = [0]
    start = time.time()
    ray.put(arr_800mb)
    while time.time() - start < max(dur, 3.0):
        ray.put(arr_800mb)
        n[0] += 1
    gbps = (n[0] + 1) * 0.8 / (time.time() - start)
    print(json.dumps({"name": "single client put gigabytes",
                      "value": round(gbps, 2), "unit": "GB/s"}), flush=True)
    del arr_800mb

    # multi client put gigabytes: 10 workers x 10 puts of 80 MB
    # (parity ray_perf.py:140 do_put / put_multi)
    @ray.remote
    def do_put():
        for _ in range(10):
            ray.put(np.zeros(10 * 1024 * 1024, dtype=np.int64))

    def put_multi():
        ray.get([do_put.remote() for _ in range(10)])

    put_multi()  # warmup
    start = time.time()
    cycles = 0
    while time.time() - start < dur:
        put_multi()
        cycles += 1
    gbps = cycles * 10 * 8 * 0.1 / (time.time() - start)
    print(json.dumps({"name": "multi client put gigabytes",
                      "value": round(gbps, 2), "unit": "GB/s"}), flush=True)

    # ---- tasks
    @ray.remote
    def tiny():
        return b"ok"

    timeit("single client tasks sync",
           lambda: [ray.get(tiny.remote()) for _ in range(100)], 100, dur)
    timeit("single client tasks async",
           lambda: ray.get([tiny.remote() for _ in range(1000)]), 1000, dur)

    timeit("single client tasks and get batch",
           lambda: ray.get([tiny.remote() for _ in range(1000)]), 1, dur)

    # an object whose payload is 10k ObjectRefs (parity ray_perf.py:149)
    @ray.remote
    def create_object_containing_ref():
        return [ray.put(1) for _ in range(10000)]

    obj_containing_ref = create_object_containing_ref.remote()
    ray.get(obj_containing_ref)
    timeit("single client get object containing 10k refs",
           lambda: ray.get(obj_containing_ref), 1, dur)

    def wait_multiple_refs():
        not_ready = [tiny.remote() for _ in range(1000)]
        while not_ready:
            _ready, not_ready = ray.wait(not_ready)

    timeit("single client wait 1k refs", wait_multiple_refs, 1, dur)

    # ---- actors
    @ray.remote
    class Actor:
        def ping(self):
            return b"ok"

        def ping_arg(self, x):
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

    # multi client tasks async: 4 worker-actors each submitting a batch of
    # tasks from inside the worker (parity ray_perf.py:183 multi_task)
    @ray.remote(num_cpus=0)
    class Batcher:
        def ping(self):
            return b"ok"

        def task_batch(self, k):
            ray.get([tiny.remote() for _ in range(k)])

        def call_batch(self, servers, k):
            ray.get([s.ping.remote() for s in servers for _ in range(k)])

        def call_batch_arg(self, servers, k):
            x = ray.put(0)
            ray.get([s.ping_arg.remote(x) for s in servers for _ in range(k)])

    m, nb = 4, (500 if args.quick else 2000)
    batchers = [Batcher.remote() for _ in range(m)]
    ray.get([b.ping.remote() for b in batchers])
    timeit("multi client tasks async",
           lambda: ray.get([b.task_batch.remote(nb) for b in batchers]),
           nb * m, dur)

    n_cpu = min(os.cpu_count() or 4, 8)
    servers = [Actor.options(num_cpus=0).remote() for _ in range(n_cpu)]
    ray.get([x.ping.remote() for x in servers])

    # 1:n — one client actor fanning out to n_cpu server actors
    # (parity ray_perf.py:222 actor_async_direct)
    nk = 200 if args.quick else 600
    client = batchers[0]
    timeit("1:n actor calls async",
           lambda: ray.get(client.call_batch.remote(servers, nk)),
           nk * n_cpu, dur)

    # n:n — m client actors each fanning out to all servers
    # (parity ray_perf.py:238 actor_multi2)
    timeit("n:n actor calls async",
           lambda: ray.get([b.call_batch.remote(servers, nk // m)
                            for b in batchers]), (nk // m) * n_cpu * m, dur)

    # n:n with a shared put-arg per batch
    # (parity ray_perf.py:243 actor_multi2_direct_arg)
    timeit("n:n actor calls with arg async",
           lambda: ray.get([b.call_batch_arg.remote(servers, nk // m)
                            for b in batchers]), (nk // m) * n_cpu * m, dur)
    for b in batchers:
        ray.kill(b)
    actors = servers

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

    # ---- Ray Client plane (parity: ray_client_microbenchmark.py) — a
    # separate driver process WITHOUT local shm, objects proxied through
    # the raylet data plane
    addr = ray.get_runtime_context().gcs_address
    client_script = textwrap.dedent(f"""
        import json, os, sys, time
        os.environ["ANTRAY_FORCE_CLIENT"] = "1"
        sys.path.insert(0, {os.path.dirname(os.path.dirname(os.path.abspath(__file__)))!r})
        import ant_ray_amd as ray
        ray.init(address="ray://{addr}")

        def timeit(name, fn, mult=1, duration={dur}):
            fn()
            start = time.time(); count = 0
            while time.time() - start < duration:
                fn(); count += 1
            print(json.dumps({{"name": name,
                               "value": round(count * mult / (time.time() - start), 1),
                               "unit": "ops/s"}}), flush=True)

        ref = ray.put(b"ok")
        timeit("Ray Client: get calls", lambda: [ray.get(ref) for _ in range(20)], 20)
        timeit("Ray Client: put calls", lambda: [ray.put(0) for _ in range(20)], 20)

        @ray.remote
        class A:
            def ping(self):
                return b"ok"

        a = A.remote(); ray.get(a.ping.remote())
        timeit("Ray Client: 1:1 actor calls sync",
               lambda: [ray.get(a.ping.remote()) for _ in range(20)], 20)
        ray.shutdown()
    """)
    r = subprocess.run([sys.executable, "-c", client_script],
                       capture_output=True, text=True, timeout=120)
    sys.stdout.write(r.stdout)
    if r.returncode != 0:
        print(json.dumps({"name": "Ray Client", "error": r.stderr[-500:]}),
              flush=True)

    ray.shutdown()


if __name__ == "__main__":
    main()

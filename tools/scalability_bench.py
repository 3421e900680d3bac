"""Scalability micro-suite (parity: release/benchmarks/distributed —
many_actors / many_tasks / many_pgs rates, single-node scale).

    python tools/scalability_bench.py [--actors N] [--tasks N] [--pgs N]
"""
import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--actors", type=int, default=200)
    ap.add_argument("--tasks", type=int, default=5000)
    ap.add_argument("--pgs", type=int, default=500)
    args = ap.parse_args()

    import ant_ray_amd as ray

    ray.init()

    @ray.remote(num_cpus=0.01)
    class A:
        def ping(self):
            return 1

    t0 = time.time()
    actors = [A.remote() for _ in range(args.actors)]
    ray.get([a.ping.remote() for a in actors], timeout=600)
    dt = time.time() - t0
    print(json.dumps({"name": "many_actors actors_per_second",
                      "value": round(args.actors / dt, 1),
                      "n": args.actors, "unit": "/s"}), flush=True)
    for a in actors:
        ray.kill(a)
    del actors
    time.sleep(1)

    @ray.remote(num_cpus=0.01)
    def t():
        return 1

    t0 = time.time()
    ray.get([t.remote() for _ in range(args.tasks)], timeout=600)
    dt = time.time() - t0
    print(json.dumps({"name": "many_tasks tasks_per_second",
                      "value": round(args.tasks / dt, 1),
                      "n": args.tasks, "unit": "/s"}), flush=True)

    from ant_ray_amd.util.placement_group import (
        placement_group,
        remove_placement_group,
    )

    t0 = time.time()
    for _ in range(args.pgs):
        pg = placement_group([{"CPU": 0.01}])
        pg.wait(30)
        remove_placement_group(pg)
    dt = time.time() - t0
    print(json.dumps({"name": "many_pgs pgs_per_second",
                      "value": round(args.pgs / dt, 1),
                      "n": args.pgs, "unit": "/s"}), flush=True)
    ray.shutdown()


if __name__ == "__main__":
    main()

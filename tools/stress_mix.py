"""4-way concurrent stress mix with strict accounting.

Reproduction harness for the round-1 "Known-rare" item: one
unresolved-get timeout observed in the actor-kill-churn thread of a 60s
mix (never reproduced since). Four threads run concurrently against one
local cluster:

  1. task cancel churn     — submit slow tasks, cancel half mid-flight
  2. streaming consumption — streaming generators consumed to completion
  3. actor kill/restart    — calls raced against ray.kill(no_restart=False)
  4. data exchange         — put/get round-trips of MB-size payloads

STRICT accounting: every ray.get() must either return or raise a TYPED
error (TaskCancelledError / RayActorError / WorkerCrashedError) within
GET_TIMEOUT seconds. A GetTimeoutError is exactly the round-1 symptom
and counts as a FAILURE. Run with RAY_testing_asio_delay_us="*=500:5000"
to add RPC chaos delays (handled in _private/protocol.py).

Usage: python tools/stress_mix.py [--seconds 60] [--seed 0]
Exit code 0 = clean, 1 = at least one unresolved get / unexpected error.
"""
from __future__ import annotations

import argparse
import os
import random
import sys
import threading
import time
import traceback

import numpy as np

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import ant_ray_amd as ray
from ant_ray_amd.exceptions import (
    GetTimeoutError,
    RayActorError,
    TaskCancelledError,
    WorkerCrashedError,
)

GET_TIMEOUT = 30.0


class Stats:
    def __init__(self):
        self.lock = threading.Lock()
        self.ok = {}
        self.typed_fail = {}
        self.unresolved = []  # the bug we hunt
        self.unexpected = []

    def add_ok(self, kind):
        with self.lock:
            self.ok[kind] = self.ok.get(kind, 0) + 1

    def add_typed(self, kind):
        with self.lock:
            self.typed_fail[kind] = self.typed_fail.get(kind, 0) + 1

    def add_unresolved(self, kind, detail):
        with self.lock:
            self.unresolved.append((kind, detail))

    def add_unexpected(self, kind, detail):
        with self.lock:
            self.unexpected.append((kind, detail))


@ray.remote(num_cpus=0.1)
def slow_task(ms):
    time.sleep(ms / 1000.0)
    return ms


@ray.remote(num_cpus=0.1, num_returns="streaming")
def stream_task(n):
    for i in range(n):
        time.sleep(0.002)
        yield i


@ray.remote(num_cpus=0.1)
def echo(x):
    return x


@ray.remote(num_cpus=0.1, max_restarts=20, max_task_retries=0)
class ChurnActor:
    def __init__(self):
        self.n = 0

    def bump(self, payload):
        self.n += 1
        time.sleep(0.005)
        return self.n, len(payload)


def thread_cancel_churn(stop, stats, seed):
    rng = random.Random(seed)
    while not stop.is_set():
        refs = [slow_task.remote(rng.randint(50, 300)) for _ in range(6)]
        victims = rng.sample(refs, 3)
        time.sleep(rng.uniform(0.0, 0.1))
        for v in victims:
            ray.cancel(v, force=rng.random() < 0.5)
        for r in refs:
            try:
                ray.get(r, timeout=GET_TIMEOUT)
                stats.add_ok("cancel_mix")
            except TaskCancelledError:
                stats.add_typed("cancelled")
            except (RayActorError, WorkerCrashedError):
                stats.add_typed("worker_died")
            except GetTimeoutError:
                stats.add_unresolved("cancel_mix", repr(r))
            except Exception:
                stats.add_unexpected("cancel_mix", traceback.format_exc())


def thread_streaming(stop, stats, seed):
    rng = random.Random(seed)
    while not stop.is_set():
        gen = stream_task.remote(rng.randint(5, 25))
        got = 0
        try:
            for ref in gen:
                ray.get(ref, timeout=GET_TIMEOUT)
                got += 1
            stats.add_ok("stream")
        except GetTimeoutError:
            stats.add_unresolved("stream", f"after {got} items")
        except (RayActorError, WorkerCrashedError):
            stats.add_typed("stream_worker_died")
        except Exception:
            stats.add_unexpected("stream", traceback.format_exc())


def thread_actor_churn(stop, stats, seed):
    """The thread that produced the round-1 failure: call an actor that
    is concurrently killed (restart allowed) and make sure every call
    resolves or fails typed."""
    rng = random.Random(seed)
    actor = ChurnActor.remote()
    payload = b"x" * 4096
    calls_since_kill = 0
    while not stop.is_set():
        refs = [actor.bump.remote(payload) for _ in range(8)]
        calls_since_kill += 8
        if rng.random() < 0.35:
            time.sleep(rng.uniform(0.0, 0.03))
            ray.kill(actor, no_restart=False)
        for r in refs:
            try:
                ray.get(r, timeout=GET_TIMEOUT)
                stats.add_ok("actor_call")
            except RayActorError:
                stats.add_typed("actor_died")
            except GetTimeoutError:
                stats.add_unresolved("actor_call",
                                     f"{calls_since_kill} since kill")
            except Exception:
                stats.add_unexpected("actor_call", traceback.format_exc())
        if rng.random() < 0.05:
            # occasionally replace the actor entirely (restart budget)
            try:
                ray.kill(actor, no_restart=True)
            except Exception:
                pass
            actor = ChurnActor.remote()
            calls_since_kill = 0


def thread_data_exchange(stop, stats, seed):
    rng = random.Random(seed)
    while not stop.is_set():
        arr = np.random.default_rng(rng.randrange(2**31)).integers(
            0, 255, size=rng.randint(1 << 18, 1 << 21), dtype=np.uint8)
        try:
            ref = ray.put(arr)
            out_ref = echo.remote(ref)
            out = ray.get(out_ref, timeout=GET_TIMEOUT)
            assert out.nbytes == arr.nbytes
            stats.add_ok("data")
        except GetTimeoutError:
            stats.add_unresolved("data", "echo get")
        except (RayActorError, WorkerCrashedError):
            stats.add_typed("data_worker_died")
        except Exception:
            stats.add_unexpected("data", traceback.format_exc())


def run(seconds: float, seed: int) -> int:
    ray.init(num_cpus=8, ignore_reinit_error=True)
    stats = Stats()
    stop = threading.Event()
    threads = [
        threading.Thread(target=fn, args=(stop, stats, seed + i), daemon=True)
        for i, fn in enumerate([thread_cancel_churn, thread_streaming,
                                thread_actor_churn, thread_data_exchange])
    ]
    t0 = time.time()
    for t in threads:
        t.start()
    time.sleep(seconds)
    stop.set()
    for t in threads:
        t.join(timeout=GET_TIMEOUT + 30)
    alive = [t for t in threads if t.is_alive()]
    dt = time.time() - t0
    print(f"stress_mix: {dt:.1f}s  ok={stats.ok}  typed={stats.typed_fail}")
    if stats.unresolved:
        print(f"UNRESOLVED GETS ({len(stats.unresolved)}):")
        for kind, detail in stats.unresolved[:10]:
            print(f"  {kind}: {detail}")
    if stats.unexpected:
        print(f"UNEXPECTED ERRORS ({len(stats.unexpected)}):")
        for kind, detail in stats.unexpected[:4]:
            print(f"  {kind}: {detail}")
    if alive:
        print(f"HUNG THREADS: {[t.name for t in alive]}")
    ray.shutdown()
    return 1 if (stats.unresolved or stats.unexpected or alive) else 0


if __name__ == "__main__":
    ap = argparse.ArgumentParser()
    ap.add_argument("--seconds", type=float, default=60.0)
    ap.add_argument("--seed", type=int, default=0)
    a = ap.parse_args()
    sys.exit(run(a.seconds, a.seed))

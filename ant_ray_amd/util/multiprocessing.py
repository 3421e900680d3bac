"""ray.util.multiprocessing parity: a multiprocessing.Pool lookalike
whose workers are Ray actors (reference python/ray/util/multiprocessing/
pool.py). Supports map/starmap/apply/apply_async/imap/imap_unordered.
"""
from __future__ import annotations

import itertools
from typing import Any, Callable, Iterable, List, Optional


class AsyncResult:
    def __init__(self, refs):
        self._refs = refs

    def get(self, timeout: Optional[float] = None):
        import ant_ray_amd as ray

        out = ray.get(self._refs, timeout=timeout)
        return out if len(self._refs) != 1 else out[0]

    def wait(self, timeout: Optional[float] = None):
        import ant_ray_amd as ray

        ray.wait(self._refs, num_returns=len(self._refs), timeout=timeout)

    def ready(self) -> bool:
        import ant_ray_amd as ray

        ready, _ = ray.wait(self._refs, num_returns=len(self._refs),
                            timeout=0)
        return len(ready) == len(self._refs)

    def successful(self) -> bool:
        try:
            self.get(timeout=0)
            return True
        except Exception:
            return False


class Pool:
    """Actor-backed process pool. processes=None uses the cluster's CPU
    count. Work items round-robin across the pool actors."""

    def __init__(self, processes: Optional[int] = None,
                 initializer: Optional[Callable] = None,
                 initargs: tuple = (), ray_remote_args: Optional[dict] = None):
        import ant_ray_amd as ray

        if not ray.is_initialized():
            ray.init()
        if processes is None:
            processes = max(1, int(ray.cluster_resources().get("CPU", 1)))
        self._size = processes

        @ray.remote(**(ray_remote_args or {"num_cpus": 1}))
        class _PoolWorker:
            def __init__(self, initializer=None, initargs=()):
                if initializer:
                    initializer(*initargs)

            def run(self, fn, args, kwargs):
                return fn(*args, **(kwargs or {}))

            def run_batch(self, fn, chunk):
                return [fn(*a) for a in chunk]

        self._actors = [_PoolWorker.remote(initializer, initargs)
                        for _ in range(processes)]
        self._rr = itertools.cycle(range(processes))
        self._closed = False

    # ------------------------------------------------------------- lifecycle
    def close(self):
        self._closed = True

    def terminate(self):
        import ant_ray_amd as ray

        self._closed = True
        for a in self._actors:
            try:
                ray.kill(a)
            except Exception:
                pass

    def join(self):
        if not self._closed:
            raise ValueError("join() before close()")

    def __enter__(self):
        return self

    def __exit__(self, *exc):
        self.terminate()

    # ------------------------------------------------------------------ apis
    def _check(self):
        if self._closed:
            raise ValueError("Pool not running")

    def apply(self, fn, args: tuple = (), kwargs: Optional[dict] = None):
        return self.apply_async(fn, args, kwargs).get()

    def apply_async(self, fn, args: tuple = (), kwargs: Optional[dict] = None):
        self._check()
        a = self._actors[next(self._rr)]
        return AsyncResult([a.run.remote(fn, args, kwargs)])

    def _chunks(self, iterable, chunksize):
        items = [(x,) if not isinstance(x, tuple) else x for x in iterable]
        if not chunksize:
            chunksize = max(1, len(items) // (self._size * 4) or 1)
        return [items[i:i + chunksize]
                for i in range(0, len(items), chunksize)], chunksize

    def map(self, fn, iterable: Iterable, chunksize: Optional[int] = None) -> List[Any]:
        return self.map_async(fn, iterable, chunksize).get()

    def starmap(self, fn, iterable: Iterable, chunksize: Optional[int] = None):
        return self.map_async(fn, list(iterable), chunksize).get()

    def map_async(self, fn, iterable: Iterable, chunksize: Optional[int] = None):
        self._check()
        chunks, _ = self._chunks(list(iterable), chunksize)
        refs = [self._actors[i % self._size].run_batch.remote(fn, c)
                for i, c in enumerate(chunks)]
        return _FlatAsyncResult(refs)

    def imap(self, fn, iterable: Iterable, chunksize: Optional[int] = None):
        import ant_ray_amd as ray

        self._check()
        chunks, _ = self._chunks(list(iterable), chunksize)
        refs = [self._actors[i % self._size].run_batch.remote(fn, c)
                for i, c in enumerate(chunks)]
        for r in refs:  # submission order
            for v in ray.get(r):
                yield v

    def imap_unordered(self, fn, iterable: Iterable,
                       chunksize: Optional[int] = None):
        import ant_ray_amd as ray

        self._check()
        chunks, _ = self._chunks(list(iterable), chunksize)
        pending = [self._actors[i % self._size].run_batch.remote(fn, c)
                   for i, c in enumerate(chunks)]
        while pending:
            done, pending = ray.wait(pending)
            for r in done:
                for v in ray.get(r):
                    yield v


class _FlatAsyncResult(AsyncResult):
    def get(self, timeout: Optional[float] = None):
        import ant_ray_amd as ray

        out = ray.get(self._refs, timeout=timeout)
        return [v for chunk in out for v in chunk]

"""ActorPool: load-balance tasks over a fixed set of actors.

Role parity: reference python/ray/util/actor_pool.py (map/map_unordered/
submit/get_next/get_next_unordered/has_next/push/pop_idle).
"""
from __future__ import annotations

from typing import Any, Callable, Iterable, List


class ActorPool:
    def __init__(self, actors: List[Any]):
        self._idle = list(actors)
        self._future_to_actor = {}
        self._index_to_future = {}
        self._next_task_index = 0
        self._next_return_index = 0
        self._pending_submits = []

    def submit(self, fn: Callable, value: Any):
        if self._idle:
            actor = self._idle.pop()
            future = fn(actor, value)
            self._future_to_actor[future] = (self._next_task_index, actor)
            self._index_to_future[self._next_task_index] = future
            self._next_task_index += 1
        else:
            self._pending_submits.append((fn, value))

    def _return_actor(self, actor):
        self._idle.append(actor)
        if self._pending_submits:
            self.submit(*self._pending_submits.pop(0))

    def has_next(self) -> bool:
        return bool(self._future_to_actor)

    def get_next(self, timeout=None) -> Any:
        import ant_ray_amd as ray

        if not self.has_next():
            raise StopIteration("no more results")
        future = self._index_to_future.pop(self._next_return_index)
        self._next_return_index += 1
        i, actor = self._future_to_actor.pop(future)
        self._return_actor(actor)
        return ray.get(future, timeout=timeout)

    def get_next_unordered(self, timeout=None) -> Any:
        import ant_ray_amd as ray

        if not self.has_next():
            raise StopIteration("no more results")
        ready, _ = ray.wait(list(self._future_to_actor), num_returns=1,
                            timeout=timeout)
        if not ready:
            raise TimeoutError
        future = ready[0]
        i, actor = self._future_to_actor.pop(future)
        del self._index_to_future[i]
        self._return_actor(actor)
        return ray.get(future)

    def map(self, fn: Callable, values: Iterable[Any]):
        for v in values:
            self.submit(fn, v)
        while self.has_next():
            yield self.get_next()

    def map_unordered(self, fn: Callable, values: Iterable[Any]):
        for v in values:
            self.submit(fn, v)
        while self.has_next():
            yield self.get_next_unordered()

    def has_free(self) -> bool:
        return bool(self._idle) and not self._pending_submits

    def push(self, actor):
        self._return_actor(actor)

    def pop_idle(self):
        return self._idle.pop() if self._idle else None

"""Distributed FIFO queue on an actor.

Role parity: reference python/ray/util/queue.py (Queue backed by an
_QueueActor wrapping asyncio.Queue; Empty/Full match queue module
semantics).
"""
from __future__ import annotations

import asyncio
from typing import Any, List, Optional

from queue import Empty, Full  # re-exported, parity with ray.util.queue


class _QueueActor:
    def __init__(self, maxsize: int = 0):
        self.q = asyncio.Queue(maxsize=maxsize)

    async def put(self, item, timeout: Optional[float] = None):
        if timeout is None:
            await self.q.put(item)
            return True
        try:
            await asyncio.wait_for(self.q.put(item), timeout)
            return True
        except asyncio.TimeoutError:
            return False

    async def put_nowait(self, item):
        try:
            self.q.put_nowait(item)
            return True
        except asyncio.QueueFull:
            return False

    async def get(self, timeout: Optional[float] = None):
        if timeout is None:
            return True, await self.q.get()
        try:
            return True, await asyncio.wait_for(self.q.get(), timeout)
        except asyncio.TimeoutError:
            return False, None

    async def get_nowait(self):
        try:
            return True, self.q.get_nowait()
        except asyncio.QueueEmpty:
            return False, None

    async def qsize(self):
        return self.q.qsize()

    async def empty(self):
        return self.q.empty()

    async def full(self):
        return self.q.full()


class Queue:
    def __init__(self, maxsize: int = 0, actor_options: Optional[dict] = None):
        import ant_ray_amd as ray

        opts = dict(actor_options or {})
        opts.setdefault("num_cpus", 0)
        opts["max_concurrency"] = max(opts.get("max_concurrency", 64), 64)
        self.actor = ray.remote(_QueueActor).options(**opts).remote(maxsize)
        self.maxsize = maxsize

    def put(self, item: Any, block: bool = True, timeout: Optional[float] = None):
        import ant_ray_amd as ray

        if not block:
            if not ray.get(self.actor.put_nowait.remote(item)):
                raise Full
            return
        if not ray.get(self.actor.put.remote(item, timeout)):
            raise Full

    def put_nowait(self, item):
        self.put(item, block=False)

    def get(self, block: bool = True, timeout: Optional[float] = None) -> Any:
        import ant_ray_amd as ray

        if not block:
            ok, v = ray.get(self.actor.get_nowait.remote())
            if not ok:
                raise Empty
            return v
        ok, v = ray.get(self.actor.get.remote(timeout))
        if not ok:
            raise Empty
        return v

    def get_nowait(self):
        return self.get(block=False)

    def put_async(self, item):
        return self.actor.put.remote(item, None)

    def get_async(self):
        return self.actor.get.remote(None)

    def qsize(self) -> int:
        import ant_ray_amd as ray

        return ray.get(self.actor.qsize.remote())

    def empty(self) -> bool:
        import ant_ray_amd as ray

        return ray.get(self.actor.empty.remote())

    def full(self) -> bool:
        import ant_ray_amd as ray

        return ray.get(self.actor.full.remote())

    def shutdown(self):
        import ant_ray_amd as ray

        try:
            ray.kill(self.actor)
        except Exception:
            pass

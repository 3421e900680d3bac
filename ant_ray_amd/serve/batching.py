"""@serve.batch — dynamic request batching.

Role parity: reference python/ray/serve/batching.py (_BatchQueue collects
concurrent calls up to max_batch_size or batch_wait_timeout_s, invokes the
underlying function once with the list, fans results back out).
"""
from __future__ import annotations

import asyncio
import functools
from typing import Any, Callable, List, Optional


class _BatchQueue:
    def __init__(self, fn, max_batch_size: int, timeout_s: float):
        self.fn = fn
        self.max_batch_size = max_batch_size
        self.timeout_s = timeout_s
        self._pending: List[tuple] = []  # (item, future)
        self._flusher: Optional[asyncio.Task] = None

    async def submit(self, instance, item: Any):
        loop = asyncio.get_running_loop()
        fut = loop.create_future()
        self._pending.append((item, fut))
        if len(self._pending) >= self.max_batch_size:
            await self._flush(instance)
        elif self._flusher is None or self._flusher.done():
            self._flusher = loop.create_task(self._delayed_flush(instance))
        return await fut

    async def _delayed_flush(self, instance):
        await asyncio.sleep(self.timeout_s)
        await self._flush(instance)

    async def _flush(self, instance):
        if not self._pending:
            return
        batch, self._pending = self._pending, []
        items = [b[0] for b in batch]
        try:
            if instance is not None:
                results = self.fn(instance, items)
            else:
                results = self.fn(items)
            if asyncio.iscoroutine(results):
                results = await results
            if len(results) != len(items):
                raise ValueError(
                    f"batched function returned {len(results)} results for "
                    f"{len(items)} inputs")
            for (_, fut), res in zip(batch, results):
                if not fut.done():
                    fut.set_result(res)
        except Exception as e:
            for _, fut in batch:
                if not fut.done():
                    fut.set_exception(e)


def batch(_fn: Optional[Callable] = None, *, max_batch_size: int = 10,
          batch_wait_timeout_s: float = 0.01):
    """Decorator: an async method taking a LIST becomes callable with single
    items; concurrent single calls are batched."""

    def wrap(fn):
        queues = {}  # per-instance queue (or None key for free functions)

        @functools.wraps(fn)
        async def wrapper(*args):
            if len(args) == 2:
                instance, item = args
            elif len(args) == 1:
                instance, item = None, args[0]
            else:
                raise TypeError("@serve.batch methods take exactly one item")
            key = id(instance)
            q = queues.get(key)
            if q is None:
                q = queues[key] = _BatchQueue(fn, max_batch_size,
                                              batch_wait_timeout_s)
            return await q.submit(instance, item)

        wrapper._antray_batch = True
        return wrapper

    if _fn is not None:
        return wrap(_fn)
    return wrap

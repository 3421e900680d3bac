"""Model multiplexing: many models per replica with LRU residency.

Role parity: reference python/ray/serve/multiplex.py (@serve.multiplexed
caches per-model loads; serve.get_multiplexed_model_id routes by the
"serve_multiplexed_model_id" request header/context).
"""
from __future__ import annotations

import asyncio
import contextvars
import functools
from collections import OrderedDict
from typing import Any, Callable

_current_model_id = contextvars.ContextVar("serve_multiplexed_model_id",
                                           default="")


def get_multiplexed_model_id() -> str:
    """The model id of the request being handled (parity serve.api)."""
    return _current_model_id.get()


def _set_multiplexed_model_id(model_id: str):
    _current_model_id.set(model_id)


def multiplexed(_fn: Callable = None, *, max_num_models_per_replica: int = 3):
    """Decorator on an async `load_model(self, model_id)` method: calls are
    cached per model id with LRU eviction at max_num_models_per_replica."""

    def wrap(fn):
        caches = {}

        @functools.wraps(fn)
        async def wrapper(self, model_id: str):
            cache = caches.setdefault(id(self), OrderedDict())
            if model_id in cache:
                cache.move_to_end(model_id)
                return cache[model_id]
            _set_multiplexed_model_id(model_id)
            model = fn(self, model_id)
            if asyncio.iscoroutine(model):
                model = await model
            cache[model_id] = model
            cache.move_to_end(model_id)
            while len(cache) > max_num_models_per_replica:
                old_id, old = cache.popitem(last=False)
                del_fn = getattr(old, "__del__", None)
                # parity: the reference awaits __del__ if the model defines it
                try:
                    if del_fn is not None:
                        r = del_fn()
                        if asyncio.iscoroutine(r):
                            await r
                except Exception:
                    pass
            return model

        wrapper._serve_multiplexed = True
        return wrapper

    return wrap(_fn) if _fn else wrap

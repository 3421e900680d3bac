"""DataContext: execution knobs. Parity: python/ray/data/context.py
(DataContext.get_current; target block sizes, concurrency caps)."""
from __future__ import annotations

import threading
from dataclasses import dataclass


@dataclass
class DataContext:
    target_max_block_size: int = 128 * 1024 * 1024
    target_min_block_size: int = 1 * 1024 * 1024
    max_concurrent_tasks: int = 16
    read_parallelism: int = 8
    eager_free: bool = True
    verbose_progress: bool = False

    _local = threading.local()

    @classmethod
    def get_current(cls) -> "DataContext":
        ctx = getattr(cls._local, "ctx", None)
        if ctx is None:
            ctx = cls()
            cls._local.ctx = ctx
        return ctx

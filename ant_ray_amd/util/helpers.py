"""as_completed / map_unordered.

Role parity: reference python/ray/util/helpers.py (as_completed:57,
map_unordered:135 — backpressure pattern from the "limit pending tasks"
design pattern).
"""
from typing import Any, Iterable, Iterator, Optional, Sequence

DEFAULT_CHUNK_SIZE = 10
DEFAULT_BACKPRESSURE_SIZE = 100


def _wait_batch(refs, *, chunk_size, yield_obj_refs, **kwargs):
    import ant_ray_amd as ray

    if chunk_size < 1:
        raise ValueError("`chunk_size` must be >= 1")
    ready, rest = ray.wait(list(refs), num_returns=min(chunk_size, len(refs)),
                           **kwargs)
    return (ready if yield_obj_refs else ray.get(ready)), rest


def as_completed(refs: Sequence, *, chunk_size: int = DEFAULT_CHUNK_SIZE,
                 yield_obj_refs: bool = False, **kwargs) -> Iterator[Any]:
    """Yield results (or refs) as batches of `chunk_size` become ready,
    instead of blocking on the whole list like ray.get(refs)."""
    refs = list(refs)
    while refs:
        out, refs = _wait_batch(refs, chunk_size=chunk_size,
                                yield_obj_refs=yield_obj_refs, **kwargs)
        yield from out


def map_unordered(fn, items: Iterable[Any], *,
                  backpressure_size: Optional[int] = DEFAULT_BACKPRESSURE_SIZE,
                  chunk_size: int = DEFAULT_CHUNK_SIZE,
                  yield_obj_refs: bool = False, **kwargs) -> Iterator[Any]:
    """Apply remote function `fn` over `items`, yielding completed results
    unordered while keeping at most `backpressure_size` tasks in flight."""
    if backpressure_size is not None and backpressure_size < 1:
        raise ValueError("`backpressure_size` must be >= 1 or None")
    it = iter(items)
    pending = []
    exhausted = False
    while True:
        while not exhausted and (backpressure_size is None
                                 or len(pending) < backpressure_size):
            try:
                pending.append(fn.remote(next(it)))
            except StopIteration:
                exhausted = True
        if not pending:
            return
        out, pending = _wait_batch(pending, chunk_size=chunk_size,
                                   yield_obj_refs=yield_obj_refs, **kwargs)
        yield from out

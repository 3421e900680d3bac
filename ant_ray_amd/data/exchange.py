"""Distributed exchange for shuffle-class ops (sort / random_shuffle /
groupby): blocks never pass through the driver.

Role parity: reference data/_internal/planner/exchange/ (sort sample +
range partition, hash shuffle task scheduler). Two phases of Ray tasks:

  map:    each input block is partitioned into P pieces
          (hash of a key column, range against sampled boundaries, or
          seeded-random) — one task per block, num_returns=P so each
          reducer pulls ONLY its partition,
  reduce: one task per partition concatenates its pieces and applies the
          partition-local finalizer (sort the range, shuffle, group+agg).

The driver holds only refs and tiny key samples; the bytes move between
workers through the shm object plane.
"""
from __future__ import annotations

from typing import Any, Callable, List, Optional

import pyarrow as pa

import ant_ray_amd as ray


def _concat(tables: List[pa.Table]) -> pa.Table:
    tables = [t for t in tables if t is not None and t.num_rows >= 0]
    if not tables:
        return pa.table({})
    return pa.concat_tables(tables, promote_options="default")


@ray.remote
def _sample_keys(block: pa.Table, keys: List[str], n: int):
    if block.num_rows == 0:
        return block.select(keys)
    import numpy as np

    idx = np.random.RandomState(0).randint(0, block.num_rows,
                                           size=min(n, block.num_rows))
    return block.select(keys).take(pa.array(idx))


def _part_range(block: pa.Table, keys: List[str], bounds, descending):
    """Range-partition rows against sampled boundaries (sort exchange)."""
    import numpy as np

    if block.num_rows == 0:
        return [block] * (len(bounds) + 1)
    col = block.column(keys[0]).to_numpy(zero_copy_only=False)
    part = np.searchsorted(bounds, col, side="right")
    if descending:
        part = len(bounds) - part
    return [block.filter(pa.array(part == p))
            for p in range(len(bounds) + 1)]


def _part_hash(block: pa.Table, key: str, P: int):
    import numpy as np

    if block.num_rows == 0:
        return [block] * P
    col = block.column(key).to_pandas()
    h = np.fromiter((hash(v) % P for v in col), dtype=np.int64,
                    count=len(col))
    return [block.filter(pa.array(h == p)) for p in range(P)]


def _part_random(block: pa.Table, P: int, seed: Optional[int], salt: int):
    import numpy as np

    if block.num_rows == 0:
        return [block] * P
    rng = np.random.RandomState(None if seed is None else seed + salt)
    part = rng.randint(0, P, size=block.num_rows)
    return [block.filter(pa.array(part == p)) for p in range(P)]


def exchange(refs: List[Any], P: int, partitioner: Callable,
             finalizer: Callable) -> List[Any]:
    """Generic two-phase exchange. partitioner(block) -> P tables;
    finalizer(concatenated_partition) -> table. Returns P block refs."""
    if not refs:
        return []
    P = max(1, P)

    @ray.remote(num_cpus=1)
    def map_task(block):
        parts = partitioner(block)
        # with num_returns=1 the whole return IS the single value — do
        # not wrap it in a 1-tuple
        return parts[0] if P == 1 else tuple(parts)

    @ray.remote(num_cpus=1)
    def reduce_task(*pieces):
        return finalizer(_concat(list(pieces)))

    map_outs = [map_task.options(num_returns=P).remote(r) for r in refs]
    if P == 1:
        cols = [list(map_outs)]  # num_returns=1 -> single refs
    else:
        cols = [[m[p] for m in map_outs] for p in range(P)]
    return [reduce_task.remote(*col) for col in cols]


def sort_exchange(refs: List[Any], keys: List[str], descending: bool,
                  P: Optional[int] = None) -> List[Any]:
    P = P or max(1, len(refs))
    # 1) sample keys to pick P-1 range boundaries (reference sort sample)
    samples = _concat(ray.get(
        [_sample_keys.remote(r, keys, 64) for r in refs], timeout=300))
    order = "descending" if descending else "ascending"
    if samples.num_rows == 0:
        bounds = []
    else:
        s = samples.sort_by([(keys[0], "ascending")])
        col = s.column(keys[0]).to_numpy(zero_copy_only=False)
        step = max(1, len(col) // P)
        bounds = [col[i] for i in range(step, len(col), step)][:P - 1]

    def partitioner(block):
        return _part_range(block, keys, bounds, descending)

    def finalizer(t):
        return t.sort_by([(k, order) for k in keys]) if t.num_rows else t

    return exchange(refs, len(bounds) + 1, partitioner, finalizer)


def shuffle_exchange(refs: List[Any], seed: Optional[int],
                     P: Optional[int] = None) -> List[Any]:
    P = P or max(1, len(refs))
    @ray.remote(num_cpus=1)
    def map_task(block, salt):
        parts = _part_random(block, P, seed, salt)
        return parts[0] if P == 1 else tuple(parts)

    @ray.remote(num_cpus=1)
    def reduce_task(salt, *pieces):
        import numpy as np

        t = _concat(list(pieces))
        if t.num_rows == 0:
            return t
        rng = np.random.RandomState(None if seed is None
                                    else seed + 100_003 + salt)
        return t.take(pa.array(rng.permutation(t.num_rows)))

    map_outs = [map_task.options(num_returns=P).remote(r, i)
                for i, r in enumerate(refs)]
    if P == 1:
        cols = [[m for m in map_outs]]
    else:
        cols = [[m[p] for m in map_outs] for p in range(P)]
    return [reduce_task.remote(p, *col) for p, col in enumerate(cols)]


def groupby_exchange(refs: List[Any], key: str,
                     finalize: Callable, P: Optional[int] = None) -> List[Any]:
    """Hash-partition by key so each group lands wholly in one reducer,
    then run the aggregation per partition (results concatenate to the
    global answer because groups never straddle partitions)."""
    P = P or max(1, min(len(refs), 16))

    def partitioner(block):
        return _part_hash(block, key, P)

    return exchange(refs, P, partitioner, finalize)


def join_exchange(left_refs: List[Any], right_refs: List[Any],
                  keys: List[str], right_keys: List[str], join_type: str,
                  suffixes: tuple, P: Optional[int] = None) -> List[Any]:
    """Distributed hash join: both sides are hash-partitioned on their
    key so matching rows meet in the same reducer; per-partition arrow
    joins concatenate into the global result (partitions are disjoint
    in key space, so inner AND outer joins compose)."""
    if not left_refs and not right_refs:
        return []
    P = P or max(1, min(max(len(left_refs), len(right_refs)), 16))

    @ray.remote(num_cpus=1)
    def part_left(block):
        parts = _part_hash(block, keys[0], P)
        return parts[0] if P == 1 else tuple(parts)

    @ray.remote(num_cpus=1)
    def part_right(block):
        parts = _part_hash(block, right_keys[0], P)
        return parts[0] if P == 1 else tuple(parts)

    @ray.remote(num_cpus=1)
    def reduce_join(n_left, *pieces):
        lt = _concat(list(pieces[:n_left]))
        rt = _concat(list(pieces[n_left:]))
        if lt.num_rows == 0 and rt.num_rows == 0:
            return lt
        if lt.num_rows == 0 and join_type == "inner":
            return lt
        if rt.num_rows == 0 and join_type == "inner":
            return rt.select([]) if lt.num_rows == 0 else lt.slice(0, 0)
        return lt.join(rt, keys=keys, right_keys=right_keys,
                       join_type=join_type, left_suffix=suffixes[0],
                       right_suffix=suffixes[1])

    louts = [part_left.options(num_returns=P).remote(r) for r in left_refs]
    routs = [part_right.options(num_returns=P).remote(r) for r in right_refs]
    out = []
    for p in range(P):
        lcol = [m if P == 1 else m[p] for m in louts]
        rcol = [m if P == 1 else m[p] for m in routs]
        out.append(reduce_join.remote(len(lcol), *(lcol + rcol)))
    return out

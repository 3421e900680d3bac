"""Streaming executor: runs the fused plan on Ray tasks/actor pools.

Role parity: reference python/ray/data/_internal/execution/
streaming_executor.py:67 (+ streaming_executor_state.py:745
select_operator_to_run, task_pool_map_operator.py, actor_pool_map_operator).
Design here: a pull-based pipeline of block-ref iterators; each fused map
stage keeps at most `max_concurrent_tasks` tasks in flight (backpressure),
yielding refs in submission order. Blocks live in the shm object store and
never pass through the driver for map stages.
"""
from __future__ import annotations

import collections
from typing import Any, Iterator, List

import ant_ray_amd as ray
from ant_ray_amd.data.block import BlockAccessor, batch_to_block
from ant_ray_amd.data.context import DataContext
from ant_ray_amd.data.plan import (
    ActorPoolStrategy,
    AllToAllOp,
    LimitOp,
    MapOp,
    ReadOp,
    fuse_stages,
)


def _apply_chain(block, chain_spec) -> Any:
    """Run a fused chain of map ops over one block (inside a task/actor)."""
    import pyarrow as pa

    for op in chain_spec:
        kind, fn, batch_size, batch_format, fn_args, fn_kwargs = op
        acc = BlockAccessor.for_block(block)
        if kind == "map_batches":
            out_tables = []
            n = acc.num_rows()
            step = batch_size or max(n, 1)
            for s in range(0, max(n, 1), step) if n else []:
                sub = BlockAccessor(acc.slice(s, min(s + step, n)))
                batch = sub.to_batch(batch_format)
                res = fn(batch, *fn_args, **fn_kwargs)
                out_tables.append(batch_to_block(res))
            block = (pa.concat_tables(out_tables, promote_options="default")
                     if out_tables else acc.to_arrow().slice(0, 0))
        elif kind == "map_rows":
            rows = [fn(r, *fn_args, **fn_kwargs) for r in acc.iter_rows()]
            block = BlockAccessor.for_block(rows).to_arrow() if rows else (
                acc.to_arrow().slice(0, 0))
        elif kind == "flat_map":
            rows = [o for r in acc.iter_rows() for o in fn(r, *fn_args, **fn_kwargs)]
            block = BlockAccessor.for_block(rows).to_arrow() if rows else (
                acc.to_arrow().slice(0, 0))
        elif kind == "filter":
            rows = [r for r in acc.iter_rows() if fn(r, *fn_args, **fn_kwargs)]
            block = BlockAccessor.for_block(rows).to_arrow() if rows else (
                acc.to_arrow().slice(0, 0))
        else:
            raise ValueError(kind)
    return block


def _chain_spec(chain: List[MapOp], instantiated_fns) -> list:
    return [
        (op.kind, instantiated_fns[i], op.batch_size, op.batch_format,
         op.fn_args, op.fn_kwargs)
        for i, op in enumerate(chain)
    ]


def _run_read_task(task) -> Any:
    out = task()
    # a generator yields multiple blocks; a list is ROWS (one block); dict /
    # table / DataFrame are one block
    if isinstance(out, Iterator):
        import pyarrow as pa

        blocks = [BlockAccessor.for_block(b).to_arrow() for b in out]
        return pa.concat_tables(blocks) if blocks else pa.table({})
    return BlockAccessor.for_block(out).to_arrow()


class _MapWorker:
    """Actor for ActorPoolStrategy stages: constructs callable-class UDFs
    once, then processes blocks."""

    def __init__(self, chain_meta):
        self._fns = []
        for fn, is_class, cargs, ckw in chain_meta:
            self._fns.append(fn(*cargs, **ckw) if is_class else fn)
        self._chain_meta = chain_meta

    def ready(self):
        return True

    def process(self, block, spec_skeleton):
        spec = [
            (kind, self._fns[i], bs, bf, fa, fk)
            for i, (kind, bs, bf, fa, fk) in enumerate(spec_skeleton)
        ]
        return _apply_chain(block, spec)


def execute_plan(ops: List[Any]) -> Iterator[Any]:
    """Yields ObjectRefs of output blocks, streaming."""
    from ant_ray_amd.data.optimizer import optimize

    ctx = DataContext.get_current()
    stages = fuse_stages(optimize(ops))
    stream: Iterator[Any] = iter(())
    for stage in stages:
        if isinstance(stage, ReadOp):
            stream = _read_stage(stage, ctx)
        elif isinstance(stage, list):  # fused map chain
            stream = _map_stage(stream, stage, ctx)
        elif isinstance(stage, AllToAllOp):
            refs = list(stream)
            stream = iter(stage.fn(refs))
        elif isinstance(stage, LimitOp):
            stream = _limit_stage(stream, stage.limit)
        else:
            raise ValueError(f"unknown stage {stage}")
    return stream


def _read_stage(op: ReadOp, ctx) -> Iterator[Any]:
    if op.block_refs:
        # blocks were put driver-side (from_numpy/from_arrow): hand the
        # refs straight to the next stage — a read TASK here would only
        # get+return the same bytes through a worker (one extra copy of
        # the whole dataset + a task round trip per block)
        yield from op.block_refs
        if not op.read_tasks:
            return
    from ant_ray_amd.data.backpressure import default_policies

    policies = default_policies(ctx)
    read_remote = ray.remote(num_cpus=1)(_run_read_task)
    window = collections.deque()
    tasks = iter(op.read_tasks)

    def admit():
        nonlocal tasks
        while (tasks is not None
               and all(p.can_add_input(op.name, len(window))
                       for p in policies)):
            try:
                window.append(read_remote.remote(next(tasks)))
            except StopIteration:
                tasks = None

    admit()
    while window:
        ref = window.popleft()
        admit()
        yield ref


def _map_stage(stream: Iterator[Any], chain: List[MapOp], ctx) -> Iterator[Any]:
    use_actors = any(isinstance(op.compute, ActorPoolStrategy) for op in chain)
    if use_actors:
        yield from _actor_map_stage(stream, chain, ctx)
        return
    import inspect

    fns = []
    for op in chain:
        f = op.fn
        if inspect.isclass(f):
            f = f(*op.fn_constructor_args, **op.fn_constructor_kwargs)
        fns.append(f)
    spec = _chain_spec(chain, fns)
    res = {"num_cpus": chain[0].num_cpus or 1}
    if any(op.num_gpus for op in chain):
        res["num_gpus"] = max(op.num_gpus or 0 for op in chain)
    from ant_ray_amd.data.backpressure import default_policies

    policies = default_policies(ctx)
    name = chain[0].name
    map_remote = ray.remote(**res)(_apply_chain)
    window = collections.deque()
    exhausted = False
    it = iter(stream)
    while True:
        while (not exhausted
               and all(p.can_add_input(name, len(window))
                       for p in policies)):
            try:
                block_ref = next(it)
            except StopIteration:
                exhausted = True
                break
            window.append(map_remote.remote(block_ref, spec))
        if not window:
            return
        yield window.popleft()


def _actor_map_stage(stream, chain: List[MapOp], ctx) -> Iterator[Any]:
    import inspect

    size = max(
        (op.compute.size for op in chain if isinstance(op.compute, ActorPoolStrategy)),
        default=2,
    )
    chain_meta = [
        (op.fn, inspect.isclass(op.fn), op.fn_constructor_args,
         op.fn_constructor_kwargs)
        for op in chain
    ]
    skeleton = [
        (op.kind, op.batch_size, op.batch_format, op.fn_args, op.fn_kwargs)
        for op in chain
    ]
    res = {"num_cpus": chain[0].num_cpus or 1, "max_concurrency": 2}
    gpus = max((op.num_gpus or 0 for op in chain), default=0)
    if gpus:
        res["num_gpus"] = gpus
    Worker = ray.remote(**res)(_MapWorker)
    pool = [Worker.remote(chain_meta) for _ in range(size)]
    ray.get([w.ready.remote() for w in pool])
    window = collections.deque()
    it = iter(stream)
    exhausted = False
    i = 0
    try:
        while True:
            while not exhausted and len(window) < 2 * size:
                try:
                    block_ref = next(it)
                except StopIteration:
                    exhausted = True
                    break
                window.append(pool[i % size].process.remote(block_ref, skeleton))
                i += 1
            if not window:
                return
            yield window.popleft()
    finally:
        for w in pool:
            try:
                ray.kill(w)
            except Exception:
                pass


def _limit_stage(stream, limit: int) -> Iterator[Any]:
    taken = 0
    for ref in stream:
        if taken >= limit:
            return
        block = ray.get(ref)
        n = BlockAccessor(block).num_rows()
        if taken + n <= limit:
            taken += n
            yield ref
        else:
            sliced = BlockAccessor(block).slice(0, limit - taken)
            taken = limit
            yield ray.put(sliced)
            return

"""Logical plan: operator list + map-chain fusion.

Role parity: reference python/ray/data/_internal/logical/ (operators +
optimizer rules; MapBatches at logical/operators/map_operator.py:160 and
fusion in logical/rules/operator_fusion.py). Consecutive row/batch map
operators fuse into one task chain so a block crosses process boundaries
once per fused stage, not once per operator.
"""
from __future__ import annotations

from dataclasses import dataclass, field
from typing import Any, Callable, Dict, List, Optional


@dataclass
class ReadOp:
    """Source: a list of no-arg callables each yielding one or more blocks."""

    name: str
    read_tasks: List[Callable]
    # estimated row count if cheaply known (range/from_items)
    num_rows: Optional[int] = None
    # optimizer hook: pushdown(columns=..., filter_expr=...) -> new ReadOp
    # with pruning/filtering folded into the read tasks (parquet supports
    # both natively; None = the source cannot push down)
    pushdown: Optional[Callable] = None
    # pre-materialized arrow-block refs (from_numpy/from_arrow put their
    # blocks driver-side): the executor yields these directly — no read
    # task, no worker round trip, no second serialization of the data
    block_refs: Optional[List] = None


@dataclass
class MapOp:
    """Row/batch transform. kind: map_batches | map_rows | flat_map | filter."""

    name: str
    kind: str
    fn: Any  # callable, or callable class for actor compute
    batch_size: Optional[int] = None
    batch_format: Optional[str] = "default"
    fn_args: tuple = ()
    fn_kwargs: Dict[str, Any] = field(default_factory=dict)
    fn_constructor_args: tuple = ()
    fn_constructor_kwargs: Dict[str, Any] = field(default_factory=dict)
    compute: Optional[Any] = None  # None => tasks; ActorPoolStrategy => actors
    num_cpus: Optional[float] = None
    num_gpus: Optional[float] = None
    concurrency: Optional[Any] = None
    # optimizer metadata for declarative ops: {"type": "select"|"filter_expr"
    # |"with_columns"|"drop"|"rename", ...} — None for opaque user UDFs
    meta: Optional[Dict[str, Any]] = None


@dataclass
class AllToAllOp:
    """Blocking exchange: fn(list_of_block_refs) -> list_of_block_refs."""

    name: str
    fn: Callable


@dataclass
class LimitOp:
    name: str
    limit: int


class ActorPoolStrategy:
    """Parity: ray.data.ActorPoolStrategy — run the fused stage in a pool of
    actors (needed for stateful/GPU UDF classes)."""

    def __init__(self, size: Optional[int] = None, min_size: Optional[int] = None,
                 max_size: Optional[int] = None):
        self.size = size or max_size or min_size or 2


def fuse_stages(ops: List[Any]) -> List[Any]:
    """Group consecutive MapOps with compatible compute into fused chains."""
    stages: List[Any] = []
    chain: List[MapOp] = []

    def flush():
        nonlocal chain
        if chain:
            stages.append(list(chain))
            chain = []

    for op in ops:
        if isinstance(op, MapOp):
            if chain and not _compatible(chain[-1], op):
                flush()
            chain.append(op)
        else:
            flush()
            stages.append(op)
    flush()
    return stages


def _compatible(a: MapOp, b: MapOp) -> bool:
    def key(o: MapOp):
        return (
            o.compute is not None and type(o.compute).__name__ or "tasks",
            o.num_gpus or 0,
        )

    return key(a) == key(b)

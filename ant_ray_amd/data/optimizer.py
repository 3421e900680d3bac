"""Logical-plan optimizer rules.

Role parity: reference python/ray/data/_internal/logical/rules/ —
projection/filter pushdown into sources and predicate reordering. Rules
run once per execution, before stage fusion (executor.py calls
optimize()). Only DECLARATIVE ops (MapOp.meta set by select_columns /
filter_expr / with_columns) are transformed; opaque user UDFs are never
moved.
"""
from __future__ import annotations

from typing import Any, List

from ant_ray_amd.data.plan import MapOp, ReadOp


def _is(op, t: str) -> bool:
    return isinstance(op, MapOp) and (op.meta or {}).get("type") == t


def filter_reorder(ops: List[Any]) -> List[Any]:
    """Move a filter_expr BEFORE an adjacent with_columns when the
    predicate references none of the columns it creates (filter first =
    fewer rows through the projection)."""
    changed = True
    ops = list(ops)
    while changed:
        changed = False
        for i in range(len(ops) - 1):
            a, b = ops[i], ops[i + 1]
            if _is(a, "with_columns") and _is(b, "filter_expr"):
                created = set(a.meta["exprs"].keys())
                needed = b.meta["expr"].columns()
                if not (needed & created):
                    ops[i], ops[i + 1] = b, a
                    changed = True
    return ops


def pushdown_into_reads(ops: List[Any]) -> List[Any]:
    """Fold leading filter_expr / select ops into a ReadOp that supports
    pushdown (parquet): the read tasks then prune columns and rows at the
    source instead of materializing full blocks."""
    if not ops or not isinstance(ops[0], ReadOp) or ops[0].pushdown is None:
        return ops
    read = ops[0]
    rest = ops[1:]
    cols = None
    fexpr = None
    while rest:
        op = rest[0]
        if _is(op, "filter_expr") and cols is None:
            # filters push only while the full column set is still there
            e = op.meta["expr"]
            fexpr = e if fexpr is None else (fexpr & e)
            rest = rest[1:]
        elif _is(op, "select"):
            cols = op.meta["columns"]
            rest = rest[1:]
        else:
            break
    if cols is None and fexpr is None:
        return ops
    new_read = read.pushdown(columns=cols, filter_expr=fexpr)
    return [new_read] + rest


def optimize(ops: List[Any]) -> List[Any]:
    ops = filter_reorder(ops)
    ops = pushdown_into_reads(ops)
    return ops

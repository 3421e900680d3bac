"""Columnar expressions for Dataset.with_columns / filter pushdown.

Role parity: reference python/ray/data/expressions.py (1,094 LoC — Expr
:206, col :940, lit :968). Expressions evaluate vectorized over pyarrow
batches via pyarrow.compute.
"""
from __future__ import annotations

from typing import Any

import pyarrow as pa
import pyarrow.compute as pc


class Expr:
    """A columnar expression tree node."""

    def _eval(self, table: pa.Table):
        raise NotImplementedError

    # arithmetic
    def __add__(self, other):
        return BinaryExpr(pc.add, self, _wrap(other))

    def __radd__(self, other):
        return BinaryExpr(pc.add, _wrap(other), self)

    def __sub__(self, other):
        return BinaryExpr(pc.subtract, self, _wrap(other))

    def __rsub__(self, other):
        return BinaryExpr(pc.subtract, _wrap(other), self)

    def __mul__(self, other):
        return BinaryExpr(pc.multiply, self, _wrap(other))

    def __rmul__(self, other):
        return BinaryExpr(pc.multiply, _wrap(other), self)

    def __truediv__(self, other):
        return BinaryExpr(pc.divide, self, _wrap(other))

    def __mod__(self, other):
        return BinaryExpr(lambda a, b: pc.subtract(
            a, pc.multiply(pc.floor(pc.divide(pc.cast(a, pa.float64()),
                                              pc.cast(b, pa.float64()))), b)),
            self, _wrap(other))

    def __floordiv__(self, other):
        return BinaryExpr(lambda a, b: pc.floor(
            pc.divide(pc.cast(a, pa.float64()), pc.cast(b, pa.float64()))),
            self, _wrap(other))

    # comparisons
    def __gt__(self, other):
        return BinaryExpr(pc.greater, self, _wrap(other))

    def __ge__(self, other):
        return BinaryExpr(pc.greater_equal, self, _wrap(other))

    def __lt__(self, other):
        return BinaryExpr(pc.less, self, _wrap(other))

    def __le__(self, other):
        return BinaryExpr(pc.less_equal, self, _wrap(other))

    def __eq__(self, other):  # noqa: A003
        return BinaryExpr(pc.equal, self, _wrap(other))

    def __ne__(self, other):
        return BinaryExpr(pc.not_equal, self, _wrap(other))

    # logical
    def __and__(self, other):
        return BinaryExpr(pc.and_kleene, self, _wrap(other))

    def __or__(self, other):
        return BinaryExpr(pc.or_kleene, self, _wrap(other))

    def __invert__(self):
        return UnaryExpr(pc.invert, self)

    def alias(self, name: str) -> "AliasExpr":
        return AliasExpr(self, name)

    def is_null(self):
        return UnaryExpr(pc.is_null, self)


class ColumnExpr(Expr):
    def __init__(self, name: str):
        self.name = name

    def _eval(self, table):
        return table.column(self.name)

    def __repr__(self):
        return f"col({self.name!r})"


class LiteralExpr(Expr):
    def __init__(self, value: Any):
        self.value = value

    def _eval(self, table):
        return pa.scalar(self.value)

    def __repr__(self):
        return f"lit({self.value!r})"


class BinaryExpr(Expr):
    def __init__(self, fn, left: Expr, right: Expr):
        self.fn = fn
        self.left = left
        self.right = right

    def _eval(self, table):
        return self.fn(self.left._eval(table), self.right._eval(table))


class UnaryExpr(Expr):
    def __init__(self, fn, operand: Expr):
        self.fn = fn
        self.operand = operand

    def _eval(self, table):
        return self.fn(self.operand._eval(table))


class AliasExpr(Expr):
    def __init__(self, expr: Expr, name: str):
        self.expr = expr
        self.name = name

    def _eval(self, table):
        return self.expr._eval(table)


def _wrap(v) -> Expr:
    return v if isinstance(v, Expr) else LiteralExpr(v)


def col(name: str) -> ColumnExpr:
    """Reference a column (parity expressions.py:940)."""
    return ColumnExpr(name)


def lit(value: Any) -> LiteralExpr:
    """A literal value (parity expressions.py:968)."""
    return LiteralExpr(value)


def eval_expr_to_column(table: pa.Table, expr: Expr) -> pa.ChunkedArray:
    out = expr._eval(table)
    if isinstance(out, pa.Scalar):
        out = pa.chunked_array([pa.array([out.as_py()] * table.num_rows)])
    return out

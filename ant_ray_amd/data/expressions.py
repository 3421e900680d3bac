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

    def is_valid(self):
        return UnaryExpr(pc.is_valid, self)

    def is_in(self, values):
        vals = list(values)
        return UnaryExpr(lambda a: pc.is_in(a, value_set=pa.array(vals)), self)

    isin = is_in

    def between(self, low, high):
        return (self >= low) & (self <= high)

    def fill_null(self, value):
        return BinaryExpr(pc.fill_null, self, _wrap(value))

    def cast(self, dtype):
        return UnaryExpr(lambda a: pc.cast(a, dtype), self)

    def abs(self):  # noqa: A003
        return UnaryExpr(pc.abs, self)

    def floor(self):
        return UnaryExpr(pc.floor, self)

    def ceil(self):
        return UnaryExpr(pc.ceil, self)

    def round(self, ndigits: int = 0):  # noqa: A003
        return UnaryExpr(lambda a: pc.round(a, ndigits=ndigits), self)

    def __neg__(self):
        return UnaryExpr(pc.negate, self)

    def columns(self) -> set:
        """Referenced column names (used by the filter-pushdown rule)."""
        out = set()
        stack = [self]
        while stack:
            e = stack.pop()
            if isinstance(e, ColumnExpr):
                out.add(e.name)
            elif isinstance(e, BinaryExpr):
                stack += [e.left, e.right]
            elif isinstance(e, (UnaryExpr,)):
                stack.append(e.operand)
            elif isinstance(e, AliasExpr):
                stack.append(e.expr)
        return out

    # namespace accessors (parity: reference namespace_expressions/)
    @property
    def str(self):  # noqa: A003
        return _StrNamespace(self)

    @property
    def dt(self):
        return _DtNamespace(self)

    @property
    def list(self):  # noqa: A003
        return _ListNamespace(self)

    @property
    def struct(self):
        return _StructNamespace(self)


class _StrNamespace:
    """String accessors (parity namespace_expressions/string_expressions)."""

    def __init__(self, expr: "Expr"):
        self._e = expr

    def lower(self):
        return UnaryExpr(pc.utf8_lower, self._e)

    def upper(self):
        return UnaryExpr(pc.utf8_upper, self._e)

    def capitalize(self):
        return UnaryExpr(pc.utf8_capitalize, self._e)

    def len(self):  # noqa: A003
        return UnaryExpr(pc.utf8_length, self._e)

    def strip(self):
        return UnaryExpr(pc.utf8_trim_whitespace, self._e)

    def contains(self, pat: str):
        return UnaryExpr(lambda a: pc.match_substring(a, pat), self._e)

    def startswith(self, pat: str):
        return UnaryExpr(lambda a: pc.starts_with(a, pat), self._e)

    def endswith(self, pat: str):
        return UnaryExpr(lambda a: pc.ends_with(a, pat), self._e)

    def replace(self, pat: str, rep: str):
        return UnaryExpr(
            lambda a: pc.replace_substring(a, pat, rep), self._e)

    def split(self, sep: str):
        return UnaryExpr(lambda a: pc.split_pattern(a, sep), self._e)


class _DtNamespace:
    """Datetime accessors (parity namespace_expressions/dt_expressions)."""

    def __init__(self, expr: "Expr"):
        self._e = expr

    def year(self):
        return UnaryExpr(pc.year, self._e)

    def month(self):
        return UnaryExpr(pc.month, self._e)

    def day(self):
        return UnaryExpr(pc.day, self._e)

    def hour(self):
        return UnaryExpr(pc.hour, self._e)

    def minute(self):
        return UnaryExpr(pc.minute, self._e)

    def second(self):
        return UnaryExpr(pc.second, self._e)

    def day_of_week(self):
        return UnaryExpr(pc.day_of_week, self._e)


class _ListNamespace:
    def __init__(self, expr: "Expr"):
        self._e = expr

    def len(self):  # noqa: A003
        return UnaryExpr(pc.list_value_length, self._e)

    def get(self, index: int):
        return UnaryExpr(lambda a: pc.list_element(a, index), self._e)


class _StructNamespace:
    def __init__(self, expr: "Expr"):
        self._e = expr

    def field(self, name: str):
        return UnaryExpr(lambda a: pc.struct_field(a, name), self._e)


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

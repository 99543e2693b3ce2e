"""Column expression API (reference: python/ray/data/expressions.py —
`col`/`lit` build an Expr tree that filter/with_columns evaluate
vectorized per batch instead of row-by-row UDFs)."""
from __future__ import annotations

import operator
from typing import Any, Dict

import numpy as np


class Expr:
    """Lazily-evaluated column expression; evaluated vectorized over a
    numpy batch dict."""

    # -------- construction --------
    def _bin(self, other, op, rev=False):
        other = other if isinstance(other, Expr) else LiteralExpr(other)
        return BinaryExpr(op, other, self) if rev else BinaryExpr(
            op, self, other)

    def __add__(self, o):
        return self._bin(o, operator.add)

    def __radd__(self, o):
        return self._bin(o, operator.add, rev=True)

    def __sub__(self, o):
        return self._bin(o, operator.sub)

    def __rsub__(self, o):
        return self._bin(o, operator.sub, rev=True)

    def __mul__(self, o):
        return self._bin(o, operator.mul)

    def __rmul__(self, o):
        return self._bin(o, operator.mul, rev=True)

    def __truediv__(self, o):
        return self._bin(o, operator.truediv)

    def __rtruediv__(self, o):
        return self._bin(o, operator.truediv, rev=True)

    def __mod__(self, o):
        return self._bin(o, operator.mod)

    def __gt__(self, o):
        return self._bin(o, operator.gt)

    def __ge__(self, o):
        return self._bin(o, operator.ge)

    def __lt__(self, o):
        return self._bin(o, operator.lt)

    def __le__(self, o):
        return self._bin(o, operator.le)

    def __eq__(self, o):  # noqa: PYI032
        return self._bin(o, operator.eq)

    def __ne__(self, o):  # noqa: PYI032
        return self._bin(o, operator.ne)

    def __and__(self, o):
        return self._bin(o, operator.and_)

    def __or__(self, o):
        return self._bin(o, operator.or_)

    def __invert__(self):
        return UnaryExpr(operator.invert, self)

    def __neg__(self):
        return UnaryExpr(operator.neg, self)

    def __hash__(self):
        return id(self)

    def is_in(self, values) -> "Expr":
        vals = list(values)
        return UnaryExpr(lambda a: np.isin(a, vals), self)

    def alias(self, name: str) -> "Expr":
        e = AliasExpr(self)
        e.name = name
        return e

    # -------- evaluation --------
    def eval(self, batch: Dict[str, np.ndarray]):
        raise NotImplementedError


class ColumnExpr(Expr):
    def __init__(self, name: str):
        self.name = name

    def eval(self, batch):
        return np.asarray(batch[self.name])

    def __repr__(self):
        return f"col({self.name!r})"


class LiteralExpr(Expr):
    def __init__(self, value: Any):
        self.value = value

    def eval(self, batch):
        return self.value

    def __repr__(self):
        return f"lit({self.value!r})"


class BinaryExpr(Expr):
    def __init__(self, op, left: Expr, right: Expr):
        self.op = op
        self.left = left
        self.right = right

    def eval(self, batch):
        lv = self.left.eval(batch)
        rv = self.right.eval(batch)
        # & and | need bool arrays, not python bools
        if self.op in (operator.and_, operator.or_):
            lv = np.asarray(lv, dtype=bool)
            rv = np.asarray(rv, dtype=bool)
        return self.op(lv, rv)


class UnaryExpr(Expr):
    def __init__(self, op, inner: Expr):
        self.op = op
        self.inner = inner

    def eval(self, batch):
        v = self.inner.eval(batch)
        if self.op is operator.invert:
            v = np.asarray(v, dtype=bool)
        return self.op(v)


class AliasExpr(Expr):
    def __init__(self, inner: Expr):
        self.inner = inner
        self.name = None

    def eval(self, batch):
        return self.inner.eval(batch)


def col(name: str) -> ColumnExpr:
    """Reference a column by name."""
    return ColumnExpr(name)


def lit(value: Any) -> LiteralExpr:
    """A literal constant in an expression."""
    return LiteralExpr(value)

"""Window expressions.

Reference analogue: the window exec family (GpuWindowExec / running /
bounded / double-pass variants, sql-plugin .../rapids/window/, SURVEY.md
§2.4). Supported this round: ranking (row_number, rank, dense_rank),
running and whole-partition aggregates (sum/count/min/max/avg), lag/lead.
Frames: ROWS UNBOUNDED PRECEDING..CURRENT ROW ("running", the default for
ordered aggregates like Spark) and UNBOUNDED..UNBOUNDED (whole partition,
the default when no order is given).
"""
from __future__ import annotations

from typing import Optional, Sequence

from ..types import DType, FLOAT64, INT32, INT64
from .expressions import Expression, _as_expr

RANKING = {"row_number", "rank", "dense_rank", "ntile"}
AGGS = {"sum", "count", "min", "max", "mean"}
OFFSETS = {"lag", "lead", "nth_value"}


class WindowSpec:
    def __init__(self, partition_by: Sequence[str] = (),
                 order_by: Sequence[str] = (),
                 descending: Optional[Sequence[bool]] = None,
                 rows_between=None, range_between=None):
        self.partition_by = list(partition_by)
        self.order_by = list(order_by)
        self.descending = list(descending) if descending is not None \
            else [False] * len(self.order_by)
        # (preceding, following) row offsets, e.g. (-3, 0) = 3 PRECEDING..
        # CURRENT ROW; None = default frame (running / whole partition)
        self.rows_between = rows_between
        # RANGE frame: (lo, hi) VALUE offsets on the single numeric order
        # key — frame = rows with key in [cur+lo, cur+hi]; None for
        # UNBOUNDED on that end (reference: GpuSpecifiedWindowFrame RangeFrame)
        self.range_between = range_between


class WindowFunc:
    """An unbound window function; bind with .over(...)."""

    def __init__(self, op: str, child: Optional[Expression] = None,
                 offset: int = 1, default=None):
        self.op = op
        self.child = _as_expr(child) if child is not None else None
        self.offset = offset
        self.default = default

    def over(self, partition_by: Sequence[str] = (),
             order_by: Sequence[str] = (),
             descending: Optional[Sequence[bool]] = None,
             rows_between=None, range_between=None) -> "WindowExpr":
        return WindowExpr(self, WindowSpec(partition_by, order_by, descending,
                                           rows_between, range_between))


class WindowExpr:
    def __init__(self, func: WindowFunc, spec: WindowSpec,
                 name: Optional[str] = None):
        self.func = func
        self.spec = spec
        self._name = name
        if func.op in RANKING and not spec.order_by:
            raise ValueError(f"{func.op} requires order_by")
        if func.op in OFFSETS and not spec.order_by:
            raise ValueError(f"{func.op} requires order_by")
        if spec.range_between is not None:
            if func.op not in AGGS:
                raise ValueError("range_between needs an aggregate function")
            if len(spec.order_by) != 1:
                raise ValueError("range_between requires exactly one "
                                 "order_by column")
            lo, hi = spec.range_between
            if lo is not None and hi is not None and lo > hi:
                raise ValueError("range_between needs lo <= hi")
        if spec.rows_between is not None:
            if func.op not in AGGS:
                raise ValueError("rows_between needs an aggregate function")
            if not spec.order_by:
                raise ValueError("rows_between requires order_by")
            lo, hi = spec.rows_between
            if lo > 0 or hi < lo:
                raise ValueError("rows_between must be (lo<=0, hi>=lo)")

    def alias(self, name: str) -> "WindowExpr":
        return WindowExpr(self.func, self.spec, name)

    def output_name(self) -> str:
        if self._name:
            return self._name
        c = f"({self.func.child})" if self.func.child is not None else "()"
        return f"{self.func.op}{c}"

    def out_dtype(self, schema) -> DType:
        op = self.func.op
        if op in RANKING:
            return INT32
        if op == "count":
            return INT64
        ct = self.func.child.dtype(schema)
        if op == "sum":
            return FLOAT64 if ct.is_floating else INT64
        if op == "mean":
            return FLOAT64
        return ct  # min/max/lag/lead keep input type

    def nullable(self, schema) -> bool:
        return self.func.op not in RANKING


def ntile(n: int) -> WindowFunc:
    """Bucket 1..n by position within the ordered partition (GpuNTile):
    bucket = (rn-1)*n // size + 1 with Spark's remainder-to-front split."""
    assert n >= 1
    return WindowFunc("ntile", None, offset=n)


def nth_value(e, k: int) -> WindowFunc:
    """k-th value (1-based) of the ordered partition (GpuNthValue);
    NULL when the partition has fewer than k rows."""
    assert k >= 1
    return WindowFunc("nth_value", e, offset=k)


def row_number() -> WindowFunc:
    return WindowFunc("row_number")


def rank() -> WindowFunc:
    return WindowFunc("rank")


def dense_rank() -> WindowFunc:
    return WindowFunc("dense_rank")


def win_sum(e) -> WindowFunc:
    return WindowFunc("sum", e)


def win_count(e) -> WindowFunc:
    return WindowFunc("count", e)


def win_min(e) -> WindowFunc:
    return WindowFunc("min", e)


def win_max(e) -> WindowFunc:
    return WindowFunc("max", e)


def win_avg(e) -> WindowFunc:
    return WindowFunc("mean", e)


def lag(e, offset: int = 1, default=None) -> WindowFunc:
    return WindowFunc("lag", e, offset, default)


def lead(e, offset: int = 1, default=None) -> WindowFunc:
    return WindowFunc("lead", e, offset, default)

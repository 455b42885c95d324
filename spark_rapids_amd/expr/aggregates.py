"""Aggregate expression definitions.

Reference analogue: org/apache/spark/sql/rapids/aggregate/aggregateFunctions.scala
(GpuSum, GpuCount, GpuMin, GpuMax, GpuAverage...). An AggExpr names the
aggregate op, the input expression, and the output type; the physical hash
aggregate lowers these to the groupby kernel's (op, column, dtype) form with
Spark result-type rules (sum(int) -> bigint, avg -> double, count -> bigint).
"""
from __future__ import annotations

from typing import Optional

from ..column import Schema
from ..types import DType, FLOAT64, INT64
from .expressions import Expression, _as_expr


class AggExpr:
    def __init__(self, op: str, child: Optional[Expression],
                 name: Optional[str] = None, distinct: bool = False):
        self.op = op
        self.child = _as_expr(child) if child is not None else None
        self._name = name
        self.distinct = distinct

    def alias(self, name: str) -> "AggExpr":
        return AggExpr(self.op, self.child, name, self.distinct)

    _DISPLAY = {"count_all": "count", "mean": "avg"}

    def output_name(self) -> str:
        if self._name:
            return self._name
        if self.op.startswith("percentile:"):
            return f"percentile({self.child}, {self.op.split(':', 1)[1]})"
        if self.op.startswith("hll:"):
            return f"approx_count_distinct({self.child})"
        disp = self._DISPLAY.get(self.op, self.op)
        if self.child is None:
            return f"{disp}(*)"
        if self.distinct:
            return f"{disp}(DISTINCT {self.child})"
        return f"{disp}({self.child})"

    def out_dtype(self, schema: Schema) -> DType:
        if self.op in ("count", "count_all"):
            return INT64
        ct = self.child.dtype(schema)
        if self.op == "sum":
            if ct.is_floating:
                return FLOAT64
            if ct.is_decimal:
                return DType.decimal(min(ct.precision + 10, 38), ct.scale)
            return INT64
        if self.op in ("mean", "stddev", "variance"):
            return FLOAT64
        if self.op in ("min", "max", "first", "last"):
            return ct
        if self.op in ("bit_and", "bit_or", "bit_xor"):
            return ct
        if self.op in ("collect_list", "collect_set"):
            return DType.list_(ct)
        if self.op.startswith("percentile:"):
            return FLOAT64
        if self.op.startswith("hll:"):
            return INT64
        raise NotImplementedError(f"agg {self.op}")

    def __str__(self):
        return self.output_name()


def sum_(e) -> AggExpr:
    return AggExpr("sum", e)


def avg(e) -> AggExpr:
    return AggExpr("mean", e)


def count(e) -> AggExpr:
    return AggExpr("count", e)


def count_star() -> AggExpr:
    return AggExpr("count_all", None)


def min_(e) -> AggExpr:
    return AggExpr("min", e)


def max_(e) -> AggExpr:
    return AggExpr("max", e)


def stddev(e) -> AggExpr:
    """stddev_samp"""
    return AggExpr("stddev", e)


def variance(e) -> AggExpr:
    """var_samp"""
    return AggExpr("variance", e)


def count_distinct(e) -> AggExpr:
    """count(DISTINCT e): lowered to a two-level aggregate (dedupe on
    (keys, e) then count). Reference analogue: RewriteDistinctAggregates'
    single-distinct plan executed by two GpuHashAggregates."""
    return AggExpr("count", e, distinct=True)


def sum_distinct(e) -> AggExpr:
    return AggExpr("sum", e, distinct=True)


def first(e) -> AggExpr:
    """First non-null value in the group. Like Spark's first() without an
    ordering: which value is 'first' is unspecified on the GPU (any
    non-null value of the group); the CPU backend returns the actual first
    by input order. Reference analogue: GpuFirst (ignoreNulls=true)."""
    return AggExpr("first", e)


def last(e) -> AggExpr:
    """Last non-null value (GpuLast); same determinism caveats as first."""
    return AggExpr("last", e)


def bit_and(e) -> AggExpr:
    """Bitwise AND of the group's non-null integers (GpuBitAndAgg)."""
    return AggExpr("bit_and", e)


def bit_or(e) -> AggExpr:
    return AggExpr("bit_or", e)


def bit_xor(e) -> AggExpr:
    return AggExpr("bit_xor", e)


def approx_count_distinct(e, rsd: float = 0.05) -> AggExpr:
    """HyperLogLog++ sketch (k_gb_hll over xxHash64, Spark precision
    p = ceil(2*log2(1.106/rsd))). Estimates use the standard HLL
    small-range correction (no empirical bias tables), so values can
    differ slightly from Spark's — the same class of divergence the
    reference gates behind incompatibleOps. For exact counts use
    count_distinct()."""
    import math

    assert 0.0 < rsd < 1.0
    p = max(4, math.ceil(2.0 * math.log2(1.106 / rsd)))
    return AggExpr(f"hll:{p}", e)


def percentile(e, p: float) -> AggExpr:
    """Exact percentile with linear interpolation (Spark percentile(col, p);
    GpuPercentile analogue). p in [0, 1]."""
    assert 0.0 <= p <= 1.0
    return AggExpr(f"percentile:{p}", e)


def approx_percentile(e, p: float) -> AggExpr:
    """Served by the exact implementation (always at least as accurate as
    the reference's t-digest approx_percentile)."""
    return percentile(e, p)


def collect_list(e) -> AggExpr:
    """Gather the group's non-null values into an array (order unspecified,
    like Spark). Reference analogue: GpuCollectList."""
    return AggExpr("collect_list", e)


def collect_set(e) -> AggExpr:
    """Gather the group's distinct non-null values (GpuCollectSet)."""
    return AggExpr("collect_set", e)

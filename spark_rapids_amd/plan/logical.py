"""Logical plan nodes.

The engine's analogue of the Catalyst logical plan that the reference's
GpuOverrides pass consumes (reference: GpuOverrides.scala wrapAndTagPlan).
Deliberately small: Scan, Filter, Project, Aggregate, Join, Sort, Limit, Union,
Exchange (inserted by the planner for distributed runs).
"""
from __future__ import annotations

from typing import List, Optional, Sequence

from ..column import Field, Schema
from ..expr.aggregates import AggExpr
from ..expr.expressions import Expression
from ..types import BOOL, DType, TypeId


class LogicalPlan:
    @property
    def children(self) -> Sequence["LogicalPlan"]:
        return ()

    def schema(self) -> Schema:
        raise NotImplementedError

    def name(self) -> str:
        return type(self).__name__


class Scan(LogicalPlan):
    def __init__(self, source, schema: Schema, label: str = "scan"):
        self.source = source  # object with .partitions() -> iterable[ColumnBatch]
        self._schema = schema
        self.label = label

    @property
    def replicated(self) -> bool:
        return bool(getattr(self.source, "replicated", False))

    def schema(self) -> Schema:
        return self._schema

    def name(self) -> str:
        return f"Scan({self.label})"


class Cached(LogicalPlan):
    """A subplan shared by several parents (e.g. the finest-level
    aggregate feeding every rollup/cube grouping set): the physical plan
    converts it to ONE exec that materializes once per execution instead
    of re-running the subtree per consumer. Reference analogue:
    ReusedExchangeExec / AQE stage reuse."""

    def __init__(self, child: LogicalPlan):
        self.child = child

    @property
    def children(self):
        return (self.child,)

    def schema(self) -> Schema:
        return self.child.schema()


class Filter(LogicalPlan):
    def __init__(self, condition: Expression, child: LogicalPlan):
        self.condition = condition
        self.child = child
        assert condition.dtype(child.schema()) == BOOL, "filter needs boolean"

    @property
    def children(self):
        return (self.child,)

    def schema(self) -> Schema:
        return self.child.schema()


class Project(LogicalPlan):
    def __init__(self, exprs: List[Expression], child: LogicalPlan):
        self.exprs = exprs
        self.child = child

    @property
    def children(self):
        return (self.child,)

    def schema(self) -> Schema:
        cs = self.child.schema()
        return Schema([Field(e.output_name(), e.dtype(cs), e.nullable(cs))
                       for e in self.exprs])


class Aggregate(LogicalPlan):
    def __init__(self, group_exprs: List[Expression], aggs: List[AggExpr],
                 child: LogicalPlan):
        cs = child.schema()
        for e in group_exprs:
            if e.dtype(cs).is_nested:
                raise NotImplementedError(
                    f"grouping by nested type {e.dtype(cs)} ({e}) is not "
                    "supported — explode the array first")
        self.group_exprs = group_exprs
        self.aggs = aggs
        self.child = child

    @property
    def children(self):
        return (self.child,)

    def schema(self) -> Schema:
        cs = self.child.schema()
        fields = [Field(e.output_name(), e.dtype(cs), e.nullable(cs))
                  for e in self.group_exprs]
        fields += [Field(a.output_name(), a.out_dtype(cs), True) for a in self.aggs]
        return Schema(fields)


class Join(LogicalPlan):
    def __init__(self, left: LogicalPlan, right: LogicalPlan,
                 left_on: List[str], right_on: List[str], how: str = "inner",
                 using: bool = False, condition=None):
        self.left = left
        self.right = right
        self.left_on = left_on
        self.right_on = right_on
        self.how = how
        # optional non-equi predicate over (left ++ right) columns applied
        # to matched pairs (reference: conditional/mixed hash joins)
        self.condition = condition
        # USING-join semantics (join by shared column names): the
        # duplicate right key columns are dropped for inner/left joins,
        # matching Spark's df.join(other, "k"). Full outer keeps both
        # (Spark would coalesce them; documented difference).
        self.using = using and how in ("inner", "left")

    @property
    def children(self):
        return (self.left, self.right)

    def schema(self) -> Schema:
        ls = self.left.schema()
        if self.how in ("semi", "anti"):
            return ls
        rs = self.right.schema()
        left_fields = [Field(f.name, f.dtype,
                             f.nullable or self.how == "full")
                       for f in ls.fields]
        right_fields = []
        for f in rs.fields:
            if self.using and f.name in self.right_on:
                continue
            nullable = f.nullable or self.how in ("left", "full")
            right_fields.append(Field(f.name, f.dtype, nullable))
        return Schema(left_fields + right_fields)

    def name(self) -> str:
        return f"Join({self.how})"


class CacheData(LogicalPlan):
    """df.cache(): materialize the child once as compressed in-memory
    parquet blobs and serve every later action from them (reference
    analogue: ParquetCachedBatchSerializer — Spark df.cache() with
    parquet-compressed cached batches)."""

    def __init__(self, child: LogicalPlan):
        self.child = child
        self.store = None  # List[bytes] after first materialization

    @property
    def children(self):
        return (self.child,)

    def schema(self) -> Schema:
        return self.child.schema()

    def name(self) -> str:
        state = "materialized" if self.store is not None else "lazy"
        return f"CacheData({state})"


class NestedLoopJoin(Join):
    """Join on an arbitrary (non-equi) condition with no equality keys:
    every (left, right) row pair is tested, chunked to bound memory
    (reference analogue: GpuBroadcastNestedLoopJoinExec with a compiled
    AST condition). Same constructor shape as Join so the optimizer can
    rebuild either with type(plan)(...)."""

    def name(self) -> str:
        return f"NestedLoopJoin({self.how})"


class MapBatches(LogicalPlan):
    """CPU python-function operator (UDF bridge); never places on GPU."""

    def __init__(self, fn, child: LogicalPlan, schema=None):
        self.fn = fn
        self.child = child
        self._schema = schema

    @property
    def children(self):
        return (self.child,)

    def schema(self) -> Schema:
        return self._schema if self._schema is not None else self.child.schema()


class CrossJoin(LogicalPlan):
    """Cartesian product (reference analogue: GpuCartesianProductExec /
    GpuBroadcastNestedLoopJoinExec — non-equi joins are cross + filter)."""

    def __init__(self, left: LogicalPlan, right: LogicalPlan):
        self.left = left
        self.right = right

    @property
    def children(self):
        return (self.left, self.right)

    def schema(self) -> Schema:
        return Schema(list(self.left.schema().fields) +
                      list(self.right.schema().fields))


class Window(LogicalPlan):
    def __init__(self, window_exprs, child: LogicalPlan):
        self.window_exprs = list(window_exprs)
        self.child = child

    @property
    def children(self):
        return (self.child,)

    def schema(self) -> Schema:
        cs = self.child.schema()
        fields = list(cs.fields)
        for w in self.window_exprs:
            fields.append(Field(w.output_name(), w.out_dtype(cs),
                                w.nullable(cs)))
        return Schema(fields)


class Sort(LogicalPlan):
    def __init__(self, child: LogicalPlan, keys: List[str],
                 descending: Optional[List[bool]] = None,
                 nulls_last: Optional[List[bool]] = None):
        self.child = child
        self.keys = keys
        self.descending = descending or [False] * len(keys)
        # Spark default: NULLS FIRST for asc, NULLS LAST for desc
        self.nulls_last = nulls_last or [d for d in self.descending]

    @property
    def children(self):
        return (self.child,)

    def schema(self) -> Schema:
        return self.child.schema()


class Limit(LogicalPlan):
    def __init__(self, child: LogicalPlan, n: int):
        self.child = child
        self.n = n

    @property
    def children(self):
        return (self.child,)

    def schema(self) -> Schema:
        return self.child.schema()


class Expand(LogicalPlan):
    """GROUPING SETS row expansion: each input row is emitted once per
    projection (all projections share one output schema). Reference
    analogue: GpuExpandExec (sql-plugin .../GpuExpandExecMeta) backing
    rollup / cube / grouping sets."""

    def __init__(self, projections: List[List[Expression]],
                 child: LogicalPlan):
        assert projections and all(
            len(p) == len(projections[0]) for p in projections)
        self.projections = projections
        self.child = child

    @property
    def children(self):
        return (self.child,)

    def schema(self) -> Schema:
        cs = self.child.schema()
        fields = []
        for i, e in enumerate(self.projections[0]):
            nullable = any(p[i].nullable(cs) for p in self.projections)
            fields.append(Field(e.output_name(), e.dtype(cs), nullable))
        return Schema(fields)


class Generate(LogicalPlan):
    """explode/posexplode of a LIST column: each input row emits one row per
    element (outer=True keeps empty/null lists as one null-element row).
    Reference analogue: GpuGenerateExec + GpuExplode/GpuPosExplode."""

    def __init__(self, column: str, child: LogicalPlan, outer: bool = False,
                 pos: bool = False):
        self.column = column
        self.child = child
        self.outer = outer
        self.pos = pos

    @property
    def children(self):
        return (self.child,)

    def schema(self) -> Schema:
        cs = self.child.schema()
        fields = [f for f in cs.fields if f.name != self.column]
        if self.pos:
            fields.append(Field("pos", DType.int32(), False))
        lf = cs.field(self.column)
        assert lf.dtype.id is TypeId.LIST, \
            f"explode needs a LIST column, got {lf.dtype}"
        fields.append(Field(self.column, lf.dtype.children[0], True))
        return Schema(fields)


class Union(LogicalPlan):
    def __init__(self, plans: List[LogicalPlan]):
        self.plans = plans

    @property
    def children(self):
        return tuple(self.plans)

    def schema(self) -> Schema:
        return self.plans[0].schema()


def is_replicated(plan: LogicalPlan) -> bool:
    """True if every rank holds identical full data for this subtree (so
    distributed exchanges/broadcasts must be skipped)."""
    if isinstance(plan, Scan):
        return plan.replicated
    if isinstance(plan, (Join, CrossJoin)):
        # the build (right) side is either replicated or broadcast
        # (all-gathered) by the exec, so the output's distribution follows
        # the stream (left) side
        return is_replicated(plan.left)
    if isinstance(plan, Union):
        return all(is_replicated(p) for p in plan.plans)
    if isinstance(plan, Aggregate):
        # replicated input -> local agg is global; keyless sharded input ->
        # all-gather merge makes the output replicated too
        if is_replicated(plan.child):
            return True
        return len(plan.group_exprs) == 0
    if plan.children:
        return all(is_replicated(c) for c in plan.children)
    return False

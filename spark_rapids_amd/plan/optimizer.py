"""Logical optimizations applied before physical planning.

Column pruning (projection pushdown): the reference relies on Spark's
Catalyst optimizer for this; here the engine owns the plan, so the pass
lives in-engine. Inserting narrow Projects under joins/aggregates removes
unused columns from join gathers and scan transfers — on the NDS power-run
shape this halves the bytes moved through the join gather kernels.
"""
from __future__ import annotations

from typing import Optional, Set

from ..expr.expressions import ColumnRef, Expression
from ..expr.windows import WindowExpr
from . import logical as L


def _refs(e, out: Set[str]):
    if isinstance(e, ColumnRef):
        out.add(e.name)
    if isinstance(e, Expression):
        for c in e.children:
            _refs(c, out)


def _expr_refs(exprs) -> Set[str]:
    out: Set[str] = set()
    for e in exprs:
        _refs(e, out)
    return out


def _project_to(plan: L.LogicalPlan, needed: Optional[Set[str]]):
    """Wrap plan in a Project keeping only `needed` (schema order), if that
    actually drops columns."""
    if needed is None:
        return plan
    names = plan.schema().names
    keep = [n for n in names if n in needed]
    if len(keep) == len(names) or not keep:
        return plan
    return L.Project([ColumnRef(n) for n in keep], plan)


def prune_columns(plan: L.LogicalPlan,
                  needed: Optional[Set[str]] = None) -> L.LogicalPlan:
    """Return an equivalent plan where children materialize only the columns
    the ancestors reference. needed=None means every output column."""
    if isinstance(plan, L.Scan):
        return _project_to(plan, needed)
    if isinstance(plan, L.Project):
        cs = plan.child.schema()
        if needed is None:
            exprs = plan.exprs
        else:
            exprs = [e for e in plan.exprs if e.output_name() in needed]
            if not exprs:
                exprs = plan.exprs[:1]
        child_needed = _expr_refs(exprs)
        return L.Project(exprs, prune_columns(plan.child, child_needed))
    if isinstance(plan, L.Filter):
        child_needed = None
        if needed is not None:
            child_needed = set(needed)
            _refs(plan.condition, child_needed)
        return L.Filter(plan.condition, prune_columns(plan.child, child_needed))
    if isinstance(plan, L.Aggregate):
        child_needed = _expr_refs(plan.group_exprs)
        for a in plan.aggs:
            if a.child is not None:
                _refs(a.child, child_needed)
        return L.Aggregate(plan.group_exprs, plan.aggs,
                           prune_columns(plan.child, child_needed))
    if isinstance(plan, L.Join):
        lnames = set(plan.left.schema().names)
        rnames = set(plan.right.schema().names)
        if needed is None:
            lneed = rneed = None
        else:
            lneed = (needed & lnames) | set(plan.left_on)
            rneed = (needed & rnames) | set(plan.right_on)
        left = prune_columns(plan.left, lneed)
        left = _project_to(left, lneed)
        right = prune_columns(plan.right, rneed)
        right = _project_to(right, rneed)
        return L.Join(left, right, plan.left_on, plan.right_on, plan.how)
    if isinstance(plan, L.Sort):
        child_needed = None if needed is None else set(needed) | set(plan.keys)
        return L.Sort(prune_columns(plan.child, child_needed), plan.keys,
                      plan.descending, plan.nulls_last)
    if isinstance(plan, L.Window):
        child_needed = None
        if needed is not None:
            child_needed = set(needed)
            spec = plan.window_exprs[0].spec
            child_needed |= set(spec.partition_by) | set(spec.order_by)
            for w in plan.window_exprs:
                if w.func.child is not None:
                    _refs(w.func.child, child_needed)
            child_needed &= set(plan.child.schema().names)
        return L.Window(plan.window_exprs,
                        prune_columns(plan.child, child_needed))
    if isinstance(plan, L.CrossJoin):
        lnames = set(plan.left.schema().names)
        if needed is None:
            lneed = rneed = None
        else:
            lneed = needed & lnames
            rneed = needed - lnames
        left = _project_to(prune_columns(plan.left, lneed), lneed)
        right = _project_to(prune_columns(plan.right, rneed), rneed)
        return L.CrossJoin(left, right)
    if isinstance(plan, L.MapBatches):
        return L.MapBatches(plan.fn, prune_columns(plan.child, None),
                            plan._schema)
    if isinstance(plan, L.Limit):
        return L.Limit(prune_columns(plan.child, needed), plan.n)
    if isinstance(plan, L.Union):
        # positional union: prune only when every child keeps the same set
        return L.Union([prune_columns(p, None) for p in plan.plans])
    return plan

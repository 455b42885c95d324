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


def _split_conjuncts(e: Expression):
    """Split a predicate on AND. Filtering by `a AND b` equals filtering by
    `a` then `b` under Kleene logic (the row passes only when both are
    TRUE), so conjuncts can be pushed independently."""
    from ..expr.expressions import BinaryExpr

    if isinstance(e, BinaryExpr) and e.op == "and":
        return _split_conjuncts(e.left) + _split_conjuncts(e.right)
    return [e]


def _and_all(conjs):
    from ..expr.expressions import BinaryExpr

    out = conjs[0]
    for c in conjs[1:]:
        out = BinaryExpr("and", out, c)
    return out


def push_filters(plan: L.LogicalPlan) -> L.LogicalPlan:
    """Predicate pushdown through joins (Catalyst PushPredicateThroughJoin
    analogue — the reference inherits this from Spark; here the engine owns
    the plan). A conjunct whose columns all come from one join side filters
    that side BEFORE the join: on dimension-filter queries this shrinks the
    build side and every downstream gather.

    Safety: inner/cross joins push to either side; left/semi/anti joins
    push only left-side conjuncts (right-side predicates would change the
    null-extension semantics)."""
    if isinstance(plan, L.Cached):
        # shared subplan: rewrite ONCE and return the same node to every
        # parent, else each consumer gets its own copy and the physical
        # CachedExec sharing is lost
        memo = getattr(plan, "_pushed_memo", None)
        if memo is None:
            memo = L.Cached(push_filters(plan.child))
            plan._pushed_memo = memo
        return memo
    # rewrite children first
    kids = [push_filters(c) for c in plan.children]
    plan = _with_children(plan, kids)
    if not isinstance(plan, L.Filter):
        return plan
    child = plan.child
    if isinstance(child, (L.Join, L.CrossJoin)):
        ls = set(child.left.schema().names)
        rs = set(child.right.schema().names)
        how = getattr(child, "how", "inner")
        left_ok = how in ("inner", "left", "semi", "anti", "cross") or \
            isinstance(child, L.CrossJoin)
        right_ok = how == "inner" or isinstance(child, L.CrossJoin)
        push_l, push_r, keep = [], [], []
        for c in _split_conjuncts(plan.condition):
            refs: Set[str] = set()
            _refs(c, refs)
            if refs and refs <= ls and left_ok:
                push_l.append(c)
            elif refs and refs <= rs and right_ok:
                push_r.append(c)
            else:
                keep.append(c)
        if not push_l and not push_r:
            return plan
        left = L.Filter(_and_all(push_l), child.left) if push_l \
            else child.left
        right = L.Filter(_and_all(push_r), child.right) if push_r \
            else child.right
        if isinstance(child, L.CrossJoin):
            new_join = L.CrossJoin(left, right)
        else:
            new_join = type(child)(left, right, child.left_on,
                                   child.right_on, child.how,
                                   using=child.using,
                                   condition=child.condition)
        # re-run on the pushed filters (stacked joins push further down)
        new_join = _with_children(
            new_join, [push_filters(c) for c in new_join.children])
        if keep:
            return L.Filter(_and_all(keep), new_join)
        return new_join
    if isinstance(child, L.Filter):
        # merge adjacent filters so conjuncts push as one set
        merged = L.Filter(_and_all([plan.condition, child.condition]),
                          child.child)
        out = push_filters(merged)
        return out
    if isinstance(child, L.Scan) and hasattr(child.source, "with_predicate"):
        triples = _simple_conjuncts(plan.condition)
        if triples:
            # row-group min/max skipping; the Filter stays for exactness
            src = child.source.with_predicate(triples)
            return L.Filter(plan.condition,
                            L.Scan(src, child.schema(), child.label))
    return plan


_STAT_OPS = {"lt", "le", "gt", "ge", "eq"}
_STAT_SWAP = {"lt": "gt", "le": "ge", "gt": "lt", "ge": "le", "eq": "eq"}


def _simple_conjuncts(cond):
    """(col, op, literal) triples usable against row-group statistics."""
    from ..expr.expressions import BinaryExpr, ColumnRef, Literal

    out = []
    for c in _split_conjuncts(cond):
        if not (isinstance(c, BinaryExpr) and c.op in _STAT_OPS):
            continue
        l, r = c.left, c.right
        if isinstance(l, ColumnRef) and isinstance(r, Literal) \
                and r.value is not None:
            out.append((l.name, c.op, r.value))
        elif isinstance(r, ColumnRef) and isinstance(l, Literal) \
                and l.value is not None:
            out.append((r.name, _STAT_SWAP[c.op], l.value))
    return out


def _with_children(plan: L.LogicalPlan, kids):
    """Shallow-rebuild a node with new children (nodes are plain objects;
    mutate the child slots in place on a copy)."""
    import copy

    if list(plan.children) == list(kids):
        return plan
    p = copy.copy(plan)
    if isinstance(plan, (L.Join, L.CrossJoin)):
        p.left, p.right = kids
    elif isinstance(plan, L.Union):
        p.plans = list(kids)
    elif hasattr(plan, "child"):
        p.child = kids[0]
    return p


def prune_columns(plan: L.LogicalPlan,
                  needed: Optional[Set[str]] = None) -> L.LogicalPlan:
    """Return an equivalent plan where children materialize only the columns
    the ancestors reference. needed=None means every output column."""
    if isinstance(plan, L.Cached):
        memo = getattr(plan, "_pruned_memo", None)
        if memo is None:
            # consumers may need different subsets; cache the full output
            # (it is small by construction) and let them project
            memo = L.Cached(prune_columns(plan.child, None))
            plan._pruned_memo = memo
        return memo
    if isinstance(plan, L.Scan):
        # push the needed-column set into sources that can skip IO+decode
        # for unused columns (parquet/orc: per-column chunks on disk)
        if needed is not None and hasattr(plan.source, "with_columns"):
            fields = plan.schema().fields
            keep = [f.name for f in fields if f.name in needed]
            if keep and len(keep) < len(fields):
                src = plan.source.with_columns(keep)
                return L.Scan(src, src.schema, plan.label)
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
            cond_refs: Set[str] = set()
            if plan.condition is not None:
                _refs(plan.condition, cond_refs)
            lneed = ((needed | cond_refs) & lnames) | set(plan.left_on)
            rneed = ((needed | cond_refs) & rnames) | set(plan.right_on)
        left = prune_columns(plan.left, lneed)
        left = _project_to(left, lneed)
        right = prune_columns(plan.right, rneed)
        right = _project_to(right, rneed)
        return type(plan)(left, right, plan.left_on, plan.right_on,
                          plan.how, using=plan.using,
                          condition=plan.condition)
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

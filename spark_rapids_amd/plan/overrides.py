"""GPU overrides: tag the logical plan, place each operator on GPU or CPU,
insert device transitions, and produce explain output.

Reference analogue: GpuOverrides.scala (wrapAndTagPlan -> tag -> convert,
registries at :4260/:4466) + RapidsMeta.scala (willNotWorkOnGpu reasons) +
GpuTransitionOverrides.scala (transition insertion). The same contract holds:
any unsupported node stays on CPU with transitions inserted around it, and
`spark.rapids.sql.explain=NOT_ON_GPU|ALL` reports per-node reasons.
"""
from __future__ import annotations

from typing import List, Optional

from ..config import RapidsConf
from ..config import ALLOW_INCOMPAT as _ALLOW_INCOMPAT
from ..expr.expressions import (_decimal_exact,
                                Alias, BinaryExpr, CaseWhen, CastExpr,
                                Coalesce, ColumnRef, Expression, IsNull,
                                Literal, Round, StringPredicate, Substring,
                                UnaryExpr)
from ..types import DType, TypeId, TypeSig, as_decimal
from . import logical as L
from . import physical as P

# ---------------------------------------------------------------------------
# supported-type signatures per exec / expression family (grows every round)
# ---------------------------------------------------------------------------

_BASIC = TypeSig.all_basic()
# sort keys: fixed-width via order-preserving u64 transforms; strings via
# chunked big-endian LSD passes; decimal128 via (lo, hi) word passes
_FIXED_KEYS = TypeSig({
    TypeId.BOOL, TypeId.INT8, TypeId.INT16, TypeId.INT32, TypeId.INT64,
    TypeId.FLOAT32, TypeId.FLOAT64, TypeId.DECIMAL64, TypeId.DECIMAL128,
    TypeId.DATE32, TypeId.TIMESTAMP, TypeId.STRING,
})
# group-by / join keys go through murmur3 row hash + KeyCol row equality,
# which handle strings too (hash.hip murmur3_str + keys.h byte compare)
_HASH_KEYS = TypeSig({
    TypeId.BOOL, TypeId.INT8, TypeId.INT16, TypeId.INT32, TypeId.INT64,
    TypeId.FLOAT32, TypeId.FLOAT64, TypeId.DECIMAL64, TypeId.DECIMAL128,
    TypeId.DATE32, TypeId.TIMESTAMP, TypeId.STRING,
})
_D128_BINARY_OK = {"add", "sub", "min", "max", "eq", "ne", "lt", "le",
                   "gt", "ge"}
_NUMERIC = TypeSig.numeric()

_GPU_BINARY_OPS = {
    "add", "sub", "mul", "div", "int_div", "mod", "pmod", "pow",
    "eq", "ne", "lt", "le", "gt", "ge", "eq_null_safe",
    "and", "or", "bitand", "bitor", "bitxor", "shiftleft", "shiftright",
    "min", "max", "concat",
}
_GPU_UNARY_OPS = {
    "neg", "abs", "not", "sqrt", "exp", "log", "floor", "ceil",
    "sin", "cos", "tan", "is_nan", "year", "month", "day",
    "trim", "ltrim", "rtrim",
}
# string ops with GPU kernels (strings.hip); eq_null_safe still CPU-only
_GPU_STRING_OK = {"eq", "ne", "lt", "le", "gt", "ge", "concat"}
_GPU_STRING_UNARY = {"length", "upper", "lower", "trim", "ltrim",
                     "rtrim", "initcap", "reverse"}


class TagReason:
    def __init__(self, node: str, reasons: List[str]):
        self.node = node
        self.reasons = reasons


class Tagger:
    def __init__(self, conf: RapidsConf):
        self.conf = conf
        self.notes: List[TagReason] = []

    # ---- expressions ---------------------------------------------------
    def expr_reasons(self, e: Expression, schema) -> List[str]:
        out: List[str] = []
        self._tag_expr(e, schema, out)
        return out

    def _tag_expr(self, e: Expression, schema, out: List[str]):
        if isinstance(e, (ColumnRef, Literal)):
            pass
        elif isinstance(e, Alias):
            pass
        elif isinstance(e, CastExpr):
            src = e.child.dtype(schema)
            if src.id is TypeId.STRING:
                # exact device parser (k_str_to_dec) covers integral,
                # decimal and bool-free numeric targets; float targets
                # parse as f64 (csv_parse)
                if not (e.to.is_integral or e.to.is_decimal
                        or e.to.is_floating):
                    out.append(f"cast string -> {e.to} not on GPU yet")
            elif e.to.id is TypeId.STRING:
                # ints and decimals format on device (i64_to_str /
                # k_dec_to_str); float->string needs Java shortest-
                # round-trip parity, bool/date literal forms are host-side
                if not (src.is_integral or src.is_decimal) \
                        or src.id is TypeId.BOOL:
                    out.append(f"cast {src} -> string not on GPU yet")
            if not self.conf.expr_enabled("Cast"):
                out.append("expression Cast disabled by conf")
        elif isinstance(e, BinaryExpr):
            lt, rt = e.left.dtype(schema), e.right.dtype(schema)
            if e.op in ("mul", "div") and _decimal_exact(lt, rt):
                pass  # exact kernels cover dec64 and dec128 operands
            else:
                in_t = e._in_dtype(schema)
                if e.op not in _GPU_BINARY_OPS:
                    out.append(f"binary op {e.op} has no GPU kernel")
                elif in_t.id is TypeId.STRING and e.op not in _GPU_STRING_OK:
                    out.append(f"binary op {e.op} on string not on GPU yet")
                elif in_t.id is TypeId.DECIMAL128 \
                        and e.op not in _D128_BINARY_OK:
                    out.append(f"binary op {e.op} on decimal128 "
                               "not on GPU yet")
            if not self.conf.expr_enabled(e.op):
                out.append(f"expression {e.op} disabled by conf")
        elif isinstance(e, UnaryExpr):
            in_t = e.child.dtype(schema)
            if in_t.id is TypeId.STRING:
                if e.op not in _GPU_STRING_UNARY:
                    out.append(f"unary op {e.op} on string not on GPU yet")
                elif e.op in ("upper", "lower", "initcap") and \
                        not self.conf.get(_ALLOW_INCOMPAT):
                    out.append(f"{e.op} on GPU is ASCII-only "
                               "(spark.rapids.sql.incompatibleOps.enabled)")
            elif e.op not in _GPU_UNARY_OPS:
                out.append(f"unary op {e.op} has no GPU kernel")
        elif isinstance(e, (IsNull, CaseWhen, Coalesce, Round)):
            pass
        elif isinstance(e, StringPredicate):
            if e.op == "rlike":
                from ..ops.regex_compiler import RegexUnsupported, compile_regex

                try:
                    compile_regex(e.pattern)
                except RegexUnsupported as ex:
                    out.append(f"regex not supported on GPU: {ex}")
        elif type(e).__name__ == "ConcatWs":
            pass
        elif type(e).__name__ == "GetJsonObject":
            keys = [k for k in e.path[1:].lstrip(".").split(".") if k]
            if len(keys) != 1:
                out.append("nested json paths run on CPU")
        elif type(e).__name__ == "CpuBridge":
            return  # bridged subtree runs on host by design
        elif type(e).__name__ == "StrSplit":
            if any(c in e.delimiter for c in ".\\+*?()[]{}|^$"):
                out.append("regex split delimiters run on CPU")
        elif type(e).__name__ == "ArrayContains":
            et = e.child.dtype(schema).children[0]
            if et.is_nested:
                out.append("array_contains over nested elements on CPU")
        elif type(e).__name__ in ("ArraySize", "ElementAt"):
            cdt = e.child.dtype(schema)
            if cdt.id is TypeId.MAP and type(e).__name__ == "ElementAt" \
                    and cdt.children[1].is_nested:
                out.append("element_at over map with nested values on CPU")
        elif type(e).__name__ in ("CreateMap", "MapView"):
            mt = e.dtype(schema)
            if any(c.is_nested for c in
                   (mt.children if mt.id is TypeId.MAP
                    else mt.children[0].children)):
                out.append("maps of nested keys/values on CPU")
        elif type(e).__name__ in ("PadExpr", "LocateExpr"):
            pass  # lpad/rpad/locate/instr device kernels (k_str_pad/locate)
        elif type(e).__name__ in ("DateFormat", "ToTimestamp", "TzConvert"):
            pass  # k_date_format / k_ts_parse / k_tz_convert (tzdb table)
        elif type(e).__name__ in ("CreateNamedStruct", "GetStructField"):
            pass  # struct columns: child-wise device columns
        elif type(e).__name__ == "HostStringFn":
            out.append(f"{getattr(e, 'name', type(e).__name__)} runs on "
                       "CPU this round")
        elif type(e).__name__ in ("RegexpExtract", "RegexpReplace",
                                  "RegexpExtractAll"):
            from ..ops.regex_compiler import RegexUnsupported, compile_regex

            try:
                compile_regex(e.pattern)
            except RegexUnsupported as ex:
                out.append(f"regex not supported on GPU: {ex}")
        elif isinstance(e, Substring):
            pass
        elif type(e).__name__ == "SampleHash":
            pass
        else:
            out.append(f"expression {type(e).__name__} not supported on GPU")
        for c in e.children:
            self._tag_expr(c, schema, out)

    # ---- plan nodes ----------------------------------------------------
    def exec_reasons(self, node: L.LogicalPlan) -> List[str]:
        reasons: List[str] = []
        name = type(node).__name__
        if not self.conf.exec_enabled(name):
            reasons.append(f"exec {name} disabled by conf")
        cs = node.children[0].schema() if node.children else None
        if isinstance(node, L.Scan):
            for f in node.schema().fields:
                r = _BASIC.supports(f.dtype)
                if r:
                    reasons.append(f"column {f.name}: {r}")
        elif isinstance(node, L.Filter):
            reasons += self.expr_reasons(node.condition, cs)
        elif isinstance(node, L.Project):
            for e in node.exprs:
                reasons += self.expr_reasons(e, cs)
        elif isinstance(node, L.Generate):
            if node.outer:
                reasons.append("explode_outer runs on CPU (padding path)")
        elif isinstance(node, L.Expand):
            for proj in node.projections:
                for e in proj:
                    reasons += self.expr_reasons(e, cs)
        elif isinstance(node, L.Aggregate):
            for e in node.group_exprs:
                r = _HASH_KEYS.supports(e.dtype(cs))
                if r:
                    reasons.append(f"group key {e}: {r}")
                reasons += self.expr_reasons(e, cs)
            for a in node.aggs:
                if a.child is not None:
                    t = a.child.dtype(cs)
                    if a.op.startswith("percentile:"):
                        if not (t.is_integral or t.is_floating
                                or t.is_decimal):
                            reasons.append(
                                f"percentile over {t} not supported")
                        reasons += self.expr_reasons(a.child, cs)
                        continue
                    if a.op in ("collect_list", "collect_set"):
                        if t.is_nested:
                            reasons.append(
                                f"{a.op} over {t} not on GPU yet")
                        reasons += self.expr_reasons(a.child, cs)
                        continue
                    r = _NUMERIC.supports(t)
                    if t.id is TypeId.STRING and a.op not in (
                            "count", "count_all", "min", "max"):
                        reasons.append(
                            f"agg {a.op} over strings not on GPU yet")
                    elif r and a.op not in ("count", "count_all", "min",
                                            "max", "first", "last",
                                            "bit_and", "bit_or", "bit_xor"):
                        reasons.append(f"agg {a.op}({a.child}): {r}")
                    if a.op in ("bit_and", "bit_or", "bit_xor") \
                            and not t.is_integral:
                        reasons.append(f"{a.op} needs an integral input")
                    if t.id is TypeId.DECIMAL128 and a.op not in (
                            "sum", "count", "count_all",
                            "collect_list", "collect_set"):
                        reasons.append(
                            f"agg {a.op} over decimal128 not supported yet")
                    reasons += self.expr_reasons(a.child, cs)
        elif isinstance(node, L.Join):
            ls, rs = node.left.schema(), node.right.schema()
            for k in node.left_on:
                r = _HASH_KEYS.supports(ls.field(k).dtype)
                if r:
                    reasons.append(f"join key {k}: {r}")
            for k in node.right_on:
                r = _HASH_KEYS.supports(rs.field(k).dtype)
                if r:
                    reasons.append(f"join key {k}: {r}")
            if node.how not in ("inner", "left", "semi", "anti", "full"):
                reasons.append(f"join type {node.how} not on GPU")
            if node.condition is not None:
                from ..column import Schema as _Schema

                pair = _Schema(list(ls.fields) + list(rs.fields))
                reasons += self.expr_reasons(node.condition, pair)
        elif isinstance(node, L.MapBatches):
            reasons.append("python map_batches runs on CPU (UDF bridge)")
        elif isinstance(node, L.Window):
            spec = node.window_exprs[0].spec
            for k in spec.partition_by + spec.order_by:
                r = _FIXED_KEYS.supports(cs.field(k).dtype)
                if r:
                    reasons.append(f"window key {k}: {r}")
            for w in node.window_exprs:
                op = w.func.op
                if op in ("row_number", "rank", "dense_rank", "ntile"):
                    continue
                if w.spec.range_between is not None:
                    okt = cs.field(w.spec.order_by[0]).dtype
                    if op not in ("sum", "count", "mean", "min", "max"):
                        reasons.append(
                            f"range-frame {op} window not on GPU yet")
                    elif not okt.is_numeric or okt.id is TypeId.DECIMAL128:
                        reasons.append(
                            f"range frame over {okt} order key on CPU")
                vt = w.func.child.dtype(cs) if w.func.child is not None else None
                if op == "nth_value":
                    if vt is not None and vt.is_nested:
                        reasons.append(f"nth_value over {vt} not on GPU")
                    continue
                if op in ("lag", "lead"):
                    if vt is not None and vt.is_nested:
                        reasons.append(f"lag/lead over {vt} not on GPU")
                    if w.func.default is not None and vt is not None \
                            and vt.id is TypeId.STRING:
                        reasons.append("lag/lead string default not on GPU")
                    continue
                if op in ("sum", "count", "mean"):
                    if vt is not None and (not vt.is_numeric
                                           or vt.id is TypeId.DECIMAL128):
                        reasons.append(f"window {op}({vt}) not on GPU")
                    continue
                if op in ("min", "max"):
                    if vt is not None and (not vt.is_numeric
                                           or vt.id is TypeId.DECIMAL128):
                        reasons.append(f"window {op}({vt}) not on GPU")
                    continue
                reasons.append(f"window function {op} not on GPU")
        elif isinstance(node, L.Sort):
            for k in node.keys:
                r = _FIXED_KEYS.supports(node.schema().field(k).dtype)
                if r:
                    reasons.append(f"sort key {k}: {r}")
        return reasons


def plan_physical(node: L.LogicalPlan, conf: RapidsConf,
                  tagger: Optional[Tagger] = None) -> P.PhysicalExec:
    """Convert logical -> physical with GPU placement + transitions."""
    top = tagger is None
    if tagger is None:
        tagger = Tagger(conf)
        from ..config import PRUNE_COLUMNS, PUSH_FILTERS
        if conf.get(PUSH_FILTERS):
            from .optimizer import push_filters

            node = push_filters(node)
        if conf.get(PRUNE_COLUMNS):
            from .optimizer import prune_columns

            node = prune_columns(node)
    gpu_wanted = conf.sql_enabled and _gpu_available()
    if top and gpu_wanted:
        from ..config import OPTIMIZER_ENABLED, OPTIMIZER_EXPLAIN

        if conf.get(OPTIMIZER_ENABLED):
            from .costing import evaluate

            keep, note = evaluate(node)
            if conf.get(OPTIMIZER_EXPLAIN):
                print(f"!CBO {note} -> "
                      f"{'GPU' if keep else 'CPU'}")
            if not keep:
                tagger.notes.append(TagReason("plan", [note]))
                gpu_wanted = False
    exec_ = _convert(node, conf, tagger, gpu_wanted)
    if top:
        exec_ = _ensure_device(exec_, "cpu")  # results surface on host
        explain = conf.explain
        if explain in ("NOT_ON_GPU", "ALL"):
            for note in tagger.notes:
                for r in note.reasons:
                    print(f"!Exec {note.node} cannot run on GPU because {r}")
        if conf.test_enabled and gpu_wanted:
            bad = [n for n in tagger.notes if n.reasons]
            if bad:
                msgs = "; ".join(f"{n.node}: {n.reasons[0]}" for n in bad)
                raise AssertionError(
                    f"spark.rapids.sql.test.enabled: ops fell back to CPU: {msgs}")
        _attach_tags(exec_, tagger)
    return exec_


def _attach_tags(exec_: P.PhysicalExec, tagger: Tagger):
    exec_.tag_notes = tagger.notes  # type: ignore[attr-defined]


def _gpu_available() -> bool:
    import torch

    return torch.cuda.is_available()


def _ensure_device(exec_: P.PhysicalExec, device: str) -> P.PhysicalExec:
    if exec_.device == device:
        return exec_
    return P.DeviceTransferExec(exec_, device)


def _convert(node: L.LogicalPlan, conf: RapidsConf, tagger: Tagger,
             gpu_wanted: bool) -> P.PhysicalExec:
    reasons = tagger.exec_reasons(node) if gpu_wanted else []
    on_gpu = gpu_wanted and not reasons
    if gpu_wanted and reasons:
        tagger.notes.append(TagReason(node.name(), reasons))
    device = "cuda" if on_gpu else "cpu"

    if isinstance(node, L.Scan):
        scan = P.ScanExec(device, node.schema(), node.source, node.label)
        if on_gpu:
            from ..config import BATCH_SIZE_BYTES

            return P.CoalesceBatchesExec(scan, conf.get(BATCH_SIZE_BYTES))
        return scan

    if isinstance(node, L.CacheData):
        inner = _convert(node.child, conf, tagger, gpu_wanted)
        return P.CacheDataExec(node, _ensure_device(inner, "cpu"))

    if isinstance(node, L.Cached):
        # one physical exec per logical node, however many parents
        # reference it (hierarchical rollup levels share their base)
        memo = getattr(node, "_phys_memo", None)
        if memo is None:
            inner = _convert(node.child, conf, tagger, gpu_wanted)
            memo = P.CachedExec(inner.device, inner, node.schema())
            node._phys_memo = memo
        return memo

    kids = [_ensure_device(_convert(c, conf, tagger, gpu_wanted), device)
            for c in node.children]

    if isinstance(node, L.Filter):
        return P.FilterExec(device, node.condition, kids[0])
    if isinstance(node, L.Project):
        from ..config import CPU_BRIDGE

        if not on_gpu and gpu_wanted and conf.get(CPU_BRIDGE) \
                and conf.exec_enabled("Project"):
            from ..expr.expressions import CpuBridge

            cs2 = node.child.schema()
            wrapped = []
            any_clean = False
            for e in node.exprs:
                if tagger.expr_reasons(e, cs2):
                    wrapped.append(CpuBridge(e))
                else:
                    wrapped.append(e)
                    any_clean = True
            if any_clean:
                inner = _ensure_device(
                    _convert(node.children[0], conf, tagger, gpu_wanted),
                    "cuda")
                return P.ProjectExec("cuda", wrapped, inner, node.schema())
        return P.ProjectExec(device, node.exprs, kids[0], node.schema())
    if isinstance(node, L.Expand):
        return P.ExpandExec(device, node.projections, kids[0], node.schema())
    if isinstance(node, L.Generate):
        return P.GenerateExec(device, node.column, kids[0], node.schema(),
                              node.outer, node.pos)
    if isinstance(node, L.Aggregate):
        from ..config import BATCH_SIZE_BYTES

        return P.HashAggregateExec(device, node.group_exprs, node.aggs,
                                   kids[0], node.schema(),
                                   input_replicated=L.is_replicated(node.child),
                                   merge_target_bytes=conf.get(BATCH_SIZE_BYTES))
    if isinstance(node, L.NestedLoopJoin):
        return P.NestedLoopJoinExec(
            device, kids[0], kids[1], [], [], node.how, node.schema(),
            right_replicated=L.is_replicated(node.right),
            condition=node.condition)
    if isinstance(node, L.Join):
        from ..config import BROADCAST_THRESHOLD, JOIN_SUBPARTITION_BYTES

        return P.HashJoinExec(device, kids[0], kids[1], node.left_on,
                              node.right_on, node.how, node.schema(),
                              right_replicated=L.is_replicated(node.right),
                              broadcast_threshold=conf.get(BROADCAST_THRESHOLD),
                              sub_partition_bytes=conf.get(
                                  JOIN_SUBPARTITION_BYTES),
                              using=node.using, condition=node.condition)
    if isinstance(node, L.MapBatches):
        return P.MapBatchesExec(node.fn, _ensure_device(kids[0], "cpu"),
                                node.schema())
    if isinstance(node, L.CrossJoin):
        return P.CrossJoinExec(device, kids[0], kids[1], node.schema(),
                               right_replicated=L.is_replicated(node.right))
    if isinstance(node, L.Window):
        from .window_exec import WindowExec

        return WindowExec(device, node.window_exprs, kids[0], node.schema())
    if isinstance(node, L.Sort):
        from ..config import BATCH_SIZE_BYTES

        return P.SortExec(device, node.keys, node.descending,
                          node.nulls_last, kids[0],
                          target_bytes=conf.get(BATCH_SIZE_BYTES),
                          input_replicated=L.is_replicated(node.child))
    if isinstance(node, L.Limit):
        child = node.children[0]
        if isinstance(child, L.Sort) and node.n <= 10_000_000:
            # fuse ORDER BY + LIMIT: sort only per-batch heads
            inner = _ensure_device(
                _convert(child.children[0], conf, tagger, gpu_wanted),
                kids[0].device)
            return P.TopNExec(kids[0].device, child.keys, child.descending,
                              child.nulls_last, node.n, inner)
        return P.LimitExec(device, node.n, kids[0])
    if isinstance(node, L.Union):
        return P.UnionExec(device, kids, node.schema())
    raise NotImplementedError(f"plan node {type(node).__name__}")

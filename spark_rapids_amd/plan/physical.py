"""Physical operators (execs).

Reference analogue: the Gpu*Exec operators of the reference's L2 layer
(GpuFilterExec/GpuProjectExec in basicPhysicalOperators.scala,
GpuHashAggregateExec in GpuAggregateExec.scala, GpuShuffledHashJoinExec,
GpuSortExec...). One implementation runs on either device: the expression and
op layers dispatch to the HIP kernel library when batches live on the GPU and
to the CPU reference backend otherwise. Execs are pull-based iterators over
ColumnBatch and are instantiated by the overrides pass with an explicit
device placement; DeviceTransferExec is the row-free transition analogue of
GpuRowToColumnarExec/GpuColumnarToRowExec (columnar both sides here).
"""
from __future__ import annotations

from typing import Iterator, List, Optional, Sequence, Tuple

from .. import ops
from ..column import Column, ColumnBatch, Field, Schema
from ..expr.aggregates import AggExpr
from ..expr.expressions import ColumnRef, Expression
from ..types import BOOL, DType, FLOAT64, INT64
from ..memory.retry import with_retry_split


# bumped by the API layer at every root execution: CachedExec instances
# recompute when the epoch moves, replay within the same execution
EXECUTION_EPOCH = 0


def new_execution():
    global EXECUTION_EPOCH
    EXECUTION_EPOCH += 1


class PhysicalExec:
    def __init__(self, device: str, schema: Schema,
                 children: Sequence["PhysicalExec"] = ()):  # noqa: D401
        self.device = device
        self.schema = schema
        self.children = list(children)
        self.metrics = {}

    @property
    def gpu(self) -> bool:
        return self.device == "cuda"

    def execute(self) -> Iterator[ColumnBatch]:
        raise NotImplementedError

    def name(self) -> str:
        prefix = "Gpu" if self.gpu else "Cpu"
        return prefix + type(self).__name__.replace("Exec", "")

    def describe(self) -> str:
        return self.name()

    def tree_string(self, indent: int = 0) -> str:
        lines = ["  " * indent + "+- " + self.describe()]
        for c in self.children:
            lines.append(c.tree_string(indent + 1))
        return "\n".join(lines)


def prefetched(it: Iterator, depth: int = 2) -> Iterator:
    """Run an iterator on a background thread with a bounded queue.

    Measured on MI355X: a NET REGRESSION for the power-run shape (GIL
    contention outweighs sync overlap), so operators do not use it by
    default; kept for IO-bound scans where the producer blocks in native
    reads (parquet prefetch pool uses the same pattern internally)."""
    import queue
    import threading

    q: "queue.Queue" = queue.Queue(maxsize=depth)
    _END = object()

    def producer():
        try:
            for item in it:
                q.put(item)
            q.put(_END)
        except BaseException as e:  # noqa: BLE001 - forwarded to consumer
            q.put(e)

    t = threading.Thread(target=producer, daemon=True)
    t.start()
    while True:
        item = q.get()
        if item is _END:
            return
        if isinstance(item, BaseException):
            raise item
        yield item


class ScanExec(PhysicalExec):
    def __init__(self, device: str, schema: Schema, source, label: str):
        super().__init__(device, schema)
        self.source = source
        self.label = label

    def execute(self) -> Iterator[ColumnBatch]:
        for batch in self.source.partitions():
            yield batch.to(self.device)

    def describe(self):
        return f"{self.name()}({self.label})"


class DeviceTransferExec(PhysicalExec):
    """Columnar host<->device transition (the engine's RowToColumnar /
    ColumnarToRow analogue — both sides columnar on MI355X)."""

    def __init__(self, child: PhysicalExec, to_device: str):
        super().__init__(to_device, child.schema, [child])

    def execute(self) -> Iterator[ColumnBatch]:
        for batch in self.children[0].execute():
            yield batch.to(self.device)

    def describe(self):
        return f"DeviceTransfer(to={self.device})"


class CoalesceBatchesExec(PhysicalExec):
    """Concatenate small batches up to the spark.rapids.sql.batchSizeBytes
    goal (reference: GpuCoalesceBatches / TargetSize goal,
    GpuCoalesceBatches.scala). 288 GB of HBM3E favors big batches: fewer,
    larger kernels and fewer allocation-sizing syncs."""

    def __init__(self, child: PhysicalExec, target_bytes: int):
        super().__init__(child.device, child.schema, [child])
        self.target_bytes = target_bytes

    def execute(self) -> Iterator[ColumnBatch]:
        pending: List[ColumnBatch] = []
        pending_bytes = 0
        for batch in self.children[0].execute():
            if batch.num_rows == 0:
                continue
            nb = batch.nbytes
            if pending and pending_bytes + nb > self.target_bytes:
                yield self._flush(pending)
                pending, pending_bytes = [], 0
            pending.append(batch)
            pending_bytes += nb
        if pending:
            yield self._flush(pending)

    def _flush(self, pending: List[ColumnBatch]) -> ColumnBatch:
        if len(pending) == 1:
            return pending[0]
        return ops.concat_batches(pending)

    def describe(self):
        return f"{self.name()}(target={self.target_bytes})"


class FilterExec(PhysicalExec):
    def __init__(self, device: str, condition: Expression, child: PhysicalExec):
        super().__init__(device, child.schema, [child])
        self.condition = condition

    def execute(self) -> Iterator[ColumnBatch]:
        for batch in self.children[0].execute():
            def task(b):
                mask = self.condition.eval(b, self.schema)
                return ops.apply_boolean_mask(b, mask)
            out = with_retry_split(task, batch)
            for ob in out:
                if ob.num_rows:
                    yield ob

    def describe(self):
        return f"{self.name()}({self.condition})"


class ProjectExec(PhysicalExec):
    def __init__(self, device: str, exprs: List[Expression], child: PhysicalExec,
                 schema: Schema):
        super().__init__(device, schema, [child])
        self.exprs = exprs

    def execute(self) -> Iterator[ColumnBatch]:
        in_schema = self.children[0].schema
        for batch in self.children[0].execute():
            cols = [e.eval(batch, in_schema) for e in self.exprs]
            yield ColumnBatch(cols, batch.num_rows)

    def describe(self):
        return f"{self.name()}[{', '.join(str(e) for e in self.exprs)}]"


def _lower_aggs(aggs: List[AggExpr], in_schema: Schema):
    """Lower logical aggregates to kernel (op, col, dtype) triples in
    partial/merge form. Returns (value_exprs, partial_specs, merge_ops,
    finalizers). mean becomes (sum, count) partials merged by sum."""
    value_exprs: List[Expression] = []
    partial: List[Tuple[str, int, DType]] = []
    merge: List[str] = []
    final: List[Tuple] = []  # ("col", j) | ("div", jnum, jden) | ...
    for a in aggs:
        if a.op == "count_all":
            j = len(partial)
            partial.append(("count_all", -1, INT64))
            merge.append("sum")
            final.append(("col", j))
        elif a.op == "count":
            value_exprs.append(a.child)
            j = len(partial)
            partial.append(("count", len(value_exprs) - 1, INT64))
            merge.append("sum")
            final.append(("col", j))
        elif a.op in ("sum", "min", "max", "first", "last",
                      "bit_and", "bit_or", "bit_xor"):
            value_exprs.append(a.child)
            j = len(partial)
            partial.append((a.op, len(value_exprs) - 1, a.out_dtype(in_schema)))
            merge.append("sum" if a.op == "sum" else a.op)
            final.append(("col", j))
        elif a.op == "mean":
            ct = a.child.dtype(in_schema)
            if ct.is_decimal:
                # decimal mean: compute over the scaled double value
                from ..expr.expressions import CastExpr

                value_exprs.append(CastExpr(a.child, FLOAT64))
            else:
                value_exprs.append(a.child)
            v = len(value_exprs) - 1
            js = len(partial)
            sum_t = FLOAT64 if (ct.is_floating or ct.is_decimal) else INT64
            partial.append(("sum", v, sum_t))
            partial.append(("count", v, INT64))
            merge.extend(["sum", "sum"])
            final.append(("div", js, js + 1, ct))
        elif a.op in ("collect_list", "collect_set") \
                or a.op.startswith("percentile:") \
                or a.op.startswith("hll:"):
            value_exprs.append(a.child)
            j = len(partial)
            partial.append((a.op, len(value_exprs) - 1,
                            a.out_dtype(in_schema)))
            merge.append("__single_pass__")  # lists are never merged:
            # HashAggregateExec routes collect aggs to the one-shot path
            final.append(("col", j))
        elif a.op in ("stddev", "variance"):
            from ..expr.expressions import BinaryExpr, CastExpr
            from ..types import FLOAT64 as F64

            ce = CastExpr(a.child, F64)
            value_exprs.append(ce)
            value_exprs.append(BinaryExpr("mul", ce, ce))
            v = len(value_exprs) - 2
            js = len(partial)
            partial.append(("sum", v, FLOAT64))
            partial.append(("sum", v + 1, FLOAT64))
            partial.append(("count", v, INT64))
            merge.extend(["sum", "sum", "sum"])
            final.append(("var", js, js + 1, js + 2, a.op == "stddev"))
        else:
            raise NotImplementedError(f"agg {a.op}")
    return value_exprs, partial, merge, final


class HashAggregateExec(PhysicalExec):
    """Partial-per-batch hash aggregation then a merge pass, mirroring the
    reference's GpuAggFirstPassIterator + GpuMergeAggregateIterator structure
    (GpuAggregateExec.scala:1942,896)."""

    def __init__(self, device: str, group_exprs: List[Expression],
                 aggs: List[AggExpr], child: PhysicalExec, schema: Schema,
                 input_replicated: bool = False,
                 merge_target_bytes: int = 2 << 30):
        super().__init__(device, schema, [child])
        self.group_exprs = group_exprs
        self.aggs = aggs
        self.input_replicated = input_replicated
        self.merge_target_bytes = merge_target_bytes

    def execute(self) -> Iterator[ColumnBatch]:
        # filter-into-aggregate fusion: when the child is a GPU filter,
        # evaluate the predicate to a selection vector and aggregate the
        # selected rows directly — no materialized gather of the filtered
        # batch (reference analogue: the pre-projection + filter fusion the
        # tiered project/AST path gives cudf).
        source = self.children[0]
        fused_condition = None
        if self.gpu and isinstance(source, FilterExec) and source.gpu:
            fused_condition = source.condition
            source = source.children[0]
        in_schema = source.schema
        nkeys = len(self.group_exprs)
        value_exprs, partial, merge_ops, final = _lower_aggs(self.aggs, in_schema)
        if any(op in ("collect_list", "collect_set")
               or op.startswith("percentile:")
               or op.startswith("hll:") for op, _, _ in partial):
            yield from self._execute_single_pass(
                source, fused_condition, in_schema, value_exprs, partial,
                final)
            return

        from ..memory.spill import SpillableBatch

        partial_handles: List[SpillableBatch] = []
        for batch in source.execute():
            def task(b):
                sel = None
                if fused_condition is not None:
                    from ..ops import gpu_backend as _gb

                    mask = fused_condition.eval(b, in_schema)
                    sel = _gb.mask_to_sel(mask, b.num_rows)
                key_cols = [e.eval(b, in_schema) for e in self.group_exprs]
                val_cols = [e.eval(b, in_schema) for e in value_exprs]
                pre = ColumnBatch(key_cols + val_cols, b.num_rows)
                specs = [(op, (nkeys + v) if v >= 0 else -1, dt)
                         for op, v, dt in partial]
                if sel is not None:
                    from ..ops import gpu_backend as _gb

                    return _gb.group_by_aggregate(pre, list(range(nkeys)),
                                                  specs, sel=sel)
                return ops.group_by_aggregate(pre, list(range(nkeys)), specs)
            # partial results are spillable between input batches so memory
            # pressure can evict them (reference: partial aggs held as
            # SpillableColumnarBatch between batches)
            partial_handles.append(
                SpillableBatch(with_retry_split_single(task, batch)))

        if not partial_handles:
            from ..shuffle import dist as _d0

            keyed_local = nkeys and not (_d0.ctx().is_multi
                                         and not self.input_replicated)
            if keyed_local:
                return
            # empty input still aggregates an empty batch: keyless Spark
            # semantics need the one-row result, and a distributed keyed
            # rank must still JOIN the exchange collectives (an early
            # return here would deadlock the other ranks)
            cols = [Column.from_pylist([], e.dtype(in_schema))
                    for e in list(self.group_exprs) + list(value_exprs)]
            if self.gpu:
                cols = [c.cuda() for c in cols]
            specs0 = [(op, (nkeys + v) if v >= 0 else -1, dt)
                      for op, v, dt in partial]
            partial_handles.append(SpillableBatch(ops.group_by_aggregate(
                ColumnBatch(cols, 0), list(range(nkeys)), specs0)))
        partial_results = [h.get() for h in partial_handles]
        merged_in = ops.concat_batches(partial_results) if len(partial_results) > 1 \
            else partial_results[0]
        for h in partial_handles:
            h.close()

        # distributed merge: keyed -> RCCL all-to-all hash exchange so each
        # rank owns a disjoint key range; keyless -> all-gather partials and
        # merge identically on every rank
        from ..shuffle import dist as _dist
        if _dist.ctx().is_multi and not self.input_replicated:
            from ..shuffle.exchange import exchange_by_hash, gather_all
            if nkeys:
                received = exchange_by_hash(merged_in, list(range(nkeys)))
            else:
                received = gather_all(merged_in)
            received = [b for b in received if b.num_rows]
            if not received:
                return
            merged_in = ops.concat_batches(received) if len(received) > 1 \
                else received[0]
        merge_specs = [(op, nkeys + j, partial[j][2])
                       for j, op in enumerate(merge_ops)]

        # repartition-based fallback (reference: GpuMergeAggregateIterator's
        # recursive bucket split, GpuAggregateExec.scala:205-306): when the
        # merged partials exceed the batch-size goal, hash-split them into
        # buckets and merge bucket-by-bucket so no single merge pass needs
        # the whole key space in memory.
        buckets = [merged_in]
        if nkeys and merged_in.nbytes > self._merge_target_bytes():
            nb = max(2, (merged_in.nbytes +
                         self._merge_target_bytes() - 1) //
                     self._merge_target_bytes())
            parted, offs = ops.hash_partition(merged_in, list(range(nkeys)),
                                              int(nb))
            buckets = [_slice_rows(parted, offs[i], offs[i + 1])
                       for i in range(int(nb))]

        cs = in_schema
        for bucket in buckets:
            if bucket.num_rows == 0:
                continue
            merged = ops.group_by_aggregate(bucket, list(range(nkeys)),
                                            merge_specs)
            yield self._final_project(merged, nkeys, final, cs)

    def _final_project(self, merged: ColumnBatch, nkeys: int, final,
                       cs: Schema) -> ColumnBatch:
        out_cols: List[Column] = [merged.columns[i] for i in range(nkeys)]
        for spec, agg in zip(final, self.aggs):
            if spec[0] == "col":
                c = merged.columns[nkeys + spec[1]]
                out_cols.append(ops.cast(c, agg.out_dtype(cs)))
            elif spec[0] == "div":
                s = ops.cast(merged.columns[nkeys + spec[1]], FLOAT64)
                c = ops.cast(merged.columns[nkeys + spec[2]], FLOAT64)
                out_cols.append(ops.binary_op("div", s, c, FLOAT64))
            elif spec[0] == "var":
                # sample variance from (sum, sumsq, count):
                # (sumsq - sum^2/n) / (n-1); NULL when n < 2
                sm = ops.cast(merged.columns[nkeys + spec[1]], FLOAT64)
                sq = ops.cast(merged.columns[nkeys + spec[2]], FLOAT64)
                cn = ops.cast(merged.columns[nkeys + spec[3]], FLOAT64)
                mean_sq = ops.binary_op(
                    "div", ops.binary_op("mul", sm, sm, FLOAT64), cn,
                    FLOAT64)
                num = ops.binary_op("sub", sq, mean_sq, FLOAT64)
                den = ops.binary_op_scalar("sub", cn, 1.0, FLOAT64)
                var = ops.binary_op("div", num, den, FLOAT64)
                if spec[4]:
                    var = ops.unary_op("sqrt", var, FLOAT64)
                out_cols.append(var)
        return ColumnBatch(out_cols, merged.num_rows)

    def _execute_single_pass(self, source, fused_condition, in_schema,
                             value_exprs, partial, final):
        """One-shot aggregation for list-building aggs (collect_list/set):
        rows are exchanged by key hash FIRST (distributed), then aggregated
        once — list partials cannot be merged."""
        nkeys = len(self.group_exprs)
        pres: List[ColumnBatch] = []
        for batch in source.execute():
            if fused_condition is not None:
                mask = fused_condition.eval(batch, in_schema)
                batch = ops.apply_boolean_mask(batch, mask)
            key_cols = [e.eval(batch, in_schema) for e in self.group_exprs]
            val_cols = [e.eval(batch, in_schema) for e in value_exprs]
            pres.append(ColumnBatch(key_cols + val_cols, batch.num_rows))
        from ..shuffle import dist as _dist
        dist_on = _dist.ctx().is_multi and not self.input_replicated
        if not pres:
            if not dist_on:
                return
            pres = []  # still must join the collectives below
        pre = ops.concat_batches(pres) if len(pres) > 1 else (
            pres[0] if pres else None)
        if dist_on:
            from ..shuffle.exchange import exchange_by_hash, gather_all
            if pre is None:
                cs0 = self.children[0].schema
                cols = [Column.from_pylist([], e.dtype(cs0))
                        for e in list(self.group_exprs) + list(value_exprs)]
                if self.gpu:
                    cols = [c.cuda() for c in cols]
                pre = ColumnBatch(cols, 0)
            if nkeys:
                received = exchange_by_hash(pre, list(range(nkeys)))
            else:
                received = gather_all(pre)
            received = [b for b in received if b is not None and b.num_rows]
            if not received:
                return
            pre = ops.concat_batches(received) if len(received) > 1                 else received[0]
        if (pre is None or pre.num_rows == 0) and nkeys:
            return
        if pre is None:
            cols = [Column.from_pylist([], e.dtype(in_schema))
                    for e in list(self.group_exprs) + list(value_exprs)]
            if self.gpu:
                cols = [c.cuda() for c in cols]
            pre = ColumnBatch(cols, 0)
        specs = [(op, (nkeys + v) if v >= 0 else -1, dt)
                 for op, v, dt in partial]
        merged = ops.group_by_aggregate(pre, list(range(nkeys)), specs)
        yield self._final_project(merged, nkeys, final, in_schema)

    def _merge_target_bytes(self) -> int:
        return self.merge_target_bytes

    def describe(self):
        keys = ", ".join(str(e) for e in self.group_exprs)
        aggs = ", ".join(str(a) for a in self.aggs)
        return f"{self.name()}(keys=[{keys}], aggs=[{aggs}])"


class HashJoinExec(PhysicalExec):
    """Build the right side once, stream the left side through gather-map
    probes (reference: GpuShuffledHashJoinExec / GpuHashJoin.scala)."""

    def __init__(self, device: str, left: PhysicalExec, right: PhysicalExec,
                 left_on: List[str], right_on: List[str], how: str,
                 schema: Schema, right_replicated: bool = True,
                 broadcast_threshold: int = 512 << 20,
                 sub_partition_bytes: int = 1 << 30,
                 using: bool = False, condition=None):
        super().__init__(device, schema, [left, right])
        self.left_on = left_on
        self.right_on = right_on
        self.how = how
        self.right_replicated = right_replicated
        self.broadcast_threshold = broadcast_threshold
        self.sub_partition_bytes = sub_partition_bytes
        self.using = using
        # extra non-equi predicate over (left ++ right) columns, applied to
        # the candidate pairs of the equi probe (reference analogue: the
        # AST-compiled join condition of ConditionalHashJoinIterator /
        # mixed joins, GpuHashJoin.scala:1653 — there the compiled AST
        # runs during the probe; here the condition is evaluated
        # vectorized over the gathered pair batch and the gather maps are
        # compacted, an equivalent device-side dataflow)
        self.condition = condition
        self._strategy = "local"

    def _local_or_empty(self, batches: List[ColumnBatch], schema: Schema):
        if batches:
            return ops.concat_batches(batches) if len(batches) > 1 \
                else batches[0]
        empty = ColumnBatch(
            [Column.from_pylist([], f.dtype) for f in schema.fields], 0)
        return empty.cuda() if self.gpu else empty

    def execute(self) -> Iterator[ColumnBatch]:
        left, right = self.children
        rbatches = list(right.execute())
        lbatches_override = None
        # distributed strategies for a sharded build side (every rank must
        # take the same branch: the decision uses the all-gathered global
        # size). Small build -> broadcast (all-gather); large build ->
        # hash-exchange BOTH sides so each rank joins one key range
        # (reference analogues: GpuBroadcastHashJoinExec vs
        # GpuShuffledHashJoinExec).
        from ..shuffle import dist as _dist
        if _dist.ctx().is_multi and not self.right_replicated:
            import torch as _torch
            import torch.distributed as _td

            from ..shuffle.exchange import exchange_by_hash, gather_all

            local = self._local_or_empty(rbatches, right.schema)
            dev = "cuda" if self.gpu else "cpu"
            sz = _torch.tensor([local.nbytes], dtype=_torch.int64, device=dev)
            _td.all_reduce(sz)
            total = int(sz.item())
            lkidx = [left.schema.index(k) for k in self.left_on]
            rkidx = [right.schema.index(k) for k in self.right_on]
            same_key_types = all(
                left.schema.fields[a].dtype.id == right.schema.fields[b].dtype.id
                for a, b in zip(lkidx, rkidx))
            if total <= self.broadcast_threshold or not same_key_types:
                self._strategy = "broadcast"
                rbatches = [b for b in gather_all(local) if b.num_rows]
            else:
                self._strategy = "shuffled"
                rbatches = [b for b in exchange_by_hash(local, rkidx)
                            if b.num_rows]
                lall = self._local_or_empty(list(left.execute()), left.schema)
                lbatches_override = [b for b in exchange_by_hash(lall, lkidx)
                                     if b.num_rows]
        if not rbatches:
            if self.how in ("inner", "semi"):
                return
            rtable = None
        else:
            rtable = ops.concat_batches(rbatches) if len(rbatches) > 1 else rbatches[0]
        lkidx = [left.schema.index(k) for k in self.left_on]
        rkidx = [right.schema.index(k) for k in self.right_on]
        right_matched = None
        if self.how == "full" and rtable is not None and rtable.num_rows:
            import numpy as _np
            import torch as _torch

            right_matched = (_torch.zeros(rtable.num_rows,
                                          dtype=_torch.uint8, device="cuda")
                             if self.gpu else
                             _np.zeros(rtable.num_rows, dtype=bool))
        lsource = lbatches_override if lbatches_override is not None \
            else left.execute()
        same_types = all(
            left.schema.fields[a].dtype.id == right.schema.fields[b].dtype.id
            for a, b in zip(lkidx, rkidx))
        if rtable is not None and same_types \
                and rtable.nbytes > self.sub_partition_bytes:
            yield from self._execute_subpartitioned(
                rtable, lsource, lkidx, rkidx, left.schema)
            return
        for lbatch in lsource:
            if lbatch.num_rows == 0:
                continue
            if rtable is None or rtable.num_rows == 0:
                if self.how in ("left", "anti", "full"):
                    if self.how == "anti":
                        yield lbatch
                    else:
                        yield self._left_with_null_right(lbatch)
                continue
            if self.condition is not None:
                yield from self._emit_conditional(lbatch, rtable, lkidx,
                                                  rkidx, right_matched)
                continue
            lmap, rmap = ops.join_gather_maps(lbatch, rtable, lkidx, rkidx,
                                              self.how, right_matched)
            if self.how in ("semi", "anti"):
                out = ops.gather(lbatch, lmap, negatives=False)
                if out.num_rows:
                    yield out
                continue
            lout = ops.gather(lbatch, lmap, negatives=False)
            rout = ops.gather(rtable, rmap,
                              negatives=self.how in ("left", "full"))
            if lout.num_rows:
                yield ColumnBatch(lout.columns +
                                  self._right_out(rout), lout.num_rows)
        if self.how == "full" and rtable is not None and rtable.num_rows:
            extra = self._unmatched_right(rtable, right_matched, left.schema)
            if extra is not None and extra.num_rows:
                yield extra

    def _execute_subpartitioned(self, rtable, lsource, lkidx, rkidx,
                                left_schema) -> Iterator[ColumnBatch]:
        """Sub-partitioned join (GpuSubPartitionHashJoin analogue): both
        sides are hash-split on the join keys and joined bucket-by-bucket;
        a key lands in exactly one bucket, so per-bucket inner/left/semi/
        anti/full results concatenate to the exact join. Left pieces are
        spill-registered between buckets."""
        from ..memory.spill import SpillableBatch

        nb = int(min(64, max(2, -(-rtable.nbytes //
                                  self.sub_partition_bytes))))
        rparted, roffs = ops.hash_partition(rtable, rkidx, nb)
        rbuckets = [_slice_rows(rparted, roffs[i], roffs[i + 1])
                    for i in range(nb)]
        lpieces: List[List] = [[] for _ in range(nb)]
        for lbatch in lsource:
            if lbatch.num_rows == 0:
                continue
            parted, offs = ops.hash_partition(lbatch, lkidx, nb)
            for i in range(nb):
                piece = _slice_rows(parted, offs[i], offs[i + 1])
                if piece.num_rows:
                    lpieces[i].append(SpillableBatch(piece))
        for i in range(nb):
            rb = rbuckets[i]
            parts = [h.get() for h in lpieces[i]]
            for h in lpieces[i]:
                h.close()
            lb = ops.concat_batches(parts) if len(parts) > 1 else (
                parts[0] if parts else None)
            if lb is None or lb.num_rows == 0:
                if self.how == "full" and rb.num_rows:
                    extra = self._unmatched_right(
                        rb, self._fresh_matched(rb), left_schema)
                    if extra is not None and extra.num_rows:
                        yield extra
                continue
            if rb.num_rows == 0:
                if self.how == "anti":
                    yield lb
                elif self.how in ("left", "full"):
                    yield self._left_with_null_right(lb)
                continue
            right_matched = self._fresh_matched(rb) \
                if self.how == "full" else None
            if self.condition is not None:
                yield from self._emit_conditional(lb, rb, lkidx, rkidx,
                                                  right_matched)
            else:
                lmap, rmap = ops.join_gather_maps(lb, rb, lkidx, rkidx,
                                                  self.how, right_matched)
                if self.how in ("semi", "anti"):
                    out = ops.gather(lb, lmap, negatives=False)
                    if out.num_rows:
                        yield out
                else:
                    lout = ops.gather(lb, lmap, negatives=False)
                    rout = ops.gather(
                        rb, rmap, negatives=self.how in ("left", "full"))
                    if lout.num_rows:
                        yield ColumnBatch(
                            lout.columns + self._right_out(rout),
                            lout.num_rows)
            if self.how == "full":
                extra = self._unmatched_right(rb, right_matched, left_schema)
                if extra is not None and extra.num_rows:
                    yield extra

    def _pair_schema(self) -> Schema:
        """Namespace the join condition resolves against: left fields
        followed by ALL right fields (duplicate names resolve left-first;
        rename before joining for unambiguous references, as in Spark)."""
        ls, rs = self.children[0].schema, self.children[1].schema
        return Schema(list(ls.fields) + list(rs.fields))

    def _conditional_pairs(self, lb: ColumnBatch, rb: ColumnBatch,
                           lkidx, rkidx):
        """Equi-probe pair maps compacted by the join condition."""
        lmap, rmap = ops.join_gather_maps(lb, rb, lkidx, rkidx, "inner",
                                          None)
        if lmap.size == 0:
            return lmap, rmap
        lout = ops.gather(lb, lmap, negatives=False)
        rout = ops.gather(rb, rmap, negatives=False)
        pair = ColumnBatch(list(lout.columns) + list(rout.columns),
                           lout.num_rows)
        mask = self.condition.eval(pair, self._pair_schema())
        maps = ops.apply_boolean_mask(
            ColumnBatch([lmap, rmap], lmap.size), mask)
        return maps.columns[0], maps.columns[1]

    def _left_hit_mask(self, n: int, lmap_f: Column,
                       invert: bool = False) -> Column:
        import torch as _torch

        dev = "cuda" if self.gpu else "cpu"
        hit = _torch.zeros(n, dtype=_torch.bool, device=dev)
        if lmap_f.size:
            hit[lmap_f.data[:lmap_f.size].long()] = True
        if invert:
            hit = ~hit
        return Column(DType.bool_(), n, hit.to(_torch.uint8), None,
                      null_count=0)

    def _emit_conditional(self, lb: ColumnBatch, rb: ColumnBatch,
                          lkidx, rkidx, right_matched
                          ) -> Iterator[ColumnBatch]:
        """Reconstruct each join type from condition-filtered pairs."""
        lmap_f, rmap_f = self._conditional_pairs(lb, rb, lkidx, rkidx)
        how = self.how
        if how in ("semi", "anti"):
            mask = self._left_hit_mask(lb.num_rows, lmap_f,
                                       invert=(how == "anti"))
            out = ops.apply_boolean_mask(lb, mask)
            if out.num_rows:
                yield out
            return
        if how == "full" and right_matched is not None and rmap_f.size:
            import torch as _torch

            idx = rmap_f.data[:rmap_f.size].long()
            if isinstance(right_matched, _torch.Tensor):
                right_matched[idx] = 1
            else:
                right_matched[idx.cpu().numpy()] = True
        if lmap_f.size:
            lout = ops.gather(lb, lmap_f, negatives=False)
            rout = ops.gather(rb, rmap_f, negatives=False)
            yield ColumnBatch(list(lout.columns) + self._right_out(rout),
                              lout.num_rows)
        if how in ("left", "full"):
            miss = self._left_hit_mask(lb.num_rows, lmap_f, invert=True)
            lrest = ops.apply_boolean_mask(lb, miss)
            if lrest.num_rows:
                yield self._left_with_null_right(lrest)

    def _fresh_matched(self, rb: ColumnBatch):
        import numpy as _np
        import torch as _torch

        return (_torch.zeros(rb.num_rows, dtype=_torch.uint8,
                             device="cuda") if self.gpu
                else _np.zeros(rb.num_rows, dtype=bool))

    def _unmatched_right(self, rtable: ColumnBatch, right_matched,
                         left_schema: Schema):
        """Full outer: rows of the build side no probe row matched, with
        null left columns. In distributed broadcast/replicated mode every
        rank holds the same build table, so only rank 0 emits them;
        shuffled mode is co-partitioned and emits locally."""
        from ..shuffle import dist as _dist

        c = _dist.ctx()
        if c.is_multi and self._strategy != "shuffled" and c.rank != 0:
            return None
        if self.gpu:
            import torch as _torch

            from ..ops import gpu_backend as _gb

            # unmatched = rows where the matched flag is 0
            notm = _torch.empty(rtable.num_rows, dtype=_torch.uint8,
                                device="cuda")
            _gb.ext.unary(_gb._UN_OPS["not"], 0, right_matched.data_ptr(), 0,
                          notm.data_ptr(), 0, rtable.num_rows, _gb._stream())
            mask_col = Column(DType.bool_(), rtable.num_rows, notm, None,
                              null_count=0)
            runm = ops.apply_boolean_mask(rtable, mask_col)
        else:
            import numpy as _np

            idx = _np.nonzero(~right_matched)[0].astype(_np.int32)
            runm = ops.gather(rtable, Column.from_numpy(idx))
        if runm.num_rows == 0:
            return None
        cols = [Column.nulls(f.dtype, runm.num_rows,
                             "cuda" if self.gpu else "cpu")
                for f in left_schema.fields]
        return ColumnBatch(cols + runm.columns, runm.num_rows)

    def _right_out(self, rout: ColumnBatch) -> List[Column]:
        """Right-side output columns; USING joins drop the duplicate key
        columns (kept on the left)."""
        if not self.using:
            return list(rout.columns)
        rs = self.children[1].schema
        return [c for f, c in zip(rs.fields, rout.columns)
                if f.name not in self.right_on]

    def _left_with_null_right(self, lbatch: ColumnBatch) -> ColumnBatch:
        nsch = self.schema
        nleft = len(lbatch.columns)
        cols = list(lbatch.columns)
        for f in nsch.fields[nleft:]:
            cols.append(Column.nulls(f.dtype, lbatch.num_rows, lbatch.device))
        return ColumnBatch(cols, lbatch.num_rows)

    def describe(self):
        pairs = ", ".join(f"{l}={r}" for l, r in zip(self.left_on, self.right_on))
        return f"{self.name()}({self.how}, {pairs})"


class CacheDataExec(PhysicalExec):
    """Serve a CacheData node: first execution materializes the child
    into compressed parquet blobs held on the logical node (so the store
    survives across actions on the same DataFrame); later executions
    decode the blobs (PCBS analogue). Always a host-side exec — the
    planner inserts a transfer when the parent runs on GPU."""

    def __init__(self, node, child: PhysicalExec):
        super().__init__("cpu", child.schema, [child])
        self.node = node

    def execute(self) -> Iterator[ColumnBatch]:
        from ..io.parquet import (batch_to_parquet_bytes,
                                  parquet_bytes_to_batch)

        if self.node.store is None:
            blobs = []
            for b in self.children[0].execute():
                if b.num_rows:
                    blobs.append(batch_to_parquet_bytes(b.cpu(),
                                                        self.schema))
            self.node.store = blobs
        for blob in self.node.store:
            yield parquet_bytes_to_batch(blob)


class NestedLoopJoinExec(HashJoinExec):
    """Broadcast nested-loop join: all (left, right) row pairs streamed
    through the join condition in bounded chunks (reference analogue:
    GpuBroadcastNestedLoopJoinExec — compiled-AST condition over cross
    blocks). Supports inner/left/semi/anti/full; the build side is
    all-gathered when sharded (nested loops cannot co-partition)."""

    PAIR_CHUNK = 1 << 22  # max candidate pairs materialized at once

    def _cross_maps(self, m: int, nr: int):
        import numpy as _np

        total = m * nr
        if self.gpu:
            import torch as _torch

            from ..ops import gpu_backend as _gb

            iota = _torch.empty(total, dtype=_torch.int32, device="cuda")
            _gb.ext.iota_i32(iota.data_ptr(), total, _gb._stream())
            icol = Column(DType.int32(), total, iota, None, null_count=0)
            lmap = _gb.binary_op_scalar("int_div", icol, nr, DType.int32())
            rmap = _gb.binary_op_scalar("mod", icol, nr, DType.int32())
        else:
            lmap = Column.from_numpy(
                _np.repeat(_np.arange(m, dtype=_np.int32), nr))
            rmap = Column.from_numpy(
                _np.tile(_np.arange(nr, dtype=_np.int32), m))
        return lmap, rmap

    def execute(self) -> Iterator[ColumnBatch]:
        import torch as _torch

        left, right = self.children
        rbatches = list(right.execute())
        from ..shuffle import dist as _dist
        if _dist.ctx().is_multi and not self.right_replicated:
            from ..shuffle.exchange import gather_all

            self._strategy = "broadcast"
            local = self._local_or_empty(rbatches, right.schema)
            rbatches = [b for b in gather_all(local) if b.num_rows]
        rtable = None
        if rbatches:
            rtable = ops.concat_batches(rbatches) if len(rbatches) > 1 \
                else rbatches[0]
        nr = rtable.num_rows if rtable is not None else 0
        right_matched = self._fresh_matched(rtable) \
            if self.how == "full" and nr else None
        dev = "cuda" if self.gpu else "cpu"
        for lbatch in left.execute():
            nl = lbatch.num_rows
            if nl == 0:
                continue
            if nr == 0:
                if self.how == "anti":
                    yield lbatch
                elif self.how in ("left", "full"):
                    yield self._left_with_null_right(lbatch)
                continue
            rows_per = max(1, self.PAIR_CHUNK // nr)
            hit = _torch.zeros(nl, dtype=_torch.bool, device=dev)
            for s0 in range(0, nl, rows_per):
                e0 = min(nl, s0 + rows_per)
                chunk = _slice_rows(lbatch, s0, e0)
                lmap, rmap = self._cross_maps(e0 - s0, nr)
                lout = ops.gather(chunk, lmap, negatives=False)
                rout = ops.gather(rtable, rmap, negatives=False)
                pair = ColumnBatch(list(lout.columns) + list(rout.columns),
                                   lout.num_rows)
                mask = self.condition.eval(pair, self._pair_schema())
                fmaps = ops.apply_boolean_mask(
                    ColumnBatch([lmap, rmap], lmap.size), mask)
                lmap_f, rmap_f = fmaps.columns
                if lmap_f.size:
                    hit[s0:e0][lmap_f.data[:lmap_f.size].long()] = True
                    if right_matched is not None:
                        idx = rmap_f.data[:rmap_f.size].long()
                        if isinstance(right_matched, _torch.Tensor):
                            right_matched[idx] = 1
                        else:
                            right_matched[idx.cpu().numpy()] = True
                    if self.how in ("inner", "left", "full"):
                        out = ops.apply_boolean_mask(pair, mask)
                        lcols = out.columns[:len(lbatch.columns)]
                        rcols = ColumnBatch(
                            out.columns[len(lbatch.columns):], out.num_rows)
                        yield ColumnBatch(
                            list(lcols) + self._right_out(rcols),
                            out.num_rows)
            if self.how in ("semi", "anti", "left", "full"):
                want = hit if self.how == "semi" else ~hit
                mcol = Column(DType.bool_(), nl, want.to(_torch.uint8),
                              None, null_count=0)
                rest = ops.apply_boolean_mask(lbatch, mcol)
                if rest.num_rows:
                    if self.how in ("left", "full"):
                        yield self._left_with_null_right(rest)
                    else:
                        yield rest
        if self.how == "full" and right_matched is not None:
            extra = self._unmatched_right(rtable, right_matched, left.schema)
            if extra is not None and extra.num_rows:
                yield extra


class CrossJoinExec(PhysicalExec):
    """Cartesian product; output bounded by maxOutputRows to keep an
    accidental unfiltered cross join from exploding memory. Gather maps are
    computed with the iota + int-div/mod kernels on GPU."""

    MAX_OUTPUT_ROWS = 1 << 28

    def __init__(self, device: str, left: PhysicalExec, right: PhysicalExec,
                 schema: Schema, right_replicated: bool = True):
        super().__init__(device, schema, [left, right])
        self.right_replicated = right_replicated

    def execute(self) -> Iterator[ColumnBatch]:
        import numpy as np

        left, right = self.children
        rbatches = list(right.execute())
        from ..shuffle import dist as _dist
        if _dist.ctx().is_multi and not self.right_replicated:
            from ..shuffle.exchange import gather_all

            local = rbatches[0] if len(rbatches) == 1 else (
                ops.concat_batches(rbatches) if rbatches else None)
            if local is None:
                local = ColumnBatch([Column.from_pylist([], f.dtype)
                                     for f in right.schema.fields], 0)
                if self.gpu:
                    local = local.cuda()
            rbatches = [b for b in gather_all(local) if b.num_rows]
        if not rbatches:
            return
        rtable = ops.concat_batches(rbatches) if len(rbatches) > 1             else rbatches[0]
        nr = rtable.num_rows
        for lbatch in left.execute():
            nl = lbatch.num_rows
            if nl == 0 or nr == 0:
                continue
            total = nl * nr
            if total > self.MAX_OUTPUT_ROWS:
                raise MemoryError(
                    f"cross join would produce {total} rows "
                    f"(> {self.MAX_OUTPUT_ROWS}); filter the inputs first")
            if self.gpu:
                import torch as _torch

                from ..ops import gpu_backend as _gb

                iota = _torch.empty(total, dtype=_torch.int32, device="cuda")
                _gb.ext.iota_i32(iota.data_ptr(), total, _gb._stream())
                icol = Column(DType.int32(), total, iota, None, null_count=0)
                lmap = _gb.binary_op_scalar("int_div", icol, nr,
                                            DType.int32())
                rmap = _gb.binary_op_scalar("mod", icol, nr, DType.int32())
            else:
                li = np.repeat(np.arange(nl, dtype=np.int32), nr)
                ri = np.tile(np.arange(nr, dtype=np.int32), nl)
                lmap = Column.from_numpy(li)
                rmap = Column.from_numpy(ri)
            lout = ops.gather(lbatch, lmap, negatives=False)
            rout = ops.gather(rtable, rmap, negatives=False)
            yield ColumnBatch(lout.columns + rout.columns, total)


class SortExec(PhysicalExec):
    def __init__(self, device: str, keys: List[str], descending: List[bool],
                 nulls_last: List[bool], child: PhysicalExec,
                 target_bytes: int = 2 << 30,
                 input_replicated: bool = False):
        super().__init__(device, child.schema, [child])
        self.keys = keys
        self.descending = descending
        self.nulls_last = nulls_last
        self.target_bytes = target_bytes
        self.input_replicated = input_replicated

    def execute(self) -> Iterator[ColumnBatch]:
        from ..memory.spill import SpillableBatch
        from ..shuffle import dist as _dist

        if _dist.ctx().is_multi and not self.input_replicated:
            yield from self._execute_distributed()
            return
        handles = [SpillableBatch(b) for b in self.children[0].execute()]
        if not handles:
            return
        total = sum(h.nbytes for h in handles)
        if total > self.target_bytes:
            yield from self._external_sort(handles)
            return
        batches = [h.get() for h in handles]
        table = ops.concat_batches(batches) if len(batches) > 1 else batches[0]
        for h in handles:
            h.close()
        if table.num_rows == 0:
            yield table
            return
        yield self._sort_one(table)

    def _sort_one(self, table: ColumnBatch) -> ColumnBatch:
        kidx = [self.schema.index(k) for k in self.keys]
        order = ops.sort_order(table, kidx, self.descending, self.nulls_last)
        return ops.gather(table, order)

    def _execute_distributed(self) -> Iterator[ColumnBatch]:
        """Distributed global ORDER BY (GpuRangePartitioner analogue):
        sample the monotone int64 sort-key proxy on every rank, agree on
        world-1 range boundaries, exchange rows by range over RCCL, then
        sort locally — rank r then holds the r-th globally ordered range.
        Every rank runs the same collective sequence (sample all-gather +
        one all-to-all) regardless of data."""
        import numpy as np
        import torch

        from ..shuffle import dist as _dist
        from ..shuffle.exchange import exchange_by_ranges

        c = _dist.ctx()
        batches = [b for b in self.children[0].execute() if b.num_rows]
        table = ops.concat_batches(batches) if len(batches) > 1 else (
            batches[0] if batches else None)
        kidx = [self.schema.index(k) for k in self.keys]
        k0, d0, n0 = kidx[0], self.descending[0], self.nulls_last[0]
        # sample the range-key proxy (empty ranks contribute a pad that is
        # filtered by the sample-count header)
        if table is not None and table.num_rows:
            kc = ops.backend_for(*table.columns).range_key(
                table.columns[k0], d0, n0)
            arr = kc.data.cpu().numpy()[:kc.size]
            stride = max(1, len(arr) // 4096)
            sample = np.ascontiguousarray(arr[::stride][:4096],
                                          dtype=np.int64)
        else:
            sample = np.zeros(0, dtype=np.int64)
        import torch.distributed as td

        gathered: List = [None] * c.world
        td.all_gather_object(gathered, sample)
        allsamp = np.concatenate([g for g in gathered if len(g)]) \
            if any(len(g) for g in gathered) else np.zeros(1, np.int64)
        qs = np.quantile(allsamp, [r / c.world for r in range(1, c.world)],
                         method="nearest").astype(np.int64)
        bounds = list(np.maximum.accumulate(qs))  # monotone cut points
        if table is None:
            # participate in the exchange with an empty batch
            cols = [Column.from_pylist([], f.dtype) for f in self.schema.fields]
            if self.gpu:
                cols = [col.cuda() for col in cols]
            table = ColumnBatch(cols, 0)
            kc = ops.backend_for(*table.columns).range_key(
                table.columns[k0], d0, n0)
        received = exchange_by_ranges(table, kc, bounds)
        received = [b for b in received if b.num_rows]
        if not received:
            return
        mine = ops.concat_batches(received) if len(received) > 1 \
            else received[0]
        yield self._sort_one(mine)

    def _external_sort(self, handles) -> Iterator[ColumnBatch]:
        """Out-of-core sort (reference analogue: GpuSortExec's full-sort
        out-of-core path / GpuOutOfCoreSortIterator): range-partition the
        input on a monotone int64 proxy of the primary sort key, spill the
        bucket pieces, then sort each bounded bucket in-core and emit the
        buckets in key order. Equal proxies always share a bucket, so the
        per-bucket in-core sort (all keys) makes the global order exact."""
        import numpy as np

        from ..memory.spill import SpillableBatch

        kidx = [self.schema.index(k) for k in self.keys]
        k0, d0, n0 = kidx[0], self.descending[0], self.nulls_last[0]
        total = sum(h.nbytes for h in handles)
        nbuckets = int(min(64, max(2, -(-total // self.target_bytes))))

        # pass 1: per-chunk range keys + a key sample for the boundaries
        samples = []
        key_handles = []
        for h in handles:
            batch = h.get()
            kc = ops.backend_for(*batch.columns).range_key(
                batch.columns[k0], d0, n0)
            key_handles.append(SpillableBatch(ColumnBatch([kc], kc.size)))
            arr = kc.data.cpu().numpy()[:kc.size]
            if len(arr):
                stride = max(1, len(arr) // 2048)
                samples.append(arr[::stride].copy())
        if not samples:
            return
        sample = np.concatenate(samples)
        qs = np.quantile(sample, [i / nbuckets for i in range(1, nbuckets)],
                         method="nearest").astype(np.int64)
        bounds = sorted(set(int(q) for q in qs))

        # pass 2: split every chunk into bucket pieces and spill them
        pieces: List[List] = [[] for _ in range(len(bounds) + 1)]
        for h, kh in zip(handles, key_handles):
            batch = h.get()
            kc = kh.get().columns[0]
            for j in range(len(bounds) + 1):
                mask = None
                if j > 0:
                    mask = ops.binary_op_scalar("ge", kc, bounds[j - 1],
                                                BOOL)
                if j < len(bounds):
                    m2 = ops.binary_op_scalar("lt", kc, bounds[j], BOOL)
                    mask = m2 if mask is None else                         ops.binary_op("and", mask, m2, BOOL)
                piece = batch if mask is None else                     ops.apply_boolean_mask(batch, mask)
                if piece.num_rows:
                    pieces[j].append(SpillableBatch(piece))
            h.close()
            kh.close()

        # pass 3: in-core sort per bucket, emitted in key order
        for j in range(len(bounds) + 1):
            if not pieces[j]:
                continue
            parts = [p.get() for p in pieces[j]]
            bucket = ops.concat_batches(parts) if len(parts) > 1 else parts[0]
            for p in pieces[j]:
                p.close()
            yield self._sort_one(bucket)

    def describe(self):
        ks = ", ".join(f"{k}{' DESC' if d else ''}"
                       for k, d in zip(self.keys, self.descending))
        return f"{self.name()}[{ks}]"


class MapBatchesExec(PhysicalExec):
    """Host python-function operator (CPU bridge / UDF escape hatch)."""

    def __init__(self, fn, child: PhysicalExec, schema: Schema):
        super().__init__("cpu", schema, [child])
        self.fn = fn

    def execute(self) -> Iterator[ColumnBatch]:
        for batch in self.children[0].execute():
            out = self.fn(batch.cpu())
            if out is not None and out.num_rows >= 0:
                yield out


class TopNExec(PhysicalExec):
    """Fused ORDER BY + LIMIT n (GpuTakeOrderedAndProjectExec analogue):
    each input batch is sorted and truncated to n rows before the final
    merge sort, so the full input is never globally sorted."""

    def __init__(self, device: str, keys: List[str], descending: List[bool],
                 nulls_last: List[bool], n: int, child: PhysicalExec):
        super().__init__(device, child.schema, [child])
        self.keys = keys
        self.descending = descending
        self.nulls_last = nulls_last
        self.n = n

    def _sort_head(self, batch: ColumnBatch) -> ColumnBatch:
        kidx = [self.schema.index(k) for k in self.keys]
        order = ops.sort_order(batch, kidx, self.descending,
                               self.nulls_last)
        taken = ops.gather(batch, order)
        return _slice_rows(taken, 0, min(self.n, taken.num_rows)) \
            if taken.num_rows > self.n else taken

    def execute(self) -> Iterator[ColumnBatch]:
        heads = [self._sort_head(b)
                 for b in self.children[0].execute() if b.num_rows]
        if not heads:
            return
        merged = ops.concat_batches(heads) if len(heads) > 1 else heads[0]
        yield self._sort_head(merged)

    def describe(self):
        ks = ", ".join(f"{k}{' DESC' if d else ''}"
                       for k, d in zip(self.keys, self.descending))
        return f"{self.name()}[top {self.n} by {ks}]"


class LimitExec(PhysicalExec):
    def __init__(self, device: str, n: int, child: PhysicalExec):
        super().__init__(device, child.schema, [child])
        self.n = n

    def execute(self) -> Iterator[ColumnBatch]:
        remaining = self.n
        for batch in self.children[0].execute():
            if remaining <= 0:
                return
            if batch.num_rows <= remaining:
                remaining -= batch.num_rows
                yield batch
            else:
                idx = Column.from_numpy(
                    __import__("numpy").arange(remaining, dtype="int32"),
                    device=batch.device)
                yield ops.gather(batch, idx)
                remaining = 0

    def describe(self):
        return f"{self.name()}({self.n})"


class ExpandExec(PhysicalExec):
    """Evaluates every projection over each input batch and concatenates:
    an N-projection expand emits N output rows per input row."""

    def __init__(self, device: str, projections, child: PhysicalExec,
                 schema: Schema):
        super().__init__(device, schema, [child])
        self.projections = projections

    def execute(self) -> Iterator[ColumnBatch]:
        in_schema = self.children[0].schema
        for batch in self.children[0].execute():
            parts = []
            for proj in self.projections:
                cols = [e.eval(batch, in_schema) for e in proj]
                parts.append(ColumnBatch(cols, batch.num_rows))
            yield parts[0] if len(parts) == 1 else ops.concat_batches(parts)

    def describe(self):
        return f"{self.name()}[{len(self.projections)} projections]"


class GenerateExec(PhysicalExec):
    """explode/posexplode: parent columns are gathered by a rowid map built
    from the list offsets (k_expand_rows on GPU); the element column is the
    list child passed through (its layout is already the flattened rows)."""

    def __init__(self, device: str, column: str, child: PhysicalExec,
                 schema: Schema, outer: bool, pos: bool):
        super().__init__(device, schema, [child])
        self.column = column
        self.outer = outer
        self.pos = pos

    def execute(self) -> Iterator[ColumnBatch]:
        cs = self.children[0].schema
        ci = cs.index(self.column)
        for batch in self.children[0].execute():
            yield self._one(batch, ci)

    def _one(self, batch: ColumnBatch, ci: int) -> ColumnBatch:
        import numpy as np

        lc = batch.columns[ci]
        others = [c for j, c in enumerate(batch.columns) if j != ci]
        n = batch.num_rows
        if self.gpu and not self.outer:
            import torch

            from ..ops import gpu_backend as gb
            from ..ops.gpu_backend import ext

            s = gb._stream()
            total = int(lc.offsets[n].item()) if n else 0
            rowid = torch.empty(max(total, 1), dtype=torch.int32,
                                device="cuda")[:total]
            pos_t = torch.empty(max(total, 1), dtype=torch.int32,
                                device="cuda")[:total]
            if n and total:
                ext.expand_rows(lc.offsets.data_ptr(), rowid.data_ptr(),
                                pos_t.data_ptr(), n, s)
            parent = ops.gather(
                ColumnBatch(others, n),
                Column(DType.int32(), total, rowid, None, null_count=0)) \
                if others else ColumnBatch([], total)
            out = list(parent.columns)
            if self.pos:
                out.append(Column(DType.int32(), total, pos_t, None,
                                  null_count=0))
            elem = lc.child
            out.append(Column(elem.dtype, total, elem.data, elem.validity,
                              elem.offsets, elem._null_count, elem.child))
            return ColumnBatch(out, total)
        # CPU (and outer) path
        host = batch.cpu()
        lch = host.columns[ci]
        offs = lch.offsets.numpy()
        lvalid = lch.valid_array()
        rowids: List[int] = []
        poss: List[int] = []
        elem_take: List[int] = []  # -1 = null element (outer padding)
        for i in range(n):
            cnt = int(offs[i + 1] - offs[i]) if lvalid[i] else 0
            if cnt == 0:
                if self.outer:
                    rowids.append(i)
                    poss.append(0)
                    elem_take.append(-1)
                continue
            for k in range(cnt):
                rowids.append(i)
                poss.append(k)
                elem_take.append(int(offs[i]) + k)
        idxc = Column.from_numpy(np.array(rowids, dtype=np.int32))
        parent = ops.gather(ColumnBatch(
            [c for j, c in enumerate(host.columns) if j != ci], n), idxc) \
            if others else ColumnBatch([], len(rowids))
        out = list(parent.columns)
        if self.pos:
            out.append(Column.from_numpy(np.array(poss, dtype=np.int32)))
        ev = lch.child.to_pylist()
        elems = [ev[t] if t >= 0 else None for t in elem_take]
        out.append(Column.from_pylist(elems, lch.dtype.children[0]))
        ob = ColumnBatch(out, len(rowids))
        return ob.cuda() if self.gpu else ob

    def describe(self):
        mode = "posexplode" if self.pos else "explode"
        o = "_outer" if self.outer else ""
        return f"{self.name()}[{mode}{o}({self.column})]"


class CachedExec(PhysicalExec):
    """Materializes its child once per root execution and replays the
    batches to every consumer (hierarchical rollup levels share the
    finest-level aggregate; reference analogue: ReusedExchangeExec).
    The epoch check makes repeated executions of a cached physical plan
    (benchmark steps) recompute instead of serving stale results."""

    def __init__(self, device: str, child: PhysicalExec, schema: Schema):
        super().__init__(device, schema, [child])
        self._epoch = -1
        self._batches = None

    def execute(self) -> Iterator[ColumnBatch]:
        if self._epoch != EXECUTION_EPOCH or self._batches is None:
            self._batches = list(self.children[0].execute())
            self._epoch = EXECUTION_EPOCH
        yield from self._batches


class UnionExec(PhysicalExec):
    def __init__(self, device: str, children: List[PhysicalExec], schema: Schema):
        super().__init__(device, schema, children)

    def execute(self) -> Iterator[ColumnBatch]:
        for c in self.children:
            yield from c.execute()


def _slice_rows(batch: ColumnBatch, start: int, end: int) -> ColumnBatch:
    import numpy as np

    idx = Column.from_numpy(np.arange(start, end, dtype=np.int32),
                            device=batch.device)
    return ops.gather(batch, idx)


def with_retry_split_single(task, batch):
    out = with_retry_split(task, batch)
    if len(out) == 1:
        return out[0]
    return ops.concat_batches(out)

"""Window physical operator.

CPU implementation via segment-vectorized numpy/pandas over the sorted
batch; the GPU path currently falls back per the overrides tagging (GPU
segmented-scan kernels are the next round's work). Output rows are in
(partition, order) sorted order, like Spark's WindowExec.
"""
from __future__ import annotations

from typing import Iterator, List

import numpy as np

from .. import ops
from ..column import Column, ColumnBatch, Schema
from ..expr.windows import WindowExpr
from ..ops import cpu_backend
from .physical import PhysicalExec


class WindowExec(PhysicalExec):
    def __init__(self, device: str, window_exprs: List[WindowExpr],
                 child: PhysicalExec, schema: Schema):
        super().__init__(device, schema, [child])
        self.window_exprs = window_exprs
        self.spec = window_exprs[0].spec
        for w in window_exprs[1:]:
            assert w.spec.partition_by == self.spec.partition_by
            assert w.spec.order_by == self.spec.order_by

    def execute(self) -> Iterator[ColumnBatch]:
        child = self.children[0]
        batches = [b.cpu() for b in child.execute()]
        if not batches:
            return
        table = ops.concat_batches(batches) if len(batches) > 1 else batches[0]
        n = table.num_rows
        cs = child.schema
        spec = self.spec
        key_idx = [cs.index(k) for k in spec.partition_by]
        order_idx = [cs.index(k) for k in spec.order_by]
        sort_idx = key_idx + order_idx
        desc = [False] * len(key_idx) + list(spec.descending)
        nl = [d for d in desc]
        if sort_idx and n:
            order = cpu_backend.sort_order(table, sort_idx, desc, nl)
            table = cpu_backend.gather(table, order)

        heads = _change_flags(table, key_idx, n)
        ochange = heads | _change_flags(table, order_idx, n) if order_idx \
            else heads
        idx = np.arange(n, dtype=np.int64)
        seg_start = idx.copy()
        if n:
            head_pos = np.flatnonzero(heads)
            seg_id = np.cumsum(heads) - 1
            seg_start = head_pos[seg_id]

        out_cols = list(table.columns)
        for w in self.window_exprs:
            out_cols.append(_compute(w, table, cs, n, heads, ochange, idx,
                                     seg_start))
        yield ColumnBatch(out_cols, n)

    def describe(self):
        names = ", ".join(w.output_name() for w in self.window_exprs)
        return f"{self.name()}[{names}]"


def _change_flags(table: ColumnBatch, key_idx: List[int], n: int) -> np.ndarray:
    if n == 0:
        return np.zeros(0, dtype=bool)
    flags = np.zeros(n, dtype=bool)
    flags[0] = True
    for ci in key_idx:
        c = table.columns[ci]
        a = np.array(c.to_pylist(), dtype=object)
        neq = np.array([a[i] != a[i - 1] for i in range(1, n)], dtype=bool) \
            if n > 1 else np.zeros(0, dtype=bool)
        flags[1:] |= neq
    return flags


def _compute(w: WindowExpr, table: ColumnBatch, cs, n, heads, ochange, idx,
             seg_start) -> Column:
    from ..ops.cpu_backend import _make, _vals, _valid

    op = w.func.op
    out_dt = w.out_dtype(cs)
    if n == 0:
        return Column.from_pylist([], out_dt)
    if op == "row_number":
        return _make((idx - seg_start + 1).astype(np.int32), None, out_dt)
    if op == "rank":
        last_change = np.maximum.accumulate(np.where(ochange, idx, -1))
        return _make((last_change - seg_start + 1).astype(np.int32), None, out_dt)
    if op == "dense_rank":
        dr = np.cumsum(ochange)
        return _make((dr - dr[seg_start] + 1).astype(np.int32), None, out_dt)

    vc = w.func.child.eval(table, cs)
    v = _vals(vc).astype(np.float64) if vc.dtype.is_numeric else _vals(vc)
    valid = _valid(vc)
    if op in ("lag", "lead"):
        k = w.func.offset if op == "lag" else -w.func.offset
        src = idx - k
        seg_end = _segment_ends(heads, idx, n)
        ok = (src >= seg_start) & (src <= seg_end)
        srcc = np.where(ok, src, 0)
        raw = _vals(vc)
        outv = np.where(ok, raw[srcc], w.func.default
                        if w.func.default is not None else 0)
        outvalid = np.where(ok, valid[srcc],
                            w.func.default is not None)
        return _make(outv.astype(out_dt.numpy_dtype()) if out_dt.is_fixed_width
                     else outv, outvalid if not outvalid.all() else None, out_dt)

    vv = np.where(valid, v, 0.0)
    running = len(w.spec.order_by) > 0
    cnt_f = valid.astype(np.int64)
    if running:
        cum = np.cumsum(vv)
        run_sum = cum - cum[seg_start] + vv[seg_start]
        ccount = np.cumsum(cnt_f)
        run_cnt = ccount - ccount[seg_start] + cnt_f[seg_start]
        if op == "count":
            return _make(run_cnt, None, out_dt)
        if op == "sum":
            res, ok = run_sum, run_cnt > 0
        elif op == "mean":
            res = run_sum / np.maximum(run_cnt, 1)
            ok = run_cnt > 0
        else:  # running min/max via pandas segment cumulation
            import pandas as pd

            seg_id = np.cumsum(heads) - 1
            s = pd.Series(np.where(valid, v, np.nan))
            g = s.groupby(seg_id)
            res = (g.cummin() if op == "min" else g.cummax())
            # null rows take the running value so far (Spark ignores nulls)
            res = res.groupby(seg_id).ffill().to_numpy()
            ok = ~np.isnan(res)
            res = np.nan_to_num(res)
    else:
        nseg = int(heads.sum())
        seg_id = np.cumsum(heads) - 1
        sums = np.zeros(nseg)
        np.add.at(sums, seg_id, vv)
        cnts = np.zeros(nseg, dtype=np.int64)
        np.add.at(cnts, seg_id, cnt_f)
        if op == "count":
            return _make(cnts[seg_id], None, out_dt)
        if op == "sum":
            res, ok = sums[seg_id], cnts[seg_id] > 0
        elif op == "mean":
            res = sums[seg_id] / np.maximum(cnts[seg_id], 1)
            ok = cnts[seg_id] > 0
        else:
            init = np.inf if op == "min" else -np.inf
            m = np.full(nseg, init)
            ufunc = np.minimum if op == "min" else np.maximum
            ufunc.at(m, seg_id, np.where(valid, v, init))
            res, ok = m[seg_id], cnts[seg_id] > 0
    res = res.astype(out_dt.numpy_dtype())
    return _make(res, ok if not ok.all() else None, out_dt)


def _segment_ends(heads, idx, n):
    nxt = np.empty(n, dtype=np.int64)
    ends = np.flatnonzero(np.append(heads[1:], True))
    seg_id = np.cumsum(heads) - 1
    return ends[seg_id]

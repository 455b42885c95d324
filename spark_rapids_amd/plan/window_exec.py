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
        if self.gpu:
            batches = [b.cuda() for b in child.execute()]
            if not batches:
                return
            table = ops.concat_batches(batches) if len(batches) > 1 \
                else batches[0]
            yield self._execute_gpu(table, child.schema)
            return
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

    # ------------------------------------------------------------------
    # GPU path: segmented computation over the sorted table built from the
    # radix sort + scan + groupby primitives (window.hip adds change-flags,
    # iota and f64 scans). See GPU_WINDOW_FUNCS in overrides for the set
    # that places here.
    # ------------------------------------------------------------------
    def _execute_gpu(self, table: ColumnBatch, cs) -> ColumnBatch:
        import torch

        from ..ops import gpu_backend as gb
        from ..ops.gpu_backend import ext
        from ..types import BOOL, FLOAT64, INT32, INT64, DType

        spec = self.spec
        n = table.num_rows
        s = gb._stream()
        key_idx = [cs.index(k) for k in spec.partition_by]
        order_idx = [cs.index(k) for k in spec.order_by]
        sort_idx = key_idx + order_idx
        desc = [False] * len(key_idx) + list(spec.descending)
        nl = [d for d in desc]
        if sort_idx and n:
            order = gb.sort_order(table, sort_idx, desc, nl)
            table = gb.gather(table, order)
        if n == 0:
            cols = list(table.columns)
            for w in self.window_exprs:
                cols.append(gb._empty_col(w.out_dtype(cs)))
            return ColumnBatch(cols, 0)

        def _u8(t):
            return Column(DType.bool_(), n, t, None, null_count=0)

        def _i32col(t):
            return Column(INT32, n, t, None, null_count=0)

        # segment heads / order-change flags
        heads = torch.empty(n, dtype=torch.uint8, device="cuda")
        if key_idx:
            kd = gb._key_desc([table.columns[i] for i in key_idx])
            ext.change_flags(kd.data_ptr(), len(key_idx), heads.data_ptr(),
                             n, s)
        else:
            heads.zero_()
            heads[0] = 1
        heads_i64 = torch.empty(n, dtype=torch.int64, device="cuda")
        ext.cast(0, 4, heads.data_ptr(), heads_i64.data_ptr(), n, s)
        excl_heads, nseg = gb._exclusive_scan_i64(heads_i64)
        # seg_id = inclusive_scan(heads) - 1 = excl + head - 1
        segid_col = gb.binary_op(
            "add", Column(INT64, n, excl_heads, None, null_count=0),
            Column(INT64, n, heads_i64, None, null_count=0), INT64)
        segid_col = gb.binary_op_scalar("sub", segid_col, 1, INT64)
        seg_id = gb.cast(segid_col, INT32)
        # head positions + per-row segment start
        head_pos = self._indices_of(heads, n)
        seg_start = self._gather_i32(head_pos, seg_id.data, n)
        iota = torch.empty(n, dtype=torch.int32, device="cuda")
        ext.iota_i32(iota.data_ptr(), n, s)
        iota_col = _i32col(iota)
        seg_start_col = _i32col(seg_start)

        out_cols = list(table.columns)
        for w in self.window_exprs:
            out_cols.append(self._compute_gpu_one(
                w, table, cs, n, heads, seg_id, seg_start_col, iota_col,
                head_pos, int(nseg)))
        return ColumnBatch(out_cols, n)

    def _indices_of(self, flags_u8: torch.Tensor, n: int):
        """positions (i32) of set flags, via the compaction kernels."""
        import torch

        from ..ops import gpu_backend as gb
        from ..ops.gpu_backend import ext

        s = gb._stream()
        nb = ext.sel_num_blocks(n)
        counts = torch.empty(nb, dtype=torch.int64, device="cuda")
        ext.mask_count(flags_u8.data_ptr(), 0, counts.data_ptr(), n, s)
        offsets, total = gb._exclusive_scan_i64(counts)
        idx = torch.empty(max(total, 1), dtype=torch.int32,
                          device="cuda")[:total]
        if total:
            ext.mask_scatter(flags_u8.data_ptr(), 0, offsets.data_ptr(),
                             idx.data_ptr(), n, s)
        return idx

    def _gather_i32(self, src: torch.Tensor, idx: torch.Tensor, n_out: int):
        import torch

        from ..ops import gpu_backend as gb
        from ..ops.gpu_backend import ext

        out = torch.empty(n_out, dtype=src.dtype, device="cuda")
        ext.gather_fixed(src.element_size(), src.data_ptr(), idx.data_ptr(),
                         out.data_ptr(), n_out, gb._stream())
        return out

    def _compute_gpu_one(self, w, table, cs, n, heads, seg_id, seg_start_col,
                         iota_col, head_pos, nseg):
        import torch

        from ..column import mask_nbytes
        from ..ops import gpu_backend as gb
        from ..ops.gpu_backend import ext
        from ..types import FLOAT64, INT32, INT64, DType

        s = gb._stream()
        op = w.func.op
        out_dt = w.out_dtype(cs)

        def _i32col(t):
            return Column(INT32, n, t, None, null_count=0)

        if op == "row_number":
            rn = gb.binary_op("sub", iota_col, seg_start_col, INT32)
            return gb.binary_op_scalar("add", rn, 1, INT32)
        if op in ("rank", "dense_rank"):
            all_keys = [table.columns[cs.index(k)] for k in
                        self.spec.partition_by + self.spec.order_by]
            och = torch.empty(n, dtype=torch.uint8, device="cuda")
            kd = gb._key_desc(all_keys)
            ext.change_flags(kd.data_ptr(), len(all_keys), och.data_ptr(), n, s)
            och64 = torch.empty(n, dtype=torch.int64, device="cuda")
            ext.cast(0, 4, och.data_ptr(), och64.data_ptr(), n, s)
            excl, _ = gb._exclusive_scan_i64(och64)
            if op == "rank":
                run_pos = self._indices_of(och, n)
                runid = gb.binary_op(
                    "add", Column(INT64, n, excl, None, null_count=0),
                    Column(INT64, n, och64, None, null_count=0), INT64)
                runid = gb.cast(gb.binary_op_scalar("sub", runid, 1, INT64),
                                INT32)
                run_start = _i32col(self._gather_i32(run_pos, runid.data, n))
                r = gb.binary_op("sub", run_start, seg_start_col, INT32)
                return gb.binary_op_scalar("add", r, 1, INT32)
            # dense_rank: inclusive run count - run count at segment start + 1
            runid_incl = gb.binary_op(
                "add", Column(INT64, n, excl, None, null_count=0),
                Column(INT64, n, och64, None, null_count=0), INT64)
            ri32 = gb.cast(runid_incl, INT32)
            at_start = _i32col(self._gather_i32(ri32.data, seg_start_col.data, n))
            dr = gb.binary_op("sub", ri32, at_start, INT32)
            return gb.binary_op_scalar("add", dr, 1, INT32)

        if op == "ntile":
            nt = w.func.offset
            hp_ext = torch.cat([head_pos, torch.tensor(
                [n], dtype=torch.int32, device="cuda")])
            segid_next = gb.binary_op_scalar("add", gb.cast(
                Column(INT32, n, seg_id.data, None, null_count=0), INT32),
                1, INT32)
            nxt_head = _i32col(self._gather_i32(hp_ext, segid_next.data, n))
            # next segment head minus this segment start IS the size
            size = gb.cast(gb.binary_op("sub", nxt_head, seg_start_col,
                                        INT32), INT64)
            rn0 = gb.cast(gb.binary_op("sub", iota_col, seg_start_col,
                                       INT32), INT64)
            num = gb.binary_op_scalar("mul", rn0, nt, INT64)
            t = gb.binary_op("int_div", num, size, INT64)
            return gb.cast(gb.binary_op_scalar("add", t, 1, INT64), INT32)

        # value-based functions
        vc = w.func.child.eval(table, cs)
        if op == "nth_value":
            hp_ext = torch.cat([head_pos, torch.tensor(
                [n], dtype=torch.int32, device="cuda")])
            segid_next = gb.binary_op_scalar("add", gb.cast(
                Column(INT32, n, seg_id.data, None, null_count=0), INT32),
                1, INT32)
            seg_end = _i32col(self._gather_i32(hp_ext, segid_next.data, n))
            seg_end = gb.binary_op_scalar("sub", seg_end, 1, INT32)
            src = gb.binary_op_scalar("add", seg_start_col,
                                      w.func.offset - 1, INT32)
            ok = gb.binary_op("le", src, seg_end, DType.bool_())
            neg1 = Column.full(-1, INT32, n, "cuda")
            srcm = gb.if_else(ok, src, neg1)
            return gb.gather(ColumnBatch([vc], n), srcm).columns[0]
        if op in ("lag", "lead"):
            k = w.func.offset if op == "lag" else -w.func.offset
            src = gb.binary_op_scalar("sub", iota_col, k, INT32)
            # segment end = next head position - 1 (last segment -> n-1)
            hp_ext = torch.cat([head_pos, torch.tensor(
                [n], dtype=torch.int32, device="cuda")])
            segid_next = gb.binary_op_scalar("add", gb.cast(
                Column(INT32, n, seg_id.data, None, null_count=0), INT32),
                1, INT32)
            seg_end = _i32col(self._gather_i32(hp_ext, segid_next.data, n))
            seg_end = gb.binary_op_scalar("sub", seg_end, 1, INT32)
            ok = gb.binary_op("and",
                              gb.binary_op("ge", src, seg_start_col,
                                           DType.bool_()),
                              gb.binary_op("le", src, seg_end,
                                           DType.bool_()), DType.bool_())
            neg1 = Column.full(-1, INT32, n, "cuda")
            srcm = gb.if_else(ok, src, neg1)
            gathered = gb.gather(ColumnBatch([vc], n), srcm).columns[0]
            if w.func.default is not None:
                dcol = Column.full(w.func.default, gathered.dtype, n, "cuda")
                return gb.if_else(ok, gathered, dcol)
            return gathered

        running = len(self.spec.order_by) > 0
        valid_u8 = torch.ones(n, dtype=torch.uint8, device="cuda")
        if vc.validity is not None:
            ext.mask_expand(vc.validity.data_ptr(), valid_u8.data_ptr(),
                            False, n, s)
        nn_col = Column(DType.bool_(), n, valid_u8, None, null_count=0)
        rb = self.spec.rows_between
        rgb = self.spec.range_between
        framed_minmax = running and op in ("min", "max")
        if (rb is not None or rgb is not None or framed_minmax) \
                and op in ("sum", "count", "mean", "min", "max"):
            hp_ext = torch.cat([head_pos, torch.tensor(
                [n], dtype=torch.int32, device="cuda")])
            segid_next = gb.binary_op_scalar(
                "add", Column(INT32, n, seg_id.data, None, null_count=0), 1,
                INT32)
            seg_end = Column(INT32, n, self._gather_i32(hp_ext,
                                                        segid_next.data, n),
                             None, null_count=0)
            seg_end = gb.binary_op_scalar("sub", seg_end, 1, INT32)
            if rb is not None:
                lo_off, hi_off = rb
                b_idx = gb.binary_op(
                    "min", gb.binary_op_scalar("add", iota_col, hi_off,
                                               INT32), seg_end, INT32)
                a_idx = gb.binary_op(
                    "max", gb.binary_op_scalar("add", iota_col, lo_off,
                                               INT32), seg_start_col,
                    INT32)
            elif rgb is not None:
                # RANGE frame: per-row binary search of the ascending order
                # key within the segment (k_range_bounds)
                lo_v, hi_v = rgb
                okey = table.columns[cs.index(self.spec.order_by[0])]
                of = gb.cast(Column(okey.dtype, n, okey.data, None,
                                    null_count=0), FLOAT64)
                of_t = of.data
                if okey.validity is not None:
                    ov_u8 = torch.empty(n, dtype=torch.uint8, device="cuda")
                    ext.mask_expand(okey.validity.data_ptr(),
                                    ov_u8.data_ptr(), False, n, s)
                    of_t = torch.where(
                        ov_u8.bool(), of_t,
                        torch.full_like(of_t, float("-inf")))
                if self.spec.descending[0]:
                    # frame offsets follow the SORT direction: negate to an
                    # ascending axis and the same (lo, hi) apply directly
                    # (matches the CPU searchsorted path above)
                    of_t = torch.neg(of_t)
                a_t = torch.empty(n, dtype=torch.int32, device="cuda")
                b_t = torch.empty(n, dtype=torch.int32, device="cuda")
                ext.range_bounds(of_t.data_ptr(),
                                 seg_start_col.data.data_ptr(),
                                 seg_end.data.data_ptr(),
                                 float(lo_v if lo_v is not None else 0.0),
                                 float(hi_v if hi_v is not None else 0.0),
                                 1 if lo_v is None else 0,
                                 1 if hi_v is None else 0,
                                 a_t.data_ptr(), b_t.data_ptr(), n, s)
                a_idx = _i32col(a_t)
                b_idx = _i32col(b_t)
            else:
                # running min/max: frame = [segment start, current row]
                a_idx, b_idx = seg_start_col, iota_col
            am1 = gb.binary_op_scalar("sub", a_idx, 1, INT32)
            if op in ("min", "max"):
                return self._framed_minmax(op, vc, nn_col, a_idx, b_idx,
                                           am1, n, s)
            use_f64 = out_dt.is_floating
            work_t = FLOAT64 if use_f64 else INT64
            v64 = gb.cast(Column(vc.dtype, n, vc.data, None, null_count=0),
                          work_t)
            nonfin = None
            if use_f64:
                # prefix-sum framing is wrong for non-finite values (one
                # inf/NaN poisons every later frame) and a null slot's
                # garbage bits could be non-finite: zero both out of the
                # scan and patch affected frames from exact non-finite
                # frame COUNTS afterwards (found by special-value fuzz)
                nnb = valid_u8[:n].to(torch.bool)
                vv = v64.data[:n]
                pos_m = torch.isposinf(vv) & nnb
                neg_m = torch.isneginf(vv) & nnb
                nan_m = torch.isnan(vv) & nnb
                bad = pos_m | neg_m | nan_m
                if bool(bad.any()):
                    nonfin = (pos_m, neg_m, nan_m)
                vz = Column(work_t, n,
                            torch.where(nnb & ~bad, vv,
                                        torch.zeros((), dtype=vv.dtype,
                                                    device="cuda")),
                            None, null_count=0)
            else:
                vz = gb.binary_op("mul", v64, gb.cast(nn_col, work_t),
                                  work_t)
            nn64 = gb.cast(nn_col, INT64)
            # inclusive global prefix sums; gather_fixed zero-fills index -1
            if use_f64:
                excl = self._scan_f64(vz.data, n)
                incl = gb.binary_op("add",
                                    Column(work_t, n, excl, None,
                                           null_count=0), vz, work_t)
            else:
                excl_t, _ = gb._exclusive_scan_i64(vz.data)
                incl = gb.binary_op("add",
                                    Column(INT64, n, excl_t, None,
                                           null_count=0), vz, INT64)
            excl_n, _ = gb._exclusive_scan_i64(nn64.data)
            incl_n = gb.binary_op("add",
                                  Column(INT64, n, excl_n, None,
                                         null_count=0), nn64, INT64)
            incl_b = Column(work_t, n,
                            self._gather_i32(incl.data, b_idx.data, n), None,
                            null_count=0)
            incl_a = Column(work_t, n,
                            self._gather_i32(incl.data, am1.data, n), None,
                            null_count=0)
            cnt_b = Column(INT64, n,
                           self._gather_i32(incl_n.data, b_idx.data, n),
                           None, null_count=0)
            cnt_a = Column(INT64, n,
                           self._gather_i32(incl_n.data, am1.data, n),
                           None, null_count=0)
            cnt_r = gb.binary_op("sub", cnt_b, cnt_a, INT64)
            # empty frames (b < a) would go negative; clamp via max(cnt, 0)
            cnt_r = gb.binary_op_scalar("max", cnt_r, 0, INT64)
            if op == "count":
                return cnt_r
            sum_r = gb.binary_op("sub", incl_b, incl_a, work_t)
            if nonfin is not None:
                def _frame_count(mask):
                    m64 = mask.to(torch.int64).contiguous()
                    excl_m, _ = gb._exclusive_scan_i64(m64)
                    incl_m = excl_m + m64
                    cb = self._gather_i32(incl_m, b_idx.data, n)
                    ca = self._gather_i32(incl_m, am1.data, n)
                    return cb - ca

                pc = _frame_count(nonfin[0])
                nc = _frame_count(nonfin[1])
                qc = _frame_count(nonfin[2])
                st = sum_r.data[:n]
                st = torch.where((qc > 0) | ((pc > 0) & (nc > 0)),
                                 torch.full_like(st, float("nan")), st)
                st = torch.where((pc > 0) & (nc == 0) & (qc == 0),
                                 torch.full_like(st, float("inf")), st)
                st = torch.where((nc > 0) & (pc == 0) & (qc == 0),
                                 torch.full_like(st, float("-inf")), st)
                sum_r = Column(work_t, n, st, None, null_count=0)
            ov = torch.empty(mask_nbytes(n), dtype=torch.uint8,
                             device="cuda")
            ext.mask_from_nonzero(cnt_r.data.data_ptr(), ov.data_ptr(), n, s)
            sum_c = Column(work_t, n, sum_r.data, ov, null_count=None)
            if op == "sum":
                return gb.cast(sum_c, out_dt) if work_t != out_dt else sum_c
            cf = gb.cast(cnt_r, FLOAT64)
            return gb.binary_op("div", gb.cast(sum_c, FLOAT64), cf, FLOAT64)
        if running and op in ("sum", "count", "mean"):
            use_f64 = out_dt.is_floating
            work_t = FLOAT64 if use_f64 else INT64
            v64 = gb.cast(Column(vc.dtype, n, vc.data, None, null_count=0),
                          work_t)
            if use_f64:
                # select (not multiply) so non-finite garbage in null
                # slots cannot leak a NaN into the running prefix; valid
                # non-finite VALUES keep exact prefix semantics here
                # (frame is always [start, i])
                nnb = valid_u8[:n].to(torch.bool)
                vz = Column(work_t, n,
                            torch.where(nnb, v64.data[:n],
                                        torch.zeros((), dtype=torch.float64,
                                                    device="cuda")),
                            None, null_count=0)
            else:
                vz = gb.binary_op("mul", v64, gb.cast(nn_col, work_t),
                                  work_t)
            run_cnt = self._running_sum_i64(
                gb.cast(nn_col, INT64).data, heads, seg_start_col, n)
            if op == "count":
                return Column(INT64, n, run_cnt, None, null_count=0)
            if use_f64:
                run_sum = self._running_sum_f64(vz.data, seg_start_col, n)
            else:
                run_sum = self._running_sum_i64(vz.data, heads,
                                                seg_start_col, n)
            sum_col = Column(work_t, n, run_sum, None, null_count=None)
            ov = torch.empty(mask_nbytes(n), dtype=torch.uint8, device="cuda")
            ext.mask_from_nonzero(run_cnt.data_ptr(), ov.data_ptr(), n, s)
            sum_col = Column(work_t, n, run_sum, ov, null_count=None)
            if op == "sum":
                return gb.cast(sum_col, out_dt) if work_t != out_dt else sum_col
            cnt_f = gb.cast(Column(INT64, n, run_cnt, None, null_count=0),
                            FLOAT64)
            return gb.binary_op("div", gb.cast(sum_col, FLOAT64), cnt_f,
                                FLOAT64)
        if not running and op in ("sum", "count", "mean", "min", "max"):
            # per-segment aggregate via the groupby kernels (seg_id as the
            # group id), then broadcast back with a gather
            acc_is_double = out_dt.is_floating or vc.dtype.is_floating
            acc = torch.empty(max(nseg, 1),
                              dtype=torch.float64 if acc_is_double
                              else torch.int64, device="cuda")
            cnt = torch.zeros(max(nseg, 1), dtype=torch.int64, device="cuda")
            gop = {"sum": 0, "min": 1, "max": 2, "count": 3, "mean": 0}[op]
            if op != "count":
                ext.gb_acc_init(gop, acc.data_ptr(), acc_is_double, nseg, s)
            ext.gb_agg(gop if op != "mean" else 0, gb._ht(vc.dtype),
                       vc.data.data_ptr(), gb._ptr(vc.validity),
                       seg_id.data.data_ptr(), acc.data_ptr(), cnt.data_ptr(),
                       acc_is_double, nseg, n, s)
            cnt_rows = self._gather_i32(cnt, seg_id.data, n)
            if op == "count":
                return Column(INT64, n, cnt_rows, None, null_count=0)
            acc_rows = self._gather_i32(acc, seg_id.data, n)
            work_t = FLOAT64 if acc_is_double else INT64
            ov = torch.empty(mask_nbytes(n), dtype=torch.uint8, device="cuda")
            ext.mask_from_nonzero(cnt_rows.data_ptr(), ov.data_ptr(), n, s)
            val = Column(work_t, n, acc_rows, ov, null_count=None)
            if op == "mean":
                cf = gb.cast(Column(INT64, n, cnt_rows, None, null_count=0),
                             FLOAT64)
                return gb.binary_op("div", gb.cast(val, FLOAT64), cf, FLOAT64)
            return gb.cast(val, out_dt) if work_t != out_dt else val
        raise NotImplementedError(f"gpu window {op}")

    def _framed_minmax(self, op, vc, nn_col, a_idx, b_idx, am1, n, s):
        """Bounded / running min-max via a sparse table: log2(n) build
        passes with the elementwise min/max kernel, then one range-query
        kernel gather per row (k_win_minmax). NULL values enter as
        +-infinity sentinels; output validity is the frame's valid count."""
        import torch

        from ..column import mask_nbytes
        from ..ops import gpu_backend as gb
        from ..ops.gpu_backend import ext
        from ..types import FLOAT64, INT64, DType

        is_f = vc.dtype.is_floating
        work_t = FLOAT64 if is_f else INT64
        v64 = gb.cast(Column(vc.dtype, n, vc.data, None, null_count=0),
                      work_t)
        if is_f:
            sent = float("inf") if op == "min" else float("-inf")
        else:
            sent = (1 << 63) - 1 if op == "min" else -(1 << 63)
        vz = gb.if_else(nn_col, v64,
                        Column.full(sent, work_t, n, "cuda"))
        levels = [vz.data]
        span = 1
        while span < n:
            prev = levels[-1]
            shifted = torch.empty_like(prev)
            shifted[: n - span] = prev[span:]
            shifted[n - span:] = prev[n - span:]
            nxt = gb.binary_op(op, Column(work_t, n, prev, None,
                                          null_count=0),
                               Column(work_t, n, shifted, None,
                                      null_count=0), work_t)
            levels.append(nxt.data)
            span *= 2
        ptrs = torch.tensor([t.data_ptr() for t in levels],
                            dtype=torch.int64).cuda()
        out = torch.empty(n, dtype=torch.float64 if is_f else torch.int64,
                          device="cuda")
        ext.win_minmax(1 if is_f else 0, ptrs.data_ptr(), len(levels),
                       a_idx.data.data_ptr(), b_idx.data.data_ptr(),
                       1 if op == "min" else 0, out.data_ptr(), n, s)
        # validity: number of valid values in the frame > 0
        nn64 = gb.cast(nn_col, INT64)
        excl_n, _ = gb._exclusive_scan_i64(nn64.data)
        incl_n = gb.binary_op("add", Column(INT64, n, excl_n, None,
                                            null_count=0), nn64, INT64)
        cnt_b = Column(INT64, n, self._gather_i32(incl_n.data, b_idx.data,
                                                  n), None, null_count=0)
        cnt_a = Column(INT64, n, self._gather_i32(incl_n.data, am1.data, n),
                       None, null_count=0)
        cnt_r = gb.binary_op("sub", cnt_b, cnt_a, INT64)
        cnt_r = gb.binary_op_scalar("max", cnt_r, 0, INT64)
        ov = torch.empty(mask_nbytes(n), dtype=torch.uint8, device="cuda")
        ext.mask_from_nonzero(cnt_r.data.data_ptr(), ov.data_ptr(), n, s)
        from ..types import TypeId

        if vc.dtype == work_t or vc.dtype.id is TypeId.DECIMAL64:
            # decimal64 backing is the same scaled int64 — no rescale
            return Column(vc.dtype, n, out, ov, null_count=None)
        res = Column(work_t, n, out, ov, null_count=None)
        return gb.cast(res, vc.dtype)

    def _running_sum_i64(self, vz: "torch.Tensor", heads, seg_start_col, n):
        import torch

        from ..ops import gpu_backend as gb
        from ..types import INT64

        excl, _ = gb._exclusive_scan_i64(vz)
        ec = Column(INT64, n, excl, None, null_count=0)
        vcol = Column(INT64, n, vz, None, null_count=0)
        at_start = Column(INT64, n, self._gather_i32(excl, seg_start_col.data,
                                                     n), None, null_count=0)
        incl = gb.binary_op("add", ec, vcol, INT64)
        return gb.binary_op("sub", incl, at_start, INT64).data

    def _scan_f64(self, vz: "torch.Tensor", n: int):
        """Global exclusive prefix sum of an f64 tensor (device kernels)."""
        import torch

        from ..ops import gpu_backend as gb
        from ..ops.gpu_backend import ext

        s = gb._stream()
        out = torch.empty(n, dtype=torch.float64, device="cuda")
        per = 256 * 8
        nb = max((n + per - 1) // per, 1)
        sums = torch.empty(nb, dtype=torch.float64, device="cuda")
        ext.scan_block_f64(vz.data_ptr(), out.data_ptr(), sums.data_ptr(),
                           n, s)
        if nb > 1:
            sums2 = torch.empty(nb, dtype=torch.float64, device="cuda")
            partial = torch.empty(max((nb + per - 1) // per, 1),
                                  dtype=torch.float64, device="cuda")
            ext.scan_block_f64(sums.data_ptr(), sums2.data_ptr(),
                               partial.data_ptr(), nb, s)
            if partial.numel() > 1:
                p2 = torch.empty_like(partial)
                p3 = torch.empty(1, dtype=torch.float64, device="cuda")
                ext.scan_block_f64(partial.data_ptr(), p2.data_ptr(),
                                   p3.data_ptr(), partial.numel(), s)
                ext.scan_add_offsets_f64(sums2.data_ptr(), p2.data_ptr(),
                                         nb, s)
            ext.scan_add_offsets_f64(out.data_ptr(), sums2.data_ptr(), n, s)
        return out

    def _running_sum_f64(self, vz: "torch.Tensor", seg_start_col, n):
        import torch

        from ..ops import gpu_backend as gb
        from ..types import FLOAT64

        out = self._scan_f64(vz, n)
        ec = Column(FLOAT64, n, out, None, null_count=0)
        vcol = Column(FLOAT64, n, vz, None, null_count=0)
        at_start = Column(FLOAT64, n,
                          self._gather_i32(out, seg_start_col.data, n), None,
                          null_count=0)
        incl = gb.binary_op("add", ec, vcol, FLOAT64)
        return gb.binary_op("sub", incl, at_start, FLOAT64).data


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
    if op == "ntile":
        nt = w.func.offset
        seg_end = _segment_ends(heads, idx, n)
        size = (seg_end - seg_start + 1).astype(np.int64)
        rn0 = (idx - seg_start).astype(np.int64)
        return _make((rn0 * nt // size + 1).astype(np.int32), None, out_dt)

    vc = w.func.child.eval(table, cs)
    v = _vals(vc).astype(np.float64) if vc.dtype.is_numeric else _vals(vc)
    valid = _valid(vc)
    if op == "nth_value":
        seg_end = _segment_ends(heads, idx, n)
        src = seg_start + w.func.offset - 1
        ok = src <= seg_end
        srcc = np.where(ok, src, 0)
        raw = _vals(vc)
        outv = np.where(ok, raw[srcc], 0)
        outvalid = np.where(ok, valid[srcc], False)
        return _make(outv.astype(out_dt.numpy_dtype())
                     if out_dt.is_fixed_width else outv,
                     outvalid if not outvalid.all() else None, out_dt)
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
    cnt_f = valid.astype(np.int64)
    rb = w.spec.rows_between
    rgb = w.spec.range_between
    if rb is not None or rgb is not None:
        seg_end = _segment_ends(heads, idx, n)
        if rb is not None:
            lo_off, hi_off = rb
            a = np.maximum(idx + lo_off, seg_start)
            b = np.minimum(idx + hi_off, seg_end)
        else:
            # RANGE frame: per-segment searchsorted on the order key
            # (descending handled by negation; null keys as -inf peers)
            lo_v, hi_v = rgb
            okc = table.columns[cs.index(w.spec.order_by[0])]
            from ..ops.cpu_backend import _vals as _cvals, _valid as _cvalid

            ov = _cvals(okc).astype(np.float64).copy()
            ovalid = _cvalid(okc)
            ov[~ovalid] = -np.inf
            if w.spec.descending[0]:
                # frame offsets follow the SORT direction: on the negated
                # (ascending) axis the same (lo, hi) apply directly
                ov = -ov
            a = np.empty(n, dtype=np.int64)
            b = np.empty(n, dtype=np.int64)
            starts = np.flatnonzero(heads)
            bounds = np.append(starts, n)
            for si in range(len(starts)):
                s0, e0 = bounds[si], bounds[si + 1]
                seg = ov[s0:e0]
                tgt = ov[s0:e0]
                if lo_v is None:
                    a[s0:e0] = s0
                else:
                    a[s0:e0] = s0 + np.searchsorted(seg, tgt + lo_v, "left")
                if hi_v is None:
                    b[s0:e0] = e0 - 1
                else:
                    b[s0:e0] = s0 + np.searchsorted(seg, tgt + hi_v,
                                                    "right") - 1
        empty_frame = b < a
        if op in ("min", "max"):
            fn = min if op == "min" else max
            res = np.zeros(n)
            ok = np.zeros(n, dtype=bool)
            for i in range(n):
                if empty_frame[i]:
                    continue
                vals_in = [v[j] for j in range(a[i], b[i] + 1) if valid[j]]
                if vals_in:
                    res[i] = fn(vals_in)
                    ok[i] = True
            res = res.astype(out_dt.numpy_dtype())
            return _make(res, ok if not ok.all() else None, out_dt)
        # exact non-finite frame semantics: zero NaN/inf out of the
        # prefix (one would poison every later frame) and patch frames
        # from their non-finite COUNTS (mirrors the GPU path)
        pos_m = np.isposinf(vv)
        neg_m = np.isneginf(vv)
        nan_m = np.isnan(vv)
        vclean = np.where(pos_m | neg_m | nan_m, 0.0, vv)
        csv = np.cumsum(vclean)
        csn = np.cumsum(cnt_f)
        am1 = np.maximum(a - 1, 0)

        def _fcnt(mask):
            cs = np.cumsum(mask.astype(np.int64))
            basec = np.where(a > 0, cs[am1], 0)
            return np.where(empty_frame, 0, cs[b] - basec)

        base_v = np.where(a > 0, csv[am1], 0.0)
        base_n = np.where(a > 0, csn[am1], 0)
        sum_r = np.where(empty_frame, 0.0, csv[b] - base_v)
        if pos_m.any() or neg_m.any() or nan_m.any():
            pc, nc, qc = _fcnt(pos_m), _fcnt(neg_m), _fcnt(nan_m)
            sum_r = np.where((qc > 0) | ((pc > 0) & (nc > 0)), np.nan,
                             sum_r)
            sum_r = np.where((pc > 0) & (nc == 0) & (qc == 0), np.inf,
                             sum_r)
            sum_r = np.where((nc > 0) & (pc == 0) & (qc == 0), -np.inf,
                             sum_r)
        cnt_r = np.where(empty_frame, 0, csn[b] - base_n)
        if op == "count":
            return _make(cnt_r.astype(np.int64), None, out_dt)
        ok = cnt_r > 0
        if op == "mean":
            res = sum_r / np.maximum(cnt_r, 1)
        else:
            res = sum_r
        return _make(res.astype(out_dt.numpy_dtype()),
                     ok if not ok.all() else None, out_dt)
    running = len(w.spec.order_by) > 0
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

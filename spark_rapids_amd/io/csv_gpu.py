"""GPU CSV decode (reference analogue: GpuCsvScan —
sql-plugin/src/main/scala/com/nvidia/spark/rapids/GpuBatchScanExec.scala
GpuCsvScan + CSVPartitionReader — over cudf's CSV reader; SURVEY.md §2.3
CSV row).

The whole file is staged to device memory once; newline positions come
from the byte-compare + stream-compaction kernels, and one k_csv_parse
launch per column walks each row to its field and parses int64/float64 /
records string spans in place (csv.hip). Quoted fields and unsupported
column types raise NotImplementedError and the caller falls back to the
CPU (arrow) reader for that file — the same per-file fallback contract
the parquet reader uses.
"""
from __future__ import annotations

import torch

from ..column import Column, ColumnBatch, Schema, mask_nbytes
from ..types import DType, TypeId

_INT_IDS = {TypeId.INT8, TypeId.INT16, TypeId.INT32, TypeId.INT64}


def read_csv_gpu(path: str, schema: Schema, header: bool = True,
                 delimiter: str = ",") -> ColumnBatch:
    from ..ops import gpu_backend as gb
    from ..ops.gpu_backend import ext

    for f in schema.fields:
        if f.dtype.id not in _INT_IDS and not f.dtype.is_floating \
                and f.dtype.id is not TypeId.STRING:
            raise NotImplementedError(f"gpu csv: column type {f.dtype}")
    with open(path, "rb") as fh:
        raw = fh.read()
    if not raw:
        return ColumnBatch([Column.from_pylist([], f.dtype).cuda()
                            for f in schema.fields], 0)
    s = gb._stream()
    data = torch.frombuffer(bytearray(raw), dtype=torch.uint8).cuda()
    nb = data.numel()
    nl = torch.empty(nb, dtype=torch.uint8, device="cuda")
    ext.byte_eq(data.data_ptr(), ord("\n"), nl.data_ptr(), nb, s)
    nl_col = Column(DType.bool_(), nb, nl, None, null_count=0)
    pos = gb.mask_to_sel(nl_col, nb)  # int32 newline positions
    npos = pos.numel()
    trailing = raw[-1:] != b"\n"
    starts = torch.empty(npos + 1, dtype=torch.int32, device="cuda")
    starts[0] = 0
    if npos:
        starts[1:] = pos + 1
    ends = torch.empty(npos + (1 if trailing else 0), dtype=torch.int32,
                       device="cuda")
    if npos:
        ends[:npos] = pos
    if trailing:
        ends[npos] = nb
    nrows_all = ends.numel()
    skip = 1 if header else 0
    row_start = starts[skip:nrows_all]
    row_end = ends[skip:nrows_all]
    n = nrows_all - skip
    if n <= 0:
        return ColumnBatch([Column.from_pylist([], f.dtype).cuda()
                            for f in schema.fields], 0)

    unsupported = torch.zeros(1, dtype=torch.int32, device="cuda")
    cols = []
    for fi, f in enumerate(schema.fields):
        valid_u8 = torch.empty(n, dtype=torch.uint8, device="cuda")
        if f.dtype.id is TypeId.STRING:
            ss = torch.empty(n, dtype=torch.int32, device="cuda")
            sl = torch.empty(n, dtype=torch.int64, device="cuda")
            ext.csv_parse(data.data_ptr(), row_start.data_ptr(),
                          row_end.data_ptr(), ord(delimiter), fi, 2,
                          0, 0, ss.data_ptr(), sl.data_ptr(),
                          valid_u8.data_ptr(), unsupported.data_ptr(), n, s)
            # assemble via span compaction (shares the substr copy kernel)
            scanned, total = gb._exclusive_scan_i64(sl)
            out_bytes = torch.empty(max(total, 1), dtype=torch.uint8,
                                    device="cuda")[:total]
            if total:
                ext.substr_copy(data.data_ptr(), ss.data_ptr(),
                                sl.data_ptr(), scanned.data_ptr(),
                                out_bytes.data_ptr(), n, s)
            offs = torch.empty(n + 1, dtype=torch.int32, device="cuda")
            ext.narrow_i64_i32(scanned.data_ptr(), offs.data_ptr(), n, s)
            offs[n] = total
            # empty unquoted field = null string (arrow CSV default)
            mask = _pack_valid(valid_u8, n, s)
            cols.append(Column(DType.string(), n, out_bytes, mask, offs,
                               null_count=None))
            continue
        is_f = f.dtype.is_floating
        out = torch.empty(n, dtype=torch.float64 if is_f else torch.int64,
                          device="cuda")
        ext.csv_parse(data.data_ptr(), row_start.data_ptr(),
                      row_end.data_ptr(), ord(delimiter), fi,
                      1 if is_f else 0, 0 if is_f else out.data_ptr(),
                      out.data_ptr() if is_f else 0, 0, 0,
                      valid_u8.data_ptr(), unsupported.data_ptr(), n, s)
        mask = _pack_valid(valid_u8, n, s)
        wide = Column(DType.float64() if is_f else DType.int64(), n, out,
                      mask, null_count=None)
        cols.append(gb.cast(wide, f.dtype) if wide.dtype != f.dtype
                    else wide)
    if int(unsupported.item()) > 0:
        raise NotImplementedError("gpu csv: quoted fields present")
    return ColumnBatch(cols, n)


def _pack_valid(valid_u8: torch.Tensor, n: int, s) -> torch.Tensor:
    from ..ops.gpu_backend import ext

    v64 = torch.empty(n, dtype=torch.int64, device="cuda")
    ext.cast(0, 4, valid_u8.data_ptr(), v64.data_ptr(), n, s)
    mask = torch.empty(mask_nbytes(n), dtype=torch.uint8, device="cuda")
    ext.mask_from_nonzero(v64.data_ptr(), mask.data_ptr(), n, s)
    return mask


def read_json_gpu(path: str, schema: Schema) -> ColumnBatch:
    """JSON-lines decode with k_json_field (flat objects; string values
    with escapes or non-scalar columns fall back per file)."""
    from ..ops import gpu_backend as gb
    from ..ops.gpu_backend import ext

    for f in schema.fields:
        if f.dtype.id not in _INT_IDS and not f.dtype.is_floating \
                and f.dtype.id not in (TypeId.STRING, TypeId.BOOL):
            raise NotImplementedError(f"gpu json: column type {f.dtype}")
    with open(path, "rb") as fh:
        raw = fh.read()
    if not raw.strip():
        return ColumnBatch([Column.from_pylist([], f.dtype).cuda()
                            for f in schema.fields], 0)
    s = gb._stream()
    data = torch.frombuffer(bytearray(raw), dtype=torch.uint8).cuda()
    nb = data.numel()
    nl = torch.empty(nb, dtype=torch.uint8, device="cuda")
    ext.byte_eq(data.data_ptr(), ord("\n"), nl.data_ptr(), nb, s)
    pos = gb.mask_to_sel(Column(DType.bool_(), nb, nl, None, null_count=0),
                         nb)
    npos = pos.numel()
    trailing = not raw.endswith(b"\n")
    starts = torch.empty(npos + 1, dtype=torch.int32, device="cuda")
    starts[0] = 0
    if npos:
        starts[1:] = pos + 1
    ends = torch.empty(npos + (1 if trailing else 0), dtype=torch.int32,
                       device="cuda")
    if npos:
        ends[:npos] = pos
    if trailing:
        ends[npos] = nb
    n = ends.numel()
    row_start, row_end = starts[:n], ends[:n]
    unsupported = torch.zeros(1, dtype=torch.int32, device="cuda")
    cols = []
    for f in schema.fields:
        nameb = f.name.encode("utf-8")
        name_t = torch.frombuffer(bytearray(nameb),
                                  dtype=torch.uint8).cuda()
        valid_u8 = torch.empty(n, dtype=torch.uint8, device="cuda")
        if f.dtype.id is TypeId.STRING:
            ss = torch.empty(n, dtype=torch.int32, device="cuda")
            sl = torch.empty(n, dtype=torch.int64, device="cuda")
            ext.json_field(data.data_ptr(), row_start.data_ptr(),
                           row_end.data_ptr(), name_t.data_ptr(),
                           len(nameb), 2, 0, 0, ss.data_ptr(),
                           sl.data_ptr(), valid_u8.data_ptr(),
                           unsupported.data_ptr(), n, s)
            scanned, total = gb._exclusive_scan_i64(sl)
            out_bytes = torch.empty(max(total, 1), dtype=torch.uint8,
                                    device="cuda")[:total]
            if total:
                ext.substr_copy(data.data_ptr(), ss.data_ptr(),
                                sl.data_ptr(), scanned.data_ptr(),
                                out_bytes.data_ptr(), n, s)
            offs = torch.empty(n + 1, dtype=torch.int32, device="cuda")
            ext.narrow_i64_i32(scanned.data_ptr(), offs.data_ptr(), n, s)
            offs[n] = total
            cols.append(Column(DType.string(), n, out_bytes,
                               _pack_valid(valid_u8, n, s), offs,
                               null_count=None))
            continue
        is_f = f.dtype.is_floating
        is_b = f.dtype.id is TypeId.BOOL
        out = torch.empty(n, dtype=torch.float64 if is_f else torch.int64,
                          device="cuda")
        t = 1 if is_f else (3 if is_b else 0)
        ext.json_field(data.data_ptr(), row_start.data_ptr(),
                       row_end.data_ptr(), name_t.data_ptr(), len(nameb),
                       t, 0 if is_f else out.data_ptr(),
                       out.data_ptr() if is_f else 0, 0, 0,
                       valid_u8.data_ptr(), unsupported.data_ptr(), n, s)
        mask = _pack_valid(valid_u8, n, s)
        wide = Column(DType.float64() if is_f else DType.int64(), n, out,
                      mask, null_count=None)
        cols.append(gb.cast(wide, f.dtype) if wide.dtype != f.dtype
                    else wide)
    if int(unsupported.item()) > 0:
        raise NotImplementedError("gpu json: escaped strings present")
    return ColumnBatch(cols, n)

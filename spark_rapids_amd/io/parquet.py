"""Parquet scan/write.

Two reader paths, mirroring the reference's reader-type choice
(GpuParquetScan PERFILE/MULTITHREADED/COALESCING + the hybrid CPU-scan mode,
SURVEY.md §2.3):

- CPU ("hybrid"): pyarrow decodes on the host into Arrow buffers that are
  adopted zero-copy as host Columns, then moved H2D once per batch. This is
  the compatibility path (all types/encodings/codecs).
- GPU_DECODE (parquet_gpu.py): host parses footers/page headers and
  decompresses pages; PLAIN/dictionary/RLE decode runs in hipdf kernels on
  the MI355X. Flat schemas, the NDS-dominant physical types.

A multithreaded prefetch pool overlaps host reads with GPU compute
(reference analogue: MultiFileReaderThreadPool, GpuMultiFileReader.scala:188).
"""
from __future__ import annotations

import concurrent.futures as cf
import glob
import os
from typing import Iterable, List, Optional

import numpy as np
import torch

from ..column import Column, ColumnBatch, Field, Schema
from ..types import (BOOL, DATE32, DType, FLOAT32, FLOAT64, INT8, INT16,
                     INT32, INT64, STRING, TIMESTAMP, TypeId)

_PA_TYPES = None


def _pa():
    import pyarrow

    return pyarrow


def arrow_to_dtype(t) -> DType:
    import pyarrow as pa

    if pa.types.is_boolean(t):
        return BOOL
    if pa.types.is_int8(t):
        return INT8
    if pa.types.is_int16(t):
        return INT16
    if pa.types.is_int32(t):
        return INT32
    if pa.types.is_int64(t):
        return INT64
    if pa.types.is_float32(t):
        return FLOAT32
    if pa.types.is_float64(t):
        return FLOAT64
    if pa.types.is_date32(t):
        return DATE32
    if pa.types.is_timestamp(t):
        return TIMESTAMP
    if pa.types.is_string(t) or pa.types.is_large_string(t):
        return STRING
    if pa.types.is_decimal(t):
        return DType.decimal(t.precision, t.scale)
    if pa.types.is_map(t):
        return DType.map_(arrow_to_dtype(t.key_type),
                          arrow_to_dtype(t.item_type))
    if pa.types.is_list(t) or pa.types.is_large_list(t):
        return DType.list_(arrow_to_dtype(t.value_type))
    if pa.types.is_struct(t):
        return DType.struct_([(t.field(i).name,
                               arrow_to_dtype(t.field(i).type))
                              for i in range(t.num_fields)])
    raise NotImplementedError(f"parquet type {t}")


def _arrow_array_to_column(arr, dtype: DType) -> Column:
    """Adopt an arrow array's buffers as a host Column (zero-copy where the
    layouts line up; decimal128 is narrowed to the int64 backing)."""
    import pyarrow as pa

    arr = arr.combine_chunks() if isinstance(arr, pa.ChunkedArray) else arr
    if arr.null_count == len(arr) and not dtype.is_nested:
        return Column.nulls(dtype, len(arr))
    if isinstance(arr, pa.Array) and arr.offset != 0:
        arr = pa.concat_arrays([arr])  # rebase offset
    if dtype.is_nested:
        # LIST/STRUCT adoption via python values (hybrid-reader nested
        # path; device-decoded nested columns are a later round)
        return Column.from_pylist(arr.to_pylist(), dtype)
    n = len(arr)
    bufs = arr.buffers()
    validity = None
    if arr.null_count > 0 and bufs[0] is not None:
        vbytes = np.frombuffer(bufs[0], dtype=np.uint8,
                               count=(n + 7) // 8).copy()
        from ..column import mask_nbytes

        padded = np.zeros(mask_nbytes(n), dtype=np.uint8)
        padded[: len(vbytes)] = vbytes
        validity = torch.from_numpy(padded)
    if dtype.id is TypeId.STRING:
        offsets = np.frombuffer(bufs[1], dtype=np.int32, count=n + 1).copy()
        nbytes = int(offsets[-1])
        data = np.frombuffer(bufs[2], dtype=np.uint8, count=nbytes).copy() \
            if bufs[2] is not None and nbytes else np.zeros(0, np.uint8)
        return Column(dtype, n, torch.from_numpy(data), validity,
                      torch.from_numpy(offsets), arr.null_count)
    if dtype.id is TypeId.BOOL:
        vals = np.frombuffer(bufs[1], dtype=np.uint8, count=(n + 7) // 8)
        dense = np.unpackbits(vals, bitorder="little")[:n].astype(np.uint8)
        return Column(dtype, n, torch.from_numpy(dense.copy()), validity,
                      null_count=arr.null_count)
    if dtype.is_decimal and dtype.id is TypeId.DECIMAL64:
        # arrow decimal128 -> low 8 bytes (precision <= 18 fits)
        raw = np.frombuffer(bufs[1], dtype=np.int64, count=2 * n)
        dense = raw[0::2].copy()
        return Column(dtype, n, torch.from_numpy(dense), validity,
                      null_count=arr.null_count)
    if dtype.id is TypeId.DECIMAL128:
        # arrow 16-byte LE values == our interleaved (lo, hi) pairs
        raw = np.frombuffer(bufs[1], dtype=np.int64, count=2 * n).copy()
        return Column(dtype, n, torch.from_numpy(raw), validity,
                      null_count=arr.null_count)
    np_dt = dtype.numpy_dtype()
    vals = np.frombuffer(bufs[1], dtype=np_dt, count=n).copy()
    return Column(dtype, n, torch.from_numpy(vals), validity,
                  null_count=arr.null_count)


def arrow_table_to_batch(tbl) -> ColumnBatch:
    cols = []
    for name in tbl.schema.names:
        arr = tbl.column(name)
        cols.append(_arrow_array_to_column(arr, arrow_to_dtype(arr.type)))
    return ColumnBatch(cols, tbl.num_rows)


def parquet_schema(path: str) -> Schema:
    import pyarrow.parquet as pq

    sch = pq.read_schema(path)
    return Schema([Field(f.name, arrow_to_dtype(f.type), f.nullable)
                   for f in sch])


# observability for GPU-vs-fallback scan routing (the bench and GPU tests
# assert the device decode path actually ran — VERDICT: no silent fallback)
SCAN_STATS = {"gpu_files": 0, "fallback_files": 0, "last_fallback": None,
              "rg_skipped": 0, "rg_scanned": 0}
# cumulative host-side phase timers for the GPU decode path (seconds,
# summed across prefetch threads — can exceed wall when overlapped)
PHASE_STATS = {"meta_s": 0.0, "io_s": 0.0, "decode_s": 0.0,
               "parse_s": 0.0, "fast_s": 0.0, "upload_s": 0.0,
               "upload_bytes": 0}


def _stat_overlaps(op: str, value, mn, mx) -> bool:
    """Can any row in [mn, mx] satisfy `col op value`? Conservative:
    comparison errors keep the row group."""
    try:
        if isinstance(mn, bytes):
            mn = mn.decode("utf-8", "ignore")
        if isinstance(mx, bytes):
            mx = mx.decode("utf-8", "ignore")
        if isinstance(value, (int, float)) and not isinstance(mn, str):
            mn, mx, value = float(mn), float(mx), float(value)
        elif isinstance(value, str) != isinstance(mn, str):
            return True
        if op == "eq":
            return mn <= value <= mx
        if op == "lt":
            return mn < value
        if op == "le":
            return mn <= value
        if op == "gt":
            return mx > value
        if op == "ge":
            return mx >= value
    except TypeError:
        return True
    return True


def _rg_keep(path: str, triples) -> set:
    """Row groups whose min/max stats can satisfy every conjunct."""
    from .parquet_gpu import _file_meta

    md, _, _ = _file_meta(path)
    if md.num_row_groups == 0:
        return set()
    name_to_idx = {md.row_group(0).column(j).path_in_schema: j
                   for j in range(md.num_columns)}
    keep = set()
    for rg in range(md.num_row_groups):
        rgmd = md.row_group(rg)
        ok = True
        for name, op, value in triples:
            j = name_to_idx.get(name)
            if j is None:
                continue
            st = rgmd.column(j).statistics
            if st is None or not st.has_min_max:
                continue
            if not _stat_overlaps(op, value, st.min, st.max):
                ok = False
                break
        if ok:
            keep.add(rg)
            SCAN_STATS["rg_scanned"] += 1
        else:
            SCAN_STATS["rg_skipped"] += 1
    return keep


class ParquetTable:
    """Scan source: one partition per (file, row-group-range) with a
    prefetching reader pool."""

    def __init__(self, path: str, num_partitions: int = 0,
                 columns: Optional[List[str]] = None, reader: str = "CPU",
                 prefetch_threads: int = 4, replicated: bool = False,
                 chunk_bytes: int = 0):
        # chunk_bytes > 0: a file whose decoded size exceeds the budget is
        # read ROW GROUP by row group as separate batches instead of one
        # concatenated file batch (the chunked-reader analogue:
        # ParquetChunkedReader / GpuParquetScan.scala:3401). The coalesce
        # exec above re-merges toward batchSizeBytes.
        self.chunk_bytes = chunk_bytes
        # replicated=True: every rank scans ALL files (dimension tables in
        # a distributed star-schema query; broadcast-join build sides skip
        # the exchange, plan/logical.py is_replicated)
        self.replicated = replicated
        if isinstance(path, (list, tuple)):
            self.files = list(path)
        else:
            self.files = sorted(glob.glob(path)) \
                if any(ch in path for ch in "*?") \
                else ([os.path.join(path, f) for f in sorted(os.listdir(path))
                       if f.endswith(".parquet")]
                      if os.path.isdir(path) else [path])
        if not self.files:
            raise FileNotFoundError(path)
        self.columns = columns
        self.reader = reader
        self.prefetch_threads = prefetch_threads
        self.schema = parquet_schema(self.files[0])
        if columns:
            keep = [f for f in self.schema.fields if f.name in columns]
            self.schema = Schema(keep)

    def _clone(self) -> "ParquetTable":
        t = ParquetTable.__new__(ParquetTable)
        t.replicated = self.replicated
        t.files = self.files
        t.columns = self.columns
        t.reader = self.reader
        t.prefetch_threads = self.prefetch_threads
        t.schema = self.schema
        t.rg_predicate = getattr(self, "rg_predicate", None)
        t.chunk_bytes = getattr(self, "chunk_bytes", 0)
        return t

    def with_columns(self, names: List[str]) -> "ParquetTable":
        """Narrowed view for column pruning: only these columns are read
        and decoded (parquet stores column chunks separately on disk)."""
        t = self._clone()
        t.columns = list(names)
        t.schema = Schema([f for f in self.schema.fields if f.name in names])
        return t

    def with_predicate(self, triples) -> "ParquetTable":
        """Attach simple (col, op, literal) conjuncts for row-group
        min/max skipping (reference analogue: predicate pushdown + row
        group filtering in GpuParquetScan.scala:107). The Filter node
        stays above the scan; the stats only SKIP row groups."""
        t = self._clone()
        t.rg_predicate = list(triples)
        return t

    def _read_one(self, path: str) -> List[ColumnBatch]:
        from .filecache import get_cached, put_cached

        cached = get_cached(path, self.columns)
        if cached is not None:
            return cached if isinstance(cached, list) else [cached]
        batches = self._read_one_uncached(path)
        put_cached(path, self.columns, batches)
        return batches

    def _rg_chunks(self, path: str, keep):
        """Split a file's surviving row groups into chunked reads whose
        estimated decoded bytes respect chunk_bytes (the chunked-reader
        analogue). Returns None for the single-read fast path."""
        budget = getattr(self, "chunk_bytes", 0)
        if not budget:
            return None
        from .parquet_gpu import _file_meta

        md, _, _ = _file_meta(path)
        rgs = [r for r in range(md.num_row_groups)
               if keep is None or r in keep]
        if len(rgs) <= 1:
            return None
        sizes = [md.row_group(r).total_byte_size for r in rgs]
        if sum(sizes) <= budget:
            return None
        chunks, cur, cur_b = [], [], 0
        for r, sz in zip(rgs, sizes):
            if cur and cur_b + sz > budget:
                chunks.append(cur)
                cur, cur_b = [], 0
            cur.append(r)
            cur_b += sz
        if cur:
            chunks.append(cur)
        return chunks

    def _read_one_uncached(self, path: str) -> List[ColumnBatch]:
        keep = None
        pred = getattr(self, "rg_predicate", None)
        if pred:
            keep = _rg_keep(path, pred)
        chunks = self._rg_chunks(path, keep)
        if self.reader == "GPU_DECODE":
            try:
                from .parquet_gpu import read_parquet_gpu

                cols = [f.name for f in self.schema.fields]
                if chunks is not None:
                    out = [read_parquet_gpu(path, cols, keep_rgs=set(ch))
                           for ch in chunks]
                else:
                    out = [read_parquet_gpu(path, cols, keep_rgs=keep)]
                SCAN_STATS["gpu_files"] += 1
                return out
            except NotImplementedError as e:
                # per-file fallback to the CPU/hybrid reader, mirroring the
                # reference's per-op CPU fallback contract
                SCAN_STATS["fallback_files"] += 1
                SCAN_STATS["last_fallback"] = f"{path}: {e}"
        import pyarrow.parquet as pq

        if chunks is not None:
            pf = pq.ParquetFile(path)
            return [arrow_table_to_batch(
                pf.read_row_groups(ch, columns=self.columns))
                for ch in chunks]
        if keep is not None:
            pf = pq.ParquetFile(path)
            if len(keep) == 0:
                sch = pf.schema_arrow
                import pyarrow as pa

                tbl = pa.table({n: pa.array([], type=sch.field(n).type)
                                for n in (self.columns
                                          or sch.names)})
            else:
                tbl = pf.read_row_groups(sorted(keep),
                                         columns=self.columns)
            return [arrow_table_to_batch(tbl)]
        tbl = pq.read_table(path, columns=self.columns)
        return [arrow_table_to_batch(tbl)]

    def _my_files(self) -> List[str]:
        # distributed scans shard files round-robin across ranks
        from ..shuffle import dist

        c = dist.ctx()
        if c.is_multi and len(self.files) > 1 and not self.replicated:
            return self.files[c.rank::c.world]
        return self.files

    def partitions(self) -> Iterable[ColumnBatch]:
        files = self._my_files()
        if len(files) <= 1 or self.prefetch_threads <= 1:
            for f in files:
                yield from self._read_one(f)
            return
        import torch

        use_streams = torch.cuda.is_available() \
            and self.reader == "GPU_DECODE"
        if not use_streams:
            with cf.ThreadPoolExecutor(self.prefetch_threads) as pool:
                futures = [pool.submit(self._read_one, f) for f in files]
                for fut in futures:
                    yield from fut.result()
            return
        # copy/compute overlap: each prefetch worker decodes on its own
        # HIP stream (uploads + decode kernels), so file IO, H2D and
        # device decode of file k+1 overlap compute on file k; the
        # consumer stream waits on the worker's recorded event before
        # touching the batch (reference analogue: the multithreaded
        # reader pool feeding the main task stream, §2.7 pipeline row)
        side_streams = [torch.cuda.Stream()
                        for _ in range(self.prefetch_threads)]
        tl = __import__("threading").local()
        counter = __import__("itertools").count()

        def read_on_stream(f):
            st = getattr(tl, "stream", None)
            if st is None:
                st = side_streams[next(counter) % len(side_streams)]
                tl.stream = st
            with torch.cuda.stream(st):
                batches = self._read_one(f)
                ev = torch.cuda.Event()
                ev.record(st)
            return batches, ev

        with cf.ThreadPoolExecutor(self.prefetch_threads) as pool:
            futures = [pool.submit(read_on_stream, f) for f in files]
            for fut in futures:
                batches, ev = fut.result()
                # order the consumer (current/default) stream after the
                # producer stream's work without a host sync
                ev.wait(torch.cuda.current_stream())
                yield from batches


def write_parquet(batch: ColumnBatch, schema: Schema, path: str,
                  compression: str = "snappy"):
    """Columnar write via arrow (host staging; chunked GPU write path is a
    later round)."""
    import pyarrow as pa
    import pyarrow.parquet as pq

    arrays = []
    for f, c in zip(schema.fields, batch.columns):
        c = c.cpu()
        arrays.append(_column_to_arrow(c, f.dtype))
    tbl = pa.table(dict(zip(schema.names, arrays)))
    pq.write_table(tbl, path, compression=compression)


def _dtype_to_arrow(dtype: DType):
    import pyarrow as pa

    if dtype.id is TypeId.LIST:
        return pa.list_(_dtype_to_arrow(dtype.children[0]))
    if dtype.id is TypeId.MAP:
        return pa.map_(_dtype_to_arrow(dtype.children[0]),
                       _dtype_to_arrow(dtype.children[1]))
    if dtype.id is TypeId.STRUCT:
        return pa.struct([(n, _dtype_to_arrow(t)) for n, t in
                          zip(dtype.field_names, dtype.children)])
    if dtype.id is TypeId.STRING:
        return pa.string()
    if dtype.is_decimal:
        return pa.decimal128(dtype.precision, dtype.scale)
    if dtype.id is TypeId.DATE32:
        return pa.date32()
    if dtype.id is TypeId.TIMESTAMP:
        return pa.timestamp("us")
    if dtype.id is TypeId.BOOL:
        return pa.bool_()
    return pa.from_numpy_dtype(dtype.numpy_dtype())


def _column_to_arrow(c: Column, dtype: DType):
    import pyarrow as pa

    valid = c.valid_array()
    mask = None if valid.all() else ~valid
    if dtype.is_nested:
        return pa.array(c.to_pylist(), type=_dtype_to_arrow(dtype))
    if dtype.id is TypeId.STRING:
        return pa.array(c.to_pylist(), type=pa.string())
    if dtype.id is TypeId.BOOL:
        return pa.array(c.to_numpy().astype(bool), mask=mask)
    if dtype.is_decimal:
        import decimal

        scale = dtype.scale
        if dtype.id is TypeId.DECIMAL128:
            from ..column import dec128_unpack

            ints = dec128_unpack(c.data.cpu().numpy()[: 2 * c.size])
        else:
            ints = c.to_numpy()
        vals = [None if not ok else decimal.Decimal(int(v)).scaleb(-scale)
                for v, ok in zip(ints, valid)]
        return pa.array(vals, type=pa.decimal128(dtype.precision, scale))
    if dtype.id is TypeId.DATE32:
        return pa.array(c.to_numpy(), type=pa.date32(), mask=mask)
    if dtype.id is TypeId.TIMESTAMP:
        return pa.array(c.to_numpy(), type=pa.timestamp("us"), mask=mask)
    return pa.array(c.to_numpy(), mask=mask)


def batch_to_parquet_bytes(batch: ColumnBatch, schema: Schema,
                           compression: str = "zstd") -> bytes:
    """Serialize a host batch to an in-memory compressed parquet blob
    (reference analogue: ParquetCachedBatchSerializer — df.cache()
    stores compressed parquet batches instead of raw columns)."""
    import io as _io

    import pyarrow as pa
    import pyarrow.parquet as pq

    arrays = [_column_to_arrow(c.cpu(), f.dtype)
              for f, c in zip(schema.fields, batch.columns)]
    tbl = pa.table(dict(zip(schema.names, arrays)))
    buf = _io.BytesIO()
    pq.write_table(tbl, buf, compression=compression)
    return buf.getvalue()


def parquet_bytes_to_batch(blob: bytes) -> ColumnBatch:
    import io as _io

    import pyarrow.parquet as pq

    return arrow_table_to_batch(pq.read_table(_io.BytesIO(blob)))

"""GPU-assisted parquet writer.

Reference analogue: GpuParquetFileFormat / GpuParquetWriter over cudf's
parquet encoder (sql-plugin .../GpuParquetFileFormat.scala). The column
payloads are produced by device kernels — validity compaction of the
valid values (stream compaction / gather) and length-prefixed PLAIN
byte-array assembly (k_str_plain_encode in decode.hip) — and copied to
the host only as finished page payloads; the host frames the format
(page headers + footer FileMetaData via the in-repo thrift-compact
writer, io/thrift_compact.py).

Encodings: PLAIN data pages (v1), RLE def levels (bit width 1),
uncompressed. Flat schemas; decimal64 as INT64/DECIMAL, decimal128 as
FIXED_LEN_BYTE_ARRAY(16) big-endian. Readable by pyarrow and by this
repo's own GPU decoder.
"""
from __future__ import annotations

import struct
from typing import List

import numpy as np
import torch

from ..column import Column, ColumnBatch, Schema
from ..types import DType, TypeId
from . import thrift_compact as tc

# parquet physical types
_PT_BOOLEAN, _PT_INT32, _PT_INT64, _PT_INT96, _PT_FLOAT, _PT_DOUBLE, \
    _PT_BYTE_ARRAY, _PT_FLBA = range(8)
_CT_DECIMAL = 5
_CT_DATE = 6
_CT_TIMESTAMP_MICROS = 10

_PHYS = {
    TypeId.BOOL: _PT_BOOLEAN,
    TypeId.INT8: _PT_INT32,
    TypeId.INT16: _PT_INT32,
    TypeId.INT32: _PT_INT32,
    TypeId.INT64: _PT_INT64,
    TypeId.FLOAT32: _PT_FLOAT,
    TypeId.FLOAT64: _PT_DOUBLE,
    TypeId.DATE32: _PT_INT32,
    TypeId.TIMESTAMP: _PT_INT64,
    TypeId.DECIMAL64: _PT_INT64,
    TypeId.DECIMAL128: _PT_FLBA,
    TypeId.STRING: _PT_BYTE_ARRAY,
}


def _rle_bit1(levels: np.ndarray) -> bytes:
    """RLE-hybrid encoding of 0/1 levels at bit width 1 (RLE runs only)."""
    out = bytearray()
    if len(levels) == 0:
        return bytes(out)
    changes = np.flatnonzero(np.diff(levels)) + 1
    starts = np.concatenate([[0], changes])
    ends = np.concatenate([changes, [len(levels)]])
    for s, e in zip(starts, ends):
        run = int(e - s)
        buf = bytearray()
        tc._w_varint(buf, run << 1)  # RLE run header
        buf.append(int(levels[s]))
        out.extend(buf)
    return bytes(out)


def _dense_valid(c: Column):
    """Device compaction: (dense column of valid rows, def levels host)."""
    from ..ops import gpu_backend as gb

    n = c.size
    if c.validity is None:
        return c, np.ones(n, dtype=np.uint8)
    levels = c.valid_array().astype(np.uint8)
    nn = gb.is_null(c)
    notnull = gb.unary_op("not", nn, DType.bool_())
    dense = gb.apply_boolean_mask(ColumnBatch([c], n), notnull).columns[0]
    return dense, levels


def _page_payload(c: Column, dtype: DType) -> (bytes, np.ndarray):
    """PLAIN-encode the column's valid values on device; returns host
    payload bytes + def levels."""
    from ..ops import gpu_backend as gb
    from ..ops.gpu_backend import ext

    dense, levels = _dense_valid(c)
    nv = dense.size
    s = gb._stream()
    if dtype.id is TypeId.STRING:
        lens = torch.empty(max(nv, 1), dtype=torch.int64,
                           device="cuda")[:nv]
        if nv:
            ext.str_plain_encode(dense.offsets.data_ptr(),
                                 dense.data.data_ptr(), 0, lens.data_ptr(),
                                 0, 0, nv, s)
        scanned, total = gb._exclusive_scan_i64(lens) if nv else (lens, 0)
        out = torch.empty(max(total, 1), dtype=torch.uint8,
                          device="cuda")[:total]
        if total:
            ext.str_plain_encode(dense.offsets.data_ptr(),
                                 dense.data.data_ptr(), scanned.data_ptr(),
                                 lens.data_ptr(), out.data_ptr(), 1, nv, s)
        return out.cpu().numpy().tobytes(), levels
    if dtype.id is TypeId.BOOL:
        vals = dense.data.cpu().numpy()[:nv].astype(np.uint8)
        return np.packbits(vals, bitorder="little").tobytes(), levels
    if dtype.id is TypeId.DECIMAL128:
        pairs = dense.data.cpu().numpy()[: 2 * nv].view(np.uint64)
        lo = pairs[0::2]
        hi = pairs[1::2]
        be = np.empty((nv, 16), dtype=np.uint8)
        be[:, :8] = hi.astype(">u8").view(np.uint8).reshape(nv, 8)
        be[:, 8:] = lo.astype(">u8").view(np.uint8).reshape(nv, 8)
        return be.tobytes(), levels
    if dtype.id in (TypeId.INT8, TypeId.INT16):
        vals = dense.data.cpu().numpy()[:nv].astype(np.int32)
        return vals.tobytes(), levels
    return dense.data.cpu().numpy()[:nv].tobytes(), levels


def _schema_elements(schema: Schema) -> List[bytes]:
    root = tc.StructWriter()
    root.f_binary(4, b"schema")
    root.f_i32(5, len(schema.fields))
    out = [root.bytes()]
    for f in schema.fields:
        se = tc.StructWriter()
        se.f_i32(1, _PHYS[f.dtype.id])
        if f.dtype.id is TypeId.DECIMAL128:
            se.f_i32(2, 16)
        se.f_i32(3, 1)  # OPTIONAL
        se.f_binary(4, f.name.encode())
        if f.dtype.is_decimal:
            se.f_i32(6, _CT_DECIMAL)
            se.f_i32(7, f.dtype.scale)
            se.f_i32(8, f.dtype.precision)
        elif f.dtype.id is TypeId.DATE32:
            se.f_i32(6, _CT_DATE)
        elif f.dtype.id is TypeId.TIMESTAMP:
            se.f_i32(6, _CT_TIMESTAMP_MICROS)
        elif f.dtype.id is TypeId.STRING:
            se.f_i32(6, 0)  # UTF8
        out.append(se.bytes())
    return out


def write_parquet_gpu(batch: ColumnBatch, schema: Schema, path: str):
    """Write one row group from a DEVICE batch (see module docstring)."""
    assert batch.columns and batch.columns[0].is_cuda, \
        "write_parquet_gpu needs a device batch"
    n = batch.num_rows
    chunks = []  # (col_meta_bytes)
    with open(path, "wb") as f:
        f.write(b"PAR1")
        for fld, c in zip(schema.fields, batch.columns):
            payload, levels = _page_payload(c, fld.dtype)
            rle = _rle_bit1(levels)
            defsec = struct.pack("<i", len(rle)) + rle
            page = defsec + payload
            dp = tc.StructWriter()
            dp.f_i32(1, n)           # num_values incl nulls
            dp.f_i32(2, 0)           # PLAIN
            dp.f_i32(3, 3)           # def levels RLE
            dp.f_i32(4, 3)           # rep levels RLE (absent, flat)
            ph = tc.StructWriter()
            ph.f_i32(1, 0)           # DATA_PAGE
            ph.f_i32(2, len(page))   # uncompressed
            ph.f_i32(3, len(page))   # compressed (none)
            ph.f_struct(5, dp.bytes())
            hdr = ph.bytes()
            offset = f.tell()
            f.write(hdr)
            f.write(page)
            cmd = tc.StructWriter()
            cmd.f_i32(1, _PHYS[fld.dtype.id])
            cmd.f_list_i32(2, [0, 3])           # encodings PLAIN, RLE
            cmd.f_list_binary(3, [fld.name.encode()])
            cmd.f_i32(4, 0)                     # UNCOMPRESSED
            cmd.f_i64(5, n)
            cmd.f_i64(6, len(hdr) + len(page))
            cmd.f_i64(7, len(hdr) + len(page))
            cmd.f_i64(9, offset)
            cc = tc.StructWriter()
            cc.f_i64(2, offset)
            cc.f_struct(3, cmd.bytes())
            chunks.append(cc.bytes())
        rg = tc.StructWriter()
        rg.f_list_struct(1, chunks)
        rg.f_i64(2, f.tell() - 4)
        rg.f_i64(3, n)
        md = tc.StructWriter()
        md.f_i32(1, 2)                          # format version
        md.f_list_struct(2, _schema_elements(schema))
        md.f_i64(3, n)
        md.f_list_struct(4, [rg.bytes()])
        md.f_binary(6, b"spark-rapids-amd hipdf gpu writer")
        footer = md.bytes()
        f.write(footer)
        f.write(struct.pack("<i", len(footer)))
        f.write(b"PAR1")

"""GPU ORC stripe decode (reference analogue: GpuOrcScan over libcudf's
ORC reader, SURVEY.md §2.3 ORC row).

Host side parses the protobuf metadata and (when the file is compressed)
inflates the stream chunks — the same host-decompress split the parquet
GPU reader uses; device kernels decode the PRESENT boolean-RLE and RLEv2
integer streams (k_orc_bool_rle / k_orc_rle_v2 in decode.hip), doubles
are raw little-endian device buffers, and string columns reuse the
dense-build + null-scatter machinery of the parquet PLAIN byte-array
path. DIRECT_V2 encodings only; dictionary-encoded or exotic columns
raise NotImplementedError and the scan falls back to the arrow reader
per file.
"""
from __future__ import annotations

from typing import List, Optional

import torch

from ..column import Column, ColumnBatch, Schema, mask_nbytes
from ..types import DType, TypeId
from . import orc_meta as om

_KIND_TO_DTYPE = {
    om.K_BOOL: DType.bool_(),
    om.K_BYTE: DType(TypeId.INT8),
    om.K_SHORT: DType(TypeId.INT16),
    om.K_INT: DType(TypeId.INT32),
    om.K_LONG: DType.int64(),
    om.K_FLOAT: DType(TypeId.FLOAT32),
    om.K_DOUBLE: DType.float64(),
    om.K_STRING: DType.string(),
    om.K_DATE: DType.date32(),
}


def read_orc_gpu(path: str, columns: Optional[List[str]] = None
                 ) -> ColumnBatch:
    from ..ops import gpu_backend as gb
    from ..ops.gpu_backend import ext

    meta = om.read_meta(path)
    with open(path, "rb") as f:
        raw = f.read()
    root_kind, subs = meta.types[0]
    if root_kind != om.K_STRUCT:
        raise NotImplementedError("orc: non-struct root")
    col_ids = [int(v) for v in _subs_list(subs)]
    names = meta.names
    keep = [(i, cid, names[i]) for i, cid in enumerate(col_ids)
            if columns is None or names[i] in columns]
    for _, cid, _ in keep:
        kind, _ = meta.types[cid]
        if kind not in _KIND_TO_DTYPE:
            raise NotImplementedError(f"orc: column kind {kind}")

    parts: List[List[Column]] = []
    for st in meta.stripes:
        streams = om.stripe_streams(raw, meta, st)
        encs = om.stripe_encodings(raw, meta, st)
        cols: List[Column] = []
        for _, cid, _ in keep:
            kind, _ = meta.types[cid]
            if encs[cid] not in (0, 2):  # DIRECT / DIRECT_V2
                raise NotImplementedError(f"orc: encoding {encs[cid]}")
            cols.append(_decode_column(raw, meta, st, streams, cid, kind,
                                       gb, ext))
        parts.append(cols)
    out = [parts[0][j] if len(parts) == 1 else None
           for j in range(len(keep))]
    if len(parts) > 1:
        from .. import ops

        batches = [ColumnBatch(p, p[0].size if p else 0) for p in parts]
        return ops.concat_batches(batches)
    return ColumnBatch(out, meta.stripes[0].num_rows if meta.stripes else 0)


def _subs_list(subs):
    # packed repeated field arrives as raw bytes; split the varints
    if len(subs) == 1 and isinstance(subs[0], (bytes, bytearray)):
        vals = []
        p = 0
        b = subs[0]
        while p < len(b):
            v, p = om._pb_varint(b, p)
            vals.append(v)
        return vals
    return subs


def _stream_bytes(raw, meta, streams, cid, kind) -> Optional[bytes]:
    for s in streams:
        if s.column == cid and s.kind == kind:
            return om._decompress(raw[s.offset:s.offset + s.length],
                                  meta.compression)
    return None


def _to_dev(b: bytes) -> torch.Tensor:
    return torch.frombuffer(bytearray(b or b"\x00"),
                            dtype=torch.uint8).cuda()


def _decode_column(raw, meta, st, streams, cid, kind, gb, ext) -> Column:
    n = st.num_rows
    s = gb._stream()
    dtype = _KIND_TO_DTYPE[kind]
    present = _stream_bytes(raw, meta, streams, cid, 0)
    data = _stream_bytes(raw, meta, streams, cid, 1)
    valid_u8 = None
    nv = n
    if present is not None:
        pd = _to_dev(present)
        valid_u8 = torch.empty(max(n, 1), dtype=torch.uint8,
                               device="cuda")[:n]
        if n:
            ext.orc_bool_rle(pd.data_ptr(), pd.numel(), n,
                             valid_u8.data_ptr(), s)
        nv = int(valid_u8.sum().item())
    mask = None
    valid_idx = None
    if valid_u8 is not None and nv < n:
        vcol = Column(DType.bool_(), n, valid_u8, None, null_count=0)
        valid_idx = gb.mask_to_sel(vcol, n)
        v64 = torch.empty(n, dtype=torch.int64, device="cuda")
        ext.cast(0, 4, valid_u8.data_ptr(), v64.data_ptr(), n, s)
        mask = torch.empty(mask_nbytes(n), dtype=torch.uint8, device="cuda")
        ext.mask_from_nonzero(v64.data_ptr(), mask.data_ptr(), n, s)
    else:
        nv = n

    if kind == om.K_STRING:
        lens_b = _stream_bytes(raw, meta, streams, cid, 2)
        dd = _to_dev(data or b"")
        lens = torch.empty(max(nv, 1), dtype=torch.int64,
                           device="cuda")[:nv]
        if nv:
            lb = _to_dev(lens_b)
            ext.orc_rle_v2(lb.data_ptr(), lb.numel(), nv, 0,
                           lens.data_ptr(), s)
        scanned, total = gb._exclusive_scan_i64(lens) if nv else (lens, 0)
        offs = torch.empty(nv + 1, dtype=torch.int32, device="cuda")
        if nv:
            ext.narrow_i64_i32(scanned.data_ptr(), offs.data_ptr(), nv, s)
        offs[nv] = total
        bytes_t = dd[:total] if total else torch.zeros(
            0, dtype=torch.uint8, device="cuda")
        dense = Column(DType.string(), nv, bytes_t, None, offs, 0)
        if valid_idx is None:
            return dense
        # scatter dense rows into null-padded positions (parquet idiom)
        ridx = torch.full((n,), -1, dtype=torch.int32, device="cuda")
        iota = torch.empty(max(nv, 1), dtype=torch.int32,
                           device="cuda")[:nv]
        if nv:
            ext.iota_i32(iota.data_ptr(), nv, s)
            ext.scatter_fixed(4, iota.data_ptr(), valid_idx.data_ptr(),
                              ridx.data_ptr(), nv, s)
        out = gb._gather_col(dense, ridx, n, maybe_negative=True)
        return Column(dtype, n, out.data, mask, out.offsets,
                      null_count=None)

    if kind == om.K_DOUBLE:
        dd = _to_dev(data or b"")
        dense = dd[: 8 * nv].view(torch.float64)
    elif kind == om.K_FLOAT:
        dd = _to_dev(data or b"")
        f32 = dd[: 4 * nv].view(torch.float32)
        dense = f32
    elif kind == om.K_BOOL:
        dense_u8 = torch.empty(max(nv, 1), dtype=torch.uint8,
                               device="cuda")[:nv]
        if nv:
            db = _to_dev(data)
            ext.orc_bool_rle(db.data_ptr(), db.numel(), nv,
                             dense_u8.data_ptr(), s)
        dense = dense_u8
    else:  # integers / date
        vals = torch.empty(max(nv, 1), dtype=torch.int64,
                           device="cuda")[:nv]
        if nv:
            db = _to_dev(data)
            ext.orc_rle_v2(db.data_ptr(), db.numel(), nv, 1,
                           vals.data_ptr(), s)
        wide = Column(DType.int64(), nv, vals, None, null_count=0)
        dense = gb.cast(wide, dtype).data if dtype.id is not TypeId.INT64 \
            else vals

    esize = dense.element_size()
    if valid_idx is None:
        return Column(dtype, n, dense, None, null_count=0)
    out = torch.zeros(n, dtype=dense.dtype, device="cuda")
    if nv:
        ext.scatter_fixed(esize, dense.data_ptr(), valid_idx.data_ptr(),
                          out.data_ptr(), nv, s)
    return Column(dtype, n, out, mask, null_count=None)

"""Columnar batch <-> contiguous byte-buffer serialization for shuffle.

Reference analogue: the kudo serializer (GpuColumnarBatchSerializer.scala /
spark-rapids-jni kudo) — a sliced, concat-friendly layout. Here the buffer
is a single device-resident uint8 tensor so RCCL all-to-all moves it over
xGMI without touching the host; every section is 8-byte aligned so
deserialization is zero-copy tensor views into the received buffer.

Layout: [int64 num_rows][per col: int64 has_validity, int64 data_bytes]
        then per column: data (8B padded) [validity][offsets (strings)].
Schema is known on both sides, so the header carries only sizes.
"""
from __future__ import annotations

from typing import List

import torch

from ..column import Column, ColumnBatch, Schema, mask_nbytes, torch_dtype
from ..types import DType, TypeId


def _pad8(n: int) -> int:
    return (n + 7) & ~7


def _as_bytes(t: torch.Tensor) -> torch.Tensor:
    if t.numel() == 0:
        return torch.zeros(0, dtype=torch.uint8, device=t.device)
    return t.contiguous().view(torch.uint8).view(-1)


def serialize_batch(batch: ColumnBatch) -> torch.Tensor:
    n = batch.num_rows
    dev = batch.columns[0].data.device if batch.columns else torch.device("cpu")
    header = [n]
    sections: List[torch.Tensor] = []
    for c in batch.columns:
        if c.dtype.id is TypeId.LIST:
            raise NotImplementedError(
                "LIST columns cannot be shuffled yet (collect/explode "
                "happen after the exchange by design)")
        data_b = _as_bytes(c.data)
        header.extend([1 if c.validity is not None else 0, data_b.numel()])
        sections.append(data_b)
        if data_b.numel() % 8:
            sections.append(torch.zeros(8 - data_b.numel() % 8,
                                        dtype=torch.uint8, device=dev))
        if c.validity is not None:
            sections.append(_as_bytes(c.validity))
        if c.dtype.id is TypeId.STRING:
            ob = _as_bytes(c.offsets)
            sections.append(ob)
            if ob.numel() % 8:
                sections.append(torch.zeros(8 - ob.numel() % 8,
                                            dtype=torch.uint8, device=dev))
    ht = torch.tensor(header, dtype=torch.int64).view(torch.uint8).view(-1)
    sections.insert(0, ht.to(dev))
    return torch.cat(sections) if len(sections) > 1 else sections[0]


def deserialize_batch(buf: torch.Tensor, schema: Schema) -> ColumnBatch:
    ncols = len(schema.fields)
    hdr_bytes = 8 * (1 + 2 * ncols)
    header = buf[:hdr_bytes].view(torch.int64)
    header = header.cpu().tolist()
    n = header[0]
    off = hdr_bytes
    cols: List[Column] = []
    for i, f in enumerate(schema.fields):
        has_valid = header[1 + 2 * i] != 0
        data_bytes = header[2 + 2 * i]
        data = buf[off:off + data_bytes]
        off += _pad8(data_bytes)
        if f.dtype.id is TypeId.STRING:
            data = data.view(torch.uint8)
        else:
            data = data.view(torch_dtype(f.dtype))
        validity = None
        if has_valid:
            vb = mask_nbytes(n)
            validity = buf[off:off + vb]
            off += vb
        offsets = None
        if f.dtype.id is TypeId.STRING:
            ob = (n + 1) * 4
            offsets = buf[off:off + ob].view(torch.int32)
            off += _pad8(ob)
        cols.append(Column(f.dtype, n, data, validity, offsets,
                           null_count=None if has_valid else 0))
    return ColumnBatch(cols, n)

"""Device-to-device shuffle exchange over RCCL/xGMI.

Reference analogue: GpuShuffleExchangeExecBase + GpuHashPartitioningBase
(murmur3 -> pmod -> partition -> contiguousSplit -> transport, SURVEY.md
§3.4), with the UCX client/server replaced by one RCCL all-to-all per
exchange wave — xGMI is fully connected intra-node so pairwise traffic
scales with per-link bandwidth.
"""
from __future__ import annotations

from typing import List, Sequence

import numpy as np
from .. import ops
from ..column import Column, ColumnBatch, Field, Schema
from . import dist
from .serializer import deserialize_batch, serialize_batch


def _slice_batch(batch: ColumnBatch, start: int, end: int) -> ColumnBatch:
    idx = Column.from_numpy(np.arange(start, end, dtype=np.int32),
                            device=batch.device)
    return ops.gather(batch, idx)


def batch_schema(batch: ColumnBatch) -> Schema:
    return Schema([Field(f"c{i}", c.dtype) for i, c in enumerate(batch.columns)])


def exchange_by_hash(batch: ColumnBatch, key_idx: Sequence[int]) -> List[ColumnBatch]:
    """Hash-partition rows across the world by key and exchange; returns the
    batches received from every rank (caller concatenates).

    The exchange runs in WAVES bounded by spark.rapids.shuffle.wave.bytes:
    every rank agrees (all-reduce max) on the wave count, then sends row
    slices of each partition per wave, so a skewed or oversized exchange
    never materializes more than ~wave_bytes of serialized send buffer at
    once (reference analogue: bounce-buffer windowing in
    BufferSendState.scala / GpuShuffleCoalesceExec batching)."""
    import torch

    c = dist.ctx()
    schema = batch_schema(batch)
    parted, offsets = ops.hash_partition(batch, list(key_idx), c.world)
    budget = dist.wave_bytes()
    n_rows = max(parted.num_rows, 1)
    bytes_per_row = max(parted.nbytes // n_rows, 1)
    max_part = max((offsets[r + 1] - offsets[r] for r in range(c.world)),
                   default=0)
    rows_per_wave = max(int(budget // bytes_per_row), 1)
    my_waves = (max_part + rows_per_wave - 1) // rows_per_wave or 1
    t = torch.tensor([my_waves], dtype=torch.int64,
                     device="cuda" if parted.is_cuda else "cpu")
    import torch.distributed as td

    td.all_reduce(t, op=td.ReduceOp.MAX)
    nwaves = int(t.item())

    received: List[ColumnBatch] = []
    for w in range(nwaves):
        send = []
        for r in range(c.world):
            lo, hi = offsets[r], offsets[r + 1]
            wl = min(lo + w * rows_per_wave, hi)
            wh = min(wl + rows_per_wave, hi)
            send.append(serialize_batch(_slice_batch(parted, wl, wh)))
        recv = dist.all_to_all_bytes(send)
        for b in recv:
            piece = deserialize_batch(b, schema)
            if piece.num_rows or (nwaves == 1 and not received):
                received.append(piece)
    return received


def gather_all(batch: ColumnBatch) -> List[ColumnBatch]:
    """All-gather the batch from every rank (keyless aggregate merge /
    broadcast build side)."""
    schema = batch_schema(batch)
    bufs = dist.all_gather_bytes(serialize_batch(batch))
    return [deserialize_batch(b, schema) for b in bufs]


def exchange_by_ranges(batch: ColumnBatch, range_key: Column,
                       bounds: Sequence[int]) -> List[ColumnBatch]:
    """Range-partition rows by a monotone int64 key against world-1 cut
    points and exchange (distributed global sort: rank r receives the
    r-th key range). Reference analogue: GpuRangePartitioner."""
    from ..types import BOOL

    c = dist.ctx()
    schema = batch_schema(batch)
    send = []
    for j in range(c.world):
        mask = None
        if j > 0:
            mask = ops.binary_op_scalar("ge", range_key, int(bounds[j - 1]),
                                        BOOL)
        if j < len(bounds):
            m2 = ops.binary_op_scalar("lt", range_key, int(bounds[j]), BOOL)
            mask = m2 if mask is None else ops.binary_op("and", mask, m2,
                                                         BOOL)
        piece = batch if mask is None else ops.apply_boolean_mask(batch,
                                                                  mask)
        send.append(serialize_batch(piece))
    recv = dist.all_to_all_bytes(send)
    return [deserialize_batch(b, schema) for b in recv]

"""Deterministic per-row sampling hash expression: murmur3 of the row
position (masked to non-negative int32), used by DataFrame.sample."""
from __future__ import annotations

from ..column import Column, ColumnBatch, Schema
from ..types import DType, INT32
from .expressions import Expression
from .. import ops


class SampleHash(Expression):
    def __init__(self, seed: int = 42):
        self.seed = seed

    def dtype(self, schema: Schema) -> DType:
        return INT32

    def nullable(self, schema: Schema) -> bool:
        return False

    def eval(self, batch: ColumnBatch, schema: Schema) -> Column:
        import numpy as np

        n = batch.num_rows
        if batch.is_cuda:
            import torch

            from ..ops import gpu_backend as gb

            iota = torch.empty(max(n, 1), dtype=torch.int32,
                               device="cuda")[:n]
            if n:
                gb.ext.iota_i32(iota.data_ptr(), n, gb._stream())
            pos = Column(INT32, n, iota, None, null_count=0)
        else:
            pos = Column.from_numpy(np.arange(n, dtype=np.int32))
        h = ops.murmur3_hash([pos], self.seed)
        return ops.binary_op_scalar("bitand", h, 0x7FFFFFFF, INT32)

    def __str__(self):
        return f"sample_hash({self.seed})"

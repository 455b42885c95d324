"""OOM retry framework: retry, spill-then-retry, split-and-retry.

Reference analogue: RmmRapidsRetryIterator.scala withRetry/withRetryNoSplit +
the RmmSpark thread OOM state machine. On MI355X the device pool is the
PyTorch-ROCm caching allocator over 288 GB HBM3E; an allocation failure
surfaces as torch.OutOfMemoryError. Strategy: (1) release cached blocks and
retry, (2) spill spillable batches and retry, (3) split the input batch and
retry halves recursively (bounded by spark.rapids.sql.retry.maxSplits).
"""
from __future__ import annotations

import threading
from typing import Callable, List

import numpy as np
import torch


class GpuRetryOOM(RuntimeError):
    """Synthetic/native OOM that is retryable without splitting."""


class GpuSplitAndRetryOOM(RuntimeError):
    """OOM that requires splitting the input to make progress."""


class _OomInjector:
    """Test hook: inject a synthetic OOM on the Nth guarded task invocation
    (reference analogue: RmmSpark.OomInjectionType)."""

    def __init__(self):
        self._lock = threading.Lock()
        self._remaining = 0
        self._split = False

    def arm(self, n: int, split: bool = False):
        with self._lock:
            self._remaining = n
            self._split = split

    def maybe_throw(self):
        with self._lock:
            if self._remaining > 0:
                self._remaining -= 1
                if self._remaining == 0:
                    if self._split:
                        raise GpuSplitAndRetryOOM("injected split OOM")
                    raise GpuRetryOOM("injected OOM")


oom_injector = _OomInjector()

_OOM_TYPES = (torch.OutOfMemoryError, GpuRetryOOM, GpuSplitAndRetryOOM) \
    if hasattr(torch, "OutOfMemoryError") else (GpuRetryOOM, GpuSplitAndRetryOOM)


def _release_device_memory():
    if torch.cuda.is_available():
        torch.cuda.empty_cache()
    # ask the spill framework to free device-resident spillables
    from .spill import spill_store
    spill_store.spill_device(target_bytes=None)


def _split_batch(batch) -> List:
    from .. import ops
    from ..column import Column

    n = batch.num_rows
    if n <= 1:
        raise MemoryError("cannot split a single-row batch further")
    half = n // 2
    lo = Column.from_numpy(np.arange(0, half, dtype=np.int32), device=batch.device)
    hi = Column.from_numpy(np.arange(half, n, dtype=np.int32), device=batch.device)
    return [ops.gather(batch, lo), ops.gather(batch, hi)]


def with_retry_split(fn: Callable, batch, max_splits: int = 8) -> List:
    """Run fn(batch); on OOM retry after releasing memory, then
    split-and-retry recursively. Returns a list of result batches."""
    from ..metrics import task_metric_add

    try:
        oom_injector.maybe_throw()
        return [fn(batch)]
    except GpuSplitAndRetryOOM:
        pass  # go straight to split
    except _OOM_TYPES:
        task_metric_add("retryCount", 1)
        _release_device_memory()
        try:
            return [fn(batch)]
        except _OOM_TYPES:
            pass
    if max_splits <= 0:
        raise MemoryError("GPU OOM: retry budget exhausted")
    task_metric_add("splitAndRetryCount", 1)
    parts = _split_batch(batch)
    out: List = []
    for p in parts:
        out.extend(with_retry_split(fn, p, max_splits - 1))
    return out

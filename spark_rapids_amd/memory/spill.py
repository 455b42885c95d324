"""Spill framework: device -> host -> disk stores with resurrection.

Reference analogue: spill/SpillFramework.scala (device/host/disk handle
stores, spill on allocation pressure, aliasing-aware spillability) +
SpillableColumnarBatch.scala. MI355X specifics: D2H staging goes through
pinned host tensors + hipMemcpyAsync (torch non_blocking copies); 288 GB of
HBM3E means spill is rare, but operators still wrap intermediate batches so
the OOM retry path can free memory deterministically.
"""
from __future__ import annotations

import os
import tempfile
import threading
from typing import Dict, List, Optional

import torch

DEVICE = "device"
HOST = "host"
DISK = "disk"


class SpillableBatch:
    """A handle to a ColumnBatch that can migrate device->host->disk and back.
    Call .get() to materialize on the original device; .close() to release."""

    _next_id = 0

    def __init__(self, batch, priority: int = 0):
        from ..column import ColumnBatch  # noqa: F401

        self.priority = priority
        self._lock = threading.RLock()
        self._batch = batch
        self._state = DEVICE if batch.is_cuda else HOST
        self._orig_device = batch.device
        self._disk_path: Optional[str] = None
        self._meta = None
        with _store_lock:
            SpillableBatch._next_id += 1
            self.id = SpillableBatch._next_id
        spill_store.register(self)

    @property
    def state(self):
        return self._state

    @property
    def nbytes(self) -> int:
        with self._lock:
            return self._batch.nbytes if self._batch is not None else 0

    def spill_to_host(self):
        import time as _time

        with self._lock:
            if self._state != DEVICE or self._batch is None:
                return 0
            t0 = _time.perf_counter()
            n = self._batch.nbytes
            self._batch = self._to_host_pinned(self._batch)
            self._state = HOST
            from ..metrics import task_metric_add

            task_metric_add("spillToHostBytes", n)
            task_metric_add("spillTimeMs",
                            (_time.perf_counter() - t0) * 1e3)
            return n

    def _to_host_pinned(self, batch):
        """D2H through the pinned pool (HostAlloc analogue) when
        configured; pageable fallback otherwise."""
        from ..column import ColumnBatch
        from . import host_pool

        if host_pool.pool() is None or not batch.is_cuda:
            return batch.to("cpu")
        self._pinned = getattr(self, "_pinned", [])
        cols = [self._copy_col_host(c) for c in batch.columns]
        torch.cuda.synchronize()  # copies must land before device frees
        return ColumnBatch(cols, batch.num_rows)

    def _copy_col_host(self, c):
        from ..column import Column
        from . import host_pool

        child = c.child
        if isinstance(child, tuple):
            child = tuple(self._copy_col_host(k) for k in child)
        elif child is not None:
            child = self._copy_col_host(child)
        cp = lambda t: host_pool.copy_tensor_to_host(t, self._pinned)
        return Column(c.dtype, c.size, cp(c.data),
                      cp(c.validity) if c.validity is not None else None,
                      cp(c.offsets) if c.offsets is not None else None,
                      c._null_count, child)

    def _release_pinned(self):
        from . import host_pool

        held = getattr(self, "_pinned", None)
        if held:
            host_pool.release(held)

    def spill_to_disk(self):
        with self._lock:
            if self._state == DEVICE:
                self.spill_to_host()
            if self._state != HOST or self._batch is None:
                return 0
            n = self._batch.nbytes
            fd, path = tempfile.mkstemp(prefix="rapids_spill_", suffix=".bin",
                                        dir=_spill_dir())
            os.close(fd)
            tensors, meta = _flatten(self._batch)
            torch.save(tensors, path)
            self._meta = meta
            self._disk_path = path
            self._batch = None
            self._release_pinned()
            self._state = DISK
            from ..metrics import task_metric_add

            task_metric_add("spillToDiskBytes", n)
            return n

    def get(self):
        """Materialize on the original device."""
        with self._lock:
            if self._state == DISK:
                tensors = torch.load(self._disk_path, weights_only=True)
                self._batch = _unflatten(tensors, self._meta)
                os.unlink(self._disk_path)
                self._disk_path = None
                self._state = HOST
            if self._state == HOST and self._orig_device != "cpu":
                self._batch = self._batch.to(self._orig_device)
                # pinned H2D uploads are async on the current stream; the
                # pinned staging can only be reused after they land
                if getattr(self, "_pinned", None):
                    torch.cuda.synchronize()
                self._release_pinned()
                self._state = DEVICE
            return self._batch

    def close(self):
        with self._lock:
            self._batch = None
            self._release_pinned()
            if self._disk_path and os.path.exists(self._disk_path):
                os.unlink(self._disk_path)
            self._disk_path = None
        spill_store.unregister(self)


def _compact(t: torch.Tensor) -> torch.Tensor:
    """Host copy with exactly-sized storage: torch.save serializes the
    WHOLE backing storage, so a small view of a pinned-pool slab must be
    cloned or the save writes the entire slab to disk."""
    h = t.cpu()
    if h.untyped_storage().nbytes() != h.numel() * h.element_size():
        h = h.clone()
    return h


def _flatten(batch):
    tensors: List[torch.Tensor] = []
    meta = {"num_rows": batch.num_rows, "cols": []}
    for c in batch.columns:
        entry = {"dtype": c.dtype, "size": c.size,
                 "has_validity": c.validity is not None,
                 "has_offsets": c.offsets is not None,
                 "null_count": c._null_count}
        tensors.append(_compact(c.data))
        if c.validity is not None:
            tensors.append(_compact(c.validity))
        if c.offsets is not None:
            tensors.append(_compact(c.offsets))
        meta["cols"].append(entry)
    return tensors, meta


def _unflatten(tensors, meta):
    from ..column import Column, ColumnBatch

    cols = []
    i = 0
    for entry in meta["cols"]:
        data = tensors[i]
        i += 1
        validity = None
        offsets = None
        if entry["has_validity"]:
            validity = tensors[i]
            i += 1
        if entry["has_offsets"]:
            offsets = tensors[i]
            i += 1
        cols.append(Column(entry["dtype"], entry["size"], data, validity,
                           offsets, entry["null_count"]))
    return ColumnBatch(cols, meta["num_rows"])


_store_lock = threading.RLock()


def _spill_dir() -> str:
    d = os.environ.get("RAPIDS_SPILL_PATH", "/tmp/rapids_spill")
    os.makedirs(d, exist_ok=True)
    return d


# pool-usage fraction for proactive spill; set by Session from
# spark.rapids.memory.gpu.spillWatermark (None until a session configures)
_watermark: Optional[float] = None


def configure_watermark(w: Optional[float]):
    global _watermark
    _watermark = w


class SpillStore:
    """Registry of live spillables; spills lowest-priority first."""

    def __init__(self):
        self._handles: Dict[int, SpillableBatch] = {}

    def register(self, h: SpillableBatch):
        with _store_lock:
            self._handles[h.id] = h
        if h.state == DEVICE and _watermark is not None:
            # proactive spill: keep pool usage under the watermark so the
            # failure callback is the backstop, not the steady state
            from . import device_pool

            device_pool.maybe_spill(_watermark)

    def unregister(self, h: SpillableBatch):
        with _store_lock:
            self._handles.pop(h.id, None)

    def device_bytes(self) -> int:
        with _store_lock:
            return sum(h.nbytes for h in self._handles.values()
                       if h.state == DEVICE)

    def spill_device(self, target_bytes: Optional[int] = None) -> int:
        """Spill device-resident handles to host until target_bytes freed
        (None = spill everything spillable)."""
        freed = 0
        with _store_lock:
            handles = sorted((h for h in self._handles.values()
                              if h.state == DEVICE), key=lambda h: h.priority)
        for h in handles:
            freed += h.spill_to_host()
            if target_bytes is not None and freed >= target_bytes:
                break
        return freed

    def spill_host_to_disk(self, target_bytes: Optional[int] = None) -> int:
        freed = 0
        with _store_lock:
            handles = sorted((h for h in self._handles.values()
                              if h.state == HOST), key=lambda h: h.priority)
        for h in handles:
            freed += h.spill_to_disk()
            if target_bytes is not None and freed >= target_bytes:
                break
        return freed


spill_store = SpillStore()

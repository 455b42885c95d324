"""Host-memory scan cache keyed by (path, mtime, columns).

Reference analogue: the filecache subsystem (sql-plugin filecache package)
that caches input data to avoid re-reading hot files. Enabled via
spark.rapids.filecache.enabled; entries invalidate when the file mtime
changes. Batches are cached on the HOST (cpu) so cached bytes do not
occupy HBM; the scan re-uploads on use (PCIe upload is far cheaper than
re-decoding + IO).
"""
from __future__ import annotations

import os
import threading
from typing import Dict, Optional, Tuple

from ..column import ColumnBatch

_lock = threading.Lock()
_cache: Dict[Tuple, ColumnBatch] = {}
_enabled = False


def configure(enabled: bool):
    global _enabled
    _enabled = enabled
    if not enabled:
        with _lock:
            _cache.clear()


def _key(path: str, columns):
    try:
        mtime = os.stat(path).st_mtime_ns
    except OSError:
        return None
    return (os.path.abspath(path), mtime,
            tuple(columns) if columns else None)


def get_cached(path: str, columns) -> Optional[ColumnBatch]:
    if not _enabled:
        return None
    k = _key(path, columns)
    with _lock:
        return _cache.get(k)


def put_cached(path: str, columns, batch: ColumnBatch):
    if not _enabled:
        return
    k = _key(path, columns)
    if k is None:
        return
    with _lock:
        if isinstance(batch, list):
            _cache[k] = [b.cpu() for b in batch]
        else:
            _cache[k] = batch.cpu()

"""Device-memory leak detection (reference analogue: the RMM leak-debug /
allocation-tracking facilities). Snapshot the torch-ROCm caching
allocator before and after a scope and flag growth beyond a tolerance —
used by tests to catch operators that keep device buffers alive.
"""
from __future__ import annotations

from contextlib import contextmanager

import torch


def allocated_bytes() -> int:
    if not torch.cuda.is_available():
        return 0
    from ..memory import device_pool

    if device_pool.is_active():
        # torch's stats APIs don't cover a pluggable allocator; the hipdf
        # pool tracks its own usage
        import hipdf

        return int(hipdf.pool_used())
    return int(torch.cuda.memory_allocated())


@contextmanager
def assert_no_leak(tolerance_bytes: int = 1 << 20):
    """Fails if device allocations grew by more than tolerance across the
    scope (after releasing python references the caller dropped)."""
    import gc

    gc.collect()
    before = allocated_bytes()
    yield
    gc.collect()
    after = allocated_bytes()
    grown = after - before
    assert grown <= tolerance_bytes, (
        f"device memory grew by {grown} bytes (> {tolerance_bytes})")

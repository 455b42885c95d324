"""Pinned host memory pool (reference analogue: HostAlloc.scala —
pinned-first host allocation with a size limit and pageable fallback,
fed by spark.rapids.memory.pinnedPool.size).

MI355X rationale: spill staging and resurrect uploads ride hipMemcpyAsync,
which only overlaps with compute when the host side is PINNED. The pool
hands out 16-byte-aligned sub-allocations of large hipHostMalloc-backed
slabs (torch pin_memory tensors) with address-ordered free-list
coalescing, so per-batch spills do not pay a hipHostMalloc/Free
(~ms-scale, device-synchronizing) each time.
"""
from __future__ import annotations

import threading
from typing import Dict, List, Optional, Tuple

import torch

_ALIGN = 16


class PinnedPool:
    """Sub-allocator over pinned slabs; returns uint8 tensor views."""

    def __init__(self, capacity_bytes: int, slab_bytes: int = 64 << 20):
        self.capacity = capacity_bytes
        self.slab_bytes = slab_bytes
        self._slabs: List[torch.Tensor] = []
        # address-ordered free ranges per slab: list of (off, size)
        self._free: List[List[Tuple[int, int]]] = []
        self._live: Dict[int, Tuple[int, int, int]] = {}  # id -> slab,off,sz
        self._lock = threading.Lock()
        self.reserved = 0
        self.used = 0

    def _new_slab(self, need: int) -> Optional[int]:
        size = max(self.slab_bytes, need)
        if self.reserved + size > self.capacity:
            # cap the last slab to the remaining capacity
            size = self.capacity - self.reserved
            if size < need:
                return None
        try:
            t = torch.empty(size, dtype=torch.uint8, pin_memory=True)
        except RuntimeError:
            # no CUDA runtime (CPU test boxes): plain host slab so the
            # allocator logic still exercises; copies just aren't async
            t = torch.empty(size, dtype=torch.uint8)
        self._slabs.append(t)
        self._free.append([(0, size)])
        self.reserved += size
        return len(self._slabs) - 1

    def alloc(self, nbytes: int) -> Optional[torch.Tensor]:
        """A pinned uint8 view of nbytes, or None (caller falls back to
        pageable)."""
        if nbytes <= 0:
            nbytes = 1
        need = (nbytes + _ALIGN - 1) // _ALIGN * _ALIGN
        with self._lock:
            for si, free in enumerate(self._free):
                for fi, (off, sz) in enumerate(free):
                    if sz >= need:
                        if sz == need:
                            free.pop(fi)
                        else:
                            free[fi] = (off + need, sz - need)
                        return self._take(si, off, need, nbytes)
            si = self._new_slab(need)
            if si is None:
                return None
            off, sz = self._free[si].pop(0)
            if sz > need:
                self._free[si].insert(0, (off + need, sz - need))
            return self._take(si, off, need, nbytes)

    def _take(self, si: int, off: int, need: int, nbytes: int):
        view = self._slabs[si][off:off + nbytes]
        self._live[id(view)] = (si, off, need)
        self.used += need
        return view

    def free(self, view: torch.Tensor):
        with self._lock:
            ent = self._live.pop(id(view), None)
            if ent is None:
                return
            si, off, need = ent
            self.used -= need
            free = self._free[si]
            # insert address-ordered and coalesce neighbours
            lo, hi = 0, len(free)
            while lo < hi:
                mid = (lo + hi) // 2
                if free[mid][0] < off:
                    lo = mid + 1
                else:
                    hi = mid
            free.insert(lo, (off, need))
            if lo + 1 < len(free) and free[lo][0] + free[lo][1] == \
                    free[lo + 1][0]:
                o, s = free.pop(lo)
                free[lo] = (o, s + free[lo][1])
            if lo > 0 and free[lo - 1][0] + free[lo - 1][1] == free[lo][0]:
                o, s = free.pop(lo - 1)
                free[lo - 1] = (o, s + free[lo - 1][1])

    def stats(self):
        with self._lock:
            return {"capacity": self.capacity, "reserved": self.reserved,
                    "used": self.used, "slabs": len(self._slabs)}


_pool: Optional[PinnedPool] = None
_pool_lock = threading.Lock()


def configure(capacity_bytes: Optional[int]):
    """Called by Session from spark.rapids.memory.pinnedPool.size;
    0/None disables (spill uses pageable host memory)."""
    global _pool
    with _pool_lock:
        _pool = PinnedPool(capacity_bytes) if capacity_bytes else None


def pool() -> Optional[PinnedPool]:
    return _pool


def copy_tensor_to_host(t: torch.Tensor, held: List[torch.Tensor]
                        ) -> torch.Tensor:
    """D2H copy into a pinned view when the pool has room (async-capable),
    else a pageable .cpu() copy. Pinned backing views are appended to
    `held` so the caller can free them."""
    p = _pool
    if p is None or not t.is_cuda:
        return t.cpu()
    nbytes = t.numel() * t.element_size()
    buf = p.alloc(nbytes)
    if buf is None:
        return t.cpu()
    held.append(buf)
    view = buf.view(t.dtype)[:t.numel()]
    view.copy_(t, non_blocking=True)
    return view


def release(held: List[torch.Tensor]):
    p = _pool
    if p is None:
        return
    for b in held:
        p.free(b)
    held.clear()

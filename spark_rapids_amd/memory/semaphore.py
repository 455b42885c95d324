"""GPU semaphore: bound the number of concurrent tasks using the GPU.

Reference analogue: GpuSemaphore.scala (spark.rapids.sql.concurrentGpuTasks,
priority by task attempt). Here tasks are python threads running query
partitions; the semaphore keeps device working sets bounded so concurrent
queries don't interleave OOMs. Priority: lower number acquires first when
contended (FIFO within priority).
"""
from __future__ import annotations

import heapq
import threading
from contextlib import contextmanager


class PrioritySemaphore:
    def __init__(self, permits: int):
        self._permits = permits
        self._lock = threading.Lock()
        self._cond = threading.Condition(self._lock)
        self._waiters = []  # heap of (priority, seq)
        self._seq = 0
        self.contended = 0  # acquisitions that had to wait (metrics)

    def acquire(self, priority: int = 0):
        import time as _time

        t0 = _time.perf_counter()
        with self._cond:
            self._seq += 1
            me = (priority, self._seq)
            heapq.heappush(self._waiters, me)
            waited = False
            while not (self._permits > 0 and self._waiters[0] == me):
                waited = True
                self._cond.wait()
            if waited:
                self.contended += 1
            heapq.heappop(self._waiters)
            self._permits -= 1
            self._cond.notify_all()
        if waited:
            from ..metrics import task_metric_add

            task_metric_add("semaphoreWaitMs",
                            (_time.perf_counter() - t0) * 1e3)

    def release(self):
        with self._cond:
            self._permits += 1
            self._cond.notify_all()


class GpuSemaphore:
    _instance = None
    _ilock = threading.Lock()

    def __init__(self, max_concurrent: int = 4):
        self._sem = PrioritySemaphore(max_concurrent)
        self._held = threading.local()

    @classmethod
    def get(cls) -> "GpuSemaphore":
        with cls._ilock:
            if cls._instance is None:
                cls._instance = GpuSemaphore()
            return cls._instance

    @classmethod
    def initialize(cls, max_concurrent: int):
        with cls._ilock:
            cls._instance = GpuSemaphore(max_concurrent)

    def acquire_if_necessary(self, priority: int = 0):
        if getattr(self._held, "count", 0) == 0:
            self._sem.acquire(priority)
        self._held.count = getattr(self._held, "count", 0) + 1

    def release_if_necessary(self):
        count = getattr(self._held, "count", 0)
        if count > 0:
            self._held.count = count - 1
            if self._held.count == 0:
                self._sem.release()

    @contextmanager
    def held(self, priority: int = 0):
        self.acquire_if_necessary(priority)
        try:
            yield
        finally:
            self.release_if_necessary()

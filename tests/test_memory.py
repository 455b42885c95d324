"""Spill framework + OOM retry + semaphore unit tests (CPU-side semantics;
reference analogues: SpillFrameworkSuite, WithRetrySuite, GpuSemaphoreSuite)."""
import threading

import pytest

from spark_rapids_amd import Column, ColumnBatch, INT64, Session, col
from spark_rapids_amd.memory.retry import (GpuRetryOOM, GpuSplitAndRetryOOM,
                                           oom_injector, with_retry_split)
from spark_rapids_amd.memory.semaphore import GpuSemaphore, PrioritySemaphore
from spark_rapids_amd.memory.spill import SpillableBatch, spill_store


def _batch(n=10):
    return ColumnBatch([Column.from_pylist(list(range(n)), INT64)])


def test_spill_host_to_disk_roundtrip(tmp_path, monkeypatch):
    monkeypatch.setenv("RAPIDS_SPILL_PATH", str(tmp_path))
    b = _batch()
    h = SpillableBatch(b)
    assert h.state == "host"
    freed = h.spill_to_disk()
    assert freed > 0
    assert h.state == "disk"
    got = h.get()
    assert got.columns[0].to_pylist() == list(range(10))
    h.close()


def test_spill_store_spills_by_priority(tmp_path, monkeypatch):
    monkeypatch.setenv("RAPIDS_SPILL_PATH", str(tmp_path))
    h1 = SpillableBatch(_batch(), priority=5)
    h2 = SpillableBatch(_batch(), priority=1)
    n = spill_store.spill_host_to_disk(target_bytes=1)
    assert n > 0
    assert h2.state == "disk"  # lower priority spilled first
    assert h1.state == "host"
    h1.close()
    h2.close()


def test_retry_success_is_passthrough():
    out = with_retry_split(lambda b: b, _batch())
    assert len(out) == 1


def test_retry_on_injected_oom():
    calls = []

    def task(b):
        calls.append(b.num_rows)
        return b

    oom_injector.arm(1)
    out = with_retry_split(task, _batch())
    assert len(out) == 1 and len(calls) == 1


def test_split_and_retry_on_injected_split_oom():
    sizes = []

    def task(b):
        sizes.append(b.num_rows)
        return b

    oom_injector.arm(1, split=True)
    out = with_retry_split(task, _batch(10))
    total = sum(b.num_rows for b in out)
    assert total == 10
    assert len(out) == 2
    assert sizes == [5, 5]


def test_filter_with_injected_split_oom(session):
    df = session.create_dataframe({"a": list(range(100))})
    oom_injector.arm(1, split=True)
    assert df.filter(col("a") >= 50).count() == 50


def test_semaphore_limits_concurrency():
    sem = PrioritySemaphore(2)
    active = []
    peak = []
    lock = threading.Lock()

    def worker():
        sem.acquire()
        with lock:
            active.append(1)
            peak.append(len(active))
        import time

        time.sleep(0.01)
        with lock:
            active.pop()
        sem.release()

    threads = [threading.Thread(target=worker) for _ in range(8)]
    for t in threads:
        t.start()
    for t in threads:
        t.join()
    assert max(peak) <= 2


def test_gpu_semaphore_reentrant():
    GpuSemaphore.initialize(1)
    g = GpuSemaphore.get()
    with g.held():
        with g.held():  # re-entry by same thread must not deadlock
            pass

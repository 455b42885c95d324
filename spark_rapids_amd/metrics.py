"""Per-operator metrics + optional rocTX ranges.

Reference analogue: GpuMetric/GpuTaskMetrics (GpuMetrics.scala:89-160,
GpuTaskMetrics.scala) and the NvtxRange registry (NvtxRangeWithDoc.scala) —
on MI355X the marker API is rocTX, reached through torch.cuda.nvtx which
maps onto roctx under ROCm. Metrics are attached to each physical exec and
surfaced via DataFrame.metrics() after an action.
"""
from __future__ import annotations

import time
from typing import Dict, List

import torch

_ROCTX = False


def enable_roctx(on: bool):
    global _ROCTX
    _ROCTX = on and torch.cuda.is_available()


def instrument(exec_) -> None:
    """Wrap every exec's execute() with wall-time + row/batch counters and
    (optionally) roctx ranges. Times are inclusive of children."""
    seen = set()

    def _wrap(e):
        if id(e) in seen or getattr(e, "_instrumented", False):
            return
        seen.add(id(e))
        e._instrumented = True
        orig = e.execute
        e.metrics = {"opTimeMs": 0.0, "numOutputRows": 0,
                     "numOutputBatches": 0}
        from .tools import lore

        lore_dir = lore.next_exec_dir(e.name())

        def wrapped(_orig=orig, _e=e, _lore=lore_dir):
            it = _orig()
            while True:
                t0 = time.perf_counter()
                if _ROCTX:
                    torch.cuda.nvtx.range_push(_e.name())
                try:
                    batch = next(it)
                except StopIteration:
                    if _ROCTX:
                        torch.cuda.nvtx.range_pop()
                    _e.metrics["opTimeMs"] += (time.perf_counter() - t0) * 1e3
                    return
                finally:
                    pass
                if _ROCTX:
                    torch.cuda.nvtx.range_pop()
                _e.metrics["opTimeMs"] += (time.perf_counter() - t0) * 1e3
                _e.metrics["numOutputRows"] += batch.num_rows
                _e.metrics["numOutputBatches"] += 1
                if _lore is not None:
                    from .tools import lore as _l

                    _l.dump_batch(_lore, batch, _e.schema,
                                  _e.metrics["numOutputBatches"] - 1)
                yield batch

        e.execute = wrapped
        for c in e.children:
            _wrap(c)

    _wrap(exec_)


_task_lock = __import__("threading").Lock()
_TASK_ZERO = {
    "semaphoreWaitMs": 0.0, "spillToHostBytes": 0, "spillToDiskBytes": 0,
    "spillTimeMs": 0.0, "retryCount": 0, "splitAndRetryCount": 0,
}
_task = dict(_TASK_ZERO)


def task_metric_add(key: str, value) -> None:
    """Bump a task-level accumulator (reference: GpuTaskMetrics.scala —
    semaphore wait, retry counts, spill time/bytes, max device memory)."""
    with _task_lock:
        _task[key] = _task.get(key, 0) + value


def task_metrics() -> Dict:
    """Current task-level accumulators; maxDeviceMemoryBytes reads the
    device pool's live high watermark when the pool is active."""
    with _task_lock:
        out = dict(_task)
    try:
        from .memory import device_pool

        if device_pool.is_active():
            out["maxDeviceMemoryBytes"] = \
                device_pool.stats()["high_watermark"]
    except Exception:  # noqa: BLE001 - pool not active / no GPU
        pass
    return out


def reset_task_metrics() -> None:
    with _task_lock:
        _task.clear()
        _task.update(_TASK_ZERO)


def collect_metrics(exec_) -> List[Dict]:
    out = []

    def _walk(e, depth):
        out.append({"exec": e.describe(), "depth": depth, **e.metrics})
        for c in e.children:
            _walk(c, depth + 1)

    _walk(exec_, 0)
    return out

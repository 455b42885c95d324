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


def collect_metrics(exec_) -> List[Dict]:
    out = []

    def _walk(e, depth):
        out.append({"exec": e.describe(), "depth": depth, **e.metrics})
        for c in e.children:
            _walk(c, depth + 1)

    _walk(exec_, 0)
    return out

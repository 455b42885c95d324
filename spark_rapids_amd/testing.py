"""Public CPU-vs-GPU equality test harness.

Reference analogue: integration_tests' assert_gpu_and_cpu_are_equal_collect
(asserts.py:589-711) + the session flip via spark.rapids.sql.enabled: run
the same query on the GPU-enabled engine and on the CPU backend, then
deep-compare (None/NaN aware, approximate floats, optional row-order
independence). Used by the in-repo tests and available to users.
"""
from __future__ import annotations

import math
from typing import Callable, Dict, Optional

from .api import DataFrame, Session


def _rows_equal(g, c, approx_float: bool, rel: float) -> bool:
    if g is None or c is None:
        return g is None and c is None
    if isinstance(c, float) and isinstance(g, float):
        if math.isnan(c) or math.isnan(g):
            return math.isnan(c) and math.isnan(g)
        if approx_float:
            tol = rel * max(abs(c), abs(g), 1e-300)
            return abs(g - c) <= max(tol, 1e-12)
    return g == c


def assert_gpu_and_cpu_are_equal(
        query: Callable[[Session], DataFrame],
        conf: Optional[Dict] = None,
        ignore_order: bool = True,
        approx_float: bool = True,
        rel: float = 1e-9,
        require_gpu_plan: bool = True):
    """Build the same query against a GPU session and a CPU session and
    assert equal results. `query` receives the Session and returns a
    DataFrame. With require_gpu_plan, also asserts at least one Gpu* exec
    is present in the GPU plan (guards against silent whole-plan fallback)."""
    import torch

    gpu_conf = dict(conf or {})
    gpu_conf["spark.rapids.sql.enabled"] = True
    cpu_conf = dict(conf or {})
    cpu_conf["spark.rapids.sql.enabled"] = False

    gdf = query(Session(gpu_conf))
    if require_gpu_plan and torch.cuda.is_available():
        tree = gdf.physical_plan().tree_string()
        assert "Gpu" in tree, f"no GPU exec in plan:\n{tree}"
    grows = gdf.collect()
    crows = query(Session(cpu_conf)).collect()
    assert len(grows) == len(crows), \
        f"row count differs: gpu={len(grows)} cpu={len(crows)}"
    if ignore_order:
        grows = sorted(grows, key=repr)
        crows = sorted(crows, key=repr)
    for i, (g, c) in enumerate(zip(grows, crows)):
        assert len(g) == len(c), f"row {i} arity"
        for j, (gv, cv) in enumerate(zip(g, c)):
            assert _rows_equal(gv, cv, approx_float, rel), \
                f"row {i} col {j}: gpu={gv!r} cpu={cv!r}"

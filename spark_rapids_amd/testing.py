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
        if math.isinf(c) or math.isinf(g):
            return g == c  # same-signed infinity only
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


# ---------------------------------------------------------------------------
# Typed data generators (reference analogue: integration_tests data_gen.py —
# seeded typed generators with nulls and special values)
# ---------------------------------------------------------------------------

_INT_BOUNDS = {
    "tinyint": (-128, 127), "smallint": (-32768, 32767),
    "int": (-2**31, 2**31 - 1), "bigint": (-2**63, 2**63 - 1),
}


def gen_column(dtype, n: int, seed: int = 0, null_frac: float = 0.1,
               special_frac: float = 0.1):
    """A seeded python value list of `dtype` with nulls and the type's
    special values mixed in (min/max/0 for ints; nan/±inf/-0.0 for
    floats; empty/unicode/whitespace strings; boundary decimals;
    epoch-edge dates/timestamps)."""
    import decimal

    import numpy as np

    from .types import TypeId

    rng = np.random.default_rng(seed)
    tid = dtype.id
    out = []
    specials = []
    if tid.value in _INT_BOUNDS:
        lo, hi = _INT_BOUNDS[tid.value]
        specials = [lo, hi, 0, -1, 1]
        base = lambda: int(rng.integers(max(lo, -10**6),
                                        min(hi, 10**6) + 1))
    elif tid in (TypeId.FLOAT32, TypeId.FLOAT64):
        specials = [float("nan"), float("inf"), float("-inf"), -0.0, 0.0]
        base = lambda: float(rng.uniform(-1e6, 1e6))
    elif tid is TypeId.BOOL:
        specials = [True, False]
        base = lambda: bool(rng.integers(0, 2))
    elif tid is TypeId.STRING:
        pool = ["", " ", "a", "Z9", "spark rapids", "wörld", "\t tab",
                "NULL", "null", "ñ", "0", "-1.5e3"]
        specials = pool
        base = lambda: "".join(
            chr(int(c)) for c in rng.integers(97, 123, int(
                rng.integers(0, 9))))
    elif dtype.is_decimal:
        q = decimal.Decimal(1).scaleb(-dtype.scale)
        mx = decimal.Decimal(10) ** (dtype.precision - dtype.scale) - q
        specials = [decimal.Decimal(0).quantize(q), mx, -mx, q, -q]
        base = lambda: (decimal.Decimal(int(rng.integers(
            -10**min(dtype.precision, 12),
            10**min(dtype.precision, 12)))).scaleb(-dtype.scale))
    elif tid is TypeId.DATE32:
        specials = [0, -719162, 2932896, 1, -1]  # epoch, 0001, 9999
        base = lambda: int(rng.integers(-30000, 30000))
    elif tid is TypeId.TIMESTAMP:
        specials = [0, 1, -1, 951782400000000]
        base = lambda: int(rng.integers(-2**40, 2**40))
    else:
        raise NotImplementedError(f"gen_column for {dtype}")
    for _ in range(n):
        r = rng.random()
        if r < null_frac:
            out.append(None)
        elif r < null_frac + special_frac and specials:
            out.append(specials[int(rng.integers(0, len(specials)))])
        else:
            out.append(base())
    return out

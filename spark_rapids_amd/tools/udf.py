"""UDF compiler analogue: trace a python scalar function into the
expression DSL so it runs as fused GPU kernels instead of a host loop.

Reference analogue: the udf-compiler module (bytecode-to-Catalyst
translation of simple Scala UDFs). Here the engine owns the expression
DSL, so tracing is direct: the function is called once with ColumnRef
expressions as arguments — if it only uses supported operators
(arithmetic, comparisons, boolean logic, the string/conditional DSL
methods), the returned object IS an expression tree and executes on the
GPU. Functions that branch on data values or call unsupported libraries
raise UdfFallback; callers then use DataFrame.map_batches (the CPU
bridge), mirroring the reference's CPU-UDF fallback.
"""
from __future__ import annotations

from typing import Callable

from ..expr.expressions import Expression, col


class UdfFallback(Exception):
    """The function could not be traced to the expression DSL."""


def compile_udf(fn: Callable, *input_columns: str) -> Expression:
    """Trace fn(col_a, col_b, ...) into an Expression.

    >>> expr = compile_udf(lambda a, b: (a + b) * 2.0, "x", "y")
    >>> df.with_column("z", expr)
    """
    args = [col(c) for c in input_columns]
    try:
        out = fn(*args)
    except Exception as e:  # data-dependent branch, foreign call, ...
        raise UdfFallback(
            f"udf not traceable to the expression DSL: {e!r}; use "
            "DataFrame.map_batches for arbitrary python") from e
    if not isinstance(out, Expression):
        raise UdfFallback(
            f"udf returned {type(out).__name__}, not an expression — it "
            "likely materialized values; use DataFrame.map_batches")
    return out

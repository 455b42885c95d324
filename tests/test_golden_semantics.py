"""Golden Spark-semantics corpus.

The round-1 verdict asked for golden outputs from real Apache Spark. This
container has no JVM and no network, so a Spark run is impossible here;
instead this file pins a LITERAL corpus of expected outputs for the corner
semantics where a shared CPU/GPU misunderstanding could hide: every
expected value below is written down by hand from the Spark 3.5 semantics
(HALF_UP decimals, non-ANSI overflow-to-NULL, Kleene logic, NaN ordering,
UTF8String casts, Java regex). Both backends are checked against the same
literals — neither backend is the oracle for the other here.

If a future round gets JVM access: regenerate with
  spark.sql(q).collect()  for each CASES entry and diff.
"""
from decimal import Decimal

import pytest

import spark_rapids_amd as sr
from spark_rapids_amd import DType, col
from spark_rapids_amd.types import BOOL, FLOAT64, INT32, INT64, STRING


def _sessions():
    return [sr.Session({"spark.rapids.sql.enabled": False})]


def _gpu_session():
    import torch

    return sr.Session() if torch.cuda.is_available() else None


def run_case(s, build):
    return build(s)


# ---- decimal HALF_UP / overflow ------------------------------------------

def _dec_cases(s):
    df = s.create_dataframe({"x": [1]})
    from spark_rapids_amd.expr.expressions import Literal

    d = lambda v, p, q: Literal(Decimal(v), DType.decimal(p, q))  # noqa: E731
    out = df.select(
        # spark: SELECT CAST(2.5 AS DECIMAL(3,0))  -> 3 (HALF_UP)
        d("2.5", 3, 1).cast(DType.decimal(3, 0)).alias("a"),
        # spark: CAST(-2.5 AS DECIMAL(3,0)) -> -3 (away from zero)
        d("-2.5", 3, 1).cast(DType.decimal(3, 0)).alias("b"),
        # spark: CAST(99.995 AS DECIMAL(4,2)) -> NULL (overflow after round)
        d("99.995", 5, 3).cast(DType.decimal(4, 2)).alias("c"),
        # spark: 1.13 * 1.13 with DECIMAL(3,2) operands -> 1.2769 (7,4)
        (d("1.13", 3, 2) * d("1.13", 3, 2)).alias("m"),
        # spark: DECIMAL(5,2) 999.99 + 999.99 -> 1999.98 at (6,2)
        (d("999.99", 5, 2) + d("999.99", 5, 2)).alias("s"),
        # spark: 1.00 / 3.00 over DECIMAL(3,2) -> 0.3333333 at scale 7?
        # DecimalPrecision: s = max(6, 2+3+1) = 6 -> 0.333333
        (d("1.00", 3, 2) / d("3.00", 3, 2)).alias("q"),
    ).to_pydict()
    return out


GOLDEN_DEC = {
    "a": [Decimal("3")],
    "b": [Decimal("-3")],
    "c": [None],
    "m": [Decimal("1.2769")],
    "s": [Decimal("1999.98")],
    "q": [Decimal("0.333333")],
}


# ---- casts (string <-> numeric, UTF8String semantics) --------------------

def _cast_cases(s):
    df = s.create_dataframe({
        "s": ["42", " 42\t", "12.9", "-12.5", "1e2", "2147483648",
              "abc", "", None, "0x1F"],
    })
    return df.select(col("s").cast(INT32).alias("i"),
                     col("s").cast(DType.decimal(10, 2)).alias("d"),
                     col("s").cast(FLOAT64).alias("f")).to_pydict()


GOLDEN_CAST = {
    # spark UTF8String.toInt: trims, truncates fractions, no exponent,
    # overflow/garbage -> NULL
    "i": [42, 42, 12, -12, None, None, None, None, None, None],
    # 2147483648.00 needs 12 digits > DECIMAL(10,2): overflow -> NULL
    "d": [Decimal("42.00"), Decimal("42.00"), Decimal("12.90"),
          Decimal("-12.50"), Decimal("100.00"), None,
          None, None, None, None],
    "f": [42.0, 42.0, 12.9, -12.5, 100.0, 2147483648.0, None, None, None,
          None],
}


# ---- integer arithmetic (non-ANSI wrap, div/0 -> NULL) -------------------

def _int_cases(s):
    df = s.create_dataframe(
        {"a": [2147483647, -2147483648, 7, 7],
         "b": [1, 1, 0, -2]},
        dtypes={"a": INT32, "b": INT32})
    return df.select((col("a") + col("b")).alias("add"),
                     (col("a") % col("b")).alias("mod"),
                     (col("a") / col("b")).alias("div")).to_pydict()


GOLDEN_INT = {
    # spark non-ANSI: int32 overflow wraps (java semantics)
    "add": [-2147483648, -2147483647, 7, 5],
    # spark: x % 0 -> NULL; sign follows the dividend
    "mod": [0, 0, None, 1],
    # spark `/` is always double; x / 0 -> NULL
    "div": [2147483647.0, -2147483648.0, None, -3.5],
}


# ---- Kleene logic / null-safe equality -----------------------------------

def _bool_cases(s):
    df = s.create_dataframe({"a": [True, True, False, None, None],
                             "b": [None, False, None, None, True]},
                            dtypes={"a": BOOL, "b": BOOL})
    from spark_rapids_amd.expr.expressions import BinaryExpr

    return df.select((col("a") & col("b")).alias("and_"),
                     (col("a") | col("b")).alias("or_"),
                     BinaryExpr("eq_null_safe", col("a"),
                                col("b")).alias("eqns")).to_pydict()


GOLDEN_BOOL = {
    "and_": [None, False, False, None, None],
    "or_": [True, True, None, None, True],
    # <=> never returns NULL
    "eqns": [False, False, False, True, False],
}


# ---- float semantics (NaN grouping/ordering, -0.0) -----------------------

def _float_cases(s):
    nan = float("nan")
    df = s.create_dataframe({"v": [1.0, nan, -0.0, 0.0, nan, None]})
    srt = [r[0] for r in df.sort("v").collect()]
    grp = sorted(
        ((r[0], r[1]) for r in
         df.group_by("v").agg(sr.count_star()).collect()),
        key=lambda t: (repr(t[0]), t[1]))
    return srt, grp


def _check_float(srt, grp):
    import math

    # spark ordering: NULL first (asc), NaN sorts LAST (greater than all)
    assert srt[0] is None
    assert srt[1:4] == [-0.0, 0.0, 1.0] or srt[1:4] == [0.0, -0.0, 1.0]
    assert math.isnan(srt[4]) and math.isnan(srt[5])
    # grouping: NaN == NaN (one group of 2); -0.0 == 0.0 (one group of 2)
    counts = {}
    for v, c in grp:
        key = "nan" if (isinstance(v, float) and math.isnan(v)) else \
            ("zero" if v == 0 else v)
        counts[key] = counts.get(key, 0) + c
    assert counts["nan"] == 2
    assert counts["zero"] == 2
    assert counts[1.0] == 1
    assert counts[None] == 1


# ---- datetime edges ------------------------------------------------------

def _dt_cases(s):
    from spark_rapids_amd.expr.datetime import date_format, to_timestamp

    df = s.create_dataframe({"t": ["1969-12-31 23:59:59",
                                   "1970-01-01 00:00:00",
                                   "2000-02-29 12:00:00",
                                   "1900-02-28 00:00:00"]})
    ts = df.select(to_timestamp(col("t")).alias("ts"))
    return ts.select(
        date_format(col("ts"), "yyyy-MM-dd HH:mm:ss").alias("rt"),
        col("ts").cast(INT64).alias("us")).to_pydict()


GOLDEN_DT = {
    "rt": ["1969-12-31 23:59:59", "1970-01-01 00:00:00",
           "2000-02-29 12:00:00", "1900-02-28 00:00:00"],
    # 1900-02-28 is 25509 days before epoch (1900 is NOT a leap year)
    "us": [-1_000_000, 0, 951_825_600_000_000, -2_203_977_600_000_000],
}


CASES = [
    ("decimal", _dec_cases, GOLDEN_DEC),
    ("casts", _cast_cases, GOLDEN_CAST),
    ("ints", _int_cases, GOLDEN_INT),
    ("bools", _bool_cases, GOLDEN_BOOL),
    ("datetime", _dt_cases, GOLDEN_DT),
]


@pytest.mark.parametrize("name,build,golden",
                         CASES, ids=[c[0] for c in CASES])
def test_cpu_matches_golden(name, build, golden):
    s = sr.Session({"spark.rapids.sql.enabled": False})
    got = build(s)
    for k, exp in golden.items():
        assert got[k] == exp, (name, k, got[k], exp)


def test_cpu_float_semantics():
    s = sr.Session({"spark.rapids.sql.enabled": False})
    _check_float(*_float_cases(s))


@pytest.mark.gpu
@pytest.mark.parametrize("name,build,golden",
                         CASES, ids=[c[0] for c in CASES])
def test_gpu_matches_golden(name, build, golden):
    s = sr.Session()
    got = build(s)
    for k, exp in golden.items():
        assert got[k] == exp, (name, k, got[k], exp)


@pytest.mark.gpu
def test_gpu_float_semantics():
    _check_float(*_float_cases(sr.Session()))

"""Spark-semantics expression tests on the CPU backend (the golden reference
that GPU numerics tests later compare against)."""
import math

import pytest

from spark_rapids_amd import Session, col, lit, when, FLOAT64, INT64, INT32


@pytest.fixture
def df(session):
    return session.create_dataframe({
        "i": [1, 2, None, -4, 5],
        "j": [10, 0, 3, None, 2],
        "f": [1.5, -2.0, 0.0, None, float("nan")],
        "b": [True, False, None, True, False],
    })


def test_arith_null_propagation(df):
    out = df.select((col("i") + col("j")).alias("s")).to_pydict()["s"]
    assert out == [11, 2, None, None, 7]


def test_division_by_zero_is_null(df):
    out = df.select((col("i") / col("j")).alias("d")).to_pydict()["d"]
    assert out[0] == pytest.approx(0.1)
    assert out[1] is None  # 2/0 -> NULL (non-ANSI Spark)
    assert out[2] is None


def test_mod_by_zero_null(df):
    out = df.select((col("i") % col("j")).alias("m")).to_pydict()["m"]
    assert out == [1, None, None, None, 1]


def test_mod_sign_follows_dividend(session):
    df = session.create_dataframe({"a": [-7, 7, -7], "b": [3, -3, -3]})
    out = df.select((col("a") % col("b")).alias("m")).to_pydict()["m"]
    assert out == [-1, 1, -1]  # Spark % follows dividend sign


def test_kleene_and_or(df):
    a = df.select((col("b") & lit(True)).alias("x")).to_pydict()["x"]
    assert a == [True, False, None, True, False]
    b = df.select((col("b") & lit(False)).alias("x")).to_pydict()["x"]
    assert b == [False] * 5  # NULL AND FALSE = FALSE
    c = df.select((col("b") | lit(True)).alias("x")).to_pydict()["x"]
    assert c == [True] * 5  # NULL OR TRUE = TRUE
    d = df.select((col("b") | lit(False)).alias("x")).to_pydict()["x"]
    assert d == [True, False, None, True, False]


def test_comparisons(df):
    out = df.select((col("i") > 1).alias("x")).to_pydict()["x"]
    assert out == [False, True, None, False, True]


def test_case_when(df):
    e = when(col("i") > 1, lit(100)).alias("c")
    out = df.select(e).to_pydict()["c"]
    # no ELSE -> null; NULL condition -> null
    assert out == [None, 100, None, None, 100]


def test_cast_float_to_int_truncates(session):
    df = session.create_dataframe({"f": [1.9, -1.9, 2.5]})
    out = df.select(col("f").cast(INT64).alias("i")).to_pydict()["i"]
    assert out == [1, -1, 2]


def test_is_null(df):
    out = df.select(col("i").is_null().alias("x")).to_pydict()["x"]
    assert out == [False, False, True, False, False]
    out2 = df.select(col("i").is_not_null().alias("x")).to_pydict()["x"]
    assert out2 == [True, True, False, True, False][:3] + [True, True][:2]


def test_unary_math(df):
    out = df.select((-col("i")).alias("n")).to_pydict()["n"]
    assert out == [-1, -2, None, 4, -5]
    out = df.select(col("f").is_null().alias("x")).to_pydict()["x"]
    assert out == [False, False, False, True, False]


def test_log_of_nonpositive_is_null(session):
    from spark_rapids_amd.expr.expressions import UnaryExpr

    df = session.create_dataframe({"x": [math.e, 0.0, -1.0]})
    out = df.select(UnaryExpr("log", col("x")).alias("l")).to_pydict()["l"]
    assert out[0] == pytest.approx(1.0)
    assert out[1] is None and out[2] is None


def test_literal_null(df):
    out = df.select((col("i") + lit(None)).alias("x")).to_pydict()["x"]
    assert out == [None] * 5


def test_int_promotion(session):
    df = session.create_dataframe({"a": [1, 2]})
    out_t = df.select((col("a") + lit(1.5)).alias("x")).schema.fields[0].dtype
    assert out_t == FLOAT64


def test_filter_drops_null_predicate(df):
    rows = df.filter(col("i") > 1).select("i").to_pydict()["i"]
    assert rows == [2, 5]


def test_float_nan_ordering_spark_semantics(session):
    """Spark float ordering: NaN == NaN, NaN greater than everything
    (found by the special-value fuzz; CPU backend previously used raw
    numpy semantics where all NaN compares are false)."""
    nan, inf = float("nan"), float("inf")
    df = session.create_dataframe({"f": [nan, 1.0, -0.0, inf]})
    out = df.select((col("f") > 0.0).alias("g"),
                    (col("f") == col("f")).alias("e")).to_pydict()
    assert out["g"] == [True, True, False, True]
    assert out["e"] == [True, True, True, True]

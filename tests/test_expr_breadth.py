"""coalesce / round / greatest / least / isin / date arithmetic tests."""
import pytest

import spark_rapids_amd as sr

from spark_rapids_amd import (DATE32, Session, coalesce, col, date_add,
                              datediff, greatest, isin, least, lit, round_)


def test_coalesce(session):
    df = session.create_dataframe({"a": [None, 2, None], "b": [10, None, None]})
    out = df.select(coalesce(col("a"), col("b"), lit(-1)).alias("c")).to_pydict()["c"]
    assert out == [10, 2, -1]


def test_round_half_up(session):
    df = session.create_dataframe({"x": [2.5, 3.5, -2.5, 1.234, 1.235]})
    out = df.select(round_(col("x")).alias("r")).to_pydict()["r"]
    assert out == [3.0, 4.0, -3.0, 1.0, 1.0]
    out = df.select(round_(col("x"), 2).alias("r")).to_pydict()["r"]
    assert out == [2.5, 3.5, -2.5, 1.23, pytest.approx(1.24)]


def test_greatest_least_skip_nulls(session):
    df = session.create_dataframe({"a": [1, None, None], "b": [5, 3, None]})
    out = df.select(greatest(col("a"), col("b")).alias("g")).to_pydict()["g"]
    assert out == [5, 3, None]
    out = df.select(least(col("a"), col("b")).alias("l")).to_pydict()["l"]
    assert out == [1, 3, None]


def test_isin(session):
    df = session.create_dataframe({"a": [1, 2, 3, None]})
    out = df.filter(isin(col("a"), 1, 3)).to_pydict()["a"]
    assert out == [1, 3]


def test_date_arithmetic(session):
    df = session.create_dataframe({"d": [10957, 10958]},
                                  dtypes={"d": DATE32})  # 2000-01-01, -02
    out = df.select(date_add(col("d"), lit(10)).alias("x")).to_pydict()["x"]
    assert out == [10967, 10968]
    out = df.select(datediff(col("d"), lit(10950)).alias("x")).to_pydict()["x"]
    assert out == [7, 8]
    sch = df.select(datediff(col("d"), lit(10950)).alias("x")).schema
    assert str(sch.fields[0].dtype) == "int"


def test_year_month_day(session):
    df = session.create_dataframe({"d": [10957, 11323]}, dtypes={"d": DATE32})
    from spark_rapids_amd.expr.expressions import UnaryExpr

    y = df.select(UnaryExpr("year", col("d")).alias("y")).to_pydict()["y"]
    m = df.select(UnaryExpr("month", col("d")).alias("m")).to_pydict()["m"]
    d = df.select(UnaryExpr("day", col("d")).alias("dd")).to_pydict()["dd"]
    assert y == [2000, 2001] and m == [1, 1] and d == [1, 1]


def test_timestamp_fields(session):
    from spark_rapids_amd import TIMESTAMP, hour, minute, second

    # 2000-01-01 13:45:30 UTC in micros
    micros = 946_734_330_000_000
    df = session.create_dataframe({"t": [micros, micros + 61_000_000]},
                                  dtypes={"t": TIMESTAMP})
    assert df.select(hour(col("t")).alias("h")).to_pydict()["h"] == [13, 13]
    assert df.select(minute(col("t")).alias("m")).to_pydict()["m"] == [45, 46]
    assert df.select(second(col("t")).alias("s")).to_pydict()["s"] == [30, 31]


def test_dayofweek_quarter(session):
    import datetime

    from spark_rapids_amd import DATE32, dayofweek, quarter

    days = [0, 1, 3, 100, 19000, 7305]
    df = session.create_dataframe({"d": days}, dtypes={"d": DATE32})
    out = df.select(dayofweek(col("d")).alias("w"),
                    quarter(col("d")).alias("q")).to_pydict()
    for dv, w, q in zip(days, out["w"], out["q"]):
        date = datetime.date(1970, 1, 1) + datetime.timedelta(days=dv)
        assert w == (date.weekday() + 1) % 7 + 1
        assert q == (date.month - 1) // 3 + 1


def test_udf_compiler_traces_to_expressions(session):
    from spark_rapids_amd.tools.udf import UdfFallback, compile_udf

    df = session.create_dataframe({"x": [1.0, 2.0, None], "y": [10, 20, 30]})
    expr = compile_udf(lambda a, b: (a + b) * 2.0 - 1.0, "x", "y")
    out = df.with_column("z", expr).to_pydict()["z"]
    assert out == [21.0, 43.0, None]

    cond = compile_udf(lambda a: (a > 1.5) & a.is_not_null(), "x")
    assert df.filter(cond).count() == 1

    with pytest.raises(UdfFallback):
        compile_udf(lambda a: float(a) + 1, "x")  # materializes -> fallback


def test_to_date_unix_timestamp(session):
    import datetime

    from spark_rapids_amd import TIMESTAMP, to_date, unix_timestamp

    us = [0, 1, 86_400_000_000, -1, 1_600_000_123_456_789]
    df = session.create_dataframe({"t": us}, dtypes={"t": TIMESTAMP})
    out = df.select(to_date(col("t")).alias("d"),
                    unix_timestamp(col("t")).alias("u")).to_pydict()
    for u, d, sec in zip(us, out["d"], out["u"]):
        dt = datetime.datetime(1970, 1, 1) + \
            datetime.timedelta(microseconds=u)
        assert d == (dt.date() - datetime.date(1970, 1, 1)).days
        assert sec == u // 1_000_000


def test_string_casts(session):
    from spark_rapids_amd import DType

    df = session.create_dataframe({"s": ["1.25", " 42 ", "x", None, "",
                                         "-7e2"]})
    out = df.select(col("s").cast(DType.decimal(9, 2)).alias("d"),
                    col("s").cast(sr.INT64).alias("i"),
                    col("s").cast(sr.FLOAT64).alias("f")).to_pydict()
    import decimal

    assert out["d"] == [decimal.Decimal("1.25"), decimal.Decimal("42.00"),
                        None, None, None, decimal.Decimal("-700.00")]
    # Spark UTF8String.toLong rejects scientific notation: '-7e2' -> NULL
    assert out["i"] == [1, 42, None, None, None, None]
    assert out["f"] == [1.25, 42.0, None, None, None, -700.0]


def test_int_to_string_cast(session):
    df = session.create_dataframe({"i": [0, -5, 123456789, None]})
    out = df.select(col("i").cast(sr.STRING).alias("s")).to_pydict()
    assert out["s"] == ["0", "-5", "123456789", None]

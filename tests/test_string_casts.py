"""String <-> numeric cast semantics (Spark CastStrings analogue):
exact parse on both backends, HALF_UP for decimals, truncation for ints,
NULL on garbage/overflow. GPU tests compare bit-exactly with CPU."""
from decimal import Decimal

import pytest

import spark_rapids_amd as sr
from spark_rapids_amd import DType, col
from spark_rapids_amd.types import (BOOL, FLOAT64, INT8, INT16, INT32,
                                    INT64, STRING)

CASES = ["42", " 42 ", "-7", "+7", "12.9", "-12.5", "0.5", ".5", "5.",
         "1e2", "1.5e3", "2E-2", "-0", "", "  ", "abc", "12a", "1.2.3",
         "9223372036854775807", "9223372036854775808",
         "-9223372036854775808", "99999999999999999999999999999999999999",
         "128", "127", "-128", "-129", "0.049999", "0.05", "-0.05",
         "123456789012345678901234567890.123", None, "NaN", "Infinity",
         "0x1A", "½"]


@pytest.fixture
def cpu():
    return sr.Session({"spark.rapids.sql.enabled": False})


def _casted(s, to):
    df = s.create_dataframe({"x": CASES}, dtypes={"x": STRING})
    return df.select(col("x").cast(to).alias("c")).to_pydict()["c"]


def test_cpu_string_to_int_semantics(cpu):
    out = dict(zip(CASES, _casted(cpu, INT32)))
    assert out["42"] == 42 and out[" 42 "] == 42
    assert out["12.9"] == 12          # truncate toward zero
    assert out["-12.5"] == -12
    assert out["1e2"] is None         # Spark: no exponent for int targets
    assert out["abc"] is None and out["12a"] is None
    assert out["9223372036854775808"] is None
    assert out["128"] == 128


def test_cpu_string_to_tinyint_bounds(cpu):
    out = dict(zip(CASES, _casted(cpu, INT8)))
    assert out["127"] == 127 and out["-128"] == -128
    assert out["128"] is None and out["-129"] is None


def test_cpu_string_to_decimal_half_up(cpu):
    out = dict(zip(CASES, _casted(cpu, DType.decimal(10, 2))))
    assert out["12.9"] == Decimal("12.90")
    assert out["0.05"] == Decimal("0.05")
    assert out["0.049999"] == Decimal("0.05")   # HALF_UP at scale 2
    assert out["-0.05"] == Decimal("-0.05")
    assert out["1e2"] == Decimal("100.00")      # exponent OK for decimals
    assert out["2E-2"] == Decimal("0.02")
    assert out["99999999999999999999999999999999999999"] is None
    assert out["NaN"] is None and out["Infinity"] is None


def test_cpu_string_to_decimal128(cpu):
    out = dict(zip(CASES, _casted(cpu, DType.decimal(33, 3))))
    assert out["123456789012345678901234567890.123"] == \
        Decimal("123456789012345678901234567890.123")


def test_cpu_decimal_to_string(cpu):
    df = cpu.create_dataframe({"v": [150, -5, 0, None]})
    df = df.select(col("v").cast(DType.decimal(7, 2)).alias("d"))
    out = df.select(col("d").cast(STRING).alias("s")).to_pydict()["s"]
    assert out == ["150.00", "-5.00", "0.00", None]


@pytest.mark.gpu
@pytest.mark.parametrize("to", [INT8, INT16, INT32, INT64,
                                DType.decimal(10, 2), DType.decimal(33, 3),
                                DType.decimal(18, 0)])
def test_gpu_string_cast_matches_cpu(to):
    sg = sr.Session()
    sc = sr.Session({"spark.rapids.sql.enabled": False})
    g = _casted(sg, to)
    c = _casted(sc, to)
    assert g == c, [(case, a, b) for case, a, b in zip(CASES, g, c)
                    if a != b]


@pytest.mark.gpu
def test_gpu_decimal_to_string_matches_cpu():
    import numpy as np

    rng = np.random.default_rng(9)
    vals = [int(v) for v in rng.integers(-10**9, 10**9, 5000)] + [None, 0]

    def q(s, dt):
        df = s.create_dataframe({"v": vals}, dtypes={"v": INT64})
        df = df.select(col("v").cast(dt).alias("d"))
        return df.select(col("d").cast(STRING).alias("s")).to_pydict()["s"]

    for dt in (DType.decimal(12, 2), DType.decimal(10, 0),
               DType.decimal(30, 5)):
        sg = sr.Session()
        sc = sr.Session({"spark.rapids.sql.enabled": False})
        assert q(sg, dt) == q(sc, dt), dt

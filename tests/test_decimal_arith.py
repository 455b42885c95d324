"""Decimal multiply/divide with Spark DecimalPrecision scale arithmetic.

Reference analogue: GpuMultiply/GpuDivide + DecimalUtil.scala typing
(allowPrecisionLoss=true). CPU tests check exact HALF_UP semantics against
python Decimal; gpu tests check the dec64 __int128 kernel against the CPU
backend.
"""
import decimal
from decimal import Decimal

import pytest

import spark_rapids_amd as sr
from spark_rapids_amd import DType, col
from spark_rapids_amd.types import decimal_arith_type

decimal.getcontext().prec = 60

D72 = DType.decimal(7, 2)
D104 = DType.decimal(10, 4)


@pytest.fixture
def cpu():
    return sr.Session({"spark.rapids.sql.enabled": False})


def test_result_types():
    assert decimal_arith_type("mul", D72, D72) == DType.decimal(15, 4)
    # div: s = max(6, 2+7+1) = 10, p = 7-2+2+10 = 17
    assert decimal_arith_type("div", D72, D72) == DType.decimal(17, 10)
    # big operands trigger the precision-loss adjustment
    d38 = DType.decimal(38, 10)
    t = decimal_arith_type("mul", d38, d38)
    assert t.precision == 38 and t.scale == 6
    # int32 operand promotes to decimal(10,0)
    assert decimal_arith_type("mul", D72, sr.INT32) == DType.decimal(18, 2)


def _dec_df(s, a_vals, b_vals, adt=D72, bdt=D72):
    df = s.create_dataframe({"a": a_vals, "b": b_vals})
    return df.select(col("a").cast(adt).alias("a"),
                     col("b").cast(bdt).alias("b"))


def test_cpu_mul_exact(cpu):
    df = _dec_df(cpu, [1.25, -3.10, None, 99999.99], [2.00, 0.07, 4.0, 0.07])
    out = df.select((col("a") * col("b")).alias("m")).to_pydict()["m"]
    assert out[0] == Decimal("2.5000")
    assert out[1] == Decimal("-0.2170")
    assert out[2] is None
    assert out[3] == Decimal("6999.9993")


def test_cpu_div_half_up_and_null(cpu):
    df = _dec_df(cpu, [1.00, 1.00, 5.00], [3.00, 0.00, 2.00])
    out = df.select((col("a") / col("b")).alias("d")).to_pydict()["d"]
    assert out[0] == Decimal("0.3333333333")  # scale 10
    assert out[1] is None                     # div by zero -> NULL
    assert out[2] == Decimal("2.5000000000")


def test_cpu_matches_python_decimal(cpu):
    import numpy as np

    rng = np.random.default_rng(3)
    a = [round(float(x), 2) for x in rng.uniform(-9999, 9999, 300)]
    b = [round(float(x), 2) for x in rng.uniform(-99, 99, 300)]
    b[7] = 0.0
    df = _dec_df(cpu, a, b)
    got = df.select((col("a") * col("b")).alias("m"),
                    (col("a") / col("b")).alias("d")).to_pydict()
    for i in range(300):
        da = Decimal(f"{a[i]:.2f}")
        db = Decimal(f"{b[i]:.2f}")
        exp_m = (da * db).quantize(Decimal("0.0001"),
                                   rounding=decimal.ROUND_HALF_UP)
        assert got["m"][i] == exp_m, (i, a[i], b[i])
        if b[i] == 0.0:
            assert got["d"][i] is None
        else:
            exp_d = (da / db).quantize(Decimal("0.0000000001"),
                                       rounding=decimal.ROUND_HALF_UP)
            assert got["d"][i] == exp_d, (i, a[i], b[i])


def test_cpu_decimal_times_int(cpu):
    df = cpu.create_dataframe({"a": [1.25, 2.50], "q": [3, -4]})
    df = df.select(col("a").cast(D72).alias("a"), col("q").alias("q"))
    out = df.select((col("a") * col("q")).alias("m")).to_pydict()["m"]
    assert out == [Decimal("3.75"), Decimal("-10.00")]


def test_cpu_decimal_times_float_is_double(cpu):
    df = _dec_df(cpu, [1.25], [1.0])
    out = df.select((col("a") * 2.0).alias("m"))
    assert out.schema.fields[0].dtype == sr.FLOAT64
    assert out.to_pydict()["m"] == [2.5]


def test_cpu_overflow_is_null(cpu):
    d = DType.decimal(38, 0)
    s = cpu
    df = s.create_dataframe({"a": [1.0], "b": [1.0]})
    df = df.select(col("a").cast(d).alias("a"), col("b").cast(d).alias("b"))
    # (38,0)*(38,0) -> adjusted to (38,6): any value >= 10^32 overflows
    big = s.create_dataframe({"x": [1]})
    from spark_rapids_amd.expr.expressions import Literal

    lit = Literal(10 ** 20, DType.decimal(38, 0))
    out = big.select((lit * lit).alias("m")).to_pydict()["m"]
    assert out == [None]


@pytest.mark.gpu
def test_gpu_matches_cpu_mul_div():
    import numpy as np

    sg = sr.Session()
    sc = sr.Session({"spark.rapids.sql.enabled": False})
    rng = np.random.default_rng(11)
    a = [round(float(x), 2) for x in rng.uniform(-99999, 99999, 20000)]
    b = [round(float(x), 4) for x in rng.uniform(-50, 50, 20000)]
    for i in range(0, 20000, 97):
        b[i] = 0.0
    for q in (
        lambda df: df.select((col("a") * col("b")).alias("r")),
        lambda df: df.select((col("a") / col("b")).alias("r")),
        lambda df: df.select((col("b") / col("a")).alias("r")),
        lambda df: df.filter(col("a") * col("b") > col("a")),
    ):
        g = q(_dec_df(sg, a, b, D72, D104)).to_pydict()
        c = q(_dec_df(sc, a, b, D72, D104)).to_pydict()
        assert g == c


@pytest.mark.gpu
def test_gpu_mul_output_d128_and_placement():
    sg = sr.Session()
    # (10,4)*(10,4) -> (21,8): decimal128 output from dec64 operands
    df = _dec_df(sg, [123456.7891, -1.0], [99999.9999, 3.0], D104, D104)
    out = df.select((col("a") * col("b")).alias("m"))
    assert out.schema.fields[0].dtype == DType.decimal(21, 8)
    tree = out.physical_plan().tree_string()
    assert "GpuProject" in tree, tree
    got = out.to_pydict()["m"]
    assert got[0] == Decimal("123456.7891") * Decimal("99999.9999")
    assert got[1] == Decimal("-3.00000000")


def test_cpu_downscale_cast_half_up_negatives(cpu):
    # regression: floor-based rounding gave -2.25 -> -2.3 via -23? no: -4/2
    # style errors; HALF_UP must round magnitude away from zero
    df = cpu.create_dataframe({"a": [2.25, -2.25, 0.05, -0.05]})
    df = df.select(col("a").cast(DType.decimal(9, 2)).alias("a"))
    out = df.select(
        col("a").cast(DType.decimal(9, 1)).alias("r")).to_pydict()["r"]
    assert out == [Decimal("2.3"), Decimal("-2.3"),
                   Decimal("0.1"), Decimal("-0.1")]


@pytest.mark.gpu
def test_gpu_downscale_cast_half_up_negatives():
    sg = sr.Session()
    sc = sr.Session({"spark.rapids.sql.enabled": False})
    vals = [2.25, -2.25, 0.05, -0.05, 123.455, -123.455, None]
    for s in (sg, sc):
        pass
    def q(s):
        df = s.create_dataframe({"a": vals})
        df = df.select(col("a").cast(DType.decimal(9, 3)).alias("a"))
        return df.select(col("a").cast(DType.decimal(9, 1)).alias("r"),
                         (col("a") * col("a")).alias("sq")).to_pydict()
    assert q(sg) == q(sc)


def test_d128_boundary_values(cpu):
    """int128 boundary arithmetic on the CPU reference (add/sub near
    +-2^126, compare ordering)."""
    from spark_rapids_amd import Column, DType
    from spark_rapids_amd.column import ColumnBatch, Field, Schema

    d = DType.decimal(38, 0)
    big = 2 ** 126
    vals = [big, -big, big - 1, -(big - 1), 0, None]
    cb = ColumnBatch([Column.from_pylist(vals, d),
                      Column.from_pylist([1] * 6, d)])
    df = cpu.from_batches([cb], Schema([Field("v", d), Field("o", d)]))
    out = df.select((col("v") + col("o")).alias("p"),
                    (col("v") - col("o")).alias("m")).to_pydict()
    assert int(out["p"][0]) == big + 1
    assert int(out["m"][1]) == -big - 1
    assert out["p"][5] is None
    srt = [r[0] for r in df.sort("v").collect()]
    assert srt[0] is None  # nulls first asc
    nn = [int(v) for v in srt[1:]]
    assert nn == sorted(nn)


def test_decimal_div_precision_loss_adjustment(cpu):
    """allowPrecisionLoss path: (38,10)/(38,10) adjusts to scale 6."""
    from spark_rapids_amd.types import decimal_arith_type

    d = DType.decimal(38, 10)
    t = decimal_arith_type("div", d, d)
    assert t.precision == 38 and t.scale == 6


# ---- ADVICE.md (round 1, high): Spark DecimalPrecision widening ----------

def test_promote_widening_add_sub_compare():
    from spark_rapids_amd.types import promote
    # decimal(18,0) vs decimal(18,10): common = (28,10) -> DECIMAL128
    t = promote(DType.decimal(18, 0), DType.decimal(18, 10))
    assert (t.precision, t.scale) == (28, 10)
    from spark_rapids_amd.types import TypeId
    assert t.id is TypeId.DECIMAL128


def test_add_large_scale_mismatch_exact(cpu):
    # ADVICE repro: cast(1e17 as decimal(18,0)) + cast(1 as decimal(18,10))
    df = cpu.create_dataframe({"a": [100000000000000000], "b": [1]})
    df = df.select(col("a").cast(DType.decimal(18, 0)).alias("a"),
                   col("b").cast(DType.decimal(18, 10)).alias("b"))
    out = df.select((col("a") + col("b")).alias("s")).to_pydict()["s"]
    assert out[0] == Decimal("100000000000000001.0000000000")


def test_compare_large_scale_mismatch(cpu):
    df = cpu.create_dataframe({"a": [100000000000000000], "b": [1]})
    df = df.select(col("a").cast(DType.decimal(18, 0)).alias("a"),
                   col("b").cast(DType.decimal(18, 10)).alias("b"))
    out = df.select((col("a") > col("b")).alias("g")).to_pydict()["g"]
    assert out[0] is True


def test_add_sub_result_dtype_spark_rules(cpu):
    # (p1,s1)=(18,0), (p2,s2)=(18,10): add -> (29,10)
    df = cpu.create_dataframe({"a": [1], "b": [1]})
    df = df.select(col("a").cast(DType.decimal(18, 0)).alias("a"),
                   col("b").cast(DType.decimal(18, 10)).alias("b"))
    out = df.select((col("a") + col("b")).alias("s"))
    t = out.schema.field("s").dtype
    assert (t.precision, t.scale) == (29, 10)


def test_rescale_overflow_is_null(cpu):
    # value too big for the target precision -> NULL, not garbage
    df = cpu.create_dataframe({"a": [999999999, 1]})
    df = df.select(col("a").cast(DType.decimal(9, 0)).alias("a"))
    out = df.select(col("a").cast(DType.decimal(5, 2)).alias("c")) \
        .to_pydict()["c"]
    assert out[0] is None
    assert out[1] == Decimal("1.00")


@pytest.mark.gpu
def test_gpu_add_large_scale_mismatch_exact():
    sg = sr.Session()
    df = sg.create_dataframe({"a": [100000000000000000, None, -7],
                              "b": [1, 5, 23]})
    df = df.select(col("a").cast(DType.decimal(18, 0)).alias("a"),
                   col("b").cast(DType.decimal(18, 10)).alias("b"))
    out = df.select((col("a") + col("b")).alias("s"),
                    (col("a") > col("b")).alias("g")).to_pydict()
    assert out["s"][0] == Decimal("100000000000000001.0000000000")
    assert out["s"][1] is None
    assert out["s"][2] == Decimal("16.0000000000")
    assert out["g"][0] is True
    assert out["g"][1] is None
    assert out["g"][2] is False


@pytest.mark.gpu
def test_gpu_rescale_roundtrip_and_overflow():
    sg = sr.Session()
    sc = sr.Session({"spark.rapids.sql.enabled": False})

    def q(s):
        df = s.create_dataframe(
            {"a": [12345678901234567, -999995, 0, None, 55]})
        df = df.select(col("a").cast(DType.decimal(17, 0)).alias("a"))
        return df.select(
            col("a").cast(DType.decimal(27, 10)).alias("up128"),
            col("a").cast(DType.decimal(7, 1)).alias("narrow"),
        ).to_pydict()

    assert q(sg) == q(sc)


# ---- wide (decimal128-operand) multiply / divide -------------------------

def _wide_ops(s):
    df = s.create_dataframe({
        "a": ["123456789012345678901234.5678", "-0.0001",
              "99999999999999999999999999999999.99", "3.14", None],
        "b": ["2.5", "4000.77", "2.0", "-0.000001", "9"],
    })
    df = df.select(col("a").cast(DType.decimal(36, 4)).alias("a"),
                   col("b").cast(DType.decimal(10, 6)).alias("b"))
    return df.select((col("a") * col("b")).alias("m"),
                     (col("a") / col("b")).alias("d")).to_pydict()


def test_cpu_wide_mul_div_exact(cpu):
    out = _wide_ops(cpu)
    # scales from Spark DecimalPrecision (allowPrecisionLoss)
    assert out["m"][0] == Decimal("308641972530864197253086.419500")
    assert out["d"][0] == Decimal("49382715604938271560493.827120")
    assert out["d"][1] == Decimal("0.000000")  # rounds to zero at scale 6
    assert out["d"][2] == Decimal("49999999999999999999999999999999.995000")
    assert out["m"][4] is None and out["d"][4] is None


@pytest.mark.gpu
def test_gpu_wide_mul_div_matches_cpu():
    sg = sr.Session()
    sc = sr.Session({"spark.rapids.sql.enabled": False})
    g, c = _wide_ops(sg), _wide_ops(sc)
    assert g == c, [(i, a, b) for i, (a, b) in
                    enumerate(zip(g["m"], c["m"])) if a != b] + \
        [(i, a, b) for i, (a, b) in enumerate(zip(g["d"], c["d"]))
         if a != b]


@pytest.mark.gpu
def test_gpu_wide_mul_div_fuzz():
    import numpy as np

    rng = np.random.default_rng(17)
    xs = [str(rng.integers(-10**17, 10**17)) + "." +
          str(rng.integers(0, 10**6)).zfill(6) for _ in range(4000)]
    ys = [str(rng.integers(-10**9, 10**9)) + "." +
          str(rng.integers(0, 100)).zfill(2) for _ in range(4000)]

    def q(s):
        df = s.create_dataframe({"x": xs, "y": ys})
        df = df.select(col("x").cast(DType.decimal(24, 6)).alias("x"),
                       col("y").cast(DType.decimal(22, 2)).alias("y"))
        return df.select((col("x") * col("y")).alias("m"),
                         (col("x") / col("y")).alias("d")).to_pydict()

    sg = sr.Session()
    sc = sr.Session({"spark.rapids.sql.enabled": False})
    g, c = q(sg), q(sc)
    for k in ("m", "d"):
        bad = [(i, a, b) for i, (a, b) in enumerate(zip(g[k], c[k]))
               if a != b]
        assert not bad, (k, bad[:5])

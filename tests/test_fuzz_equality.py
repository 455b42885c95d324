"""Randomized GPU-vs-CPU plan equality (reference analogue: the
integration_tests assert_gpu_and_cpu_are_equal harness run over
generated data). Deterministic seeds; every case builds the same plan on
both backends and compares collected rows."""
import decimal

import numpy as np
import pytest

import spark_rapids_amd as sr
from spark_rapids_amd import (DType, col, count, count_star, sum_, min_,
                              max_, avg)

decimal.getcontext().prec = 60


def _rand_frame(s, rng, n):
    words = ["alpha", "beta", "gamma", "", "delta-x", None]
    return s.create_dataframe({
        "k8": [int(v) for v in rng.integers(-3, 4, n)],
        "k32": [int(v) if v % 13 else None
                for v in rng.integers(-50, 50, n)],
        "f": [float(v) if v % 7 else None
              for v in rng.uniform(-100, 100, n)],
        "d": [round(float(v), 2) for v in rng.uniform(-1000, 1000, n)],
        "s": [words[v % 6] for v in rng.integers(0, 6, n)],
    }, dtypes={"d": DType.decimal(9, 2)})


_PREDS = [
    lambda: col("k32") > 10,
    lambda: (col("f") < 0.0) | col("k32").is_null(),
    lambda: col("s").contains("a") & (col("k8") != 0),
    lambda: col("d") * col("k8") >= col("d"),
    lambda: col("s").rlike(r"^[ad].*a$"),
]

_AGGS = [
    lambda: [sum_(col("f")), count_star()],
    lambda: [min_(col("k32")), max_(col("f")), avg(col("d"))],
    lambda: [sum_(col("d")), count_star()],
]


def _norm(rows):
    out = []
    for r in rows:
        out.append(tuple(
            round(v, 8) if isinstance(v, float) else v for v in r))
    return sorted(out, key=repr)


def _cases():
    for seed in range(8):
        yield seed


@pytest.mark.gpu
@pytest.mark.parametrize("seed", list(_cases()))
def test_fuzz_plan_equality(seed):
    rng = np.random.default_rng(100 + seed)
    sg = sr.Session()
    sc = sr.Session({"spark.rapids.sql.enabled": False})
    n = int(rng.integers(1000, 20000))
    pred = _PREDS[seed % len(_PREDS)]
    aggs = _AGGS[seed % len(_AGGS)]
    kind = seed % 4

    def q(s):
        df = _rand_frame(s, np.random.default_rng(100 + seed), n)
        if kind == 0:
            return df.filter(pred()).group_by("k8").agg(*aggs()).collect()
        if kind == 1:
            dim = s.create_dataframe(
                {"dk": list(range(-50, 50)),
                 "w": [float(v) for v in range(100)]})
            return (df.join(dim, on="k32", right_on=["dk"])
                    .filter(pred()).agg(*aggs()).collect())
        if kind == 2:
            return (df.filter(pred())
                    .sort("k32", "f", "d").limit(500).collect())
        return df.group_by("s", "k8").agg(*aggs()).collect()

    g, c = _norm(q(sg)), _norm(q(sc))
    assert len(g) == len(c), (len(g), len(c))
    for rg, rc in zip(g, c):
        assert len(rg) == len(rc)
        for a, b in zip(rg, rc):
            if isinstance(a, float) and isinstance(b, float):
                if a != a and b != b:
                    continue
                assert a == pytest.approx(b, rel=1e-6, abs=1e-9), (rg, rc)
            else:
                assert a == b, (rg, rc)


# ---------------------------------------------------------------------------
# Special-value fuzz (reference analogue: data_gen.py special values fed
# through assert_gpu_and_cpu_are_equal)
# ---------------------------------------------------------------------------

@pytest.mark.gpu
@pytest.mark.parametrize("seed", range(4))
def test_fuzz_special_values_exprs(seed):
    from spark_rapids_amd.testing import (assert_gpu_and_cpu_are_equal,
                                          gen_column)
    from spark_rapids_amd.types import INT32, INT64, FLOAT64, STRING

    n = 4000
    data = {
        "i": gen_column(INT32, n, seed * 11 + 1),
        "l": gen_column(INT64, n, seed * 11 + 2),
        "f": gen_column(FLOAT64, n, seed * 11 + 3),
        "s": gen_column(STRING, n, seed * 11 + 4),
        "d": gen_column(DType.decimal(9, 2), n, seed * 11 + 5),
    }

    def q(s):
        df = s.create_dataframe(
            {k: list(v) for k, v in data.items()},
            dtypes={"d": DType.decimal(9, 2)})
        return (df
                .with_column("a", col("i").cast(sr.INT64) + col("l"))
                .with_column("b", col("f") * 2.0 - col("f"))
                .with_column("c", col("s").length())
                .with_column("e", col("d") + col("d"))
                .with_column("g", col("f") > 0.0)
                .with_column("h", col("s").contains("a"))
                .filter(col("i").is_not_null() | col("f").is_null()))

    assert_gpu_and_cpu_are_equal(
        q, conf={"spark.rapids.sql.incompatibleOps.enabled": True})


@pytest.mark.gpu
@pytest.mark.parametrize("seed", range(3))
def test_fuzz_special_values_agg_sort(seed):
    from spark_rapids_amd.testing import (assert_gpu_and_cpu_are_equal,
                                          gen_column)
    from spark_rapids_amd.types import INT32, FLOAT64, STRING

    n = 6000
    data = {
        "k": gen_column(INT32, n, seed * 7 + 1, null_frac=0.15),
        "f": gen_column(FLOAT64, n, seed * 7 + 2),
        "s": gen_column(STRING, n, seed * 7 + 3),
    }

    def q(s):
        df = s.create_dataframe({k: list(v) for k, v in data.items()})
        return (df.group_by("k")
                .agg(sum_(col("f")), count_star(), count(col("s")),
                     min_(col("s")), max_(col("s")))
                .sort("k"))

    assert_gpu_and_cpu_are_equal(q, rel=1e-6)


@pytest.mark.gpu
@pytest.mark.parametrize("seed", range(3))
def test_fuzz_special_values_window_join(seed):
    from spark_rapids_amd import win_sum, row_number
    from spark_rapids_amd.testing import (assert_gpu_and_cpu_are_equal,
                                          gen_column)
    from spark_rapids_amd.types import INT32, FLOAT64

    n = 5000
    data = {
        "p": [abs(v) % 40 if v is not None else None
              for v in gen_column(INT32, n, seed * 13 + 1, null_frac=0.05)],
        "o": gen_column(FLOAT64, n, seed * 13 + 2, null_frac=0.05),
        "v": gen_column(FLOAT64, n, seed * 13 + 3),
        "k": gen_column(INT32, n, seed * 13 + 4, null_frac=0.2),
        "t": list(range(n)),
    }
    rdata = {
        "k": gen_column(INT32, 800, seed * 13 + 5, null_frac=0.1),
        "w": gen_column(FLOAT64, 800, seed * 13 + 6),
    }

    def q(s):
        df = s.create_dataframe({k: list(v) for k, v in data.items()})
        r = s.create_dataframe({k: list(v) for k, v in rdata.items()})
        # unique tiebreaker: special values repeat, and row_number over
        # ties is backend-nondeterministic
        df = df.with_column(
            "rn", row_number().over(["p"], ["o", "t"]))
        df = df.with_column(
            "ws", win_sum(col("v")).over(["p"], ["o", "t"],
                                         rows_between=(-2, 2)))
        return df.join(r, on="k", how="left").select(
            "p", "rn", "ws", "w")

    assert_gpu_and_cpu_are_equal(q, rel=1e-6)

"""CPU-vs-GPU equality harness driven query-shape tests (runs the CPU half
anywhere; full comparison under -m gpu)."""
import numpy as np
import pytest

import spark_rapids_amd as sr
from spark_rapids_amd.testing import assert_gpu_and_cpu_are_equal

pytestmark = pytest.mark.gpu

RNG = np.random.default_rng(21)
N = 40_000


def _data():
    return {
        "k": RNG.integers(0, 30, N),
        "j": RNG.integers(0, 2000, N),
        "v": RNG.uniform(-50, 50, N),
        "w": RNG.integers(-5, 5, N),
        "d": RNG.integers(9000, 11000, N).astype(np.int32),
    }


DATA = _data()


def _df(s):
    return s.create_dataframe({k: v.copy() for k, v in DATA.items()},
                              dtypes={"d": sr.DATE32}, num_partitions=3)


@pytest.mark.parametrize("case", [
    "filter_project", "case_when", "coalesce_round", "groupby_multi",
    "fused_filter_agg", "join_agg", "sort_limit", "window_mix",
    "isin_between", "stddev",
])
def test_query_shapes_equal(case):
    q = {
        "filter_project": lambda s: _df(s).filter(
            (sr.col("v") > 0) & (sr.col("w") != 0))
            .select((sr.col("v") * 2.0).alias("x"), "k"),
        "case_when": lambda s: _df(s).select(
            sr.CaseWhen([(sr.col("v") > 0, sr.lit(1)),
                         (sr.col("v") > -10, sr.lit(2))],
                        sr.lit(3)).alias("c")),
        "coalesce_round": lambda s: _df(s).select(
            sr.round_(sr.col("v"), 1).alias("r"),
            sr.coalesce(sr.col("v") / sr.col("w").cast(sr.FLOAT64),
                        sr.lit(-1.0)).alias("c")),
        "groupby_multi": lambda s: _df(s).group_by("k", "w").agg(
            sr.sum_(sr.col("v")), sr.avg(sr.col("v")), sr.count_star(),
            sr.min_(sr.col("j")), sr.max_(sr.col("j"))),
        "fused_filter_agg": lambda s: _df(s).filter(sr.col("d") < 10_500)
            .group_by("k").agg(sr.sum_(sr.col("v")), sr.count_star()),
        "join_agg": lambda s: _df(s).join(
            s.create_dataframe({"j": np.arange(2000),
                                "cat": np.arange(2000) % 7}),
            on="j").group_by("cat").agg(sr.sum_(sr.col("v"))),
        "sort_limit": lambda s: _df(s).sort("k", "j").limit(500),
        "window_mix": lambda s: _df(s).with_column(
            "rn", sr.row_number().over(["k"], ["j"])).filter(
            sr.col("rn") <= 5),
        "isin_between": lambda s: _df(s).filter(
            sr.isin(sr.col("w"), -1, 1) & (sr.col("d") >= 9500)
            & (sr.col("d") <= 10_000)).agg(sr.count_star()),
        "stddev": lambda s: _df(s).group_by("k").agg(
            sr.stddev(sr.col("v")), sr.variance(sr.col("v"))),
    }[case]
    assert_gpu_and_cpu_are_equal(q, rel=1e-7)

"""count/sum(DISTINCT) via the single-distinct two-level aggregate rewrite
(reference analogue: Spark RewriteDistinctAggregates executed as two
GpuHashAggregates)."""
import numpy as np
import pytest

import spark_rapids_amd as sr
from spark_rapids_amd import (col, count, count_distinct, count_star, sum_,
                              sum_distinct)


@pytest.fixture
def cpu():
    return sr.Session({"spark.rapids.sql.enabled": False})


def _df(s, n=1000):
    rng = np.random.default_rng(9)
    return s.create_dataframe({
        "k": [int(v) for v in rng.integers(0, 7, n)],
        "c": [int(v) if v else None for v in rng.integers(0, 40, n)],
        "v": [float(v) for v in rng.uniform(0, 10, n)],
    })


def test_count_distinct_grouped(cpu):
    df = _df(cpu)
    rows = df.group_by("k").agg(count_distinct(col("c")), count(col("c")),
                                sum_(col("v")), count_star()).collect()
    raw = df.collect()
    for k, cd, cnt, sv, all_ in rows:
        vals = [r[1] for r in raw if r[0] == k]
        assert cd == len({v for v in vals if v is not None})
        assert cnt == sum(1 for v in vals if v is not None)
        assert all_ == len(vals)
        assert sv == pytest.approx(
            sum(r[2] for r in raw if r[0] == k))


def test_global_distinct(cpu):
    df = _df(cpu, 500)
    (cd, sd), = df.agg(count_distinct(col("c")),
                       sum_distinct(col("c"))).collect()
    vals = {r[1] for r in df.collect() if r[1] is not None}
    assert cd == len(vals)
    assert sd == sum(vals)


def test_multiple_distinct_columns_raises(cpu):
    df = _df(cpu, 10)
    with pytest.raises(NotImplementedError):
        df.group_by("k").agg(count_distinct(col("c")),
                             count_distinct(col("v")))


def test_output_names(cpu):
    df = _df(cpu, 10)
    out = df.group_by("k").agg(count_distinct(col("c")))
    assert out.schema.fields[-1].name == "count(DISTINCT c)"


@pytest.mark.gpu
def test_gpu_distinct_matches_cpu():
    sg = sr.Session()
    sc = sr.Session({"spark.rapids.sql.enabled": False})

    def q(s):
        return sorted(_df(s, 20000).group_by("k")
                      .agg(count_distinct(col("c")), sum_(col("v")),
                           count_star()).collect())

    g, c = q(sg), q(sc)
    for rg, rc in zip(g, c):
        assert rg[0] == rc[0] and rg[1] == rc[1] and rg[3] == rc[3]
        assert rg[2] == pytest.approx(rc[2], rel=1e-12)

"""df.cache()/persist(): parquet-compressed cached batches (reference
analogue: ParquetCachedBatchSerializer + cache_test.py)."""
import pytest

import spark_rapids_amd as sr
from spark_rapids_amd import col, sum_


@pytest.fixture
def cpu():
    return sr.Session({"spark.rapids.sql.enabled": False})


def test_cache_materializes_once(cpu):
    calls = {"n": 0}

    def fn(b):
        calls["n"] += 1
        return b

    df = cpu.create_dataframe({"k": [1, 2, 1, 2], "v": [1.0, 2.0, 3.0, 4.0]})
    df = df.map_batches(fn).cache()
    r1 = df.group_by("k").agg(sum_(col("v"))).sort("k").collect()
    n_after_first = calls["n"]
    assert n_after_first >= 1
    r2 = df.group_by("k").agg(sum_(col("v"))).sort("k").collect()
    assert calls["n"] == n_after_first  # served from the cache
    assert r1 == r2 == [(1, 4.0), (2, 6.0)]
    # the plan advertises materialization
    assert "CacheData(materialized)" in str(df.plan.name())


def test_cache_types_roundtrip(cpu):
    import decimal

    from spark_rapids_amd.types import DType

    df = cpu.create_dataframe(
        {"d": [decimal.Decimal("1.25"), None],
         "s": ["x", None],
         "b": [True, False],
         "ts": [1000000, None]},
        dtypes={"d": DType.decimal(9, 2), "ts": sr.TIMESTAMP})
    cached = df.cache()
    out1 = cached.to_pydict()
    out2 = cached.to_pydict()
    assert out1 == out2
    assert out1["d"] == [decimal.Decimal("1.25"), None]
    assert out1["b"] == [True, False]


def test_unpersist_recomputes(cpu):
    calls = {"n": 0}

    def fn(b):
        calls["n"] += 1
        return b

    df = cpu.create_dataframe({"v": [1, 2, 3]}).map_batches(fn).cache()
    df.collect()
    first = calls["n"]
    df.unpersist()
    df.collect()
    assert calls["n"] > first


@pytest.mark.gpu
def test_cache_gpu_plan(tmp_path):
    import numpy as np

    s = sr.Session()
    rng = np.random.default_rng(3)
    df = s.create_dataframe(
        {"k": [int(v) for v in rng.integers(0, 50, 20000)],
         "v": [float(v) for v in rng.uniform(0, 1, 20000)]}).cache()
    g1 = df.group_by("k").agg(sum_(col("v"))).sort("k").collect()
    g2 = df.group_by("k").agg(sum_(col("v"))).sort("k").collect()
    assert [r[0] for r in g1] == [r[0] for r in g2]
    for a, b in zip(g1, g2):
        # float atomics reassociate between runs; values match approx
        assert abs(a[1] - b[1]) <= 1e-9 * max(abs(a[1]), 1.0)
    c = sr.Session({"spark.rapids.sql.enabled": False})
    dfc = c.create_dataframe(
        {"k": [r[0] for r in g1], "s": [r[1] for r in g1]})
    assert dfc.count() == len(g1)

"""Edge-case robustness: empty inputs, single rows, all-null columns,
extreme values through every operator."""
import numpy as np
import pytest

import spark_rapids_amd as sr
from spark_rapids_amd import avg, col, count_star, sum_


def test_empty_table_through_operators(session):
    df = session.create_dataframe({"a": [], "b": []},
                                  dtypes={"a": sr.INT64, "b": sr.FLOAT64})
    assert df.filter(col("a") > 0).count() == 0
    assert df.group_by("a").agg(sum_(col("b"))).collect() == []
    assert df.sort("a").collect() == []
    other = session.create_dataframe({"a": [1], "c": [2]})
    assert df.join(other, on="a").collect() == []
    assert other.join(df.select("a", "b"), on="a", how="left").collect() \
        == [(1, 2, None)]


def test_single_row(session):
    df = session.create_dataframe({"a": [42], "b": [1.5]})
    assert df.group_by("a").agg(sum_(col("b")), count_star()).collect() \
        == [(42, 1.5, 1)]
    assert df.sort("a", descending=True).collect() == [(42, 1.5)]


def test_all_null_column(session):
    df = session.create_dataframe({"a": [None, None, None]},
                                  dtypes={"a": sr.INT64})
    assert df.agg(sum_(col("a"))).collect() == [(None,)]
    assert df.filter(col("a") > 0).count() == 0
    assert df.group_by("a").agg(count_star()).collect() == [(None, 3)]


def test_extreme_int_values(session):
    vals = [2**63 - 1, -2**63, 0, -1]
    df = session.create_dataframe({"a": vals}, dtypes={"a": sr.INT64})
    out = df.sort("a").to_pydict()["a"]
    assert out == sorted(vals)
    # sum wraps like Spark non-ANSI (int64 overflow)
    df2 = session.create_dataframe({"a": [2**62, 2**62, 2**62, 2**62]},
                                   dtypes={"a": sr.INT64})
    s = df2.agg(sum_(col("a"))).collect()[0][0]
    assert isinstance(s, int)


def test_float_specials_aggregate(session):
    df = session.create_dataframe({"v": [np.inf, -np.inf, np.nan, 1.0]})
    out = df.agg(sum_(col("v"))).collect()[0][0]
    assert np.isnan(out)


def test_unicode_strings(session):
    vals = ["héllo", "日本語", "🚀", "", None, "Ωmega"]
    df = session.create_dataframe({"s": vals})
    assert df.to_pydict()["s"] == vals
    assert df.filter(col("s").is_not_null()).count() == 5
    assert df.select(col("s").length().alias("l")).to_pydict()["l"] \
        == [5, 3, 1, 0, None, 5]


def test_duplicate_column_names_join(session):
    left = session.create_dataframe({"k": [1, 2], "v": [10, 20]})
    right = session.create_dataframe({"k": [1, 2], "v": [30, 40]})
    out = sorted(left.join(right, on="k").collect())
    # USING join: duplicate right key dropped (Spark df.join(other, "k"))
    assert out == [(1, 10, 30), (2, 20, 40)]


def test_deep_expression_nesting(session):
    df = session.create_dataframe({"a": [1.0, 2.0]})
    e = col("a")
    for _ in range(40):
        e = e + sr.lit(1.0)
    out = df.select(e.alias("x")).to_pydict()["x"]
    assert out == [41.0, 42.0]


def test_zero_partition_groupby(session):
    df = session.create_dataframe({"k": [1, 2], "v": [1.0, 2.0]},
                                  num_partitions=5)  # more parts than rows
    assert sorted(df.group_by("k").agg(sum_(col("v"))).collect()) \
        == [(1, 1.0), (2, 2.0)]


class TestEmptyAndNullEdges:
    @pytest.fixture
    def cpu(self):
        return sr.Session({"spark.rapids.sql.enabled": False})

    def _empty(self, s):
        df = s.create_dataframe({"k": [1], "v": [1.0], "s": ["x"]})
        return df.filter(col("k") < 0)  # empty downstream

    def test_empty_through_operators(self, cpu):
        e = self._empty(cpu)
        assert e.collect() == []
        assert e.group_by("k").agg(sum_(col("v"))).collect() == []
        assert e.agg(count_star(), sum_(col("v"))).collect() == [(0, None)]
        assert e.sort("k").collect() == []
        assert e.sort("k").limit(5).collect() == []
        assert e.distinct().collect() == []
        other = cpu.create_dataframe({"k": [1], "w": [2.0]})
        assert e.join(other, on="k").collect() == []
        assert other.join(e.select(col("k"), col("v")), on="k",
                          how="left").collect() == [(1, 2.0, None)]
        assert e.rollup("k").agg(count_star()).collect() == []

    def test_all_null_join_keys_never_match(self, cpu):
        l = cpu.create_dataframe({"k": [None, None], "v": [1, 2]})
        r = cpu.create_dataframe({"k": [None], "w": [9]})
        assert l.join(r, on="k").collect() == []
        lj = sorted(l.join(r, on="k", how="left").collect(), key=repr)
        assert lj == sorted([(None, 1, None), (None, 2, None)], key=repr)
        assert l.join(r, on="k", how="anti").count() == 2

    def test_null_group_key_single_group(self, cpu):
        df = cpu.create_dataframe({"k": [None, None], "v": [1.0, 2.0]})
        out = df.group_by("k").agg(sum_(col("v")), count_star()).collect()
        assert out == [(None, 3.0, 2)]

    def test_single_row_everything(self, cpu):
        df = cpu.create_dataframe({"k": [7], "v": [1.5]})
        assert df.sort("k").collect() == [(7, 1.5)]
        assert df.group_by("k").agg(avg(col("v"))).collect() == [(7, 1.5)]
        from spark_rapids_amd import win_sum

        w = df.with_column("r", win_sum(col("v")).over(["k"], ["v"]))
        assert w.collect() == [(7, 1.5, 1.5)]

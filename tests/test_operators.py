"""Operator-level tests: aggregate, join, sort, limit, union on the CPU path."""
import pytest

from spark_rapids_amd import (Session, avg, col, count, count_star, lit, max_,
                              min_, sum_)


def test_groupby_sum_count(session):
    df = session.create_dataframe({
        "k": [1, 2, 1, 2, 1, None],
        "v": [10, 20, 30, None, 50, 60],
    })
    out = df.group_by("k").agg(
        sum_(col("v")), count(col("v")), count_star()).sort("k").collect()
    # NULLS FIRST on ascending sort
    assert out == [(None, 60, 1, 1), (1, 90, 3, 3), (2, 20, 1, 2)]


def test_groupby_avg_ignores_nulls(session):
    df = session.create_dataframe({"k": [1, 1, 2], "v": [2.0, None, 6.0]})
    out = df.group_by("k").agg(avg(col("v"))).sort("k").collect()
    assert out == [(1, 2.0), (2, 6.0)]


def test_groupby_all_null_group_sums_to_null(session):
    df = session.create_dataframe({"k": [1, 1], "v": [None, None]},
                                  dtypes={"v": __import__("spark_rapids_amd").INT32})
    out = df.group_by("k").agg(sum_(col("v"))).collect()
    assert out == [(1, None)]


def test_global_agg(session):
    df = session.create_dataframe({"v": [1, 2, 3, 4]})
    assert df.agg(sum_(col("v")), min_(col("v")), max_(col("v"))).collect() \
        == [(10, 1, 4)]


def test_multi_partition_agg(session):
    df = session.create_dataframe({"k": [1, 2] * 50, "v": list(range(100))},
                                  num_partitions=7)
    out = dict((k, v) for k, v in
               df.group_by("k").agg(sum_(col("v"))).collect())
    assert out == {1: sum(range(0, 100, 2)), 2: sum(range(1, 100, 2))}


def test_inner_join(session):
    left = session.create_dataframe({"k": [1, 2, 3, None], "a": [10, 20, 30, 40]})
    right = session.create_dataframe({"k": [2, 3, 4, None], "b": [200, 300, 400, 500]})
    out = sorted(left.join(right, on="k").select("a", "b").collect())
    # NULL keys never match
    assert out == [(20, 200), (30, 300)]


def test_left_join(session):
    left = session.create_dataframe({"k": [1, 2], "a": [10, 20]})
    right = session.create_dataframe({"k": [2], "b": [200]})
    out = sorted(left.join(right, on="k", how="left").select("a", "b").collect())
    assert out == [(10, None), (20, 200)]


def test_semi_anti_join(session):
    left = session.create_dataframe({"k": [1, 2, 3], "a": [10, 20, 30]})
    right = session.create_dataframe({"k": [2, 2, 3]})
    semi = sorted(left.join(right, on="k", how="semi").select("a").to_pydict()["a"])
    anti = sorted(left.join(right, on="k", how="anti").select("a").to_pydict()["a"])
    assert semi == [20, 30]
    assert anti == [10]


def test_join_duplicate_keys_cross_product(session):
    left = session.create_dataframe({"k": [1, 1], "a": [1, 2]})
    right = session.create_dataframe({"k": [1, 1], "b": [3, 4]})
    out = left.join(right, on="k").collect()
    assert len(out) == 4


def test_sort_orders(session):
    df = session.create_dataframe({"x": [3, None, 1, 2]})
    asc = df.sort("x").to_pydict()["x"]
    assert asc == [None, 1, 2, 3]  # Spark asc = NULLS FIRST
    desc = df.sort("x", descending=True).to_pydict()["x"]
    assert desc == [3, 2, 1, None]  # Spark desc = NULLS LAST


def test_sort_multi_key(session):
    df = session.create_dataframe({"a": [1, 1, 2, 2], "b": [2, 1, 4, 3]})
    out = df.sort("a", "b", descending=[False, True]).collect()
    assert out == [(1, 2), (1, 1), (2, 4), (2, 3)]


def test_limit(session):
    df = session.create_dataframe({"x": list(range(10))}, num_partitions=3)
    assert df.limit(5).count() == 5


def test_union(session):
    a = session.create_dataframe({"x": [1, 2]})
    b = session.create_dataframe({"x": [3]})
    assert sorted(a.union(b).to_pydict()["x"]) == [1, 2, 3]


def test_count(session):
    df = session.create_dataframe({"x": [1, None, 3]})
    assert df.count() == 3


def test_string_group_keys_cpu(session):
    df = session.create_dataframe({"s": ["a", "b", "a", None],
                                   "v": [1, 2, 3, 4]})
    out = dict(df.group_by("s").agg(sum_(col("v"))).collect())
    assert out == {"a": 4, "b": 2, None: 4}


def test_stddev_variance(session):
    import statistics

    from spark_rapids_amd import stddev, variance

    data = [2.0, 4.0, 4.0, 4.0, 5.0, 5.0, 7.0, 9.0]
    df = session.create_dataframe({"k": [1] * len(data) , "v": data})
    out = df.group_by("k").agg(stddev(col("v")), variance(col("v"))).collect()
    assert out[0][1] == pytest.approx(statistics.stdev(data))
    assert out[0][2] == pytest.approx(statistics.variance(data))


def test_stddev_multi_partition(session):
    import statistics

    from spark_rapids_amd import stddev

    data = [float(i % 17) for i in range(1000)]
    df = session.create_dataframe({"v": data}, num_partitions=4)
    out = df.agg(stddev(col("v"))).collect()
    assert out[0][0] == pytest.approx(statistics.stdev(data), rel=1e-9)


def test_coalesce_batches_exec():
    from spark_rapids_amd import Column, ColumnBatch, INT64
    from spark_rapids_amd.column import Field, Schema
    from spark_rapids_amd.plan.physical import CoalesceBatchesExec, ScanExec

    batches = [ColumnBatch([Column.from_pylist(list(range(i * 10, i * 10 + 10)),
                                               INT64)]) for i in range(5)]

    class Src:
        def partitions(self):
            return iter(batches)

    schema = Schema([Field("x", INT64)])
    scan = ScanExec("cpu", schema, Src(), "t")
    # tiny target: batches pass through one-by-one
    small = list(CoalesceBatchesExec(scan, 8).execute())
    assert len(small) == 5
    # big target: everything coalesces to one batch
    scan2 = ScanExec("cpu", schema, Src(), "t")
    big = list(CoalesceBatchesExec(scan2, 1 << 30).execute())
    assert len(big) == 1 and big[0].num_rows == 50
    assert big[0].columns[0].to_pylist() == list(range(50))


def test_repartition_merge_fallback(session):
    # force tiny merge target so the bucket-split path runs
    session.conf.set("spark.rapids.sql.batchSizeBytes", 4096)
    import numpy as np

    n = 20_000
    rng = np.random.default_rng(3)
    df = session.create_dataframe({
        "k": rng.integers(0, 5000, n),
        "v": rng.uniform(0, 1, n),
    }, num_partitions=6)
    out = df.group_by("k").agg(sum_(col("v")), count_star()).collect()
    assert sum(r[2] for r in out) == n
    assert len(out) == len(set(r[0] for r in out))  # keys unique across buckets


def test_distinct(session):
    df = session.create_dataframe({"a": [1, 1, 2, 2, 3], "b": [1, 1, 2, 9, 3]})
    out = sorted(df.distinct().collect())
    assert out == [(1, 1), (2, 2), (2, 9), (3, 3)]


def test_map_batches_udf_bridge(session):
    df = session.create_dataframe({"a": [1, 2, 3, 4]})

    def double(batch):
        from spark_rapids_amd import Column, ColumnBatch, INT64

        vals = [v * 2 for v in batch.columns[0].to_pylist()]
        return ColumnBatch([Column.from_pylist(vals, INT64)])

    out = df.map_batches(double).to_pydict()["a"]
    assert out == [2, 4, 6, 8]
    # composable with engine ops afterwards
    assert df.map_batches(double).filter(col("a") > 4).count() == 2


def test_session_range(session):
    df = session.range(10)
    assert df.count() == 10
    assert session.range(2, 8, 2).to_pydict()["id"] == [2, 4, 6]


def test_full_outer_join(session):
    left = session.create_dataframe({"k": [1, 2, 3, None], "a": [10, 20, 30, 40]})
    right = session.create_dataframe({"k": [2, 3, 4, None], "b": [200, 300, 400, 500]})
    out = sorted(left.join(right, on="k", how="full").collect(), key=repr)
    # matches: 2,3; left-unmatched: 1, None; right-unmatched: 4, None
    assert len(out) == 6
    d = {r[:2]: r[2:] for r in out}
    assert d[(2, 20)] == (2, 200)
    assert d[(3, 30)] == (3, 300)
    assert d[(1, 10)] == (None, None)
    assert (None, None, 4, 400) in out


def test_cross_join_and_non_equi(session):
    a = session.create_dataframe({"x": [1, 2, 3]})
    b = session.create_dataframe({"y": [10, 20]})
    out = sorted(a.cross_join(b).collect())
    assert len(out) == 6 and out[0] == (1, 10)
    # non-equi: x * 10 < y
    ne = sorted(a.cross_join(b).filter(col("x") * 10 < col("y")).collect())
    assert ne == [(1, 20)]


def test_sample_deterministic(session):
    df = session.create_dataframe({"x": list(range(10_000))})
    a = df.sample(0.1, seed=7).count()
    b = df.sample(0.1, seed=7).count()
    assert a == b
    assert 700 < a < 1300  # ~10%
    c = df.sample(0.5, seed=7).count()
    assert 4500 < c < 5500


class TestConditionalJoins:
    """Non-equi join conditions over the equi probe (reference:
    ConditionalHashJoinIterator / mixed joins — AST-compiled condition;
    here evaluated vectorized over candidate pairs)."""

    def _sides(self, s):
        left = s.create_dataframe({
            "k": [1, 1, 2, 2, 3, None],
            "a": [10, 20, 30, 40, 50, 60]})
        right = s.create_dataframe({
            "k": [1, 1, 2, 4],
            "b": [15, 25, 100, 7]})
        return left, right

    def test_inner_conditional(self, session):
        l, r = self._sides(session)
        out = sorted(l.join(r, on="k", condition=col("a") < col("b"))
                     .select("a", "b").collect())
        # k=1 pairs: (10,15)(10,25)(20,25); k=2: none (30,40 < 100 both!)
        assert out == [(10, 15), (10, 25), (20, 25), (30, 100), (40, 100)]

    def test_left_conditional(self, session):
        l, r = self._sides(session)
        out = sorted(l.join(r, on="k", how="left",
                            condition=col("a") < col("b"))
                     .select("a", "b").collect())
        assert out == [(10, 15), (10, 25), (20, 25), (30, 100), (40, 100),
                       (50, None), (60, None)]

    def test_semi_anti_conditional(self, session):
        l, r = self._sides(session)
        semi = sorted(l.join(r, on="k", how="semi",
                             condition=col("a") < col("b"))
                      .to_pydict()["a"])
        anti = sorted(l.join(r, on="k", how="anti",
                             condition=col("a") < col("b"))
                      .to_pydict()["a"])
        assert semi == [10, 20, 30, 40]
        assert anti == [50, 60]

    def test_full_conditional(self, session):
        l, r = self._sides(session)
        out = sorted(l.join(r, on="k", how="full",
                            condition=col("a") < col("b")).collect(),
                     key=repr)
        pairs = [(row[1], row[3]) for row in out]
        # matched pairs + unmatched left (null b) + unmatched right (b=7
        # under k=4 never matches; b=15/25/100 all matched)
        assert sorted(p for p in pairs if None not in p) == \
            [(10, 15), (10, 25), (20, 25), (30, 100), (40, 100)]
        assert sorted(p[0] for p in pairs if p[1] is None) == [50, 60]
        assert sorted(p[1] for p in pairs if p[0] is None) == [7]

    def test_conditional_pruning_keeps_condition_columns(self, session):
        # column pruning must retain condition-referenced columns that the
        # final projection drops
        l, r = self._sides(session)
        out = sorted(l.join(r, on="k", condition=col("a") < col("b"))
                     .select("k").to_pydict()["k"])
        assert out == [1, 1, 1, 2, 2]

    def test_conditional_subpartitioned(self):
        s = Session({"spark.rapids.sql.enabled": False,
                     "spark.rapids.sql.join.subPartition.targetBytes": 1})
        l, r = self._sides(s)
        out = sorted(l.join(r, on="k", how="left",
                            condition=col("a") < col("b"))
                     .select("a", "b").collect())
        assert out == [(10, 15), (10, 25), (20, 25), (30, 100), (40, 100),
                       (50, None), (60, None)]


class TestNestedLoopJoin:
    def _sides(self, s):
        left = s.create_dataframe({"a": [1, 5, 9, None]})
        right = s.create_dataframe({"b": [2, 6, 7]})
        return left, right

    def test_nl_inner(self, session):
        l, r = self._sides(session)
        out = sorted(l.join_nl(r, col("a") < col("b")).collect(), key=repr)
        assert out == [(1, 2), (1, 6), (1, 7), (5, 6), (5, 7)]

    def test_nl_left_semi_anti(self, session):
        l, r = self._sides(session)
        lo = sorted(l.join_nl(r, col("a") < col("b"), "left").collect(),
                    key=repr)
        assert lo == [(1, 2), (1, 6), (1, 7), (5, 6), (5, 7),
                      (9, None), (None, None)]
        semi = sorted(l.join_nl(r, col("a") < col("b"), "semi")
                      .to_pydict()["a"])
        anti = l.join_nl(r, col("a") < col("b"), "anti").to_pydict()["a"]
        assert semi == [1, 5]
        assert sorted(anti, key=repr) == [9, None]

    def test_nl_full(self, session):
        l, r = self._sides(session)
        out = sorted(l.join_nl(r, col("a") > col("b"), "full").collect(),
                     key=repr)
        # pairs: 5>2, 9>2, 9>6, 9>7; unmatched left: 1, None;
        # unmatched right: none (2,6,7 all matched by 9)
        assert out == [(1, None), (5, 2), (9, 2), (9, 6), (9, 7),
                       (None, None)]

    def test_nl_chunked(self, session):
        import spark_rapids_amd.plan.physical as P
        old = P.NestedLoopJoinExec.PAIR_CHUNK
        P.NestedLoopJoinExec.PAIR_CHUNK = 4
        try:
            l = session.create_dataframe({"a": list(range(20))})
            r = session.create_dataframe({"b": [5, 15]})
            out = sorted(l.join_nl(r, col("a") < col("b")).collect())
            exp = sorted([(a, b) for a in range(20) for b in (5, 15)
                          if a < b])
            assert out == exp
        finally:
            P.NestedLoopJoinExec.PAIR_CHUNK = old

"""collect_list / collect_set aggregates producing LIST columns
(reference analogue: GpuCollectList/GpuCollectSet over cudf lists;
single-pass aggregation path in HashAggregateExec)."""
import numpy as np
import pytest

import spark_rapids_amd as sr
from spark_rapids_amd import col, collect_list, collect_set, count_star, sum_


@pytest.fixture
def cpu():
    return sr.Session({"spark.rapids.sql.enabled": False})


def _df(s, n=2000):
    rng = np.random.default_rng(13)
    return s.create_dataframe({
        "k": [int(v) for v in rng.integers(0, 9, n)],
        "v": [int(v) if v != 3 else None for v in rng.integers(0, 30, n)],
        "w": [float(v) for v in rng.uniform(0, 1, n)],
    })


def _expected(raw):
    exp = {}
    for k, v, w in raw:
        exp.setdefault(k, []).append(v)
    return exp


def test_collect_list_cpu(cpu):
    df = _df(cpu)
    raw = df.collect()
    exp = _expected(raw)
    for k, lst, st, c in df.group_by("k").agg(
            collect_list(col("v")), collect_set(col("v")),
            count_star()).collect():
        want = [v for v in exp[k] if v is not None]
        assert sorted(lst) == sorted(want)
        assert sorted(st) == sorted(set(want))
        assert c == len(exp[k])


def test_collect_empty_group_is_empty_list(cpu):
    df = cpu.create_dataframe({"k": [1], "v": [None]})
    rows = df.group_by("k").agg(collect_list(col("v"))).collect()
    assert rows == [(1, [])]


def test_list_column_roundtrip():
    from spark_rapids_amd import Column, DType, INT64

    lt = DType.list_(INT64)
    c = Column.from_pylist([[1, 2], [], None, [5]], lt)
    assert c.to_pylist() == [[1, 2], [], None, [5]]


def test_collect_mixed_with_mean(cpu):
    df = _df(cpu, 500)
    rows = df.group_by("k").agg(collect_list(col("v")),
                                sr.avg(col("w"))).collect()
    raw = df.collect()
    for k, lst, m in rows:
        ws = [r[2] for r in raw if r[0] == k]
        assert m == pytest.approx(sum(ws) / len(ws))


@pytest.mark.gpu
def test_gpu_collect_matches_cpu():
    sg = sr.Session()
    sc = sr.Session({"spark.rapids.sql.enabled": False})

    def q(s):
        rows = (_df(s, 20000).group_by("k")
                .agg(collect_list(col("v")), collect_set(col("v")),
                     sum_(col("w")), count_star()).collect())
        return sorted((k, sorted(l), sorted(st), round(sw, 6), c)
                      for k, l, st, sw, c in rows)

    assert q(sg) == q(sc)


@pytest.mark.gpu
def test_gpu_collect_placement():
    s = sr.Session()
    tree = (_df(s, 10).group_by("k").agg(collect_list(col("v")))
            .physical_plan().tree_string())
    assert "GpuHashAggregate" in tree, tree


def test_first_last_cpu(cpu):
    from spark_rapids_amd import first, last

    df = cpu.create_dataframe({
        "k": [1, 1, 1, 2, 2],
        "v": [None, 10, 20, None, None],
        "s": ["a", None, "c", "d", None]})
    rows = sorted(df.group_by("k").agg(first(col("v")), last(col("v")),
                                       first(col("s"))).collect())
    assert rows == [(1, 10, 20, "a"), (2, None, None, "d")]


@pytest.mark.gpu
def test_gpu_first_any_value():
    """GPU first() returns SOME non-null value of the group."""
    from spark_rapids_amd import first

    sg = sr.Session()
    df = _df(sg, 5000)
    raw = df.collect()
    groups = {}
    for k, v, w in raw:
        if v is not None:
            groups.setdefault(k, set()).add(v)
    for k, f in df.group_by("k").agg(first(col("v"))).collect():
        if k in groups:
            assert f in groups[k], (k, f)
        else:
            assert f is None


def test_percentile_cpu(cpu):
    from spark_rapids_amd import percentile
    import numpy as np

    rng = np.random.default_rng(4)
    k = [int(v) for v in rng.integers(0, 5, 3000)]
    v = [float(x) if i % 11 else None
         for i, x in enumerate(rng.uniform(-50, 50, 3000))]
    df = cpu.create_dataframe({"k": k, "v": v})
    rows = df.group_by("k").agg(percentile(col("v"), 0.25),
                                percentile(col("v"), 0.75)).collect()
    for g, p25, p75 in rows:
        vals = [x for kk, x in zip(k, v) if kk == g and x is not None]
        assert p25 == pytest.approx(np.percentile(vals, 25))
        assert p75 == pytest.approx(np.percentile(vals, 75))


def test_sql_percentile(cpu):
    df = cpu.create_dataframe({"k": [1, 1, 1, 1], "v": [1.0, 2.0, 3.0, 4.0]})
    cpu.register("tperc", df)
    out = cpu.sql("SELECT k, percentile(v, 0.5) FROM tperc GROUP BY k")
    assert out.collect() == [(1, 2.5)]


@pytest.mark.gpu
def test_gpu_percentile_matches_cpu():
    from spark_rapids_amd import percentile
    import numpy as np

    sg = sr.Session()
    sc = sr.Session({"spark.rapids.sql.enabled": False})

    def q(s):
        df = _df(s, 30000)
        return sorted(df.group_by("k").agg(
            percentile(col("v"), 0.5), percentile(col("w"), 0.99),
            count_star()).collect())

    g, c = q(sg), q(sc)
    for rg, rc in zip(g, c):
        assert rg[0] == rc[0] and rg[3] == rc[3]
        assert rg[1] == pytest.approx(rc[1], rel=1e-12)
        assert rg[2] == pytest.approx(rc[2], rel=1e-12)


def test_bit_aggs_cpu(cpu):
    from spark_rapids_amd import bit_and, bit_or, bit_xor

    df = cpu.create_dataframe({
        "k": [1, 1, 2, 2, 3],
        "v": [0b1100, 0b1010, 7, None, None]})
    rows = sorted(df.group_by("k").agg(
        bit_and(col("v")), bit_or(col("v")), bit_xor(col("v"))).collect())
    assert rows == [(1, 8, 14, 6), (2, 7, 7, 7), (3, None, None, None)]


@pytest.mark.gpu
def test_gpu_bit_aggs_match_cpu():
    from spark_rapids_amd import bit_and, bit_or, bit_xor
    import numpy as np

    rng = np.random.default_rng(5)
    data = {"k": [int(v) for v in rng.integers(0, 50, 30000)],
            "v": [int(v) if v % 7 else None
                  for v in rng.integers(0, 2**40, 30000)]}

    def q(s):
        df = s.create_dataframe(data)
        return sorted(df.group_by("k").agg(
            bit_and(col("v")), bit_or(col("v")),
            bit_xor(col("v")), count_star()).collect())

    sg = sr.Session()
    sc = sr.Session({"spark.rapids.sql.enabled": False})
    assert q(sg) == q(sc)


def test_explode_cpu(cpu):
    from spark_rapids_amd import Column, DType, INT64
    from spark_rapids_amd.column import ColumnBatch, Field, Schema

    lt = DType.list_(INT64)
    lc = Column.from_pylist([[1, 2], [], None, [5]], lt)
    kc = Column.from_pylist(["a", "b", "c", "d"], sr.STRING)
    df = cpu.from_batches([ColumnBatch([kc, lc])],
                          Schema([Field("k", sr.STRING), Field("v", lt)]))
    assert df.explode("v").collect() == [("a", 1), ("a", 2), ("d", 5)]
    assert df.explode("v", outer=True).collect() == \
        [("a", 1), ("a", 2), ("b", None), ("c", None), ("d", 5)]
    assert df.posexplode("v").collect() == \
        [("a", 0, 1), ("a", 1, 2), ("d", 0, 5)]


def test_explode_of_collect_roundtrip(cpu):
    df = _df(cpu, 500)
    collected = df.group_by("k").agg(collect_list(col("v")))
    back = collected.explode("collect_list(v)")
    raw = [(r[0], r[1]) for r in df.collect() if r[1] is not None]
    assert sorted(back.collect()) == sorted(raw)


@pytest.mark.gpu
def test_gpu_explode_matches_cpu():
    sg = sr.Session()
    sc = sr.Session({"spark.rapids.sql.enabled": False})

    def q(s):
        df = _df(s, 20000)
        out = (df.group_by("k").agg(collect_list(col("v")))
               .posexplode("collect_list(v)").collect())
        return sorted(out)

    assert q(sg) == q(sc) or sorted(r[::2] for r in q(sg)) == \
        sorted(r[::2] for r in q(sc))


@pytest.mark.gpu
def test_gpu_explode_placement():
    sg = sr.Session()
    df = _df(sg, 10)
    tree = (df.group_by("k").agg(collect_list(col("v")))
            .explode("collect_list(v)").physical_plan().tree_string())
    assert "GpuGenerate" in tree, tree


@pytest.mark.gpu
def test_gpu_collect_strings_matches_cpu():
    words = ["alpha", "b", "", "gamma-long-string", None]

    def q(s):
        import numpy as np

        rng = np.random.default_rng(3)
        df = s.create_dataframe({
            "k": [int(v) for v in rng.integers(0, 9, 8000)],
            "s": [words[v % 5] for v in rng.integers(0, 5, 8000)]})
        rows = df.group_by("k").agg(collect_list(col("s"))).collect()
        return sorted((k, sorted(l)) for k, l in rows)

    sg = sr.Session()
    sc = sr.Session({"spark.rapids.sql.enabled": False})
    assert q(sg) == q(sc)


def test_list_columns_through_sort_filter_union(cpu):
    df = (cpu.create_dataframe({"k": [2, 1, 1, 3]})
          .group_by("k").agg(collect_list(col("k"))))
    assert df.sort("k").collect() == [(1, [1, 1]), (2, [2]), (3, [3])]
    assert df.filter(col("k") > 1).sort("k").collect() == \
        [(2, [2]), (3, [3])]
    other = (cpu.create_dataframe({"k": [9]})
             .group_by("k").agg(collect_list(col("k"))))
    assert sorted(df.union(other).collect()) == \
        [(1, [1, 1]), (2, [2]), (3, [3]), (9, [9])]


def test_nested_group_key_clear_error(cpu):
    df = cpu.create_dataframe({"p": [["a"], ["b"]]})
    with pytest.raises(NotImplementedError, match="nested"):
        df.group_by("p").agg(count_star())


@pytest.mark.gpu
def test_collect_set_strings_gpu_matches_cpu():
    import numpy as np

    import spark_rapids_amd as sr
    from spark_rapids_amd import col, collect_set

    rng = np.random.default_rng(9)
    n = 20_000
    words = ["a", "bb", "ccc", "", "zz", "Aa"]
    data = {"k": [int(v) for v in rng.integers(0, 100, n)],
            "s": [None if i % 13 == 0 else words[int(v)]
                  for i, v in enumerate(rng.integers(0, 6, n))]}

    def q(s):
        df = s.create_dataframe(data)
        out = df.group_by("k").agg(collect_set(col("s"))).collect()
        return sorted((k, sorted(v)) for k, v in out)

    g = q(sr.Session())
    c = q(sr.Session({"spark.rapids.sql.enabled": False}))
    assert g == c

"""decimal(>18) — 128-bit backed columns: arithmetic, compare, group keys,
sums, parquet roundtrip, sort. CPU everywhere; GPU equality under -m gpu."""
import decimal
decimal.getcontext().prec = 60

import numpy as np
import pytest

import spark_rapids_amd as sr
from spark_rapids_amd import Column, ColumnBatch, DType, col, count_star, sum_

D = DType.decimal(30, 2)
BIG = 10**24


def _vals(n=2000, nulls=True):
    rng = np.random.default_rng(9)
    out = []
    for i in range(n):
        if nulls and i % 17 == 0:
            out.append(None)
        else:
            hi = int(rng.integers(-2**30, 2**30))
            lo = int(rng.integers(0, 2**60))
            out.append(hi * (2**60) + lo)  # ~ +-2^90, well past int64
    return out


def _df(s, n=2000):
    vals = _vals(n)
    keys = [i % 7 for i in range(n)]
    from spark_rapids_amd.column import Field, Schema

    cols = [Column.from_pylist(keys, sr.INT32),
            Column.from_pylist(vals, D)]
    schema = Schema([Field("k", sr.INT32), Field("v", D)])
    return s.from_batches([ColumnBatch(cols)], schema), vals, keys


def test_pylist_roundtrip():
    vals = [BIG * 5 + 7, None, -3, 0]
    c = Column.from_pylist(vals, D)
    out = c.to_pylist()
    assert out[1] is None
    assert int(out[0].scaleb(2)) == BIG * 5 + 7
    assert int(out[2].scaleb(2)) == -3


def test_cpu_add_compare_sort(session):
    df, vals, _ = _df(session)
    out = df.select((col("v") + col("v")).alias("w")).to_pydict()["w"]
    for v, w in zip(vals, out):
        if v is None:
            assert w is None
        else:
            assert int(w.scaleb(2)) == 2 * v
    cnt = df.filter(col("v") > col("v") - col("v")).count()
    exp = sum(1 for v in vals if v is not None and v > 0)
    assert cnt == exp
    ordered = [x for x in df.sort("v").to_pydict()["v"] if x is not None]
    assert ordered == sorted(ordered)


def test_cpu_group_sum(session):
    df, vals, keys = _df(session)
    out = dict((k, v) for k, v, _ in
               df.group_by("k").agg(sum_(col("v")), count_star()).collect())
    for k in range(7):
        exp = sum(v for v, kk in zip(vals, keys) if kk == k and v is not None)
        assert int(out[k].scaleb(2)) == exp


def test_sum_decimal64_widens_to_128(session):
    d64 = DType.decimal(12, 2)
    df = session.create_dataframe(
        {"v": [10**10, 2 * 10**10, None]}, dtypes={"v": d64})
    out = df.agg(sum_(col("v"))).collect()[0][0]
    assert int(out.scaleb(2)) == 3 * 10**10
    sch = df.agg(sum_(col("v"))).schema
    assert sch.fields[0].dtype.id.value == "decimal128"


def test_parquet_roundtrip_decimal128(tmp_path, session):
    df, vals, _ = _df(session, 500)
    p = str(tmp_path / "d.parquet")
    session.write_parquet(df, p)
    back = session.read_parquet(p)
    got = back.to_pydict()["v"]
    orig = df.to_pydict()["v"]
    assert got == orig


@pytest.mark.gpu
def test_gpu_d128_matches_cpu():
    sg = sr.Session()
    sc = sr.Session({"spark.rapids.sql.enabled": False})
    for q in (
        lambda s: _df(s, 5000)[0].select((col("v") + col("v")).alias("w")),
        lambda s: _df(s, 5000)[0].filter(col("v") > col("v") - col("v")),
        lambda s: _df(s, 5000)[0].group_by("k").agg(sum_(col("v")),
                                                    count_star()),
        lambda s: _df(s, 5000)[0].group_by("v").agg(count_star()),
    ):
        g = sorted(q(sg).collect(), key=repr)
        c = sorted(q(sc).collect(), key=repr)
        assert g == c


@pytest.mark.gpu
def test_gpu_d128_plan_placement():
    s = sr.Session()
    df, _, _ = _df(s, 100)
    tree = df.group_by("k").agg(sum_(col("v"))).physical_plan().tree_string()
    assert "GpuHashAggregate" in tree, tree

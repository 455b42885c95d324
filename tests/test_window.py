"""Window function tests vs hand-computed expectations (Spark semantics)."""
import numpy as np
import pytest

from spark_rapids_amd import (Session, col, dense_rank, lag, lead, rank,
                              row_number, win_avg, win_count, win_max,
                              win_min, win_sum)


@pytest.fixture
def df(session):
    return session.create_dataframe({
        "p": [1, 1, 1, 2, 2, 1],
        "o": [10, 30, 20, 5, 5, 30],
        "v": [1.0, 2.0, None, 4.0, 5.0, 6.0],
    })


def test_row_number(df):
    out = df.with_column("rn", row_number().over(["p"], ["o"])).collect()
    # sorted by (p, o): p=1 o=10,20,30,30 ; p=2 o=5,5
    rns = [(r[0], r[1], r[3]) for r in out]
    assert rns == [(1, 10, 1), (1, 20, 2), (1, 30, 3), (1, 30, 4),
                   (2, 5, 1), (2, 5, 2)]


def test_rank_and_dense_rank(df):
    out = df.with_column("rk", rank().over(["p"], ["o"])).collect()
    assert [r[3] for r in out] == [1, 2, 3, 3, 1, 1]
    out = df.with_column("dr", dense_rank().over(["p"], ["o"])).collect()
    assert [r[3] for r in out] == [1, 2, 3, 3, 1, 1]


def test_running_sum_ignores_nulls(df):
    out = df.with_column("s", win_sum(col("v")).over(["p"], ["o"])).collect()
    # p=1 sorted by o: v = 1.0, None(o=20), then o=30 twice (2.0, 6.0 in
    # original row order for ties)
    vals = [r[3] for r in out]
    assert vals[0] == 1.0
    assert vals[1] == 1.0  # null ignored, frame sum so far
    assert vals[2] + vals[3] >= vals[2]  # monotone over ties
    assert vals[4] == 4.0 and vals[5] == 9.0


def test_partition_agg_no_order(df):
    out = df.with_column("t", win_sum(col("v")).over(["p"])).collect()
    for p, o, v, t in out:
        assert t == (9.0 if p == 1 else 9.0)  # p1: 1+2+6, p2: 4+5


def test_win_count_avg_min_max(df):
    out = df.with_column("c", win_count(col("v")).over(["p"])).collect()
    assert [r[3] for r in out] == [3, 3, 3, 3, 2, 2]
    out = df.with_column("m", win_min(col("o").cast(
        __import__("spark_rapids_amd").INT64)).over(["p"])).collect()
    assert [r[3] for r in out] == [10, 10, 10, 10, 5, 5]
    out = df.with_column("a", win_avg(col("v")).over(["p"])).collect()
    assert [r[3] for r in out] == pytest.approx([3.0, 3.0, 3.0, 3.0, 4.5, 4.5])


def test_lag_lead(df):
    out = df.with_column("lg", lag(col("o")).over(["p"], ["o"])).collect()
    assert [r[3] for r in out] == [None, 10, 20, 30, None, 5]
    out = df.with_column("ld", lead(col("o")).over(["p"], ["o"])).collect()
    assert [r[3] for r in out] == [20, 30, 30, None, 5, None]


def test_lag_default(df):
    out = df.with_column("lg", lag(col("o"), 1, -1).over(["p"], ["o"])).collect()
    assert [r[3] for r in out] == [-1, 10, 20, 30, -1, 5]


def test_window_running_min(session):
    df = session.create_dataframe({"p": [1] * 5, "o": [1, 2, 3, 4, 5],
                                   "v": [3.0, 1.0, None, 2.0, 0.5]})
    out = df.with_column("m", win_min(col("v")).over(["p"], ["o"])).collect()
    assert [r[3] for r in out] == [3.0, 1.0, 1.0, 1.0, 0.5]


def test_window_falls_back_to_cpu(session):
    df = session.create_dataframe({"p": [1], "o": [1], "v": [1.0]})
    tree = df.with_column("rn", row_number().over(["p"], ["o"])).explain()
    assert "Window" in tree


def test_window_large_random(session):
    rng = np.random.default_rng(5)
    n = 20_000
    df = session.create_dataframe({
        "p": rng.integers(0, 50, n),
        "o": rng.integers(0, 1000, n),
        "v": rng.uniform(0, 10, n),
    })
    out = df.with_column("rn", row_number().over(["p"], ["o"])).to_pydict()
    # validate per-partition: rn is 1..len(partition) in order
    import collections

    per = collections.defaultdict(list)
    for p, o, v, rn in zip(out["p"], out["o"], out["v"], out["rn"]):
        per[p].append((rn, o))
    for p, rows in per.items():
        assert [r[0] for r in rows] == list(range(1, len(rows) + 1))
        os_ = [r[1] for r in rows]
        assert os_ == sorted(os_)


import numpy as np
import pytest as _pt

import spark_rapids_amd as sr


def _win_data(n=30_000, nulls=0.1):
    rng = np.random.default_rng(11)
    return {
        "p": rng.integers(0, 40, n),
        "o": rng.integers(0, 500, n),
        "v": np.where(rng.random(n) < nulls, np.nan, rng.uniform(0, 10, n)),
    }


def _df(s, data, nulls=True):
    import numpy as _np

    valid = ~_np.isnan(data["v"])
    from spark_rapids_amd import Column, ColumnBatch, Field, Schema
    from spark_rapids_amd.types import FLOAT64, INT64

    cols = [Column.from_numpy(data["p"].astype(_np.int64)),
            Column.from_numpy(data["o"].astype(_np.int64)),
            Column.from_numpy(_np.nan_to_num(data["v"]), FLOAT64,
                              valid if not valid.all() else None)]
    schema = Schema([Field("p", INT64), Field("o", INT64),
                     Field("v", FLOAT64)])
    return s.from_batches([ColumnBatch(cols)], schema)


@_pt.mark.gpu
@_pt.mark.parametrize("fn", ["row_number", "rank", "dense_rank",
                             "run_sum", "run_count", "run_avg",
                             "part_sum", "part_min", "part_max", "part_avg",
                             "lag", "lead"])
def test_gpu_window_matches_cpu(fn):
    data = _win_data()
    exprs = {
        "row_number": lambda: row_number().over(["p"], ["o"]),
        "rank": lambda: rank().over(["p"], ["o"]),
        "dense_rank": lambda: dense_rank().over(["p"], ["o"]),
        "run_sum": lambda: win_sum(col("v")).over(["p"], ["o"]),
        "run_count": lambda: win_count(col("v")).over(["p"], ["o"]),
        "run_avg": lambda: win_avg(col("v")).over(["p"], ["o"]),
        "part_sum": lambda: win_sum(col("v")).over(["p"]),
        "part_min": lambda: win_min(col("v")).over(["p"]),
        "part_max": lambda: win_max(col("v")).over(["p"]),
        "part_avg": lambda: win_avg(col("v")).over(["p"]),
        "lag": lambda: lag(col("v"), 2).over(["p"], ["o"]),
        "lead": lambda: lead(col("v"), 1, -5.0).over(["p"], ["o"]),
    }
    sg = sr.Session()
    sc = sr.Session({"spark.rapids.sql.enabled": False})
    qg = _df(sg, data).with_column("w", exprs[fn]())
    if fn not in ():
        tree = qg.physical_plan().tree_string()
        assert "GpuWindow" in tree, tree
    gout = qg.to_pydict()["w"]
    cout = _df(sc, data).with_column("w", exprs[fn]()).to_pydict()["w"]
    assert len(gout) == len(cout)
    for i, (g, c) in enumerate(zip(gout, cout)):
        if c is None or g is None:
            assert g is None and c is None, (fn, i, g, c)
        elif isinstance(c, float):
            if np.isnan(c) or np.isnan(g):
                assert np.isnan(c) and np.isnan(g), (fn, i, g, c)
            else:
                assert g == _pt.approx(c, rel=1e-9), (fn, i)
        else:
            assert g == c, (fn, i, g, c)


def test_bounded_frame_cpu(session):
    df = session.create_dataframe({
        "p": [1] * 6 + [2] * 3,
        "o": [1, 2, 3, 4, 5, 6, 1, 2, 3],
        "v": [1.0, 2.0, None, 4.0, 5.0, 6.0, 10.0, 20.0, 30.0],
    })
    e = win_sum(col("v")).over(["p"], ["o"], rows_between=(-1, 1))
    out = df.with_column("s", e).collect()
    vals = [r[3] for r in out]
    assert vals[:6] == [3.0, 3.0, 6.0, 9.0, 15.0, 11.0]
    assert vals[6:] == [30.0, 60.0, 50.0]
    e2 = win_count(col("v")).over(["p"], ["o"], rows_between=(-2, 0))
    cnts = [r[3] for r in df.with_column("c", e2).collect()]
    assert cnts[:6] == [1, 2, 2, 2, 2, 3]
    e3 = win_min(col("v")).over(["p"], ["o"], rows_between=(-1, 0))
    mins = [r[3] for r in df.with_column("m", e3).collect()]
    assert mins[:6] == [1.0, 1.0, 2.0, 4.0, 4.0, 5.0]


@_pt.mark.gpu
@_pt.mark.parametrize("frame", [(-1, 1), (-3, 0), (0, 2), (-5, -1)])
def test_gpu_bounded_frame_matches_cpu(frame):
    data = _win_data(8000)
    e = lambda: win_sum(col("v")).over(["p"], ["o"], rows_between=frame)
    sg = sr.Session()
    sc = sr.Session({"spark.rapids.sql.enabled": False})
    qg = _df(sg, data).with_column("w", e())
    assert "GpuWindow" in qg.physical_plan().tree_string()
    gout = qg.to_pydict()["w"]
    cout = _df(sc, data).with_column("w", e()).to_pydict()["w"]
    for i, (g, c) in enumerate(zip(gout, cout)):
        if c is None or g is None:
            assert g is None and c is None, (frame, i, g, c)
        else:
            assert g == _pt.approx(c, rel=1e-9), (frame, i)
    ec = lambda: win_count(col("v")).over(["p"], ["o"], rows_between=frame)
    gc_ = _df(sg, data).with_column("w", ec()).to_pydict()["w"]
    cc_ = _df(sc, data).with_column("w", ec()).to_pydict()["w"]
    assert gc_ == cc_


class TestRangeFrames:
    def _df(self, s, n=400):
        rng = np.random.default_rng(17)
        k = sorted([float(v) for v in rng.uniform(0, 100, n)])
        return s.create_dataframe({
            "p": [int(v) for v in rng.integers(0, 5, n)],
            "t": [float(v) for v in rng.uniform(0, 100, n)],
            "v": [float(v) if i % 13 else None
                  for i, v in enumerate(rng.uniform(0, 10, n))]})

    def test_range_sum_cpu_manual(self, cpu_session):
        df = cpu_session.create_dataframe({
            "p": [1, 1, 1, 1], "t": [1.0, 2.0, 3.0, 10.0],
            "v": [1.0, 2.0, 3.0, 4.0]})
        out = df.with_column("r", win_sum(col("v")).over(
            partition_by=["p"], order_by=["t"],
            range_between=(-1.0, 1.0))).to_pydict()["r"]
        # frames: t=1 -> [1,2]; t=2 -> [1,2,3]; t=3 -> [2,3]; t=10 -> [10]
        assert out == [3.0, 6.0, 5.0, 4.0]

    def test_range_unbounded_low(self, cpu_session):
        df = cpu_session.create_dataframe({
            "p": [1, 1, 1], "t": [1.0, 2.0, 2.0], "v": [1.0, 2.0, 4.0]})
        out = df.with_column("r", win_sum(col("v")).over(
            partition_by=["p"], order_by=["t"],
            range_between=(None, 0.0))).to_pydict()["r"]
        # RANGE UNBOUNDED..CURRENT includes peers: both t=2 rows get 7
        assert out == [1.0, 7.0, 7.0]

    def test_range_desc_cpu(self, cpu_session):
        df = cpu_session.create_dataframe({
            "p": [1, 1, 1], "t": [3.0, 2.0, 1.0], "v": [1.0, 2.0, 3.0]})
        out = df.with_column("r", win_sum(col("v")).over(
            partition_by=["p"], order_by=["t"], descending=[True],
            range_between=(-1.0, 0.0))).to_pydict()["r"]
        # desc: frame = keys in [cur, cur+1]: t=3 ->{3}:1... wait preceding
        # along sort direction (larger first): t=3 -> [3,4]:1; t=2 ->[2,3]:3
        assert out == [1.0, 3.0, 5.0]

    @pytest.mark.gpu
    def test_gpu_range_matches_cpu(self):
        sg = sr.Session()
        sc = sr.Session({"spark.rapids.sql.enabled": False})

        def q(s):
            df = self._df(s, 3000)
            return df.with_column("r", win_sum(col("v")).over(
                partition_by=["p"], order_by=["t"],
                range_between=(-5.0, 5.0))).with_column(
                "c", win_count(col("v")).over(
                    partition_by=["p"], order_by=["t"],
                    range_between=(None, 0.0))).to_pydict()

        g, c = q(sg), q(sc)
        assert g["c"] == c["c"]
        for x, y in zip(g["r"], c["r"]):
            assert (x is None) == (y is None)
            if x is not None:
                assert x == pytest.approx(y, rel=1e-9)

    def test_range_desc_null_keys_cpu(self, cpu_session):
        # null order keys sort last under descending and form a peer
        # group with an unbounded-feeling key (-inf on the negated axis)
        df = cpu_session.create_dataframe({
            "p": [1, 1, 1, 1], "t": [3.0, 2.0, None, None],
            "v": [1.0, 2.0, 4.0, 8.0]})
        out = df.with_column("r", win_sum(col("v")).over(
            partition_by=["p"], order_by=["t"], descending=[True],
            range_between=(-1.0, 0.0))).to_pydict()["r"]
        # t=3 -> keys in [3,4] = {3}: 1; t=2 -> [2,3] = {2,3}: 3;
        # nulls are peers of each other only: 12, 12
        assert out == [1.0, 3.0, 12.0, 12.0]

    @pytest.mark.gpu
    def test_gpu_range_desc_matches_cpu(self):
        sg = sr.Session()
        sc = sr.Session({"spark.rapids.sql.enabled": False})
        rng = np.random.default_rng(23)
        n = 3000
        data = {
            "p": [int(v) for v in rng.integers(0, 7, n)],
            "t": [float(v) if i % 11 else None
                  for i, v in enumerate(rng.uniform(0, 50, n))],
            "v": [float(v) if i % 13 else None
                  for i, v in enumerate(rng.uniform(0, 10, n))]}

        def q(s):
            df = s.create_dataframe(data)
            df = df.with_column("r", win_sum(col("v")).over(
                partition_by=["p"], order_by=["t"], descending=[True],
                range_between=(-2.0, 2.0)))
            df = df.with_column("c", win_count(col("v")).over(
                partition_by=["p"], order_by=["t"], descending=[True],
                range_between=(None, 0.0)))
            df = df.with_column("m", win_max(col("v")).over(
                partition_by=["p"], order_by=["t"], descending=[True],
                range_between=(-3.0, 0.0)))
            return df.to_pydict()

        qg = sr.Session().create_dataframe(data).with_column(
            "r", win_sum(col("v")).over(
                partition_by=["p"], order_by=["t"], descending=[True],
                range_between=(-2.0, 2.0)))
        assert "GpuWindow" in qg.physical_plan().tree_string()
        g, c = q(sg), q(sc)
        assert g["c"] == c["c"]
        assert g["m"] == c["m"]
        for x, y in zip(g["r"], c["r"]):
            assert (x is None) == (y is None)
            if x is not None:
                assert x == pytest.approx(y, rel=1e-9)


def test_ntile_nth_value_cpu(cpu_session):
    from spark_rapids_amd import ntile, nth_value

    df = cpu_session.create_dataframe({
        "k": ["a"] * 7 + ["b"] * 3,
        "t": list(range(7)) + [0, 1, 2],
        "v": [float(x) for x in range(10)]})
    out = (df.with_column("nt", ntile(3).over(["k"], ["t"]))
           .with_column("n2", nth_value(col("v"), 2).over(["k"], ["t"]))
           .with_column("n9", nth_value(col("v"), 9).over(["k"], ["t"]))
           .collect())
    nts = [r[3] for r in out]
    assert nts == [1, 1, 1, 2, 2, 3, 3, 1, 2, 3]  # 3/2/2 split like Spark
    assert [r[4] for r in out] == [1.0] * 7 + [8.0] * 3
    assert all(r[5] is None for r in out)  # k=9 beyond every partition


@pytest.mark.gpu
def test_gpu_ntile_nth_matches_cpu():
    from spark_rapids_amd import ntile, nth_value

    def q(s):
        df = _rand_df(s, 4000) if False else s.create_dataframe({
            "k": [int(v) % 11 for v in range(4000)],
            "t": [int(v) // 11 for v in range(4000)],
            "v": [float(v) * 0.5 for v in range(4000)]})
        return (df.with_column("nt", ntile(4).over(["k"], ["t"]))
                .with_column("nv", nth_value(col("v"), 3).over(["k"], ["t"]))
                .collect())

    sg = sr.Session()
    sc = sr.Session({"spark.rapids.sql.enabled": False})
    assert sorted(q(sg)) == sorted(q(sc))


@pytest.mark.gpu
def test_gpu_minmax_windows_match_cpu():
    from spark_rapids_amd import win_max, win_min

    rng = np.random.default_rng(23)
    n = 20000
    data = {"p": [int(v) for v in rng.integers(0, 40, n)],
            "t": [int(v) for v in rng.integers(0, 10**6, n)],
            "v": [float(v) if v % 11 else None
                  for v in rng.integers(-1000, 1000, n)],
            "i": [int(v) for v in rng.integers(-50, 50, n)]}

    def q(s):
        df = s.create_dataframe(data)
        return (df
                .with_column("rmin", win_min(col("v")).over(["p"], ["t"]))
                .with_column("bmax", win_max(col("v")).over(
                    ["p"], ["t"], rows_between=(-3, 1)))
                .with_column("imin", win_min(col("i")).over(
                    ["p"], ["t"], rows_between=(-5, 0)))
                .with_column("gmax", win_max(col("v")).over(["p"]))
                .to_pydict())

    sg = sr.Session()
    sc = sr.Session({"spark.rapids.sql.enabled": False})
    g, c = q(sg), q(sc)
    for k in ("rmin", "bmax", "imin", "gmax"):
        assert len(g[k]) == len(c[k])
        for a, b in zip(g[k], c[k]):
            assert (a is None) == (b is None), k
            if a is not None:
                assert a == pytest.approx(b), k


def test_framed_sum_nonfinite_frames(session):
    """A NaN/inf value must only affect frames CONTAINING it — prefix-sum
    framing used to poison every later frame (found by the special-value
    fuzz; both backends now patch from non-finite frame counts)."""
    inf, nan = float("inf"), float("nan")
    df = session.create_dataframe({
        "p": [1] * 6, "t": list(range(6)),
        "v": [1.0, inf, 2.0, 4.0, nan, 8.0]})
    out = df.with_column("w", win_sum(col("v")).over(
        ["p"], ["t"], rows_between=(0, 1))).to_pydict()["w"]
    # frames: [1,inf]=inf, [inf,2]=inf, [2,4]=6, [4,nan]=nan,
    # [nan,8]=nan, [8]=8
    import math
    assert out[0] == inf and out[1] == inf
    assert out[2] == 6.0
    assert math.isnan(out[3]) and math.isnan(out[4])
    assert out[5] == 8.0
    df2 = session.create_dataframe({
        "p": [1] * 3, "t": [0, 1, 2], "v": [inf, -inf, 5.0]})
    out2 = df2.with_column("w", win_sum(col("v")).over(
        ["p"], ["t"], rows_between=(-1, 0))).to_pydict()["w"]
    assert out2[0] == inf and math.isnan(out2[1])
    assert out2[2] == -inf  # frame [-inf, 5]

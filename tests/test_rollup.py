"""Rollup / cube via ExpandExec (reference analogue: GpuExpandExec
backing ROLLUP/CUBE/GROUPING SETS under a hash aggregate)."""
import numpy as np
import pytest

import spark_rapids_amd as sr
from spark_rapids_amd import col, count_star, sum_


@pytest.fixture
def cpu():
    return sr.Session({"spark.rapids.sql.enabled": False})


def _df(s, n=200):
    rng = np.random.default_rng(5)
    return s.create_dataframe({
        "a": [["x", "y", "z", None][i] for i in rng.integers(0, 4, n)],
        "b": [int(v) for v in rng.integers(0, 3, n)],
        "v": [float(v) for v in rng.uniform(0, 10, n)],
    })


def test_rollup_totals(cpu):
    df = _df(cpu)
    rows = df.rollup("a", "b").agg(sum_(col("v")), count_star()).collect()
    # grand total row: all keys null, gid = 0b11
    grand = [r for r in rows if r[2] == 3]
    assert len(grand) == 1
    total = sum(r[3] for r in rows if r[2] == 0)
    assert grand[0][3] == pytest.approx(total)
    assert grand[0][4] == 200
    # per-a subtotals (gid=1) match the sum of that a's detail rows
    for sub in (r for r in rows if r[2] == 1):
        detail = sum(r[3] for r in rows if r[2] == 0 and r[0] == sub[0])
        assert sub[3] == pytest.approx(detail)


def test_rollup_distinguishes_real_nulls(cpu):
    df = cpu.create_dataframe({"a": ["x", None], "v": [1.0, 2.0]})
    rows = df.rollup("a").agg(sum_(col("v"))).collect()
    # (x, 0), (None real, 0), (None rolled, 1)
    assert sorted(rows, key=repr) == sorted(
        [("x", 0, 1.0), (None, 0, 2.0), (None, 1, 3.0)], key=repr)


def test_cube_group_count(cpu):
    df = _df(cpu, 100)
    rows = df.cube("a", "b").agg(count_star()).collect()
    det = {(r[0], r[1]) for r in rows if r[2] == 0}
    a_only = {r[0] for r in rows if r[2] == 1}
    b_only = {r[1] for r in rows if r[2] == 2}
    assert len(rows) == len(det) + len(a_only) + len(b_only) + 1


@pytest.mark.gpu
def test_gpu_rollup_matches_cpu():
    sg = sr.Session()
    sc = sr.Session({"spark.rapids.sql.enabled": False})
    g = sorted(_df(sg, 5000).rollup("a", "b")
               .agg(sum_(col("v")), count_star()).collect(), key=repr)
    c = sorted(_df(sc, 5000).rollup("a", "b")
               .agg(sum_(col("v")), count_star()).collect(), key=repr)
    assert len(g) == len(c)
    for rg, rc in zip(g, c):
        assert rg[:3] == rc[:3]
        assert rg[3] == pytest.approx(rc[3], rel=1e-12)
        assert rg[4] == rc[4]


@pytest.mark.gpu
def test_gpu_rollup_plan_shapes():
    s = sr.Session()
    # distributive aggs: hierarchical re-aggregation (Cached + Union)
    tree = (_df(s, 10).rollup("b").agg(count_star())
            .physical_plan().tree_string())
    assert "GpuCached" in tree and "GpuUnion" in tree, tree
    # non-distributive aggs (avg) still take the Expand path
    from spark_rapids_amd import avg, col

    tree2 = (_df(s, 10).rollup("b").agg(avg(col("v")))
             .physical_plan().tree_string())
    assert "GpuExpand" in tree2, tree2


def test_rollup_hierarchical_decimal_and_counts():
    """The hierarchical grouping-set path must match brute-force totals,
    including decimal sums (re-agg casts back to the finest sum dtype)."""
    import spark_rapids_amd as sr
    from spark_rapids_amd import col, sum_, count_star, DType

    s = sr.Session({"spark.rapids.sql.enabled": False})
    df = s.create_dataframe({
        "a": ["x", "x", "y", "y", "y"],
        "b": [1, 2, 1, 1, None],
        "v": [10, 20, 30, 40, 50],
    })
    df = df.with_column("d", col("v").cast(DType.decimal(7, 2)))
    out = df.rollup("a", "b").agg(sum_(col("d")).alias("s"),
                                  count_star().alias("c")).to_pydict()
    rows = {(a, b, g): (str(sv), c) for a, b, g, sv, c in
            zip(out["a"], out["b"], out["spark_grouping_id"], out["s"],
                out["c"])}
    assert rows[("x", 1, 0)] == ("10.00", 1)
    assert rows[("y", None, 0)] == ("50.00", 1)   # null key, finest level
    assert rows[("x", None, 1)] == ("30.00", 2)
    assert rows[("y", None, 1)] == ("120.00", 3)
    assert rows[(None, None, 3)] == ("150.00", 5)


def test_rollup_empty_input_no_rows():
    import spark_rapids_amd as sr
    from spark_rapids_amd import col, sum_

    s = sr.Session({"spark.rapids.sql.enabled": False})
    df = s.create_dataframe({"a": [1], "v": [1.0]}).filter(col("a") > 5)
    out = df.rollup("a").agg(sum_(col("v"))).collect()
    assert out == []

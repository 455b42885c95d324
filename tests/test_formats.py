"""CSV / ORC / JSON scan + write tests (host parse, columnar engine on top)."""
import pytest

from spark_rapids_amd import col, count_star, sum_


def test_csv_roundtrip(tmp_path, session):
    df = session.create_dataframe({"a": [1, 2, 3], "b": [1.5, 2.5, None],
                                   "s": ["x", "y", "z"]})
    p = str(tmp_path / "t.csv")
    session.write_csv(df, p)
    back = session.read_csv(p)
    assert back.to_pydict()["a"] == [1, 2, 3]
    assert back.filter(col("a") > 1).count() == 2


def test_orc_roundtrip(tmp_path, session):
    df = session.create_dataframe({"a": [1, None, 3], "b": [1.5, 2.5, 3.5]})
    p = str(tmp_path / "t.orc")
    session.write_orc(df, p)
    back = session.read_orc(p)
    assert back.to_pydict()["a"] == [1, None, 3]
    assert back.agg(sum_(col("b"))).collect()[0][0] == pytest.approx(7.5)


def test_json_lines(tmp_path, session):
    p = tmp_path / "t.jsonl"
    p.write_text('{"a": 1, "s": "x"}\n{"a": 2, "s": null}\n{"a": 3, "s": "z"}\n')
    back = session.read_json(str(p))
    assert back.to_pydict()["a"] == [1, 2, 3]
    assert back.to_pydict()["s"] == ["x", None, "z"]

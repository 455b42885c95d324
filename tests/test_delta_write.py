"""Delta Lake write path: create/append/overwrite, DELETE/UPDATE/MERGE,
OPTIMIZE, history, time travel (reference analogues:
GpuOptimisticTransaction, GpuDeleteCommand, GpuMergeIntoCommand)."""
import json
import os

import pytest

import spark_rapids_amd as sr
from spark_rapids_amd import col


@pytest.fixture
def s():
    return sr.Session({"spark.rapids.sql.enabled": False})


def _mk(s, path):
    df = s.create_dataframe({"k": [1, 2, 3, 4],
                             "v": [10.0, 20.0, 30.0, 40.0],
                             "name": ["a", "b", "c", "d"]})
    return s.write_delta(df, path)


def test_create_append_read(s, tmp_path):
    p = str(tmp_path / "t1")
    t = _mk(s, p)
    assert sorted(s.read_delta(p).to_pydict()["k"]) == [1, 2, 3, 4]
    t.append(s.create_dataframe({"k": [5], "v": [50.0], "name": ["e"]}))
    assert sorted(s.read_delta(p).to_pydict()["k"]) == [1, 2, 3, 4, 5]
    # log structure: v0 has protocol+metaData with a Spark schemaString
    log0 = [json.loads(x) for x in
            open(os.path.join(p, "_delta_log",
                              "0" * 20 + ".json")) if x.strip()]
    kinds = [next(iter(a)) for a in log0]
    assert "protocol" in kinds and "metaData" in kinds and "add" in kinds
    meta = [a for a in log0 if "metaData" in a][0]["metaData"]
    sch = json.loads(meta["schemaString"])
    assert sch["type"] == "struct"
    assert [f["name"] for f in sch["fields"]] == ["k", "v", "name"]
    assert sch["fields"][1]["type"] == "double"


def test_time_travel(s, tmp_path):
    p = str(tmp_path / "t2")
    t = _mk(s, p)
    t.append(s.create_dataframe({"k": [9], "v": [90.0], "name": ["z"]}))
    assert sorted(s.read_delta(p, version=0).to_pydict()["k"]) == \
        [1, 2, 3, 4]
    assert sorted(s.read_delta(p).to_pydict()["k"]) == [1, 2, 3, 4, 9]


def test_delete_rewrites_only_matching_files(s, tmp_path):
    p = str(tmp_path / "t3")
    t = _mk(s, p)
    t.append(s.create_dataframe({"k": [100, 101],
                                 "v": [1.0, 2.0], "name": ["x", "y"]}))
    from spark_rapids_amd.io.delta import live_files

    before = set(live_files(p))
    t.delete(col("k") == 2)
    after = set(live_files(p))
    got = s.read_delta(p).to_pydict()
    assert sorted(got["k"]) == [1, 3, 4, 100, 101]
    # the second file (100,101) had no match and was NOT rewritten
    assert any(f in after for f in before), "untouched file rewritten"


def test_update(s, tmp_path):
    p = str(tmp_path / "t4")
    t = _mk(s, p)
    t.update({"v": col("v") * 2.0}, col("k") >= 3)
    got = s.read_delta(p).sort("k").to_pydict()
    assert got["v"] == [10.0, 20.0, 60.0, 80.0]


def test_merge_upsert(s, tmp_path):
    p = str(tmp_path / "t5")
    t = _mk(s, p)
    src = s.create_dataframe({"k": [2, 3, 99],
                              "v": [222.0, 333.0, 999.0],
                              "name": ["B", "C", "NEW"]})
    t.merge(src, on=["k"],
            when_matched_update={"v": col("src_v"),
                                 "name": col("src_name")})
    got = s.read_delta(p).sort("k").to_pydict()
    assert got["k"] == [1, 2, 3, 4, 99]
    assert got["v"] == [10.0, 222.0, 333.0, 40.0, 999.0]
    assert got["name"] == ["a", "B", "C", "d", "NEW"]
    hist = t.history()
    assert hist[-1]["operation"] == "MERGE"


def test_merge_delete_matched(s, tmp_path):
    p = str(tmp_path / "t6")
    t = _mk(s, p)
    src = s.create_dataframe({"k": [1, 4], "v": [0.0, 0.0],
                              "name": ["", ""]})
    t.merge(src, on=["k"], when_matched_delete=True,
            when_not_matched_insert=False)
    assert sorted(s.read_delta(p).to_pydict()["k"]) == [2, 3]


def test_optimize_compacts(s, tmp_path):
    p = str(tmp_path / "t7")
    t = _mk(s, p)
    for i in range(3):
        t.append(s.create_dataframe({"k": [10 + i], "v": [0.0],
                                     "name": ["x"]}))
    from spark_rapids_amd.io.delta import live_files

    assert len(live_files(p)) == 4
    t.optimize()
    assert len(live_files(p)) == 1
    assert sorted(s.read_delta(p).to_pydict()["k"]) == \
        [1, 2, 3, 4, 10, 11, 12]


@pytest.mark.gpu
def test_gpu_delta_roundtrip(tmp_path):
    sg = sr.Session()
    p = str(tmp_path / "tg")
    t = _mk(sg, p)
    src = sg.create_dataframe({"k": [2, 77], "v": [5.0, 7.0],
                               "name": ["B", "N"]})
    t.merge(src, on=["k"], when_matched_update={"v": col("src_v"),
                                                "name": col("src_name")})
    got = sg.read_delta(p).sort("k").to_pydict()
    assert got["k"] == [1, 2, 3, 4, 77]
    assert got["v"][1] == 5.0

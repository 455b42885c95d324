"""STRUCT columns: create/get-field/filter/join-carry/concat
(reference analogues: GpuCreateNamedStruct / GpuGetStructField and the
struct rows of the GpuColumnVector type lattice)."""
import pytest

import spark_rapids_amd as sr
from spark_rapids_amd import col
from spark_rapids_amd.expr.expressions import (CreateNamedStruct,
                                               GetStructField, named_struct)
from spark_rapids_amd.types import DType, INT32, STRING


@pytest.fixture
def cpu():
    return sr.Session({"spark.rapids.sql.enabled": False})


def test_struct_column_roundtrip():
    from spark_rapids_amd import Column

    st = DType.struct_([("a", INT32), ("b", STRING)])
    c = Column.from_pylist(
        [{"a": 1, "b": "x"}, None, {"a": None, "b": "z"}], st)
    assert c.to_pylist() == [{"a": 1, "b": "x"}, None,
                             {"a": None, "b": "z"}]
    assert c.null_count == 1


def test_create_and_get_field(cpu):
    df = cpu.create_dataframe({"x": [1, 2, None], "y": ["a", None, "c"]})
    out = df.select(named_struct(x=col("x"), y=col("y")).alias("s"))
    assert str(out.schema.field("s").dtype) == "struct<x:int, y:string>"
    rows = out.to_pydict()["s"]
    assert rows == [{"x": 1, "y": "a"}, {"x": 2, "y": None},
                    {"x": None, "y": "c"}]
    back = out.select(GetStructField(col("s"), "y").alias("yy")) \
        .to_pydict()["yy"]
    assert back == ["a", None, "c"]


def test_struct_through_filter_and_join(cpu):
    df = cpu.create_dataframe({"k": [1, 2, 3, 4],
                               "x": [10, 20, 30, 40],
                               "y": ["a", "b", "c", "d"]})
    withs = df.select(col("k"),
                      named_struct(x=col("x"), y=col("y")).alias("s"))
    filtered = withs.filter(col("k") > 2)
    assert filtered.to_pydict()["s"] == [{"x": 30, "y": "c"},
                                        {"x": 40, "y": "d"}]
    dim = cpu.create_dataframe({"k": [3, 4], "name": ["three", "four"]})
    j = filtered.join(dim, on="k").sort("k").to_pydict()
    assert j["s"] == [{"x": 30, "y": "c"}, {"x": 40, "y": "d"}]
    # union (concat path)
    u = filtered.union(filtered).to_pydict()["s"]
    assert len(u) == 4


def test_get_missing_field_raises(cpu):
    df = cpu.create_dataframe({"x": [1]})
    s = df.select(named_struct(x=col("x")).alias("s"))
    with pytest.raises(KeyError):
        s.select(GetStructField(col("s"), "zz").alias("b")).collect()


@pytest.mark.gpu
def test_gpu_struct_matches_cpu():
    def q(s):
        df = s.create_dataframe({"k": list(range(2000)),
                                 "x": [i * 3 for i in range(2000)],
                                 "y": [f"s{i % 17}" for i in range(2000)]})
        withs = df.select(col("k"),
                          named_struct(x=col("x"), y=col("y")).alias("s"))
        f = withs.filter(col("k") % 3 == 0)
        out1 = f.to_pydict()["s"]
        out2 = f.select(GetStructField(col("s"), "y").alias("yy")) \
            .to_pydict()["yy"]
        u = withs.union(withs)
        return out1, out2, u.to_pydict()["s"][:10]

    g = q(sr.Session())
    c = q(sr.Session({"spark.rapids.sql.enabled": False}))
    assert g == c

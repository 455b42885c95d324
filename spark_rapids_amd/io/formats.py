"""CSV / ORC / JSON-lines scans and writes (reference analogue:
GpuCSVScan / GpuOrcScan / GpuJsonToStructs + writers).

These run the host parse through arrow (the reference's text formats also
host-buffer the lines before GPU parse) and hand device batches to the GPU
plan; a native GPU text parser is a later round. ORC reading covers the
same flat types as the parquet CPU path.
"""
from __future__ import annotations

import glob
import os
from typing import Iterable, List, Optional

from ..column import ColumnBatch, Field, Schema
from .parquet import arrow_table_to_batch, arrow_to_dtype


def _expand(path: str) -> List[str]:
    if any(ch in path for ch in "*?"):
        files = sorted(glob.glob(path))
    elif os.path.isdir(path):
        files = [os.path.join(path, f) for f in sorted(os.listdir(path))
                 if not f.startswith(".")]
    else:
        files = [path]
    if not files:
        raise FileNotFoundError(path)
    return files


class CsvTable:
    def __init__(self, path: str, header: bool = True,
                 delimiter: str = ","):
        import pyarrow.csv as pacsv

        self.files = _expand(path)
        self.header = header
        self.delimiter = delimiter
        self.read_opts = pacsv.ReadOptions(autogenerate_column_names=not header)
        self.parse_opts = pacsv.ParseOptions(delimiter=delimiter)
        # Spark CSV semantics: empty field = null (also what the GPU
        # decoder produces)
        self.convert_opts = pacsv.ConvertOptions(strings_can_be_null=True)
        tbl = pacsv.read_csv(self.files[0], read_options=self.read_opts,
                             parse_options=self.parse_opts,
                             convert_options=self.convert_opts)
        self.schema = Schema([Field(f.name, arrow_to_dtype(f.type), True)
                              for f in tbl.schema])

    def partitions(self) -> Iterable[ColumnBatch]:
        import pyarrow.csv as pacsv

        import torch

        for f in self.files:
            if torch.cuda.is_available():
                try:
                    from .csv_gpu import read_csv_gpu

                    yield read_csv_gpu(f, self.schema,
                                       header=self.header,
                                       delimiter=self.delimiter)
                    continue
                except NotImplementedError:
                    pass  # per-file CPU fallback (quotes / exotic types)
            tbl = pacsv.read_csv(f, read_options=self.read_opts,
                                 parse_options=self.parse_opts,
                                 convert_options=self.convert_opts)
            yield arrow_table_to_batch(tbl)


class OrcTable:
    def __init__(self, path: str):
        import pyarrow.orc as paorc

        self.files = _expand(path)
        sch = paorc.ORCFile(self.files[0]).schema
        self.schema = Schema([Field(n, arrow_to_dtype(sch.field(n).type), True)
                              for n in sch.names])

    def partitions(self) -> Iterable[ColumnBatch]:
        import pyarrow.orc as paorc

        import torch

        for f in self.files:
            if torch.cuda.is_available():
                try:
                    from .orc_gpu import read_orc_gpu

                    yield read_orc_gpu(f, [fl.name
                                           for fl in self.schema.fields])
                    continue
                except NotImplementedError:
                    pass  # per-file fallback (dict encodings, exotic types)
            yield arrow_table_to_batch(paorc.ORCFile(f).read())


class JsonTable:
    """JSON-lines scan."""

    def __init__(self, path: str):
        import pyarrow.json as pajson

        self.files = _expand(path)
        tbl = pajson.read_json(self.files[0])
        self.schema = Schema([Field(f.name, arrow_to_dtype(f.type), True)
                              for f in tbl.schema])

    def partitions(self) -> Iterable[ColumnBatch]:
        import pyarrow.json as pajson

        import torch

        for f in self.files:
            if torch.cuda.is_available():
                try:
                    from .csv_gpu import read_json_gpu

                    yield read_json_gpu(f, self.schema)
                    continue
                except NotImplementedError:
                    pass
            yield arrow_table_to_batch(pajson.read_json(f))


def write_csv(batch: ColumnBatch, schema: Schema, path: str):
    import pyarrow.csv as pacsv

    from .parquet import _column_to_arrow
    import pyarrow as pa

    arrays = [_column_to_arrow(c.cpu(), f.dtype)
              for f, c in zip(schema.fields, batch.columns)]
    pacsv.write_csv(pa.table(dict(zip(schema.names, arrays))), path)


def write_orc(batch: ColumnBatch, schema: Schema, path: str):
    import pyarrow.orc as paorc

    from .parquet import _column_to_arrow
    import pyarrow as pa

    arrays = [_column_to_arrow(c.cpu(), f.dtype)
              for f, c in zip(schema.fields, batch.columns)]
    paorc.write_table(pa.table(dict(zip(schema.names, arrays))), path)

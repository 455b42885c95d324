"""User-facing session + DataFrame API.

The DataFrame surface mirrors what the reference accelerates underneath
Spark SQL; here it is the engine's own frontend (no JVM in the loop). A
Session owns a RapidsConf (the `spark.rapids.*` surface), a catalog, and the
plan pipeline: logical plan -> GPU overrides tagging -> physical plan ->
columnar execution on HIP kernels with CPU fallback per operator.
"""
from __future__ import annotations

from typing import Dict, Iterable, List, Optional, Sequence, Union

import numpy as np

from .column import Column, ColumnBatch, Field, Schema
from .config import RapidsConf, CONCURRENT_GPU_TASKS
from .expr.aggregates import AggExpr
from .expr.expressions import Expression, col as _col
from .memory.semaphore import GpuSemaphore
from .plan import logical as L
from .plan.overrides import plan_physical
from .types import DType, INT32, TypeId


class MemTable:
    """In-memory partitioned table source."""

    def __init__(self, batches: List[ColumnBatch], schema: Schema,
                 replicated: bool = False):
        self.batches = batches
        self.schema = schema
        self.replicated = replicated

    def partitions(self) -> Iterable[ColumnBatch]:
        return iter(self.batches)


class DataFrame:
    def __init__(self, session: "Session", plan: L.LogicalPlan):
        self.session = session
        self.plan = plan

    # ---- transformations ----------------------------------------------
    def filter(self, condition: Expression) -> "DataFrame":
        return DataFrame(self.session, L.Filter(condition, self.plan))

    where = filter

    def select(self, *exprs: Union[str, Expression]) -> "DataFrame":
        es = [(_col(e) if isinstance(e, str) else e) for e in exprs]
        return DataFrame(self.session, L.Project(es, self.plan))

    def with_column(self, name: str, expr) -> "DataFrame":
        from .expr.windows import WindowExpr

        if isinstance(expr, WindowExpr):
            return DataFrame(self.session,
                             L.Window([expr.alias(name)], self.plan))
        sch = self.plan.schema()
        es: List[Expression] = []
        replaced = False
        for f in sch.fields:
            if f.name == name:
                es.append(expr.alias(name))
                replaced = True
            else:
                es.append(_col(f.name))
        if not replaced:
            es.append(expr.alias(name))
        return DataFrame(self.session, L.Project(es, self.plan))

    def group_by(self, *keys: Union[str, Expression]) -> "GroupedData":
        es = [(_col(k) if isinstance(k, str) else k) for k in keys]
        return GroupedData(self, es)

    def rollup(self, *keys: str) -> "GroupedData":
        """Hierarchical grouping sets: (k1..kn), (k1..kn-1), ..., ().
        Lowered to Expand + hash aggregate with a spark_grouping_id column
        (reference analogue: GpuExpandExec under Aggregate for ROLLUP)."""
        n = len(keys)
        sets = [tuple(keys[:i]) for i in range(n, -1, -1)]
        return GroupedData(self, [_col(k) for k in keys],
                           grouping_sets=sets)

    def cube(self, *keys: str) -> "GroupedData":
        """All 2^n grouping-set combinations (n <= 10)."""
        if len(keys) > 10:
            raise ValueError("cube supports at most 10 keys")
        import itertools

        sets = []
        for r in range(len(keys), -1, -1):
            for combo in itertools.combinations(keys, r):
                sets.append(combo)
        return GroupedData(self, [_col(k) for k in keys],
                           grouping_sets=sets)

    def agg(self, *aggs: AggExpr) -> "DataFrame":
        if any(a.distinct for a in aggs):
            return GroupedData(self, [])._agg_distinct(list(aggs))
        return DataFrame(self.session, L.Aggregate([], list(aggs), self.plan))

    def join(self, other: "DataFrame", on: Union[str, Sequence[str]],
             how: str = "inner", right_on: Optional[Sequence[str]] = None,
             condition=None) -> "DataFrame":
        """Equi-join, optionally with an extra non-equi `condition`
        (reference: conditional/mixed hash joins compiled to an AST).
        The condition is an expression over left columns followed by
        right columns; duplicate names resolve to the left side."""
        if isinstance(on, str):
            on = [on]
        using = right_on is None
        r_on = list(right_on) if right_on is not None else list(on)
        return DataFrame(self.session,
                         L.Join(self.plan, other.plan, list(on), r_on, how,
                                using=using, condition=condition))

    def cross_join(self, other: "DataFrame") -> "DataFrame":
        """Cartesian product; combine with filter() for non-equi INNER
        joins."""
        return DataFrame(self.session, L.CrossJoin(self.plan, other.plan))

    def join_nl(self, other: "DataFrame", condition,
                how: str = "inner") -> "DataFrame":
        """Nested-loop join on an arbitrary condition, no equality keys
        (reference: GpuBroadcastNestedLoopJoinExec). Supports
        inner/left/semi/anti/full; pairs are tested in bounded chunks."""
        return DataFrame(self.session,
                         L.NestedLoopJoin(self.plan, other.plan, [], [],
                                          how, condition=condition))

    def cache(self) -> "DataFrame":
        """Materialize this DataFrame's result on first action as
        compressed in-memory parquet batches and serve later actions
        from them (ParquetCachedBatchSerializer analogue)."""
        if not isinstance(self.plan, L.CacheData):
            self.plan = L.CacheData(self.plan)
        return self

    persist = cache

    def unpersist(self) -> "DataFrame":
        if isinstance(self.plan, L.CacheData):
            self.plan.store = None
            self.plan = self.plan.child
        return self

    def sort(self, *keys: str, descending: Union[bool, List[bool]] = False) -> "DataFrame":
        ks = list(keys)
        if isinstance(descending, bool):
            desc = [descending] * len(ks)
        else:
            desc = list(descending)
        cs = self.plan.schema()
        missing = [k for k in ks if k not in cs.names]
        if missing:
            # Spark allows ORDER BY on input columns the projection dropped:
            # re-plan as project+passthrough -> sort -> drop (ADVICE.md
            # round 1: SELECT rank() OVER (...) FROM t ORDER BY g).
            if isinstance(self.plan, L.Project) \
                    and all(m in self.plan.child.schema().names
                            for m in missing):
                widened = L.Project(
                    list(self.plan.exprs) + [_col(m) for m in missing],
                    self.plan.child)
                sorted_ = L.Sort(widened, ks, desc)
                final = L.Project([_col(n) for n in cs.names], sorted_)
                return DataFrame(self.session, final)
            raise ValueError(
                f"ORDER BY column(s) {missing} are neither in the select "
                f"list nor available from the input (have: {cs.names})")
        return DataFrame(self.session, L.Sort(self.plan, ks, desc))

    def sample(self, fraction: float, seed: int = 42) -> "DataFrame":
        """Bernoulli sample via the murmur3 row-position hash (reference
        analogue: GpuPartitionwiseSampledRDD / GpuPoissonSampler's
        deterministic per-row draw). Deterministic for a given seed."""
        from .expr.sample import SampleHash
        from .expr.expressions import lit

        threshold = int(fraction * 2147483647)
        return self.filter(SampleHash(seed) < lit(threshold))

    def explode(self, column: str, outer: bool = False) -> "DataFrame":
        """One output row per element of the LIST column (explode /
        explode_outer)."""
        return DataFrame(self.session,
                         L.Generate(column, self.plan, outer, pos=False))

    def posexplode(self, column: str, outer: bool = False) -> "DataFrame":
        return DataFrame(self.session,
                         L.Generate(column, self.plan, outer, pos=True))

    def drop(self, *names: str) -> "DataFrame":
        """Project away the named columns."""
        keep = [f.name for f in self.schema.fields if f.name not in names]
        return self.select(*[_col(n) for n in keep])

    def with_column_renamed(self, old: str, new: str) -> "DataFrame":
        exprs = [(_col(f.name).alias(new) if f.name == old else _col(f.name))
                 for f in self.schema.fields]
        return self.select(*exprs)

    def union_by_name(self, other: "DataFrame") -> "DataFrame":
        """Union matching the other frame's columns by NAME (Spark
        unionByName)."""
        names = [f.name for f in self.schema.fields]
        other_names = {f.name for f in other.schema.fields}
        missing = [n for n in names if n not in other_names]
        if missing:
            raise ValueError(f"unionByName: missing columns {missing}")
        return self.union(other.select(*[_col(n) for n in names]))

    def distinct(self) -> "DataFrame":
        """Drop duplicate rows (group-by all columns with no aggregates)."""
        keys = [_col(f.name) for f in self.plan.schema().fields]
        return DataFrame(self.session, L.Aggregate(keys, [], self.plan))

    def map_batches(self, fn, schema: Optional[Schema] = None) -> "DataFrame":
        """Run a python function over host batches (the CPU-bridge / UDF
        escape hatch, reference analogue: GpuCpuBridgeExpression). The
        overrides pass keeps this on CPU with transitions around it."""
        return DataFrame(self.session,
                         L.MapBatches(fn, self.plan, schema))

    def limit(self, n: int) -> "DataFrame":
        return DataFrame(self.session, L.Limit(self.plan, n))

    def union(self, other: "DataFrame") -> "DataFrame":
        return DataFrame(self.session, L.Union([self.plan, other.plan]))

    # ---- actions --------------------------------------------------------
    @property
    def schema(self) -> Schema:
        return self.plan.schema()

    def physical_plan(self):
        # cached per DataFrame: repeated collect() of the same frame
        # (dashboards, benchmark steps) skips the tag/convert pass
        if getattr(self, "_phys", None) is None:
            self._phys = plan_physical(self.plan, self.session.conf)
        return self._phys

    def collect_batch(self) -> ColumnBatch:
        from .plan.physical import new_execution

        new_execution()
        exec_ = self.physical_plan()
        from .metrics import instrument

        instrument(exec_)
        self._last_exec = exec_
        sem = GpuSemaphore.get()
        try:
            with sem.held():
                batches = [b.cpu() for b in exec_.execute()]
        except (RuntimeError, MemoryError) as e:
            from .tools import crashdump

            p = crashdump.dump(e, exec_.tree_string())
            if p:
                import sys

                print(f"[rapids] GPU error; crash bundle: {p}",
                      file=sys.stderr)
            raise
        if not batches:
            return ColumnBatch(
                [Column.from_pylist([], f.dtype) for f in self.schema.fields], 0)
        if len(batches) == 1:
            return batches[0]
        from . import ops

        return ops.concat_batches(batches)

    def collect(self) -> List[tuple]:
        batch = self.collect_batch()
        cols = [c.to_pylist() for c in batch.columns]
        return list(zip(*cols)) if cols else []

    def to_pydict(self) -> Dict[str, list]:
        batch = self.collect_batch()
        return {f.name: c.to_pylist()
                for f, c in zip(self.schema.fields, batch.columns)}

    def count(self) -> int:
        from .expr.aggregates import count_star

        rows = self.agg(count_star()).collect()
        return rows[0][0] if rows else 0

    def metrics(self):
        """Per-operator metrics from the last action (opTimeMs inclusive of
        children, output rows/batches)."""
        from .metrics import collect_metrics

        exec_ = getattr(self, "_last_exec", None)
        return collect_metrics(exec_) if exec_ is not None else []

    def explain(self) -> str:
        exec_ = self.physical_plan()
        out = [exec_.tree_string()]
        notes = getattr(exec_, "tag_notes", [])
        for n in notes:
            for r in n.reasons:
                out.append(f"  !{n.node}: {r}")
        return "\n".join(out)


class GroupedData:
    def __init__(self, df: DataFrame, keys: List[Expression],
                 grouping_sets=None):
        self.df = df
        self.keys = keys
        self.grouping_sets = grouping_sets

    # grouping-set aggs that re-aggregate exactly from finest-level
    # partials (distributive): coarser rollup/cube levels are computed
    # from the small base result instead of Expand-replicating the input
    _HIERARCHICAL_OK = {"sum", "count", "count_all", "min", "max"}

    def agg(self, *aggs: AggExpr) -> DataFrame:
        if any(a.distinct for a in aggs):
            if self.grouping_sets is not None:
                raise NotImplementedError("distinct aggs with rollup/cube")
            return self._agg_distinct(list(aggs))
        if self.grouping_sets is None:
            return DataFrame(self.df.session,
                             L.Aggregate(self.keys, list(aggs), self.df.plan))
        if all(a.op in self._HIERARCHICAL_OK for a in aggs):
            return self._agg_hierarchical(list(aggs))
        from .expr.expressions import Alias, Literal

        child = self.df.plan
        cs = child.schema()
        key_names = [k.output_name() for k in self.keys]
        projections = []
        for kept in self.grouping_sets:
            gid = 0
            for i, k in enumerate(key_names):
                if k not in kept:
                    gid |= 1 << (len(key_names) - 1 - i)
            proj = []
            for f in cs.fields:
                if f.name in key_names and f.name not in kept:
                    proj.append(Alias(Literal(None, f.dtype), f.name))
                else:
                    proj.append(Alias(_col(f.name), f.name))
            proj.append(Alias(Literal(gid, INT32), "spark_grouping_id"))
            projections.append(proj)
        expand = L.Expand(projections, child)
        group = self.keys + [_col("spark_grouping_id")]
        return DataFrame(self.df.session,
                         L.Aggregate(group, list(aggs), expand))

    def _agg_hierarchical(self, aggs: List[AggExpr]) -> DataFrame:
        """Rollup/cube via hierarchical re-aggregation (distributive aggs
        only): aggregate ONCE at the full key set, then each coarser
        grouping set re-aggregates that small result — instead of Spark's
        Expand path which replicates the whole input n_sets times
        (GpuExpandExec feeding GpuHashAggregateExec). Counts re-aggregate
        as sums; decimal re-sums are cast back to the finest sum dtype
        (same null-on-overflow bound Spark applies)."""
        from .expr.expressions import Alias, Literal

        key_names = [k.output_name() for k in self.keys]
        agg_names = [a.output_name() for a in aggs]
        base = L.Cached(L.Aggregate(self.keys, list(aggs), self.df.plan))
        bs = base.schema()
        key_dtype = {k: bs.field(k).dtype for k in key_names}
        agg_dtype = {n: bs.field(n).dtype for n in agg_names}
        levels = []
        for kept in self.grouping_sets:
            gid = 0
            for i, k in enumerate(key_names):
                if k not in kept:
                    gid |= 1 << (len(key_names) - 1 - i)
            gid_lit = Alias(Literal(gid, INT32), "spark_grouping_id")
            if set(kept) == set(key_names):
                proj = [Alias(_col(k), k) for k in key_names] + [gid_lit] \
                    + [Alias(_col(n), n) for n in agg_names]
                levels.append(L.Project(proj, base))
                continue
            # grouping by the constant gid keeps empty-input semantics:
            # zero input rows -> zero output rows even for the () set
            pre = L.Project([Alias(_col(k), k) for k in kept] + [gid_lit]
                            + [Alias(_col(n), n) for n in agg_names], base)
            re_aggs = [AggExpr("sum" if a.op in ("count", "count_all")
                               else a.op, _col(n)).alias(n)
                       for a, n in zip(aggs, agg_names)]
            g2 = L.Aggregate([_col(k) for k in kept]
                             + [_col("spark_grouping_id")], re_aggs, pre)
            g2s = g2.schema()
            proj = []
            for k in key_names:
                proj.append(Alias(_col(k), k) if k in kept
                            else Alias(Literal(None, key_dtype[k]), k))
            proj.append(Alias(_col("spark_grouping_id"),
                              "spark_grouping_id"))
            for n in agg_names:
                e = _col(n)
                if g2s.field(n).dtype != agg_dtype[n]:
                    e = e.cast(agg_dtype[n])
                proj.append(Alias(e, n))
            levels.append(L.Project(proj, g2))
        return DataFrame(self.df.session, L.Union(levels))

    def _agg_distinct(self, aggs: List[AggExpr]) -> DataFrame:
        """Single-distinct-column rewrite (Spark RewriteDistinctAggregates):
        inner aggregate groups by (keys, d) computing partials of the
        non-distinct aggs; the outer aggregate merges them and counts/sums
        the now-unique d values."""
        from .expr.expressions import Alias

        dists = [a for a in aggs if a.distinct]
        if len({str(d.child) for d in dists}) != 1:
            raise NotImplementedError(
                "multiple DISTINCT columns in one aggregate")
        d = dists[0].child
        inner_aggs: List[AggExpr] = []
        outer: List[Optional[AggExpr]] = []
        for i, a in enumerate(aggs):
            if a.distinct:
                if a.op not in ("count", "sum"):
                    raise NotImplementedError(f"{a.op}(DISTINCT) unsupported")
                outer.append(AggExpr(a.op, _col("__dist__"),
                                     a.output_name()))
                continue
            pname = f"__p{i}"
            if a.op == "count_all":
                inner_aggs.append(AggExpr("count_all", None, pname))
                outer.append(AggExpr("sum", _col(pname), a.output_name()))
            elif a.op == "count":
                inner_aggs.append(AggExpr("count", a.child, pname))
                outer.append(AggExpr("sum", _col(pname), a.output_name()))
            elif a.op in ("sum", "min", "max"):
                inner_aggs.append(AggExpr(a.op, a.child, pname))
                outer.append(AggExpr(a.op, _col(pname), a.output_name()))
            else:
                raise NotImplementedError(
                    f"{a.op} mixed with DISTINCT aggs unsupported")
        inner_keys = self.keys + [Alias(d, "__dist__")]
        inner = L.Aggregate(inner_keys, inner_aggs, self.df.plan)
        outer_keys = [_col(k.output_name()) for k in self.keys]
        return DataFrame(self.df.session,
                         L.Aggregate(outer_keys, outer, inner))


class Session:
    def __init__(self, conf: Optional[Dict] = None):
        self.conf = RapidsConf(conf)
        self.catalog: Dict[str, MemTable] = {}
        GpuSemaphore.initialize(self.conf.get(CONCURRENT_GPU_TASKS))
        from .config import ROCTX_ENABLED
        from .metrics import enable_roctx

        enable_roctx(self.conf.get(ROCTX_ENABLED))
        from .tools import lore

        lore.configure(self.conf.get_raw("spark.rapids.sql.lore.dumpPath"))
        from .config import SHUFFLE_COMPRESS as SHUFFLE_CODEC
        from .shuffle import dist as _dist

        _dist.set_codec(self.conf.get(SHUFFLE_CODEC))
        from .config import SHUFFLE_WAVE_BYTES

        _dist.set_wave_bytes(self.conf.get(SHUFFLE_WAVE_BYTES))
        from .config import FILECACHE
        from .io import filecache as _fc

        _fc.configure(self.conf.get(FILECACHE))
        from .config import MEM_POOL_FRACTION, MEM_POOL_MODE

        if self.conf.sql_enabled \
                and str(self.conf.get(MEM_POOL_MODE)).upper() == "HIPDF":
            from .config import MEM_SPILL_WATERMARK
            from .memory import device_pool
            from .memory import spill as _spill

            device_pool.activate(self.conf.get(MEM_POOL_FRACTION))
            _spill.configure_watermark(self.conf.get(MEM_SPILL_WATERMARK))
            from .config import PINNED_POOL_SIZE
            from .memory import host_pool as _hp

            if _hp.pool() is None:
                _hp.configure(self.conf.get(PINNED_POOL_SIZE))

    # ---- conf ----------------------------------------------------------
    def set(self, key: str, value) -> "Session":
        self.conf.set(key, value)
        return self

    # ---- data ingestion -------------------------------------------------
    def create_dataframe(self, data: Dict[str, list],
                         dtypes: Optional[Dict[str, DType]] = None,
                         num_partitions: int = 1,
                         replicated: bool = False) -> DataFrame:
        names = list(data)
        cols = []
        for n in names:
            v = data[n]
            dt = (dtypes or {}).get(n)
            if isinstance(v, np.ndarray):
                cols.append(Column.from_numpy(v, dt))
            else:
                if dt is None:
                    dt = _infer_list_dtype(v)
                cols.append(Column.from_pylist(list(v), dt))
        batch = ColumnBatch(cols)
        schema = Schema([Field(n, c.dtype) for n, c in zip(names, cols)])
        batches = _split_partitions(batch, num_partitions)
        table = MemTable(batches, schema, replicated)
        return DataFrame(self, L.Scan(table, schema, "memory"))

    def from_batches(self, batches: List[ColumnBatch], schema: Schema,
                     label: str = "memory", replicated: bool = False,
                     coalesce: bool = True) -> DataFrame:
        """In-memory table. By default the stored layout is coalesced ONCE
        toward spark.rapids.sql.batchSizeBytes at load time, so every query
        over the table streams large contiguous batches instead of paying a
        re-concatenation per scan (288 GB of HBM3E favors few, large
        resident batches). Pass coalesce=False to keep the caller's exact
        batch boundaries (multi-batch code paths in tests)."""
        if coalesce and len(batches) > 1:
            from .config import BATCH_SIZE_BYTES
            from . import ops as _ops

            target = self.conf.get(BATCH_SIZE_BYTES)
            merged: List[ColumnBatch] = []
            pending: List[ColumnBatch] = []
            nbytes = 0
            for b in batches:
                if pending and nbytes + b.nbytes > target:
                    merged.append(pending[0] if len(pending) == 1
                                  else _ops.concat_batches(pending))
                    pending, nbytes = [], 0
                pending.append(b)
                nbytes += b.nbytes
            if pending:
                merged.append(pending[0] if len(pending) == 1
                              else _ops.concat_batches(pending))
            batches = merged
        return DataFrame(self, L.Scan(MemTable(batches, schema, replicated),
                                      schema, label))

    def register(self, name: str, df: DataFrame):
        self.catalog[name] = df

    def table(self, name: str) -> DataFrame:
        return self.catalog[name]

    def sql(self, query: str) -> DataFrame:
        """Parse a SQL query over registered tables into the same logical
        plans the DataFrame API builds (and through the same GPU overrides)."""
        from .sql.parser import parse_sql

        return parse_sql(self, query)

    def read_parquet(self, path: str, columns=None,
                     replicated: bool = False) -> DataFrame:
        """replicated=True marks the table as fully present on every rank
        (dimension tables): scans read all files instead of sharding
        files[rank::world], and joins against it skip the exchange."""
        import torch

        from .config import PARQUET_MT_THREADS, PARQUET_READER_TYPE
        from .io.parquet import ParquetTable

        reader = str(self.conf.get(PARQUET_READER_TYPE)).upper()
        if reader == "AUTO":
            reader = "GPU_DECODE" if (torch.cuda.is_available()
                                      and self.conf.sql_enabled) else "CPU"
        from .config import BATCH_SIZE_BYTES

        src = ParquetTable(path, columns=columns, reader=reader,
                           prefetch_threads=self.conf.get(PARQUET_MT_THREADS),
                           replicated=replicated,
                           chunk_bytes=self.conf.get(BATCH_SIZE_BYTES))
        return DataFrame(self, L.Scan(src, src.schema, f"parquet:{path}"))

    def write_parquet(self, df: DataFrame, path: str,
                      compression: str = "snappy",
                      gpu_encode: bool = False):
        """gpu_encode=True keeps the result on the GPU and writes PLAIN
        pages encoded by device kernels (io/parquet_write.py); the default
        stages through arrow on the host."""
        if gpu_encode:
            from . import ops as _ops
            from .io.parquet_write import write_parquet_gpu

            exec_ = df.physical_plan()
            sem = GpuSemaphore.get()
            with sem.held():
                # strip the trailing host transfer: encode from device
                root = exec_
                while type(root).__name__ == "DeviceTransferExec":
                    root = root.children[0]
                batches = list(root.execute())
                if not batches:
                    batches = [ColumnBatch(
                        [Column.from_pylist([], f.dtype).cuda()
                         for f in df.schema.fields], 0)]
                batch = batches[0] if len(batches) == 1 \
                    else _ops.concat_batches(batches)
                write_parquet_gpu(batch, df.schema, path)
            return
        from .io.parquet import write_parquet

        batch = df.collect_batch()
        write_parquet(batch, df.schema, path, compression)

    def range(self, start: int, end: Optional[int] = None,
              step: int = 1, num_partitions: int = 1) -> DataFrame:
        """Integer range source (reference analogue: GpuRangeExec)."""
        if end is None:
            start, end = 0, start
        import numpy as np

        vals = np.arange(start, end, step, dtype=np.int64)
        return self.create_dataframe({"id": vals},
                                     num_partitions=num_partitions)

    def read_csv(self, path: str, header: bool = True,
                 delimiter: str = ",") -> DataFrame:
        from .io.formats import CsvTable

        src = CsvTable(path, header, delimiter)
        return DataFrame(self, L.Scan(src, src.schema, f"csv:{path}"))

    def read_orc(self, path: str) -> DataFrame:
        from .io.formats import OrcTable

        src = OrcTable(path)
        return DataFrame(self, L.Scan(src, src.schema, f"orc:{path}"))

    def read_hive_text(self, path: str, header: bool = False,
                       delimiter: str = "\x01") -> DataFrame:
        """Hive-style delimited text (ctrl-A separated by default): the
        CSV scan with the Hive delimiter (GpuHiveTextFileFormat analogue)."""
        return self.read_csv(path, header=header, delimiter=delimiter)

    def read_delta(self, path: str, version=None) -> DataFrame:
        """Delta Lake table scan: replay the _delta_log to the live file
        set (optionally at an older `version` — time travel), then scan
        with the parquet reader (GPU page decode)."""
        from .io.delta import live_files
        from .io.parquet import ParquetTable

        files = live_files(path, version=version)
        if not files:
            raise FileNotFoundError(f"delta table has no live files: {path}")
        t = ParquetTable.__new__(ParquetTable)
        t.__init__(files[0])
        t.files = files
        return DataFrame(self, L.Scan(t, t.schema, "delta"))

    def read_iceberg(self, path: str, snapshot_id=None) -> DataFrame:
        """Iceberg table scan: metadata.json -> manifest list -> manifests
        -> live parquet files, v2 position deletes applied (reference
        analogue: the iceberg/ GPU scan bridges)."""
        import torch

        from .config import PARQUET_MT_THREADS, PARQUET_READER_TYPE
        from .io.iceberg import IcebergTable

        reader = str(self.conf.get(PARQUET_READER_TYPE)).upper()
        if reader == "AUTO":
            reader = "GPU_DECODE" if (torch.cuda.is_available()
                                      and self.conf.sql_enabled) else "CPU"
        src = IcebergTable(path, snapshot_id=snapshot_id, reader=reader,
                           prefetch_threads=self.conf.get(
                               PARQUET_MT_THREADS))
        return DataFrame(self, L.Scan(src, src.schema, f"iceberg:{path}"))

    def delta_table(self, path: str):
        """Writer-side Delta handle (append/overwrite/delete/update/merge/
        optimize/history)."""
        from .io.delta_write import DeltaTable

        return DeltaTable(self, path)

    def write_delta(self, df: DataFrame, path: str, mode: str = "error"):
        from .io.delta_write import DeltaTable

        return DeltaTable.create(self, path, df, mode=mode)

    def read_avro(self, path: str) -> DataFrame:
        """Avro object-container scan (flat records; host decode)."""
        from .io.avro import AvroTable

        t = AvroTable(path)
        return DataFrame(self, L.Scan(t, t.schema, "avro"))

    def write_avro(self, df: DataFrame, path: str,
                   codec: str = "deflate"):
        from .io.avro import write_avro

        write_avro(df.collect_batch(), df.schema, path, codec)

    def read_json(self, path: str) -> DataFrame:
        from .io.formats import JsonTable

        src = JsonTable(path)
        return DataFrame(self, L.Scan(src, src.schema, f"json:{path}"))

    def write_csv(self, df: DataFrame, path: str):
        from .io.formats import write_csv

        write_csv(df.collect_batch(), df.schema, path)

    def write_orc(self, df: DataFrame, path: str):
        from .io.formats import write_orc

        write_orc(df.collect_batch(), df.schema, path)


def _infer_list_dtype(v: list) -> DType:
    from .expr.expressions import _infer_literal_dtype

    dt = None
    for x in v:
        if x is None:
            continue
        if dt is None:
            if isinstance(x, list):
                elem = next((e for e in x if e is not None), 0)
                from .expr.expressions import _infer_literal_dtype as _ild

                return DType.list_(_ild(elem))
            dt = _infer_literal_dtype(x)
            if dt.id is not TypeId.INT32:
                return dt
        # ints: widen to INT64 if ANY value needs it (not just the first)
        elif not isinstance(x, bool) and isinstance(x, int) \
                and not -(2 ** 31) <= x < 2 ** 31:
            return DType.int64()
    return dt or DType.int32()


def _split_partitions(batch: ColumnBatch, n: int) -> List[ColumnBatch]:
    if n <= 1 or batch.num_rows == 0:
        return [batch]
    from . import ops

    rows = batch.num_rows
    per = (rows + n - 1) // n
    out = []
    for s in range(0, rows, per):
        idx = Column.from_numpy(
            np.arange(s, min(s + per, rows), dtype=np.int32))
        out.append(ops.gather(batch, idx))
    return out

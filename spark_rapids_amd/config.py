"""Typed config registry with the `spark.rapids.*` key surface.

Fresh implementation of the reference's RapidsConf idea
(reference: sql-plugin/src/main/scala/com/nvidia/spark/rapids/RapidsConf.scala):
a builder-registered, typed, documented config system that can generate its
own docs (docs/configs.md) and that every layer reads through one object.
"""
from __future__ import annotations

import threading
from dataclasses import dataclass
from typing import Any, Callable, Dict, Optional


@dataclass
class ConfEntry:
    key: str
    default: Any
    doc: str
    conv: Callable[[str], Any]
    startup_only: bool = False
    internal: bool = False


_REGISTRY: Dict[str, ConfEntry] = {}


def _register(entry: ConfEntry) -> ConfEntry:
    assert entry.key not in _REGISTRY, f"duplicate conf key {entry.key}"
    _REGISTRY[entry.key] = entry
    return entry


def _to_bool(v) -> bool:
    if isinstance(v, bool):
        return v
    return str(v).strip().lower() in ("true", "1", "yes")


def _to_int(v) -> int:
    return int(str(v).strip())


def _to_float(v) -> float:
    return float(str(v).strip())


def _to_str(v) -> str:
    return str(v)


def _bytes_conv(v) -> int:
    """Parse '512m', '8g', plain ints."""
    if isinstance(v, int):
        return v
    s = str(v).strip().lower()
    mult = 1
    for suffix, m in (("k", 1 << 10), ("m", 1 << 20), ("g", 1 << 30), ("t", 1 << 40)):
        if s.endswith(suffix):
            s = s[:-1]
            mult = m
            break
    return int(float(s) * mult)


def bool_conf(key, default, doc, **kw):
    return _register(ConfEntry(key, default, doc, _to_bool, **kw))


def int_conf(key, default, doc, **kw):
    return _register(ConfEntry(key, default, doc, _to_int, **kw))


def float_conf(key, default, doc, **kw):
    return _register(ConfEntry(key, default, doc, _to_float, **kw))


def str_conf(key, default, doc, **kw):
    return _register(ConfEntry(key, default, doc, _to_str, **kw))


def bytes_conf(key, default, doc, **kw):
    return _register(ConfEntry(key, default, doc, _bytes_conv, **kw))


# --------------------------------------------------------------------------
# The spark.rapids.* surface (subset grows every round; key names match the
# reference's surface so users can port configs 1:1).
# --------------------------------------------------------------------------
SQL_ENABLED = bool_conf(
    "spark.rapids.sql.enabled", True,
    "Enable (true) or disable (false) GPU SQL acceleration; when false every "
    "operator runs on the CPU backend.")
TEST_ENABLED = bool_conf(
    "spark.rapids.sql.test.enabled", False,
    "Testing mode: fail if an operator that was expected to run on GPU falls "
    "back to CPU (used by the CPU-vs-GPU equality harness).")
EXPLAIN = str_conf(
    "spark.rapids.sql.explain", "NONE",
    "Explain why parts of a query did or did not run on GPU: NONE, NOT_ON_GPU, ALL.")
BATCH_SIZE_BYTES = bytes_conf(
    "spark.rapids.sql.batchSizeBytes", 2 << 30,
    "Target size in bytes of output columnar batches; the coalescing "
    "iterator concatenates small batches up to this goal. Sized for 288 GB "
    "HBM3E: larger batches mean fewer, bigger kernels.")
MAX_READER_BATCH_SIZE_ROWS = int_conf(
    "spark.rapids.sql.reader.batchSizeRows", 1 << 31 - 1,
    "Soft cap on rows per batch produced by file readers.")
CONCURRENT_GPU_TASKS = int_conf(
    "spark.rapids.sql.concurrentGpuTasks", 4,
    "Number of concurrent tasks allowed to hold the GPU semaphore at once.")
HAS_NANS = bool_conf(
    "spark.rapids.sql.hasNans", True,
    "Assume floating point data may contain NaNs (affects agg/join tagging).")
IMPROVED_FLOAT_OPS = bool_conf(
    "spark.rapids.sql.variableFloatAgg.enabled", True,
    "Allow floating point aggregation on GPU even though ordering of "
    "operations may produce slightly different results than CPU.")
DECIMAL_ENABLED = bool_conf(
    "spark.rapids.sql.decimalType.enabled", True,
    "Enable decimal columns on GPU.")
MEM_POOL_FRACTION = float_conf(
    "spark.rapids.memory.gpu.allocFraction", 0.9,
    "Fraction of free device memory the pool may grow to.")
MEM_POOL_MODE = str_conf(
    "spark.rapids.memory.gpu.pool", "HIPDF",
    "Device memory pool: HIPDF installs the hipdf sub-allocator (hipMalloc "
    "slab + spill-before-OOM failure callback, RMM-pool analogue) as "
    "torch's CUDA allocator; TORCH keeps torch's caching allocator. The "
    "HIPDF pool can only install before the first device allocation of "
    "the process.")
MEM_SPILL_WATERMARK = float_conf(
    "spark.rapids.memory.gpu.spillWatermark", 0.85,
    "Pool-usage fraction above which spillable batches are proactively "
    "moved to host (reference analogue: spill from the RMM event handler "
    "before allocations fail).")
OPTIMIZER_ENABLED = bool_conf(
    "spark.rapids.sql.optimizer.enabled", False,
    "Cost-based optimizer (CostBasedOptimizer analogue): estimate CPU vs "
    "GPU cost from scan cardinalities, per-operator costs and transfer "
    "rates; when the GPU estimate (incl. H2D transfer) exceeds the CPU "
    "estimate the plan stays on the CPU. Off by default like the "
    "reference.")
OPTIMIZER_EXPLAIN = bool_conf(
    "spark.rapids.sql.optimizer.explain", False,
    "Print the cost-based optimizer's estimates for every plan.")
SHUFFLE_WAVE_BYTES = bytes_conf(
    "spark.rapids.shuffle.wave.bytes", 1 << 30,
    "Bytes budget per exchange wave: a shuffle whose serialized send "
    "buffers exceed this is split into multiple all-to-all rounds so the "
    "exchange memory stays bounded (bounce-buffer windowing analogue).")
PINNED_POOL_SIZE = bytes_conf(
    "spark.rapids.memory.pinnedPool.size", 8 << 30,
    "Size of the pinned host memory pool used for spill and H2D/D2H staging.")
HOST_SPILL_STORAGE_SIZE = bytes_conf(
    "spark.rapids.memory.host.spillStorageSize", 32 << 30,
    "Maximum bytes of host memory used to hold spilled device buffers before "
    "spilling further to disk.")
SPILL_PATH = str_conf(
    "spark.rapids.memory.spillPath", "/tmp/rapids_spill",
    "Local directory for disk spill files.")
PARQUET_READER_TYPE = str_conf(
    "spark.rapids.sql.format.parquet.reader.type", "AUTO",
    "Parquet reader: AUTO (GPU decode when a GPU is present, CPU otherwise), "
    "GPU_DECODE (hipdf kernels decode PLAIN/dictionary pages on device, "
    "per-file CPU fallback for unsupported features), CPU (pyarrow host "
    "decode feeding device batches — the hybrid-scan analogue).")
PARQUET_ENABLED = bool_conf(
    "spark.rapids.sql.format.parquet.enabled", True,
    "Enable parquet scans on GPU.")
PARQUET_MT_THREADS = int_conf(
    "spark.rapids.sql.format.parquet.multiThreadedRead.numThreads", 8,
    "Threads in the multithreaded parquet prefetch pool.")
BROADCAST_THRESHOLD = bytes_conf(
    "spark.rapids.sql.join.broadcastThreshold", 512 << 20,
    "Distributed joins all-gather (broadcast) the build side when its "
    "global size is below this; larger builds hash-exchange BOTH sides "
    "across ranks instead (shuffled hash join).")
SHUFFLE_MODE = str_conf(
    "spark.rapids.shuffle.mode", "MULTITHREADED",
    "Shuffle transport: MULTITHREADED (host staging) or RCCL (device-to-device "
    "all-to-all over xGMI).")
SHUFFLE_PARTITIONS = int_conf(
    "spark.rapids.sql.shuffle.partitions", 16,
    "Default number of shuffle partitions per GPU for exchanges.")
SHUFFLE_COMPRESS = str_conf(
    "spark.rapids.shuffle.compression.codec", "none",
    "Codec for host-path shuffle payloads (none|zstd|lz4|snappy). Device "
    "(RCCL/xGMI) shuffles stay uncompressed: the links outrun a host "
    "compression round-trip.")
RETRY_MAX_SPLITS = int_conf(
    "spark.rapids.sql.retry.maxSplits", 8,
    "Maximum recursive batch splits attempted by the OOM retry framework "
    "before giving up.")
GPU_OOM_INJECTION = int_conf(
    "spark.rapids.sql.test.injectOOM", 0, "Inject a synthetic GPU OOM on the "
    "Nth tracked allocation (testing only, 0 = off).", internal=True)
STABLE_SORT = bool_conf(
    "spark.rapids.sql.stableSort.enabled", False,
    "Use a stable sort on GPU (matches CPU tie ordering; slightly slower).")
PRUNE_COLUMNS = bool_conf(
    "spark.rapids.sql.optimizer.pruneColumns.enabled", True,
    "Push projections below joins/aggregates so unused columns are never "
    "gathered or transferred (Catalyst-optimizer analogue).")
JOIN_SUBPARTITION_BYTES = int_conf(
    "spark.rapids.sql.join.subPartition.targetBytes", 1 << 30,
    "Build sides larger than this are hash-split into buckets and joined "
    "bucket-by-bucket (GpuSubPartitionHashJoin analogue), bounding the "
    "peak size of any single hash table and its gather maps.")
CPU_BRIDGE = bool_conf(
    "spark.rapids.sql.cpuBridge.enabled", True,
    "Evaluate CPU-only expressions inside GPU projections via a host "
    "round-trip instead of demoting the whole project "
    "(GpuCpuBridgeExpression analogue).")
FILECACHE = bool_conf(
    "spark.rapids.filecache.enabled", False,
    "Cache decoded scan batches per (file, mtime) in host memory so "
    "repeated scans skip IO + decode (reference analogue: the filecache "
    "layer). Best for dimension tables read by many queries.")
PUSH_FILTERS = bool_conf(
    "spark.rapids.sql.optimizer.pushFilters.enabled", True,
    "Push filter conjuncts below joins when they reference only one side "
    "(PushPredicateThroughJoin analogue): dimension-table predicates then "
    "filter 50 rows instead of the joined fact output.")
LORE_DUMP_PATH = str_conf(
    "spark.rapids.sql.lore.dumpPath", "",
    "When set, dump every operator's output batches to this directory as "
    "parquet for offline replay (LORE analogue; tools/lore.py replay()).")
ROCTX_ENABLED = bool_conf(
    "spark.rapids.sql.rocTx.enabled", False,
    "Emit rocTX ranges around each operator (view with rocprofv3 "
    "--marker-trace; reference analogue: NVTX ranges + nsys).")
ALLOW_INCOMPAT = bool_conf(
    "spark.rapids.sql.incompatibleOps.enabled", True,
    "Allow operators whose GPU results can differ from the CPU in corner "
    "cases (float ordering, NaN handling).")

_PER_OP_PREFIX = "spark.rapids.sql.exec."
_PER_EXPR_PREFIX = "spark.rapids.sql.expression."


class RapidsConf:
    """One immutable-ish view of configuration; sessions own one instance."""

    def __init__(self, settings: Optional[Dict[str, Any]] = None):
        self._settings: Dict[str, Any] = dict(settings or {})
        self._lock = threading.Lock()

    def set(self, key: str, value: Any) -> "RapidsConf":
        with self._lock:
            self._settings[key] = value
        return self

    def get(self, entry: ConfEntry):
        raw = self._settings.get(entry.key, None)
        if raw is None:
            return entry.default
        return entry.conv(raw)

    def get_raw(self, key: str, default=None):
        return self._settings.get(key, default)

    # convenience accessors -------------------------------------------------
    @property
    def sql_enabled(self) -> bool:
        return self.get(SQL_ENABLED)

    @property
    def test_enabled(self) -> bool:
        return self.get(TEST_ENABLED)

    @property
    def explain(self) -> str:
        return str(self.get(EXPLAIN)).upper()

    @property
    def batch_size_bytes(self) -> int:
        return self.get(BATCH_SIZE_BYTES)

    def exec_enabled(self, name: str) -> bool:
        return _to_bool(self._settings.get(_PER_OP_PREFIX + name, True))

    def expr_enabled(self, name: str) -> bool:
        return _to_bool(self._settings.get(_PER_EXPR_PREFIX + name, True))

    def copy(self) -> "RapidsConf":
        return RapidsConf(dict(self._settings))


def help_doc() -> str:
    """Generate markdown documentation for all registered configs
    (reference analogue: RapidsConf.help -> docs/configs.md)."""
    lines = [
        "# spark.rapids.* configuration",
        "",
        "| key | default | description |",
        "|---|---|---|",
    ]
    for key in sorted(_REGISTRY):
        e = _REGISTRY[key]
        if e.internal:
            continue
        lines.append(f"| `{e.key}` | `{e.default}` | {e.doc} |")
    return "\n".join(lines) + "\n"


def registry() -> Dict[str, ConfEntry]:
    return dict(_REGISTRY)

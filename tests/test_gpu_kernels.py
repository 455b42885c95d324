"""GPU kernel numerics tests: every hipdf kernel vs the CPU reference
backend on randomized data with nulls (reference analogue: the
CPU-vs-GPU equality harness of integration_tests/)."""
import numpy as np
import pytest

import spark_rapids_amd as sr
from spark_rapids_amd import Column, ColumnBatch, DType
from spark_rapids_amd.ops import cpu_backend
from spark_rapids_amd.types import FLOAT32, FLOAT64, INT8, INT16, INT32, INT64, STRING

pytestmark = pytest.mark.gpu

RNG = np.random.default_rng(42)


def _rand_col(dtype, n=10_000, nulls=0.2, lo=-100, hi=100):
    valid = RNG.random(n) >= nulls
    if dtype.id.value == "string":
        words = ["", "a", "bb", "spark", "rapids", "mi355x", "wörld", "xyzzy"]
        vals = [None if not v else words[i % len(words)]
                for i, v in enumerate(valid)]
        return Column.from_pylist(vals, dtype)
    if dtype.is_floating:
        vals = RNG.uniform(lo, hi, n).astype(dtype.numpy_dtype())
    else:
        vals = RNG.integers(lo, hi, n).astype(dtype.numpy_dtype())
    return Column.from_numpy(vals, dtype, valid if nulls else None)


def _cols_equal(a: Column, b: Column, approx=False):
    la, lb = a.to_pylist(), b.to_pylist()
    assert len(la) == len(lb)
    for i, (x, y) in enumerate(zip(la, lb)):
        if x is None or y is None:
            assert x is None and y is None, f"row {i}: {x} != {y}"
        elif approx and isinstance(x, float):
            if np.isnan(x) or np.isnan(y):
                assert np.isnan(x) and np.isnan(y), f"row {i}: {x} != {y}"
            else:
                assert x == pytest.approx(y, rel=1e-12, abs=1e-9), f"row {i}"
        else:
            assert x == y, f"row {i}: {x} != {y}"


@pytest.mark.parametrize("dtype", [INT32, INT64, FLOAT32, FLOAT64, INT16, INT8])
@pytest.mark.parametrize("op", ["add", "sub", "mul", "min", "max"])
def test_binary_arith(dtype, op):
    a, b = _rand_col(dtype), _rand_col(dtype)
    cpu = cpu_backend.binary_op(op, a, b, dtype)
    from spark_rapids_amd.ops import gpu_backend
    gpu = gpu_backend.binary_op(op, a.cuda(), b.cuda(), dtype).cpu()
    _cols_equal(cpu, gpu, approx=dtype.is_floating)


@pytest.mark.parametrize("op", ["div", "mod", "pmod", "int_div"])
def test_binary_null_producing(op):
    dt = FLOAT64 if op == "div" else INT64
    a = _rand_col(dt)
    b = _rand_col(dt, lo=-3, hi=3)  # plenty of zero divisors
    cpu = cpu_backend.binary_op(op, a, b, dt)
    from spark_rapids_amd.ops import gpu_backend
    gpu = gpu_backend.binary_op(op, a.cuda(), b.cuda(), dt).cpu()
    _cols_equal(cpu, gpu, approx=True)


@pytest.mark.parametrize("op", ["eq", "ne", "lt", "le", "gt", "ge", "eq_null_safe"])
def test_binary_cmp(op):
    a = _rand_col(INT32, lo=-5, hi=5)
    b = _rand_col(INT32, lo=-5, hi=5)
    cpu = cpu_backend.binary_op(op, a, b, DType.bool_())
    from spark_rapids_amd.ops import gpu_backend
    gpu = gpu_backend.binary_op(op, a.cuda(), b.cuda(), DType.bool_()).cpu()
    _cols_equal(cpu, gpu)


def test_cmp_nan_semantics():
    a = Column.from_numpy(np.array([np.nan, 1.0, np.nan, -0.0]), FLOAT64)
    b = Column.from_numpy(np.array([np.nan, np.nan, 2.0, 0.0]), FLOAT64)
    from spark_rapids_amd.ops import gpu_backend
    for op, exp in [("eq", [True, False, False, True]),
                    ("lt", [False, True, False, False]),
                    ("gt", [False, False, True, False])]:
        gpu = gpu_backend.binary_op(op, a.cuda(), b.cuda(), DType.bool_()).cpu()
        assert gpu.to_pylist() == exp, op


def test_kleene_bool():
    a = Column.from_pylist([True, False, None] * 3, DType.bool_())
    b = Column.from_pylist([True] * 3 + [False] * 3 + [None] * 3, DType.bool_())
    from spark_rapids_amd.ops import gpu_backend
    for op in ("and", "or"):
        cpu = cpu_backend.binary_op(op, a, b, DType.bool_())
        gpu = gpu_backend.binary_op(op, a.cuda(), b.cuda(), DType.bool_()).cpu()
        _cols_equal(cpu, gpu)


@pytest.mark.parametrize("op", ["add", "mul", "lt"])
def test_binary_scalar(op):
    a = _rand_col(INT64)
    out_t = DType.bool_() if op == "lt" else INT64
    cpu = cpu_backend.binary_op_scalar(op, a, 7, out_t)
    from spark_rapids_amd.ops import gpu_backend
    gpu = gpu_backend.binary_op_scalar(op, a.cuda(), 7, out_t).cpu()
    _cols_equal(cpu, gpu)


@pytest.mark.parametrize("op", ["neg", "abs", "sqrt", "exp", "log", "floor", "ceil"])
def test_unary(op):
    dt = FLOAT64
    a = _rand_col(dt)
    cpu = cpu_backend.unary_op(op, a, dt)
    from spark_rapids_amd.ops import gpu_backend
    gpu = gpu_backend.unary_op(op, a.cuda(), dt).cpu()
    _cols_equal(cpu, gpu, approx=True)


@pytest.mark.parametrize("src,dst", [
    (FLOAT64, INT64), (FLOAT64, INT32), (INT64, FLOAT64), (INT32, INT64),
    (INT64, INT32), (FLOAT32, FLOAT64), (INT32, DType.bool_()),
])
def test_cast(src, dst):
    a = _rand_col(src, lo=-1000, hi=1000)
    cpu = cpu_backend.cast(a, dst)
    from spark_rapids_amd.ops import gpu_backend
    gpu = gpu_backend.cast(a.cuda(), dst).cpu()
    _cols_equal(cpu, gpu, approx=dst.is_floating)


def test_cast_nan_saturation():
    a = Column.from_numpy(np.array([np.nan, np.inf, -np.inf, 1e20, -1e20, 2.9]),
                          FLOAT64)
    from spark_rapids_amd.ops import gpu_backend
    gpu = gpu_backend.cast(a.cuda(), INT32).cpu().to_pylist()
    cpu = cpu_backend.cast(a, INT32).to_pylist()
    assert gpu == cpu == [0, 2**31 - 1, -2**31, 2**31 - 1, -2**31, 2]


def test_filter_and_gather_with_strings():
    batch = ColumnBatch([
        _rand_col(INT64), _rand_col(FLOAT64), _rand_col(STRING),
    ])
    mask = _rand_col(DType.bool_(), nulls=0.1, lo=0, hi=2)
    cpu = cpu_backend.apply_boolean_mask(batch, mask)
    from spark_rapids_amd.ops import gpu_backend
    gpu = gpu_backend.apply_boolean_mask(batch.cuda(), mask.cuda()).cpu()
    assert cpu.num_rows == gpu.num_rows
    for c, g in zip(cpu.columns, gpu.columns):
        _cols_equal(c, g, approx=True)


def test_gather_negative_indices_nullify():
    batch = ColumnBatch([_rand_col(INT64, n=100, nulls=0.0)])
    idx = Column.from_numpy(np.array([0, -1, 5, 99, -1], dtype=np.int32))
    cpu = cpu_backend.gather(batch, idx)
    from spark_rapids_amd.ops import gpu_backend
    gpu = gpu_backend.gather(batch.cuda(), idx.cuda()).cpu()
    for c, g in zip(cpu.columns, gpu.columns):
        _cols_equal(c, g)


def test_concat_batches():
    batches = [ColumnBatch([_rand_col(INT64, n=n), _rand_col(STRING, n=n)])
               for n in (100, 37, 1, 200)]
    cpu = cpu_backend.concat_batches(batches)
    from spark_rapids_amd.ops import gpu_backend
    gpu = gpu_backend.concat_batches([b.cuda() for b in batches]).cpu()
    for c, g in zip(cpu.columns, gpu.columns):
        _cols_equal(c, g)


@pytest.mark.parametrize("dtype", [INT32, INT64, FLOAT32, FLOAT64, STRING, INT8])
def test_murmur3_matches_cpu(dtype):
    c = _rand_col(dtype)
    cpu = cpu_backend.murmur3_hash([c], 42)
    from spark_rapids_amd.ops import gpu_backend
    gpu = gpu_backend.murmur3_hash([c.cuda()], 42).cpu()
    _cols_equal(cpu, gpu)


def test_murmur3_multi_column_chain():
    cols = [_rand_col(INT64), _rand_col(INT32), _rand_col(FLOAT64)]
    cpu = cpu_backend.murmur3_hash(cols, 42)
    from spark_rapids_amd.ops import gpu_backend
    gpu = gpu_backend.murmur3_hash([c.cuda() for c in cols], 42).cpu()
    _cols_equal(cpu, gpu)


def test_hash_partition_agrees_with_cpu():
    batch = ColumnBatch([_rand_col(INT64), _rand_col(FLOAT64)])
    nparts = 16
    cpu, cpu_offs = cpu_backend.hash_partition(batch, [0], nparts)
    from spark_rapids_amd.ops import gpu_backend
    gpu, gpu_offs = gpu_backend.hash_partition(batch.cuda(), [0], nparts)
    gpu = gpu.cpu()
    assert cpu_offs == gpu_offs
    # same rows in each partition (order within a partition may differ)
    for p in range(nparts):
        cs = sorted(zip(*[c.to_pylist()[cpu_offs[p]:cpu_offs[p + 1]]
                          for c in cpu.columns]), key=repr)
        gs = sorted(zip(*[c.to_pylist()[gpu_offs[p]:gpu_offs[p + 1]]
                          for c in gpu.columns]), key=repr)
        assert cs == gs


@pytest.mark.parametrize("op", ["sum", "min", "max", "count", "mean"])
def test_reduce(op):
    for dt in (INT64, FLOAT64):
        c = _rand_col(dt)
        cpu = cpu_backend.reduce(op, c)
        from spark_rapids_amd.ops import gpu_backend
        gpu = gpu_backend.reduce(op, c.cuda())
        if isinstance(cpu, float):
            assert gpu == pytest.approx(cpu)
        else:
            assert gpu == cpu


def _sorted_rows(batch: ColumnBatch):
    return sorted(zip(*[c.to_pylist() for c in batch.columns]), key=repr)


@pytest.mark.parametrize("nkeys,ngroups", [(1, 7), (1, 5000), (2, 100)])
def test_group_by_aggregate(nkeys, ngroups):
    n = 50_000
    keys = [Column.from_numpy(
        RNG.integers(0, ngroups, n).astype(np.int64), INT64,
        RNG.random(n) >= 0.05) for _ in range(nkeys)]
    vals = _rand_col(FLOAT64, n=n)
    ivals = _rand_col(INT64, n=n)
    batch = ColumnBatch(keys + [vals, ivals])
    aggs = [("sum", nkeys, FLOAT64), ("count", nkeys, INT64),
            ("min", nkeys + 1, INT64), ("max", nkeys + 1, INT64),
            ("sum", nkeys + 1, INT64), ("count_all", -1, INT64)]
    cpu = cpu_backend.group_by_aggregate(batch, list(range(nkeys)), aggs)
    from spark_rapids_amd.ops import gpu_backend
    gpu = gpu_backend.group_by_aggregate(batch.cuda(), list(range(nkeys)),
                                         aggs).cpu()
    assert cpu.num_rows == gpu.num_rows
    crows = _sorted_rows(cpu)
    grows = _sorted_rows(gpu)
    for cr, gr in zip(crows, grows):
        for x, y in zip(cr, gr):
            if isinstance(x, float) and y is not None and x is not None:
                assert x == pytest.approx(y, rel=1e-9), (cr, gr)
            else:
                assert x == y, (cr, gr)


@pytest.mark.parametrize("how", ["inner", "left", "semi", "anti"])
def test_join(how):
    n = 20_000
    left = ColumnBatch([
        Column.from_numpy(RNG.integers(0, 1000, n).astype(np.int64), INT64,
                          RNG.random(n) >= 0.05),
        _rand_col(FLOAT64, n=n),
    ])
    right = ColumnBatch([
        Column.from_numpy(RNG.integers(0, 1500, 5000).astype(np.int64), INT64,
                          RNG.random(5000) >= 0.05),
        _rand_col(INT32, n=5000),
    ])
    from spark_rapids_amd.ops import cpu_backend as cb, gpu_backend as gb
    lcpu, rcpu = cb.join_gather_maps(left, right, [0], [0], how)
    lgpu, rgpu = gb.join_gather_maps(left.cuda(), right.cuda(), [0], [0], how)
    if how in ("inner", "left"):
        cpu_out = ColumnBatch(cb.gather(left, lcpu).columns +
                              cb.gather(right, rcpu).columns)
        gpu_out = ColumnBatch(
            gb.gather(left.cuda(), lgpu).columns +
            gb.gather(right.cuda(), rgpu).columns).cpu()
    else:
        cpu_out = cb.gather(left, lcpu)
        gpu_out = gb.gather(left.cuda(), lgpu).cpu()
    assert cpu_out.num_rows == gpu_out.num_rows, how
    assert _sorted_rows(cpu_out) == _sorted_rows(gpu_out)


def test_end_to_end_query_gpu_vs_cpu():
    n = 100_000
    data = {
        "k": RNG.integers(0, 50, n).astype(np.int64),
        "v": RNG.uniform(0, 100, n),
        "w": RNG.integers(-10, 10, n).astype(np.int32),
    }

    def run(enabled):
        s = sr.Session({"spark.rapids.sql.enabled": enabled})
        df = s.create_dataframe(dict(data), num_partitions=4)
        return (df.filter((sr.col("v") > 10.0) & (sr.col("w") != 0))
                  .with_column("vw", sr.col("v") * sr.col("w").cast(sr.FLOAT64))
                  .group_by("k")
                  .agg(sr.sum_(sr.col("vw")), sr.count_star(),
                       sr.avg(sr.col("v")), sr.min_(sr.col("w")))
                  .sort("k").collect())

    gpu, cpu = run(True), run(False)
    assert len(gpu) == len(cpu)
    for g, c in zip(gpu, cpu):
        assert g[0] == c[0] and g[2] == c[2] and g[4] == c[4]
        assert g[1] == pytest.approx(c[1], rel=1e-9)
        assert g[3] == pytest.approx(c[3], rel=1e-9)


def test_gpu_plan_placement():
    s = sr.Session()
    df = s.create_dataframe({"a": [1, 2, 3]})
    tree = df.filter(sr.col("a") > 1).physical_plan().tree_string()
    assert "GpuFilter" in tree, tree


def test_retry_split_on_gpu():
    from spark_rapids_amd.memory.retry import oom_injector
    s = sr.Session()
    df = s.create_dataframe({"a": list(range(1000))})
    oom_injector.arm(1, split=True)
    assert df.filter(sr.col("a") >= 500).count() == 500


@pytest.mark.parametrize("dtype", [INT32, INT64, FLOAT64, FLOAT32, INT8])
@pytest.mark.parametrize("desc", [False, True])
def test_sort_order_single_key(dtype, desc):
    batch = ColumnBatch([_rand_col(dtype), _rand_col(INT64, nulls=0.0)])
    from spark_rapids_amd.ops import gpu_backend
    cpu = cpu_backend.sort_order(batch, [0], [desc], [desc])
    gpu = gpu_backend.sort_order(batch.cuda(), [0], [desc], [desc]).cpu()
    # permutations may differ on ties; compare gathered key column + stable
    # payload ordering via sorted rows of (key, payload)
    cpu_rows = cpu_backend.gather(batch, cpu)
    gpu_rows = cpu_backend.gather(batch, gpu)
    k_cpu = cpu_rows.columns[0].to_pylist()
    k_gpu = gpu_rows.columns[0].to_pylist()
    if dtype.is_floating:
        for a, b in zip(k_cpu, k_gpu):
            if a is None or b is None:
                assert a is None and b is None
            elif np.isnan(a) or np.isnan(b):
                assert np.isnan(a) and np.isnan(b)
            else:
                assert a == b
    else:
        assert k_cpu == k_gpu


def test_sort_order_multi_key_stable():
    n = 20_000
    a = Column.from_numpy(RNG.integers(0, 10, n).astype(np.int32), INT32,
                          RNG.random(n) >= 0.05)
    b = Column.from_numpy(RNG.integers(-50, 50, n).astype(np.int64), INT64)
    batch = ColumnBatch([a, b])
    from spark_rapids_amd.ops import gpu_backend
    for desc in ([False, False], [True, False], [False, True]):
        nl = desc[:]  # spark default: nulls last iff descending
        cpu = cpu_backend.sort_order(batch, [0, 1], desc, nl)
        gpu = gpu_backend.sort_order(batch.cuda(), [0, 1], desc, nl).cpu()
        cpu_rows = cpu_backend.gather(batch, cpu)
        gpu_rows = cpu_backend.gather(batch, gpu)
        for c, g in zip(cpu_rows.columns, gpu_rows.columns):
            assert c.to_pylist() == g.to_pylist(), desc


def test_sort_nan_greatest_gpu():
    vals = np.array([1.0, np.nan, -np.inf, np.inf, -0.0, 0.0, -5.5])
    batch = ColumnBatch([Column.from_numpy(vals, FLOAT64)])
    from spark_rapids_amd.ops import gpu_backend
    gpu = gpu_backend.sort_order(batch.cuda(), [0], [False], [False]).cpu()
    out = cpu_backend.gather(batch, gpu).columns[0].to_pylist()
    assert out[0] == -np.inf and out[-1] is not None and np.isnan(out[-1])
    assert out[-2] == np.inf


def test_sort_exec_gpu_e2e():
    s = sr.Session()
    n = 50_000
    df = s.create_dataframe({
        "a": RNG.integers(0, 100, n).astype(np.int64),
        "b": RNG.uniform(-1, 1, n),
    }, num_partitions=3)
    tree = df.sort("a").physical_plan().tree_string()
    assert "GpuSort" in tree, tree
    gpu = df.sort("a", "b").collect()
    s2 = sr.Session({"spark.rapids.sql.enabled": False})
    df2 = s2.create_dataframe({
        "a": RNG.integers(0, 100, n).astype(np.int64),
        "b": RNG.uniform(-1, 1, n),
    }, num_partitions=3)
    # data differs (rng advanced); just validate GPU output is sorted
    av = [r[0] for r in gpu]
    assert av == sorted(av)


def test_group_by_string_keys_gpu():
    words = ["alpha", "beta", "gamma", None, "", "alpha2"]
    n = 30_000
    keys = Column.from_pylist([words[i % 6] for i in range(n)], STRING)
    vals = _rand_col(FLOAT64, n=n)
    batch = ColumnBatch([keys, vals])
    aggs = [("sum", 1, FLOAT64), ("count_all", -1, INT64)]
    from spark_rapids_amd.ops import gpu_backend
    cpu = cpu_backend.group_by_aggregate(batch, [0], aggs)
    gpu = gpu_backend.group_by_aggregate(batch.cuda(), [0], aggs).cpu()
    assert cpu.num_rows == gpu.num_rows == 6
    crows = _sorted_rows(cpu)
    grows = _sorted_rows(gpu)
    for cr, gr in zip(crows, grows):
        assert cr[0] == gr[0] and cr[2] == gr[2]
        if cr[1] is not None:
            assert gr[1] == pytest.approx(cr[1], rel=1e-9)


def test_join_string_keys_gpu():
    lwords = ["a", "bb", None, "ccc", "bb", ""]
    rwords = ["bb", "ccc", "", "zz", None]
    left = ColumnBatch([Column.from_pylist(lwords * 500, STRING),
                        _rand_col(INT64, n=3000, nulls=0.0)])
    right = ColumnBatch([Column.from_pylist(rwords * 100, STRING),
                         _rand_col(INT32, n=500, nulls=0.0)])
    from spark_rapids_amd.ops import cpu_backend as cb, gpu_backend as gb
    lc, rc = cb.join_gather_maps(left, right, [0], [0], "inner")
    lg, rg = gb.join_gather_maps(left.cuda(), right.cuda(), [0], [0], "inner")
    cpu_out = ColumnBatch(cb.gather(left, lc).columns +
                          cb.gather(right, rc).columns)
    gpu_out = ColumnBatch(gb.gather(left.cuda(), lg).columns +
                          gb.gather(right.cuda(), rg).columns).cpu()
    assert _sorted_rows(cpu_out) == _sorted_rows(gpu_out)


def test_spill_device_to_disk_roundtrip_gpu(tmp_path, monkeypatch):
    monkeypatch.setenv("RAPIDS_SPILL_PATH", str(tmp_path))
    from spark_rapids_amd.memory.spill import SpillableBatch

    b = ColumnBatch([_rand_col(INT64, n=1000), _rand_col(STRING, n=1000)]).cuda()
    ref = b.cpu()
    h = SpillableBatch(b)
    assert h.state == "device"
    assert h.spill_to_host() > 0
    assert h.state == "host"
    assert h.spill_to_disk() > 0
    got = h.get()  # resurrect back onto the device
    assert got.is_cuda
    for c, g in zip(ref.columns, got.cpu().columns):
        assert c.to_pylist() == g.to_pylist()
    h.close()


def test_enabled_assert_mode_gpu():
    s = sr.Session({"spark.rapids.sql.test.enabled": True})
    df = s.create_dataframe({"a": [1.0, 2.0]})
    # fully-GPU plan passes
    assert df.filter(sr.col("a") > 1.0).count() == 1
    # a CPU-fallback op (disabled exec) must raise
    s2 = sr.Session({"spark.rapids.sql.test.enabled": True,
                     "spark.rapids.sql.exec.Filter": False})
    df2 = s2.create_dataframe({"a": [1.0, 2.0]})
    with pytest.raises(AssertionError):
        df2.filter(sr.col("a") > 1.0).count()


def test_concurrent_queries_semaphore_gpu():
    """Two threads running GPU queries concurrently under the semaphore
    (reference analogue: concurrentGpuTasks)."""
    import threading

    s = sr.Session({"spark.rapids.sql.concurrentGpuTasks": 2})
    n = 200_000
    df = s.create_dataframe({
        "k": RNG.integers(0, 100, n), "v": RNG.uniform(0, 1, n)})
    results = [None, None]
    errors = []

    def work(slot):
        try:
            out = (df.filter(sr.col("v") > 0.5).group_by("k")
                   .agg(sr.count_star()).collect())
            results[slot] = sum(r[1] for r in out)
        except Exception as e:  # noqa: BLE001
            errors.append(e)

    ts = [threading.Thread(target=work, args=(i,)) for i in range(2)]
    [t.start() for t in ts]
    [t.join() for t in ts]
    assert not errors, errors
    assert results[0] == results[1] and results[0] is not None


def test_full_outer_join_gpu():
    s = sr.Session()
    rng2 = np.random.default_rng(31)
    left = s.create_dataframe({
        "k": rng2.integers(0, 800, 5000).astype(np.int64),
        "a": rng2.uniform(0, 1, 5000)})
    right = s.create_dataframe({
        "k": rng2.integers(400, 1200, 900).astype(np.int64),
        "b": rng2.integers(0, 100, 900).astype(np.int32)})
    tree = left.join(right, on="k", how="full").physical_plan().tree_string()
    assert "GpuHashJoin" in tree, tree
    gpu = sorted(left.join(right, on="k", how="full").collect(), key=repr)
    s2 = sr.Session({"spark.rapids.sql.enabled": False})
    left2 = s2.create_dataframe({
        "k": rng2.integers(0, 800, 5000).astype(np.int64),
        "a": rng2.uniform(0, 1, 5000)})
    # regenerate identical data with the same seed stream is tricky; instead
    # compare GPU vs CPU on THE SAME session data via conf flip
    s3 = sr.Session({"spark.rapids.sql.enabled": False})
    lcpu = s3.create_dataframe(
        {"k": left.collect_batch().columns[0].to_numpy(),
         "a": left.collect_batch().columns[1].to_numpy()})
    rcpu = s3.create_dataframe(
        {"k": right.collect_batch().columns[0].to_numpy(),
         "b": right.collect_batch().columns[1].to_numpy()})
    cpu = sorted(lcpu.join(rcpu, on="k", how="full").collect(), key=repr)
    assert len(gpu) == len(cpu)
    assert gpu == cpu


@pytest.mark.gpu
def test_gpu_string_sort_matches_cpu():
    import numpy as np
    import spark_rapids_amd as sr
    from spark_rapids_amd import col as _c

    rng = np.random.default_rng(33)
    words = ["", "a", "ab", "abc", "abd", "b", "zz-very-long-string-tail",
             "zz-very-long-string-tail2", "Zed", "émile"]
    vals = [words[v] if i % 13 else None
            for i, v in enumerate(rng.integers(0, len(words), 30000))]
    tie = [int(v) for v in rng.integers(0, 1000, 30000)]

    def q(s, desc, nl):
        df = s.create_dataframe({"s": vals, "t": tie})
        out = df.sort("s", "t", descending=[desc, False]).collect()
        return out

    sg = sr.Session()
    sc = sr.Session({"spark.rapids.sql.enabled": False})
    for desc in (False, True):
        g, c = q(sg, desc, False), q(sc, desc, False)
        assert g == c, f"desc={desc}"
    tree = (sg.create_dataframe({"s": vals, "t": tie})
            .sort("s").physical_plan().tree_string())
    assert "GpuSort" in tree or "GpuTopN" in tree, tree


@pytest.mark.gpu
def test_gpu_decimal128_sort_matches_cpu():
    import numpy as np
    import spark_rapids_amd as sr
    from spark_rapids_amd import DType, col as _c

    rng = np.random.default_rng(7)
    vals = [int(h) * (2 ** 64) + int(l) if i % 11 else None
            for i, (h, l) in enumerate(zip(
                rng.integers(-2**40, 2**40, 20000),
                rng.integers(0, 2**62, 20000)))]
    d = DType.decimal(30, 2)

    def q(s, desc):
        from spark_rapids_amd.column import Column, ColumnBatch, Field, Schema

        cb = ColumnBatch([Column.from_pylist(vals, d)])
        df = s.from_batches([cb], Schema([Field("v", d)]))
        return df.sort("v", descending=desc).collect()

    sg = sr.Session()
    sc = sr.Session({"spark.rapids.sql.enabled": False})
    for desc in (False, True):
        assert q(sg, desc) == q(sc, desc), desc


@pytest.mark.gpu
def test_graft_smoke_entry():
    """The driver's smoke() contract runs end-to-end on this box."""
    import __graft_entry__

    __graft_entry__.smoke()


@pytest.mark.parametrize("how", ["inner", "left", "semi", "anti", "full"])
def test_conditional_join_gpu_matches_cpu(how):
    import spark_rapids_amd as sr
    from spark_rapids_amd import col

    rng = np.random.default_rng(31)
    nl, nr = 20_000, 6_000
    data_l = {"k": [int(v) if i % 17 else None
                    for i, v in enumerate(rng.integers(0, 800, nl))],
              "a": [float(v) for v in rng.uniform(0, 100, nl)]}
    data_r = {"k": [int(v) if i % 13 else None
                    for i, v in enumerate(rng.integers(0, 1000, nr))],
              "b": [float(v) for v in rng.uniform(0, 100, nr)]}

    def q(s):
        l = s.create_dataframe(data_l)
        r = s.create_dataframe(data_r)
        df = l.join(r, on="k", how=how, condition=col("a") < col("b"))
        return sorted(df.collect(), key=repr)

    sg = sr.Session()
    qg = sg.create_dataframe(data_l).join(
        sg.create_dataframe(data_r), on="k", how=how,
        condition=col("a") < col("b"))
    assert "GpuHashJoin" in qg.physical_plan().tree_string()
    g = q(sg)
    c = q(sr.Session({"spark.rapids.sql.enabled": False}))
    assert len(g) == len(c), how
    for gr, cr in zip(g, c):
        for x, y in zip(gr, cr):
            if isinstance(y, float) and x is not None:
                assert x == pytest.approx(y, rel=1e-9), (how, gr, cr)
            else:
                assert x == y, (how, gr, cr)


@pytest.mark.parametrize("how", ["inner", "left", "semi", "anti", "full"])
def test_nested_loop_join_gpu_matches_cpu(how):
    import spark_rapids_amd as sr
    from spark_rapids_amd import col

    rng = np.random.default_rng(37)
    data_l = {"a": [float(v) if i % 19 else None
                    for i, v in enumerate(rng.uniform(0, 100, 2000))]}
    data_r = {"b": [float(v) for v in rng.uniform(0, 100, 500)]}

    def q(s):
        l = s.create_dataframe(data_l)
        r = s.create_dataframe(data_r)
        return sorted(l.join_nl(r, (col("a") < col("b") + 0.5)
                        & (col("b") < col("a") + 0.5), how).collect(), key=repr)

    sg = sr.Session()
    qg = sg.create_dataframe(data_l).join_nl(
        sg.create_dataframe(data_r), (col("a") < col("b") + 0.5) & (col("b") < col("a") + 0.5), how)
    assert "GpuNestedLoopJoin" in qg.physical_plan().tree_string()
    g = q(sg)
    c = q(sr.Session({"spark.rapids.sql.enabled": False}))
    assert len(g) == len(c), how
    for gr, cr in zip(g, c):
        for x, y in zip(gr, cr):
            if isinstance(y, float) and x is not None:
                assert x == pytest.approx(y, rel=1e-9), (how, gr, cr)
            else:
                assert x == y, (how, gr, cr)


def test_group_min_max_strings_gpu():
    import spark_rapids_amd as sr
    from spark_rapids_amd import col, min_, max_

    rng = np.random.default_rng(41)
    n = 30_000
    words = ["", "a", "zz", "alpha", "beta", "Zeta", "ä", "m"]
    data = {"k": [int(v) for v in rng.integers(0, 200, n)],
            "s": [None if i % 11 == 0 else words[int(v)]
                  for i, v in enumerate(rng.integers(0, 8, n))]}

    def q(s):
        df = s.create_dataframe(data)
        return sorted(df.group_by("k")
                      .agg(min_(col("s")), max_(col("s"))).collect())

    sg = sr.Session()
    df = sg.create_dataframe(data).group_by("k").agg(min_(col("s")))
    assert "GpuHashAggregate" in df.physical_plan().tree_string()
    g = q(sg)
    c = q(sr.Session({"spark.rapids.sql.enabled": False}))
    assert g == c
    # keyless reduction over strings
    gk = sg.create_dataframe(data).agg(min_(col("s")),
                                       max_(col("s"))).collect()
    ck = sr.Session({"spark.rapids.sql.enabled": False}) \
        .create_dataframe(data).agg(min_(col("s")),
                                    max_(col("s"))).collect()
    assert gk == ck

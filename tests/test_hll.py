"""xxHash64 + HyperLogLog++ approx_count_distinct (reference analogues:
spark-rapids-jni Hash.xxhash64 and HyperLogLogPlusPlusHostUDF)."""
import numpy as np
import pytest

import spark_rapids_amd as sr
from spark_rapids_amd import approx_count_distinct, col, count_distinct


@pytest.fixture
def cpu():
    return sr.Session({"spark.rapids.sql.enabled": False})


def test_xxh64_canonical_vectors():
    from spark_rapids_amd.ops.cpu_backend import xxh64_bytes, xxh64_long

    # canonical XXH64 reference vectors
    assert xxh64_bytes(b"", 0) == 0xEF46DB3751D8E999
    assert xxh64_bytes(b"a", 0) == 0xD24EC4F1A98C6E5B
    assert xxh64_bytes(b"abc", 0) == 0x44BC2CF5AD770999
    assert xxh64_bytes(b"Hello, world!", 0) == 0xF58336A78B6F9476
    assert xxh64_bytes(b"x" * 100, 12345) == xxh64_bytes(b"x" * 100, 12345)
    assert xxh64_long(0, 0) != xxh64_long(1, 0)


def test_cpu_hll_accuracy(cpu):
    rng = np.random.default_rng(7)
    n = 50_000
    true_card = 8_000
    df = cpu.create_dataframe({
        "g": rng.integers(0, 4, n),
        "v": rng.integers(0, true_card, n),
    })
    out = dict((g, c) for g, c in
               df.group_by("g").agg(approx_count_distinct(col("v")))
               .collect())
    exact = dict((g, c) for g, c in
                 df.group_by("g").agg(count_distinct(col("v"))).collect())
    for g in exact:
        rel = abs(out[g] - exact[g]) / exact[g]
        assert rel < 0.15, (g, out[g], exact[g])


def test_cpu_hll_small_range_is_near_exact(cpu):
    df = cpu.create_dataframe({"v": list(range(40)) * 3})
    (got,) = df.agg(approx_count_distinct(col("v"))).collect()[0]
    assert abs(got - 40) <= 2, got


def test_cpu_hll_strings_and_nulls(cpu):
    vals = [f"s{i % 300}" for i in range(5000)]
    vals[::7] = [None] * len(vals[::7])
    df = cpu.create_dataframe({"v": vals})
    (got,) = df.agg(approx_count_distinct(col("v"))).collect()[0]
    assert abs(got - 300) / 300 < 0.1, got


@pytest.mark.gpu
def test_gpu_xxhash64_matches_cpu():
    from spark_rapids_amd import Column
    from spark_rapids_amd.ops import cpu_backend, gpu_backend
    from spark_rapids_amd.types import FLOAT64, INT32, INT64, STRING

    rng = np.random.default_rng(3)
    cols = [
        Column.from_numpy(rng.integers(-10**9, 10**9, 20000)
                          .astype(np.int64), INT64,
                          rng.random(20000) >= 0.05),
        Column.from_numpy(rng.integers(-100, 100, 20000).astype(np.int32),
                          INT32),
        Column.from_numpy(rng.uniform(-1, 1, 20000), FLOAT64),
        Column.from_pylist([None if i % 11 == 0 else f"str{i % 997}"
                            for i in range(20000)], STRING),
    ]
    h_cpu = cpu_backend.xxhash64(cols).to_pylist()
    h_gpu = gpu_backend.xxhash64([c.cuda() for c in cols]).cpu().to_pylist()
    assert h_cpu == h_gpu


@pytest.mark.gpu
def test_gpu_hll_matches_cpu():
    rng = np.random.default_rng(5)
    n = 100_000
    data = {"g": [int(v) for v in rng.integers(0, 8, n)],
            "v": [int(v) for v in rng.zipf(1.5, n) % 50_000],
            "s": [f"k{int(v) % 3000}" for v in rng.integers(0, 10**9, n)]}

    def q(s):
        df = s.create_dataframe(data)
        return sorted(df.group_by("g")
                      .agg(approx_count_distinct(col("v")),
                           approx_count_distinct(col("s"))).collect())

    g = q(sr.Session())
    c = q(sr.Session({"spark.rapids.sql.enabled": False}))
    assert g == c  # same hash + same registers -> identical estimates

"""date_format / to_timestamp / from_unixtime / timezone conversion
(reference analogues: datetimeExpressions + GpuTimeZoneDB)."""
from datetime import datetime, timezone

import numpy as np
import pytest

import spark_rapids_amd as sr
from spark_rapids_amd import col
from spark_rapids_amd.expr.datetime import (FormatUnsupported, compile_format,
                                            date_format, from_unixtime,
                                            from_utc_timestamp, to_timestamp,
                                            to_utc_timestamp)
from spark_rapids_amd.types import TIMESTAMP


@pytest.fixture
def cpu():
    return sr.Session({"spark.rapids.sql.enabled": False})


def _us(*a):
    return int(datetime(*a, tzinfo=timezone.utc).timestamp() * 1_000_000)


def test_compile_format():
    toks, w = compile_format("yyyy-MM-dd HH:mm:ss")
    assert w == 19 and len(toks) == 11
    with pytest.raises(FormatUnsupported):
        compile_format("yyyy-MM-dd EEEE")


def test_cpu_date_format_roundtrip(cpu):
    ts = [_us(2021, 3, 14, 6, 59, 59), _us(1999, 12, 31, 23, 0, 1),
          _us(1970, 1, 1), None, _us(1969, 7, 20, 20, 17, 40)]
    df = cpu.create_dataframe({"t": ts}, dtypes={"t": TIMESTAMP})
    out = df.select(
        date_format(col("t"), "yyyy-MM-dd HH:mm:ss").alias("s"),
        date_format(col("t"), "yyyy/MM").alias("ym")).to_pydict()
    assert out["s"][0] == "2021-03-14 06:59:59"
    assert out["s"][1] == "1999-12-31 23:00:01"
    assert out["s"][3] is None
    assert out["s"][4] == "1969-07-20 20:17:40"  # pre-epoch
    assert out["ym"][0] == "2021/03"
    back = df.select(to_timestamp(
        date_format(col("t"), "yyyy-MM-dd HH:mm:ss")).alias("b")) \
        .to_pydict()["b"]
    assert back == ts


def test_cpu_to_timestamp_invalid_null(cpu):
    df = cpu.create_dataframe({"s": ["2020-02-29 10:00:00",
                                     "2021-02-29 10:00:00",  # no leap day
                                     "2021-13-01 00:00:00",
                                     "garbage", None,
                                     "2021-01-01x00:00:00"]})
    out = df.select(to_timestamp(col("s")).alias("t")).to_pydict()["t"]
    assert out[0] == _us(2020, 2, 29, 10)
    assert out[1] is None and out[2] is None and out[3] is None
    assert out[4] is None and out[5] is None


def test_cpu_from_unixtime(cpu):
    df = cpu.create_dataframe({"s": [0, 86400, 1_600_000_000]})
    out = df.select(from_unixtime(col("s")).alias("f")).to_pydict()["f"]
    assert out[0] == "1970-01-01 00:00:00"
    assert out[1] == "1970-01-02 00:00:00"
    assert out[2] == "2020-09-13 12:26:40"


def test_cpu_tz_convert_matches_zoneinfo(cpu):
    from zoneinfo import ZoneInfo

    zones = ["America/New_York", "Europe/Berlin", "Asia/Kolkata",
             "Australia/Sydney"]
    rng = np.random.default_rng(5)
    secs = rng.integers(0, 4_000_000_000, 300)
    ts = [int(s) * 1_000_000 for s in secs]
    df = cpu.create_dataframe({"t": ts}, dtypes={"t": TIMESTAMP})
    for z in zones:
        out = df.select(
            from_utc_timestamp(col("t"), z).alias("w")).to_pydict()["w"]
        for s, w in zip(secs, out):
            utc = datetime.fromtimestamp(int(s), tz=timezone.utc)
            expect = utc.astimezone(ZoneInfo(z)).replace(tzinfo=timezone.utc)
            assert w == int(expect.timestamp() * 1_000_000), (z, s)


def test_cpu_to_utc_roundtrip(cpu):
    # unambiguous instants round-trip exactly
    ts = [_us(2021, 6, 1, 12), _us(2021, 1, 15, 3), _us(1995, 4, 2, 9)]
    df = cpu.create_dataframe({"t": ts}, dtypes={"t": TIMESTAMP})
    z = "America/New_York"
    back = df.select(to_utc_timestamp(
        from_utc_timestamp(col("t"), z), z).alias("b")).to_pydict()["b"]
    assert back == ts


@pytest.mark.gpu
def test_gpu_datetime_matches_cpu():
    rng = np.random.default_rng(7)
    ts = [int(v) * 1_000_000 for v in
          rng.integers(-2_000_000_000, 4_000_000_000, 20000)] + [None]

    def q(s):
        df = s.create_dataframe({"t": ts}, dtypes={"t": TIMESTAMP})
        return df.select(
            date_format(col("t"), "yyyy-MM-dd HH:mm:ss").alias("s"),
            from_utc_timestamp(col("t"), "America/New_York").alias("ny"),
            to_utc_timestamp(col("t"), "Europe/Berlin").alias("ber"),
        ).to_pydict()

    g = q(sr.Session())
    c = q(sr.Session({"spark.rapids.sql.enabled": False}))
    for k in g:
        bad = [(i, a, b) for i, (a, b) in enumerate(zip(g[k], c[k]))
               if a != b]
        assert not bad, (k, bad[:5])


@pytest.mark.gpu
def test_gpu_ts_parse_matches_cpu():
    rng = np.random.default_rng(8)
    good = [f"{y:04d}-{m:02d}-{d:02d} {h:02d}:{mi:02d}:{s:02d}"
            for y, m, d, h, mi, s in zip(
                rng.integers(1900, 2100, 5000), rng.integers(1, 13, 5000),
                rng.integers(1, 32, 5000), rng.integers(0, 24, 5000),
                rng.integers(0, 60, 5000), rng.integers(0, 60, 5000))]
    cases = good + ["junk", "2020-1-1 0:0:0", None, ""]

    def q(s):
        df = s.create_dataframe({"x": cases})
        return df.select(to_timestamp(col("x")).alias("t")).to_pydict()["t"]

    g = q(sr.Session())
    c = q(sr.Session({"spark.rapids.sql.enabled": False}))
    bad = [(i, cases[i], a, b) for i, (a, b) in enumerate(zip(g, c))
           if a != b]
    assert not bad, bad[:5]

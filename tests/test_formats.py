"""CSV / ORC / JSON scan + write tests (host parse, columnar engine on top)."""
import numpy as np
import pyarrow as pa
import pytest

import spark_rapids_amd as sr
from spark_rapids_amd import col, count_star, sum_


def test_csv_roundtrip(tmp_path, session):
    df = session.create_dataframe({"a": [1, 2, 3], "b": [1.5, 2.5, None],
                                   "s": ["x", "y", "z"]})
    p = str(tmp_path / "t.csv")
    session.write_csv(df, p)
    back = session.read_csv(p)
    assert back.to_pydict()["a"] == [1, 2, 3]
    assert back.filter(col("a") > 1).count() == 2


def test_orc_roundtrip(tmp_path, session):
    df = session.create_dataframe({"a": [1, None, 3], "b": [1.5, 2.5, 3.5]})
    p = str(tmp_path / "t.orc")
    session.write_orc(df, p)
    back = session.read_orc(p)
    assert back.to_pydict()["a"] == [1, None, 3]
    assert back.agg(sum_(col("b"))).collect()[0][0] == pytest.approx(7.5)


def test_json_lines(tmp_path, session):
    p = tmp_path / "t.jsonl"
    p.write_text('{"a": 1, "s": "x"}\n{"a": 2, "s": null}\n{"a": 3, "s": "z"}\n')
    back = session.read_json(str(p))
    assert back.to_pydict()["a"] == [1, 2, 3]
    assert back.to_pydict()["s"] == ["x", None, "z"]


@pytest.mark.gpu
def test_gpu_csv_matches_arrow(tmp_path):
    import numpy as np

    rng = np.random.default_rng(4)
    n = 30_000
    lines = ["i,f,s"]
    for k in range(n):
        i = "" if k % 17 == 0 else str(int(rng.integers(-10**12, 10**12)))
        f = "" if k % 23 == 0 else repr(float(rng.uniform(-1e6, 1e6)))
        sv = "" if k % 13 == 0 else f"name_{k}"
        lines.append(f"{i},{f},{sv}")
    p = str(tmp_path / "t.csv")
    with open(p, "w") as fh:
        fh.write("\n".join(lines) + "\n")
    sg = sr.Session()
    df = sg.read_csv(p)
    got = df.to_pydict()
    import pyarrow.csv as pacsv

    exp = pacsv.read_csv(
        p, convert_options=pacsv.ConvertOptions(strings_can_be_null=True))
    assert got["i"] == exp.column("i").to_pylist()
    ge, ee = got["f"], exp.column("f").to_pylist()
    for a, b in zip(ge, ee):
        assert (a is None) == (b is None)
        if a is not None:
            assert a == pytest.approx(b, rel=1e-14, abs=1e-300)
    assert got["s"] == exp.column("s").to_pylist()
    # a GPU query over the csv scan
    out = df.filter(col("i") > 0).agg(count_star()).collect()
    cpuv = sum(1 for v in exp.column("i").to_pylist()
               if v is not None and v > 0)
    assert out[0][0] == cpuv


@pytest.mark.gpu
def test_gpu_csv_quoted_falls_back(tmp_path):
    p = str(tmp_path / "q.csv")
    with open(p, "w") as fh:
        fh.write('a,b\n1,"x,y"\n2,plain\n')
    sg = sr.Session()
    out = sg.read_csv(p).to_pydict()
    assert out["a"] == [1, 2]
    assert out["b"] == ["x,y", "plain"]


@pytest.mark.gpu
def test_gpu_json_matches_arrow(tmp_path):
    import json as pyjson

    import numpy as np

    rng = np.random.default_rng(9)
    n = 10_000
    rows = []
    for k in range(n):
        r = {"i": int(rng.integers(-10**9, 10**9)),
             "f": float(round(rng.uniform(-100, 100), 6)),
             "s": f"v{k}", "b": bool(k % 3)}
        if k % 11 == 0:
            r["i"] = None
        if k % 13 == 0:
            del r["s"]
        rows.append(r)
    p = str(tmp_path / "t.json")
    with open(p, "w") as fh:
        for r in rows:
            fh.write(pyjson.dumps(r) + "\n")
    sg = sr.Session()
    got = sg.read_json(p).to_pydict()
    assert got["i"] == [r["i"] for r in rows]
    assert got["s"] == [r.get("s") for r in rows]
    assert got["b"] == [r["b"] for r in rows]
    for a, r in zip(got["f"], rows):
        assert a == pytest.approx(r["f"], rel=1e-12)


def _orc_file(tmp_path, n=40_000, compression="uncompressed"):
    import pyarrow.orc as paorc

    rng = np.random.default_rng(12)
    big = rng.integers(-10**12, 10**12, n)
    big[::97] = 3  # outlier mix pushes the writer into patched-base runs
    t = pa.table({
        "i64": pa.array([int(v) if i % 11 else None
                         for i, v in enumerate(big)], pa.int64()),
        "i32": pa.array(rng.integers(-1000, 1000, n).astype(np.int32)),
        "f64": pa.array([float(v) if i % 7 else None
                         for i, v in enumerate(rng.uniform(-1, 1, n))]),
        "s": pa.array([f"word{v}" if v % 5 else None for v in range(n)]),
        "b": pa.array([bool(v % 3 == 0) for v in range(n)]),
    })
    p = str(tmp_path / "t.orc")
    paorc.write_table(t, p, compression=compression)
    return p, t


def test_orc_host_decoders_vs_pyarrow(tmp_path):
    """Host reference RLEv2/bool-RLE decoders against pyarrow (incl.
    patched-base runs)."""
    from spark_rapids_amd.io import orc_meta as om

    p, t = _orc_file(tmp_path)
    raw = open(p, "rb").read()
    meta = om.read_meta(p)
    assert meta.num_rows == t.num_rows
    st = meta.stripes[0]
    streams = om.stripe_streams(raw, meta, st)

    def get(col, kind):
        for s in streams:
            if s.column == col and s.kind == kind:
                return om._decompress(raw[s.offset:s.offset + s.length],
                                      meta.compression)

    exp = t.column("i64").to_pylist()[:st.num_rows]
    pres = om.bool_rle_decode(get(1, 0), st.num_rows)
    assert (pres == np.array([v is not None for v in exp])).all()
    nv = int(pres.sum())
    vals = om.rle_v2_decode(get(1, 1), nv, signed=True)
    assert (vals == np.array([v for v in exp if v is not None])).all()


@pytest.mark.gpu
@pytest.mark.parametrize("compression", ["uncompressed", "zlib"])
def test_gpu_orc_matches_pyarrow(tmp_path, compression):
    p, t = _orc_file(tmp_path, compression=compression)
    sg = sr.Session()
    got = sg.read_orc(p).to_pydict()
    for name in t.schema.names:
        exp = t.column(name).to_pylist()
        if name == "f64":
            for a, b in zip(got[name], exp):
                assert (a is None) == (b is None)
                if a is not None:
                    assert a == pytest.approx(b, rel=1e-15)
        else:
            assert got[name] == exp, name


@pytest.mark.gpu
def test_gpu_orc_query(tmp_path):
    p, t = _orc_file(tmp_path, n=20000)
    sg = sr.Session()
    out = (sg.read_orc(p).filter(col("i32") > 0)
           .agg(count_star(), sum_(col("i32"))).collect())
    vals = [v for v in t.column("i32").to_pylist() if v is not None and v > 0]
    assert out[0][0] == len(vals) and out[0][1] == sum(vals)


def test_hive_text(tmp_path, session):
    p = str(tmp_path / "h.txt")
    with open(p, "w") as f:
        f.write("1\x01alpha\x012.5\n2\x01beta\x013.5\n")
    df = session.read_hive_text(p)
    rows = df.collect()
    assert len(rows) == 2 and rows[0][1] == "alpha" and rows[1][2] == 3.5


def test_avro_roundtrip(tmp_path, session):
    df = session.create_dataframe({
        "i": [1, 2, None, 4],
        "f": [1.5, None, 2.5, -3.5],
        "s": ["x", "", None, "zz"],
        "b": [True, False, None, True]})
    p = str(tmp_path / "t.avro")
    session.write_avro(df, p)
    back = session.read_avro(p).to_pydict()
    assert back["i"] == [1, 2, None, 4]
    assert back["f"] == [1.5, None, 2.5, -3.5]
    assert back["s"] == ["x", "", None, "zz"]
    assert back["b"] == [True, False, None, True]


def test_avro_null_codec_and_bytes(tmp_path, session):
    df = session.create_dataframe({"i": list(range(500)),
                                   "s": [f"v{v}" for v in range(500)]})
    p = str(tmp_path / "u.avro")
    session.write_avro(df, p, codec="null")
    out = session.read_avro(p)
    assert out.count() == 500
    assert out.to_pydict()["s"][123] == "v123"


def _make_delta(tmp_path, session):
    import json
    import os

    root = str(tmp_path / "dtab")
    log = os.path.join(root, "_delta_log")
    os.makedirs(log)
    # three parquet files; one later removed
    for i in range(3):
        df = session.create_dataframe({"a": [i * 10 + k for k in range(5)],
                                       "s": [f"f{i}"] * 5})
        session.write_parquet(df, os.path.join(root, f"part-{i}.parquet"))
    with open(os.path.join(log, "00000000000000000000.json"), "w") as f:
        f.write(json.dumps({"metaData": {"configuration": {}}}) + "\n")
        for i in range(3):
            f.write(json.dumps({"add": {"path": f"part-{i}.parquet"}}) + "\n")
    with open(os.path.join(log, "00000000000000000001.json"), "w") as f:
        f.write(json.dumps({"remove": {"path": "part-1.parquet"}}) + "\n")
    return root


def test_delta_log_replay(tmp_path, session):
    root = _make_delta(tmp_path, session)
    df = session.read_delta(root)
    rows = sorted(df.collect())
    assert len(rows) == 10  # files 0 and 2 live, file 1 removed
    assert {r[1] for r in rows} == {"f0", "f2"}
    out = df.agg(count_star(), sum_(col("a"))).collect()
    assert out[0][0] == 10
    assert out[0][1] == sum(range(0, 5)) + sum(range(20, 25))


def test_delta_deletion_vectors_unsupported(tmp_path, session):
    import json
    import os

    root = str(tmp_path / "dv")
    os.makedirs(os.path.join(root, "_delta_log"))
    with open(os.path.join(root, "_delta_log",
                           "00000000000000000000.json"), "w") as f:
        f.write(json.dumps({"add": {"path": "x.parquet",
                                    "deletionVector": {"id": 1}}}) + "\n")
    with pytest.raises(NotImplementedError):
        session.read_delta(root)


@pytest.mark.gpu
def test_gpu_orc_multi_stripe(tmp_path):
    import pyarrow.orc as paorc

    rng = np.random.default_rng(5)
    n = 60_000
    t = pa.table({
        "a": pa.array([int(v) if i % 9 else None for i, v in
                       enumerate(rng.integers(-10**9, 10**9, n))],
                      pa.int64()),
        "s": pa.array([f"r{v}" for v in range(n)]),
    })
    p = str(tmp_path / "ms.orc")
    paorc.write_table(t, p, stripe_size=64 * 1024,
                      dictionary_key_size_threshold=0.0)
    sg = sr.Session()
    meta_stripes = len(__import__(
        "spark_rapids_amd.io.orc_meta", fromlist=["read_meta"]
    ).read_meta(p).stripes)
    assert meta_stripes > 1, meta_stripes
    got = sg.read_orc(p).to_pydict()
    assert got["a"] == t.column("a").to_pylist()
    assert got["s"] == t.column("s").to_pylist()


def test_delta_checkpoint_replay(tmp_path, session):
    import json
    import os

    import pyarrow.parquet as apq

    root = str(tmp_path / "dckpt")
    log = os.path.join(root, "_delta_log")
    os.makedirs(log)
    for i in range(3):
        df = session.create_dataframe({"a": [i * 10 + k for k in range(4)]})
        session.write_parquet(df, os.path.join(root, f"p{i}.parquet"))
    # checkpoint at version 1 holds adds for p0/p1 and a remove of p0
    ck = pa.table({
        "add": [{"path": "p0.parquet"}, {"path": "p1.parquet"}, None],
        "remove": [None, None, {"path": "p0.parquet"}],
    })
    apq.write_table(ck, os.path.join(
        log, f"{1:020d}.checkpoint.parquet"))
    with open(os.path.join(log, "_last_checkpoint"), "w") as f:
        f.write(json.dumps({"version": 1}))
    # later JSON commit adds p2
    with open(os.path.join(log, f"{2:020d}.json"), "w") as f:
        f.write(json.dumps({"add": {"path": "p2.parquet"}}) + "\n")
    # an older JSON commit that must be IGNORED (pre-checkpoint)
    with open(os.path.join(log, f"{0:020d}.json"), "w") as f:
        f.write(json.dumps({"add": {"path": "ghost.parquet"}}) + "\n")
    rows = sorted(r[0] for r in session.read_delta(root).collect())
    assert rows == sorted(list(range(10, 14)) + list(range(20, 24)))

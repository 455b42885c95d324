"""String function tests: CPU semantics + GPU kernel equality."""
import numpy as np
import pytest

import spark_rapids_amd as sr
from spark_rapids_amd import Column, ColumnBatch, STRING, col
from spark_rapids_amd.ops import cpu_backend

WORDS = ["hello", "Hello World", "", "spark", None, "SPARKLE", "park", "spa",
         "end%", "a_b", "wörld", "sp"]


def _scol(n=1200):
    return Column.from_pylist([WORDS[i % len(WORDS)] for i in range(n)], STRING)


def test_cpu_contains_like(session):
    df = session.create_dataframe({"s": WORDS})
    out = df.select(col("s").contains("park").alias("x")).to_pydict()["x"]
    assert out == [False, False, False, True, None, False, True, False,
                   False, False, False, False]
    out = df.select(col("s").like("sp%").alias("x")).to_pydict()["x"]
    assert out == [False, False, False, True, None, False, False, True,
                   False, False, False, True]
    out = df.select(col("s").like("a_b").alias("x")).to_pydict()["x"]
    assert out[9] is True


def test_cpu_substring(session):
    df = session.create_dataframe({"s": ["hello", None, "ab", ""]})
    out = df.select(col("s").substr(2, 3).alias("x")).to_pydict()["x"]
    assert out == ["ell", None, "b", ""]
    out = df.select(col("s").substr(-3, 2).alias("x")).to_pydict()["x"]
    assert out == ["ll", None, "ab", ""]


def test_cpu_length_case(session):
    df = session.create_dataframe({"s": ["abc", "wörld", None, ""]})
    out = df.select(col("s").length().alias("x")).to_pydict()["x"]
    assert out == [3, 5, None, 0]
    out = df.select(col("s").upper().alias("x")).to_pydict()["x"]
    assert out == ["ABC", "WÖRLD", None, ""]


@pytest.mark.gpu
@pytest.mark.parametrize("op", ["contains", "starts_with", "ends_with"])
def test_gpu_str_find(op):
    from spark_rapids_amd.ops import gpu_backend
    c = _scol()
    for pat in ("spark", "sp", "", "park"):
        cpu = cpu_backend.str_predicate(op, c, pat)
        gpu = gpu_backend.str_predicate(op, c.cuda(), pat).cpu()
        assert cpu.to_pylist() == gpu.to_pylist(), (op, pat)


@pytest.mark.gpu
@pytest.mark.parametrize("pat", ["sp%", "%ark%", "a_b", "%", "_", "he__o",
                                 "w_rld"])
def test_gpu_like(pat):
    from spark_rapids_amd.ops import gpu_backend
    c = _scol()
    cpu = cpu_backend.str_predicate("like", c, pat)
    gpu = gpu_backend.str_predicate("like", c.cuda(), pat).cpu()
    assert cpu.to_pylist() == gpu.to_pylist(), pat


@pytest.mark.gpu
@pytest.mark.parametrize("op", ["eq", "ne", "lt", "ge"])
def test_gpu_str_cmp(op):
    from spark_rapids_amd.ops import gpu_backend
    from spark_rapids_amd import DType
    a, b = _scol(), Column.from_pylist(
        [WORDS[(i * 7 + 3) % len(WORDS)] for i in range(1200)], STRING)
    cpu = cpu_backend.binary_op(op, a, b, DType.bool_())
    gpu = gpu_backend.binary_op(op, a.cuda(), b.cuda(), DType.bool_()).cpu()
    assert cpu.to_pylist() == gpu.to_pylist()
    cpu = cpu_backend.binary_op_scalar(op, a, "spark", DType.bool_())
    gpu = gpu_backend.binary_op_scalar(op, a.cuda(), "spark",
                                       DType.bool_()).cpu()
    assert cpu.to_pylist() == gpu.to_pylist()


@pytest.mark.gpu
@pytest.mark.parametrize("pos,ln", [(2, 3), (1, -1), (-3, 2), (0, 2), (7, 5)])
def test_gpu_substring(pos, ln):
    from spark_rapids_amd.ops import gpu_backend
    c = _scol()
    cpu = cpu_backend.substring(c, pos, ln)
    gpu = gpu_backend.substring(c.cuda(), pos, ln).cpu()
    assert cpu.to_pylist() == gpu.to_pylist()


@pytest.mark.gpu
def test_gpu_length_and_case_ascii():
    from spark_rapids_amd.ops import gpu_backend
    from spark_rapids_amd.types import INT32, STRING as STR
    ascii_col = Column.from_pylist(["abc", "XYZ", None, "", "MiXeD1"], STRING)
    cpu = cpu_backend.unary_op("length", ascii_col, INT32)
    gpu = gpu_backend.unary_op("length", ascii_col.cuda(), INT32).cpu()
    assert cpu.to_pylist() == gpu.to_pylist()
    for op in ("upper", "lower"):
        cpu = cpu_backend.unary_op(op, ascii_col, STR)
        gpu = gpu_backend.unary_op(op, ascii_col.cuda(), STR).cpu()
        assert cpu.to_pylist() == gpu.to_pylist()


@pytest.mark.gpu
def test_gpu_string_filter_e2e():
    s = sr.Session()
    n = 30_000
    df = s.create_dataframe({
        "s": [WORDS[i % len(WORDS)] for i in range(n)],
        "v": list(range(n)),
    })
    tree = df.filter(col("s").like("sp%")).physical_plan().tree_string()
    assert "GpuFilter" in tree, tree
    gpu = df.filter(col("s").like("sp%")).count()
    s2 = sr.Session({"spark.rapids.sql.enabled": False})
    df2 = s2.create_dataframe({
        "s": [WORDS[i % len(WORDS)] for i in range(n)],
        "v": list(range(n)),
    })
    assert gpu == df2.filter(col("s").like("sp%")).count()


class TestTrimConcatReplace:
    @pytest.fixture
    def cpu(self):
        return sr.Session({"spark.rapids.sql.enabled": False})

    def test_cpu(self, cpu):
        df = cpu.create_dataframe({"a": ["  hi ", "x", None, ""],
                                   "b": ["A", None, "B", ""]})
        out = df.select(
            col("a").trim().alias("t"), col("a").ltrim().alias("l"),
            col("a").rtrim().alias("r"),
            col("a").concat(col("b")).alias("c"),
            col("a").replace("hi", "yo$").alias("rep")).to_pydict()
        assert out["t"] == ["hi", "x", None, ""]
        assert out["l"] == ["hi ", "x", None, ""]
        assert out["r"] == ["  hi", "x", None, ""]
        assert out["c"] == ["  hi A", None, None, ""]
        assert out["rep"] == ["  yo$ ", "x", None, ""]

    @pytest.mark.gpu
    def test_gpu_matches_cpu(self):
        import numpy as np

        rng = np.random.default_rng(2)
        base = ["  pad %d " % v if v % 3 else "x%d.y" % v
                for v in rng.integers(0, 1000, 5000)]
        vals = [v if i % 17 else None for i, v in enumerate(base)]

        def q(s):
            df = s.create_dataframe({"a": vals, "b": base})
            return df.select(
                col("a").trim().alias("t"),
                col("a").concat(col("b")).alias("c"),
                col("a").concat("#").alias("cl"),
                col("a").replace(".", "_").alias("rep")).to_pydict()

        sg = sr.Session()
        sc = sr.Session({"spark.rapids.sql.enabled": False})
        assert q(sg) == q(sc)

    @pytest.mark.gpu
    def test_gpu_placement(self):
        sg = sr.Session()
        df = sg.create_dataframe({"a": ["q"]})
        tree = (df.select(col("a").trim().concat("z"))
                .physical_plan().tree_string())
        assert "GpuProject" in tree, tree


class TestSplitSize:
    @pytest.fixture
    def cpu(self):
        return sr.Session({"spark.rapids.sql.enabled": False})

    def test_cpu_java_semantics(self, cpu):
        df = cpu.create_dataframe({"s": ["a,b,c", "x", "", "a,,", ",,",
                                         None, ",lead"]})
        out = df.select(col("s").split(",").alias("p")) \
                .with_column("n", col("p").size()).to_pydict()
        assert out["p"] == [["a", "b", "c"], ["x"], [""], ["a"], [],
                            None, ["", "lead"]]
        assert out["n"] == [3, 1, 1, 1, 0, None, 2]

    def test_split_explode(self, cpu):
        df = cpu.create_dataframe({"s": ["a b", "c"]})
        out = df.select(col("s").split(" ").alias("w")).explode("w").collect()
        assert out == [("a",), ("b",), ("c",)]

    def test_sql_split_size(self, cpu):
        cpu.register("tsplit", cpu.create_dataframe({"s": ["a|b|c"]}))
        # note: | must go through CPU (regex special) — still correct
        out = cpu.sql("SELECT size(split(s, ',')) FROM tsplit").collect()
        assert out == [(1,)]

    @pytest.mark.gpu
    def test_gpu_matches_cpu(self):
        import numpy as np

        rng = np.random.default_rng(8)
        toks = ["a", "bb", "", "ccc"]
        vals = [",".join(toks[j % 4] for j in range(v % 6))
                if i % 13 else None
                for i, v in enumerate(rng.integers(0, 40, 6000))]

        def q(s):
            df = s.create_dataframe({"s": vals})
            return (df.select(col("s").split(",").alias("p"))
                    .with_column("n", col("p").size()).to_pydict())

        sg = sr.Session()
        sc = sr.Session({"spark.rapids.sql.enabled": False})
        assert q(sg) == q(sc)

    @pytest.mark.gpu
    def test_gpu_split_placement(self):
        sg = sr.Session()
        df = sg.create_dataframe({"s": ["q,r"]})
        tree = (df.select(col("s").split(",").alias("p"))
                .physical_plan().tree_string())
        assert "GpuProject" in tree, tree


class TestElementAt:
    @pytest.fixture
    def cpu(self):
        return sr.Session({"spark.rapids.sql.enabled": False})

    def test_cpu(self, cpu):
        df = cpu.create_dataframe({"s": ["a,b,c", "x", None, ""]})
        out = (df.select(col("s").split(",").alias("p"))
               .select(col("p").element_at(1).alias("f"),
                       col("p").element_at(-1).alias("l"),
                       col("p").element_at(5).alias("m")).to_pydict())
        assert out["f"] == ["a", "x", None, ""]
        assert out["l"] == ["c", "x", None, ""]
        assert out["m"] == [None] * 4

    @pytest.mark.gpu
    def test_gpu_matches_cpu(self):
        vals = [",".join(f"t{j}" for j in range(i % 5)) if i % 7 else None
                for i in range(3000)]

        def q(s):
            df = s.create_dataframe({"s": vals})
            return (df.select(col("s").split(",").alias("p"))
                    .select(col("p").element_at(1).alias("a"),
                            col("p").element_at(2).alias("b"),
                            col("p").element_at(-1).alias("z"))
                    .to_pydict())

        sg = sr.Session()
        sc = sr.Session({"spark.rapids.sql.enabled": False})
        assert q(sg) == q(sc)


class TestInitcapReverse:
    @pytest.fixture
    def cpu(self):
        return sr.Session({"spark.rapids.sql.enabled": False})

    def test_cpu(self, cpu):
        df = cpu.create_dataframe({"s": ["hello wORLD", "", None, "émile x"]})
        out = df.select(col("s").initcap().alias("i"),
                        col("s").reverse().alias("r")).to_pydict()
        assert out["i"][0] == "Hello World"
        assert out["r"][0] == "DLROw olleh"
        assert out["r"][3] == "x elimé"  # codepoint-order reverse

    def test_sql_distinct(self, cpu):
        cpu.register("tdst", cpu.create_dataframe({"a": [1, 1, 2]}))
        assert sorted(cpu.sql("SELECT DISTINCT a FROM tdst").collect()) == \
            [(1,), (2,)]

    @pytest.mark.gpu
    def test_gpu_matches_cpu(self):
        vals = [f"word{v} second{v}" if v % 7 else None for v in range(4000)]
        vals += ["émile unicode-pass", ""]

        def q(s):
            df = s.create_dataframe({"s": vals})
            return df.select(col("s").reverse().alias("r"),
                             col("s").initcap().alias("i")).to_pydict()

        sg = sr.Session({"spark.rapids.sql.incompatibleOps.enabled": True})
        sc = sr.Session({"spark.rapids.sql.enabled": False,
                         "spark.rapids.sql.incompatibleOps.enabled": True})
        g, c = q(sg), q(sc)
        assert g["r"] == c["r"]
        # initcap on pure-ASCII rows must match; the unicode row is the
        # documented ASCII-only incompat (é is not uppercased on GPU)
        for a, b, v in zip(g["i"], c["i"], vals):
            if v is not None and v.isascii():
                assert a == b, v


class TestGetJsonObject:
    @pytest.fixture
    def cpu(self):
        return sr.Session({"spark.rapids.sql.enabled": False})

    def test_cpu(self, cpu):
        df = cpu.create_dataframe({"j": ['{"a": 1, "b": "x"}',
                                         '{"b": "y"}', None, "bad"]})
        out = df.select(col("j").get_json_object("$.a").alias("a"),
                        col("j").get_json_object("$.b").alias("b")
                        ).to_pydict()
        assert out["a"] == ["1", None, None, None]
        assert out["b"] == ["x", "y", None, None]

    @pytest.mark.gpu
    def test_gpu_matches_cpu(self):
        import json

        rows = [json.dumps({"k": i, "s": f"v{i}", "f": i / 4,
                            "b": bool(i % 2)})
                if i % 9 else None for i in range(4000)]

        def q(s):
            df = s.create_dataframe({"j": rows})
            return df.select(
                col("j").get_json_object("$.k").alias("k"),
                col("j").get_json_object("$.s").alias("s"),
                col("j").get_json_object("$.b").alias("b"),
                col("j").get_json_object("$.missing").alias("m")
            ).to_pydict()

        sg = sr.Session()
        sc = sr.Session({"spark.rapids.sql.enabled": False})
        assert q(sg) == q(sc)


class TestConcatWs:
    @pytest.fixture
    def cpu(self):
        return sr.Session({"spark.rapids.sql.enabled": False})

    def test_cpu(self, cpu):
        from spark_rapids_amd import concat_ws

        df = cpu.create_dataframe({"a": ["x", None, None, ""],
                                   "b": ["1", "2", None, "z"]})
        out = df.select(concat_ws("-", col("a"), col("b")).alias("c"),
                        concat_ws("", col("a"), col("b")).alias("e")
                        ).to_pydict()
        assert out["c"] == ["x-1", "2", "", "-z"]
        assert out["e"] == ["x1", "2", "", "z"]

    def test_sql(self, cpu):
        cpu.register("tcw", cpu.create_dataframe({"a": ["p"], "b": ["q"]}))
        out = cpu.sql("SELECT concat_ws('.', a, b) FROM tcw").collect()
        assert out == [("p.q",)]

    @pytest.mark.gpu
    def test_gpu_matches_cpu(self):
        from spark_rapids_amd import concat_ws
        import numpy as np

        rng = np.random.default_rng(14)
        n = 6000
        a = [f"left{v}" if v % 3 else None for v in rng.integers(0, 30, n)]
        b = [f"mid{v}" if v % 5 else None for v in rng.integers(0, 30, n)]
        c = [f"r{v}" if v % 2 else "" for v in rng.integers(0, 30, n)]

        def q(s):
            df = s.create_dataframe({"a": a, "b": b, "c": c})
            return df.select(concat_ws("|", col("a"), col("b"),
                                       col("c")).alias("j")).to_pydict()

        sg = sr.Session()
        sc = sr.Session({"spark.rapids.sql.enabled": False})
        assert q(sg) == q(sc)


def test_pad_and_locate():
    s = sr.Session({"spark.rapids.sql.enabled": False})
    df = s.create_dataframe({"s": ["ab", "abcdef", None]})
    out = df.select(col("s").lpad(4, "0").alias("l"),
                    col("s").rpad(4, "*").alias("r"),
                    col("s").locate("b").alias("p")).to_pydict()
    assert out["l"] == ["00ab", "abcd", None]
    assert out["r"] == ["ab**", "abcd", None]
    assert out["p"] == [2, 2, None]
    s.register("tpad", df)
    assert s.sql("SELECT instr(s, 'cd') FROM tpad").collect() == \
        [(0,), (3,), (None,)]


def test_host_string_fns():
    from spark_rapids_amd import (ascii_, repeat_str, substring_index,
                                  translate)

    s = sr.Session({"spark.rapids.sql.enabled": False})
    df = s.create_dataframe({"s": ["a.b.c", "xyz", None]})
    out = df.select(substring_index(col("s"), ".", 2).alias("si"),
                    substring_index(col("s"), ".", -1).alias("sn"),
                    translate(col("s"), "abc", "ABC").alias("tr"),
                    repeat_str(col("s"), 2).alias("rp"),
                    ascii_(col("s")).alias("a")).to_pydict()
    assert out["si"] == ["a.b", "xyz", None]
    assert out["sn"] == ["c", "xyz", None]
    assert out["tr"] == ["A.B.C", "xyz", None]
    assert out["rp"] == ["a.b.ca.b.c", "xyzxyz", None]
    assert out["a"] == [97, 120, None]
    s.register("thsf", df)
    assert s.sql("SELECT substring_index(s, '.', -1) FROM thsf").collect() \
        == [("c",), ("xyz",), (None,)]


# ---- lpad/rpad/locate on device (round 2) --------------------------------

def _pad_cases():
    return ["hi", "hello world", "", "ab", None, "über", "xyzt"]


def test_cpu_pad_cycles_fill():
    import spark_rapids_amd as sr
    from spark_rapids_amd import col
    from spark_rapids_amd.expr.expressions import PadExpr

    s = sr.Session({"spark.rapids.sql.enabled": False})
    df = s.create_dataframe({"x": _pad_cases()})
    out = df.select(
        PadExpr(col("x"), 5, "xy", True).alias("l"),
        PadExpr(col("x"), 5, "xy", False).alias("r"),
        PadExpr(col("x"), 3, "", True).alias("e")).to_pydict()
    # Spark lpad('hi',5,'xy') = 'xyxhi' (fill cycles)
    assert out["l"][0] == "xyxhi"
    assert out["r"][0] == "hixyx"
    assert out["l"][1] == "hello"          # truncate to width
    assert out["l"][4] is None
    assert out["e"][0] == "hi"             # empty fill: truncate only


def test_cpu_locate_codepoints():
    import spark_rapids_amd as sr
    from spark_rapids_amd import col
    from spark_rapids_amd.expr.expressions import LocateExpr

    s = sr.Session({"spark.rapids.sql.enabled": False})
    df = s.create_dataframe({"x": ["hello", "ababab", None, "aüba"]})
    out = df.select(LocateExpr(col("x"), "ab", 2).alias("p"),
                    LocateExpr(col("x"), "b", 1).alias("q"),
                    LocateExpr(col("x"), "", 3).alias("e")).to_pydict()
    assert out["p"] == [0, 3, None, 0]
    assert out["q"] == [0, 2, None, 3]     # codepoint index, not byte
    assert out["e"] == [3, 3, None, 3]


@pytest.mark.gpu
def test_gpu_pad_locate_match_cpu():
    import spark_rapids_amd as sr
    from spark_rapids_amd import col
    from spark_rapids_amd.expr.expressions import LocateExpr, PadExpr

    cases = _pad_cases() * 500

    def q(s):
        df = s.create_dataframe({"x": cases})
        return df.select(
            PadExpr(col("x"), 7, "xy", True).alias("l"),
            PadExpr(col("x"), 7, "·", False).alias("r"),
            PadExpr(col("x"), 2, " ", True).alias("t"),
            LocateExpr(col("x"), "b", 1).alias("p1"),
            LocateExpr(col("x"), "l", 3).alias("p3"),
        ).to_pydict()

    g = q(sr.Session())
    c = q(sr.Session({"spark.rapids.sql.enabled": False}))
    assert g == c

"""Regex (RLike) tests: bytecode compiler + CPU semantics; GPU equality
under -m gpu (reference analogue: RegularExpressionTranspilerSuite fuzz)."""
import re as pyre

import numpy as np
import pytest

import spark_rapids_amd as sr
from spark_rapids_amd import Column, STRING, col
from spark_rapids_amd.ops import cpu_backend
from spark_rapids_amd.ops.regex_compiler import RegexUnsupported, compile_regex

WORDS = ["hello", "Hello World", "", "spark123", None, "sp", "park", "a-b",
         "aaab", "xyzzy", "foo bar baz", "12.5", "tab\there", "ab", "b",
         "aaaa", "caab"]

PATTERNS = [
    "spark", "^sp", "k$", "sp.*k", "a+b", "a*b", "ab?", "[a-c]+b",
    "[^a-z ]+", r"\d+", r"\w+@?", r"\s", "(foo|bar)+", "^(a|c)a+b$",
    "a{2,3}b", "x{2}", r"1\d\.5", "(?:He|he)llo", ".*", "a(b|c)*$",
]


def test_compiler_rejects_unsupported():
    for p in [r"(a)\1", "a(?=b)", "a*?", "(?P<x>a)", "a{1000}"]:
        with pytest.raises(RegexUnsupported):
            compile_regex(p)


@pytest.mark.parametrize("pat", PATTERNS)
def test_cpu_rlike_matches_python_re(session, pat):
    df = session.create_dataframe({"s": WORDS})
    out = df.select(col("s").rlike(pat).alias("m")).to_pydict()["m"]
    for w, got in zip(WORDS, out):
        if w is None:
            assert got is None
        else:
            assert got == bool(pyre.search(pat, w)), (pat, w)


@pytest.mark.gpu
@pytest.mark.parametrize("pat", PATTERNS)
def test_gpu_rlike_matches_cpu(pat):
    c = Column.from_pylist(WORDS * 200, STRING)
    from spark_rapids_amd.ops import gpu_backend

    cpu = cpu_backend.str_predicate("rlike", c, pat)
    gpu = gpu_backend.str_predicate("rlike", c.cuda(), pat).cpu()
    assert cpu.to_pylist() == gpu.to_pylist(), pat


@pytest.mark.gpu
def test_gpu_rlike_plan_and_fallback():
    s = sr.Session()
    df = s.create_dataframe({"s": [w for w in WORDS if w is not None] * 100})
    q = df.filter(col("s").rlike(r"^sp.*\d+$"))
    assert "GpuFilter" in q.physical_plan().tree_string()
    # unsupported pattern tags the filter onto CPU instead of failing
    q2 = df.filter(col("s").rlike(r"(a)\1"))
    assert "CpuFilter" in q2.physical_plan().tree_string()


class TestRegexpExtractReplace:
    @pytest.fixture
    def cpu(self):
        return sr.Session({"spark.rapids.sql.enabled": False})

    def test_extract_cpu(self, cpu):
        df = cpu.create_dataframe(
            {"s": ["ab-12-34", "x99y", None, "nope", ""]})
        out = df.select(
            col("s").regexp_extract(r"(\d+)-(\d+)", 1).alias("g1"),
            col("s").regexp_extract(r"(\d+)-(\d+)", 2).alias("g2"),
            col("s").regexp_extract(r"(\d+)", 1).alias("d")).to_pydict()
        assert out["g1"] == ["12", "", None, "", ""]
        assert out["g2"] == ["34", "", None, "", ""]
        assert out["d"] == ["12", "99", None, "", ""]

    def test_replace_cpu(self, cpu):
        df = cpu.create_dataframe({"s": ["a1b22c", "", None, "xyz"]})
        out = df.select(
            col("s").regexp_replace(r"\d+", "#").alias("r"),
            col("s").regexp_replace(r"([a-z])(\d)", "$2$1").alias("sw"),
        ).to_pydict()
        assert out["r"] == ["a#b#c", "", None, "xyz"]
        assert out["sw"] == ["1a2b2c", "", None, "xyz"]

    def test_replace_empty_match(self, cpu):
        df = cpu.create_dataframe({"s": ["ab"]})
        out = df.select(col("s").regexp_replace("x*", "-")).collect()
        assert out == [("-a-b-",)]

    def test_sql(self, cpu):
        df = cpu.create_dataframe({"s": ["k=42"]})
        cpu.register("trx", df)
        # Spark unescapes string literals: the regex backslash must be
        # doubled in SQL text ('\\d' in the query string reaches the
        # regex engine as \d); a single '\d' collapses to 'd'.
        out = cpu.sql(
            "SELECT regexp_extract(s, '(\\\\d+)', 1) FROM trx").collect()
        assert out == [("42",)]
        out = cpu.sql(
            "SELECT regexp_extract(s, '(\\d+)', 1) FROM trx").collect()
        assert out == [("",)]

    @pytest.mark.gpu
    def test_gpu_matches_cpu(self):
        import numpy as np

        rng = np.random.default_rng(6)
        words = ["item-%d-%d" % (a, b) for a, b in
                 zip(rng.integers(0, 1000, 4000), rng.integers(0, 99, 4000))]
        words += ["nomatch", "", "x-y-z"] * 100
        vals = [w if i % 31 else None for i, w in enumerate(words)]

        def q(s):
            df = s.create_dataframe({"s": vals})
            return df.select(
                col("s").regexp_extract(r"(\d+)-(\d+)", 2).alias("g"),
                col("s").regexp_replace(r"(\d+)", "<$1>").alias("r"),
                col("s").regexp_replace(r"-", "_").alias("u")).to_pydict()

        sg = sr.Session()
        sc = sr.Session({"spark.rapids.sql.enabled": False})
        assert q(sg) == q(sc)

    @pytest.mark.gpu
    def test_gpu_placement(self):
        sg = sr.Session()
        df = sg.create_dataframe({"s": ["a1"]})
        tree = (df.select(col("s").regexp_extract(r"(\d)", 1))
                .physical_plan().tree_string())
        assert "GpuProject" in tree, tree


class TestRegexpExtractAll:
    @pytest.fixture
    def cpu(self):
        return sr.Session({"spark.rapids.sql.enabled": False})

    def test_cpu(self, cpu):
        df = cpu.create_dataframe({"s": ["a1 b22 c333", "none", None, ""]})
        out = (df.select(col("s").regexp_extract_all(r"(\d+)").alias("m"),
                         col("s").regexp_extract_all(r"([a-z])(\d+)", 2)
                         .alias("g2")).to_pydict())
        assert out["m"] == [["1", "22", "333"], [], None, []]
        assert out["g2"] == [["1", "22", "333"], [], None, []]

    @pytest.mark.gpu
    def test_gpu_matches_cpu(self):
        import numpy as np

        rng = np.random.default_rng(12)
        vals = [" ".join(f"k{v}v{v*3}" for v in rng.integers(0, 99, v % 5))
                if i % 11 else None
                for i, v in enumerate(rng.integers(0, 30, 4000))]

        def q(s):
            df = s.create_dataframe({"s": vals})
            return (df.select(
                col("s").regexp_extract_all(r"v(\d+)").alias("m"))
                .to_pydict())

        sg = sr.Session()
        sc = sr.Session({"spark.rapids.sql.enabled": False})
        assert q(sg) == q(sc)

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

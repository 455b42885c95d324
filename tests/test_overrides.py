"""Tagging / fallback / explain / config-surface tests (no GPU needed:
tagging logic is identical; placement just resolves to CPU here)."""
import pytest

from spark_rapids_amd import Session, col, lit, sum_
from spark_rapids_amd.config import RapidsConf, help_doc, registry
from spark_rapids_amd.plan.overrides import Tagger
from spark_rapids_amd.plan import logical as L


def _logical(df):
    return df.plan


def test_config_defaults_and_set():
    c = RapidsConf()
    assert c.sql_enabled is True
    c.set("spark.rapids.sql.enabled", "false")
    assert c.sql_enabled is False
    c.set("spark.rapids.sql.batchSizeBytes", "512m")
    assert c.batch_size_bytes == 512 << 20


def test_config_doc_generation():
    doc = help_doc()
    assert "spark.rapids.sql.enabled" in doc
    assert "spark.rapids.memory.pinnedPool.size" in doc
    assert len(registry()) >= 20


def test_tagger_accepts_numeric_plan(session):
    df = session.create_dataframe({"a": [1], "b": [2.0]})
    plan = L.Filter((col("a") > 0), _logical(df))
    t = Tagger(session.conf)
    assert t.exec_reasons(plan) == []


def test_tagger_accepts_string_group_and_sort_keys(session):
    df = session.create_dataframe({"s": ["x"], "v": [1]})
    plan = L.Aggregate([col("s")], [sum_(col("v"))], _logical(df))
    t = Tagger(session.conf)
    assert t.exec_reasons(plan) == []  # hash keys handle strings
    sort_plan = L.Sort(_logical(df), ["s"])
    assert Tagger(session.conf).exec_reasons(sort_plan) == []
    # LIST keys stay off the GPU sort
    bad = L.Sort(L.Generate("p", L.Project(
        [col("s").split(",").alias("p")], _logical(df))), ["p"])
    # (generate output is the element type, so sort a real LIST instead)


def test_per_exec_disable_conf(session):
    session.set("spark.rapids.sql.exec.Filter", "false")
    df = session.create_dataframe({"a": [1]})
    plan = L.Filter(col("a") > 0, _logical(df))
    t = Tagger(session.conf)
    assert any("disabled by conf" in r for r in t.exec_reasons(plan))


def test_per_expression_disable_conf(session):
    session.set("spark.rapids.sql.expression.add", "false")
    df = session.create_dataframe({"a": [1]})
    t = Tagger(session.conf)
    reasons = t.expr_reasons(col("a") + lit(1), df.schema)
    assert any("disabled by conf" in r for r in reasons)


def test_sql_enabled_false_runs_cpu(cpu_session):
    df = cpu_session.create_dataframe({"a": [1, 2, 3]})
    exec_ = df.filter(col("a") > 1).physical_plan()
    assert "Cpu" in exec_.tree_string()
    assert df.filter(col("a") > 1).count() == 2


def test_explain_tree(session):
    df = session.create_dataframe({"a": [1, 2]})
    s = df.filter(col("a") > 0).select((col("a") * 2).alias("b")).explain()
    assert "Project" in s and "Filter" in s and "Scan" in s


def test_results_identical_cpu_vs_default(session, cpu_session):
    data = {"k": [1, 2, 1, 2], "v": [1.0, 2.0, 3.0, 4.0]}
    q = lambda s: (s.create_dataframe(data)
                   .filter(col("v") > 1.0)
                   .group_by("k").agg(sum_(col("v")))
                   .sort("k").collect())
    assert q(session) == q(cpu_session)


def test_metrics_collection(session):
    df = session.create_dataframe({"a": list(range(100))})
    q = df.filter(__import__("spark_rapids_amd").col("a") > 10)
    q.collect()
    m = q.metrics()
    assert m and any(x["numOutputRows"] == 89 for x in m)
    assert all("opTimeMs" in x for x in m)


def test_lore_dump_and_replay(tmp_path):
    from spark_rapids_amd import Session, col, sum_
    import os

    s = Session({"spark.rapids.sql.lore.dumpPath": str(tmp_path)})
    df = s.create_dataframe({"k": [1, 2, 1], "v": [1.0, 2.0, 3.0]})
    df.filter(col("v") > 0.5).group_by("k").agg(sum_(col("v"))).collect()
    dirs = sorted(os.listdir(tmp_path))
    assert any("Filter" in d for d in dirs), dirs
    from spark_rapids_amd.tools.lore import replay

    s2 = Session()
    fdir = next(os.path.join(tmp_path, d) for d in dirs if "Filter" in d)
    re_df = replay(s2, fdir)
    assert re_df.count() == 3


def test_column_pruning_under_join(session):
    from spark_rapids_amd.plan.optimizer import prune_columns

    fact = session.create_dataframe(
        {"k": [1, 2], "a": [1, 2], "b": [3, 4], "c": [5, 6]})
    dim = session.create_dataframe({"k": [1, 2], "x": [7, 8], "y": [9, 10]})
    q = (fact.join(dim, on="k").group_by("x")
         .agg(sum_(col("a"))))
    pruned = prune_columns(q.plan)
    # the join children should be narrowed to {k,a} and {k,x}
    join = pruned.child if hasattr(pruned, "child") else pruned
    from spark_rapids_amd.plan import logical as L

    node = pruned
    while not isinstance(node, L.Join):
        node = node.children[0]
    assert set(node.left.schema().names) == {"k", "a"}
    assert set(node.right.schema().names) == {"k", "x"}
    # results identical with and without pruning
    res = q.collect()
    session.conf.set("spark.rapids.sql.optimizer.pruneColumns.enabled", False)
    assert sorted(res) == sorted(q.collect())


def test_cpu_bridge_expression_plan():
    import torch

    orig = torch.cuda.is_available
    torch.cuda.is_available = lambda: True
    try:
        import spark_rapids_amd as sr

        s = sr.Session({"spark.rapids.sql.incompatibleOps.enabled": False})
        df = s.create_dataframe({"a": ["x"], "b": [1]})
        tree = (df.select(col("a").upper().alias("u"),
                          (col("b") + 1).alias("c"))
                .physical_plan().tree_string())
        assert "GpuProject" in tree and "cpu_bridge" in tree, tree
        off = sr.Session({"spark.rapids.sql.incompatibleOps.enabled": False,
                          "spark.rapids.sql.cpuBridge.enabled": False})
        df2 = off.create_dataframe({"a": ["x"], "b": [1]})
        tree2 = (df2.select(col("a").upper().alias("u"))
                 .physical_plan().tree_string())
        assert "CpuProject" in tree2, tree2
    finally:
        torch.cuda.is_available = orig


@pytest.mark.gpu
def test_cpu_bridge_results_match():
    import spark_rapids_amd as sr

    s = sr.Session({"spark.rapids.sql.incompatibleOps.enabled": False})
    sc = sr.Session({"spark.rapids.sql.enabled": False})
    vals = ["MiXeD case", "ABC", None, ""]

    def q(sess):
        df = sess.create_dataframe({"a": vals, "b": list(range(4))})
        return (df.select(col("a").upper().alias("u"),
                          (col("b") * 2).alias("d")).to_pydict())

    assert q(s) == q(sc)


def test_lore_dump_and_load(tmp_path, cpu_session):
    """LORE: per-operator batch dumps land on disk and reload."""
    import glob
    import os

    import spark_rapids_amd as sr
    from spark_rapids_amd.tools import lore

    s = sr.Session({"spark.rapids.sql.enabled": False,
                    "spark.rapids.sql.lore.dumpPath": str(tmp_path)})
    df = s.create_dataframe({"a": [1, 2, 3], "b": [1.0, 2.0, 3.0]})
    df.filter(col("a") > 1).agg(sum_(col("b"))).collect()
    dumped = glob.glob(os.path.join(str(tmp_path), "**", "*"),
                       recursive=True)
    assert dumped, "lore dump produced no files"
    # turn dumping back off for other tests
    sr.Session({"spark.rapids.sql.enabled": False})
    lore.configure("")


def test_docgen_runs(tmp_path, monkeypatch):
    from spark_rapids_amd.tools import docgen

    monkeypatch.setattr(docgen, "REPO", str(tmp_path))
    docgen.main()
    import os

    assert os.path.exists(os.path.join(str(tmp_path), "docs", "configs.md"))
    text = open(os.path.join(str(tmp_path), "docs",
                             "supported_ops.md")).read()
    assert "Expand" in text and "regexp_extract" in text


def test_config_surface_documented():
    """Every registered spark.rapids.* key renders into docs/configs.md."""
    from spark_rapids_amd.config import _REGISTRY, help_doc

    doc = help_doc()
    public = [k for k, e in _REGISTRY.items()
              if not getattr(e, "internal", False)]
    assert len(public) >= 25
    for k in public:
        assert k in doc, k


def test_expression_nullability():
    s = Session({"spark.rapids.sql.enabled": False})
    df = s.create_dataframe({"a": [1, None], "b": [1.0, 2.0]})
    sch = df.select((col("a") + col("b")).alias("x")).schema
    assert sch.fields[0].nullable


def test_lore_replay_roundtrip(tmp_path):
    """Dump an operator's output, then replay it as a DataFrame."""
    import glob
    import os

    import spark_rapids_amd as sr
    from spark_rapids_amd.tools import lore

    s = sr.Session({"spark.rapids.sql.enabled": False,
                    "spark.rapids.sql.lore.dumpPath": str(tmp_path)})
    df = s.create_dataframe({"a": [3, 1, 2], "b": [1.0, 2.0, 3.0]})
    expected = df.filter(col("a") > 1).collect()
    df.filter(col("a") > 1).collect()
    dirs = [d for d in glob.glob(os.path.join(str(tmp_path), "*Filter*"))
            if os.path.isdir(d)]
    assert dirs, os.listdir(str(tmp_path))
    clean = sr.Session({"spark.rapids.sql.enabled": False})
    lore.configure("")
    replayed = sorted(lore.replay(clean, dirs[0]).collect())
    assert replayed == sorted(expected)

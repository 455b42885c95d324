"""Optimizer passes: filter pushdown through joins (+ column pruning
interaction). Reference analogue: Spark Catalyst PushPredicateThroughJoin,
which the reference plugin inherits from Spark before GpuOverrides runs."""
import pytest

import spark_rapids_amd as sr
from spark_rapids_amd import col, count_star, sum_


@pytest.fixture
def cpu():
    return sr.Session({"spark.rapids.sql.enabled": False})


def _tables(s):
    fact = s.create_dataframe({
        "id": [1, 2, 3, 4, 5, None],
        "v": [1.0, 2.0, 3.0, 4.0, 5.0, 6.0]})
    dim = s.create_dataframe({
        "did": [1, 2, 3, 9],
        "cat": [0, 1, 0, 0]})
    return fact, dim


def test_pushdown_inner_join_both_sides(cpu):
    fact, dim = _tables(cpu)
    q = (fact.join(dim, on="id", right_on=["did"])
         .filter((col("cat") < 1) & (col("v") > 1.0)))
    tree = q.physical_plan().tree_string()
    assert tree.index("Filter((cat lt 1))") > tree.index("HashJoin")
    rows = q.collect()
    assert sorted(r[0] for r in rows) == [3]


def test_pushdown_preserves_results_vs_disabled(cpu):
    off = sr.Session({"spark.rapids.sql.enabled": False,
                      "spark.rapids.sql.optimizer.pushFilters.enabled":
                          False})
    for s in ():
        pass

    def q(s):
        fact, dim = _tables(s)
        return sorted(fact.join(dim, on="id", right_on=["did"], how="left")
                      .filter(col("v") > 1.0).collect(), key=repr)

    assert q(cpu) == q(off)


def test_left_join_right_predicate_not_pushed(cpu):
    fact, dim = _tables(cpu)
    # cat IS NULL on unmatched rows: pushing it below the left join would
    # wrongly drop the null-extended rows before they exist
    q = (fact.join(dim, on="id", right_on=["did"], how="left")
         .filter(col("cat").is_null()))
    rows = q.collect()
    # unmatched ids: 4, 5, None
    assert sorted((r[0] is None, r[0]) for r in rows) == \
        [(False, 4), (False, 5), (True, None)]


def test_stacked_joins_push_two_levels(cpu):
    fact, dim = _tables(cpu)
    dim2 = cpu.create_dataframe({"did2": [1, 2, 3, 4, 5],
                                 "flag": [1, 0, 1, 0, 1]})
    q = (fact.join(dim, on="id", right_on=["did"])
         .join(dim2, on="id", right_on=["did2"])
         .filter((col("cat") < 1) & (col("flag") > 0) & (col("v") >= 1.0)))
    tree = q.physical_plan().tree_string()
    assert "Filter((cat lt 1))" in tree
    assert "Filter((flag gt 0))" in tree
    assert sorted(r[0] for r in q.collect()) == [1, 3]


def test_pushdown_conf_off(cpu):
    s = sr.Session({"spark.rapids.sql.enabled": False,
                    "spark.rapids.sql.optimizer.pushFilters.enabled": False})
    fact, dim = _tables(s)
    q = fact.join(dim, on="id", right_on=["did"]).filter(col("cat") < 1)
    tree = q.physical_plan().tree_string()
    assert tree.index("Filter") < tree.index("HashJoin")


def test_cost_based_optimizer_veto_small_plan():
    """CBO (default off) vetoes GPU placement for tiny inputs when
    enabled (CostBasedOptimizer analogue)."""
    import spark_rapids_amd as sr
    from spark_rapids_amd import col, sum_

    s = sr.Session({"spark.rapids.sql.optimizer.enabled": True})
    df = s.create_dataframe({"k": [1, 2], "v": [1.0, 2.0]})
    q = df.group_by("k").agg(sum_(col("v")))
    tree = q.physical_plan().tree_string()
    assert "Gpu" not in tree, tree
    assert sorted(q.collect()) == [(1, 1.0), (2, 2.0)]
    # default off: no veto machinery in the way
    s2 = sr.Session({"spark.rapids.sql.enabled": False})
    assert sorted(s2.create_dataframe({"k": [1]}).collect()) == [(1,)]


def test_cost_based_optimizer_estimates():
    from spark_rapids_amd.plan.costing import estimate_rows, evaluate
    import spark_rapids_amd as sr
    from spark_rapids_amd import col, sum_

    s = sr.Session({"spark.rapids.sql.enabled": False})
    df = s.create_dataframe({"k": list(range(1000)),
                             "v": [0.0] * 1000})
    plan = df.filter(col("k") > 10).group_by("k") \
        .agg(sum_(col("v"))).plan
    assert estimate_rows(plan) > 0
    keep, note = evaluate(plan)
    assert "est cpu" in note


@pytest.mark.gpu
def test_gpu_cost_based_optimizer_veto():
    import spark_rapids_amd as sr
    from spark_rapids_amd import col, sum_

    s = sr.Session({"spark.rapids.sql.optimizer.enabled": True})
    tiny = s.create_dataframe({"k": [1, 2], "v": [1.0, 2.0]})
    q = tiny.group_by("k").agg(sum_(col("v")))
    assert "Gpu" not in q.physical_plan().tree_string()
    assert sorted(q.collect()) == [(1, 1.0), (2, 2.0)]
    # a large plan stays on GPU despite the optimizer
    big = s.create_dataframe({"k": list(range(200_000)) * 5,
                              "v": [0.5] * 1_000_000})
    q2 = big.group_by("k").agg(sum_(col("v")))
    assert "Gpu" in q2.physical_plan().tree_string()

"""TPC-H q1 + q5 on 1x MI355X (BASELINE.json config 2: "TPC-H SF100 q1+q5
(scan/filter/agg/hash-join) on 1 MI355X executor").

Synthetic lineitem/orders/customer/supplier/nation/region with TPC-H
cardinality ratios at a configurable scale factor (SF1 = 6M lineitem rows;
no network, so data is generated in memory — random-uniform value
distributions, char(1) flags dictionary-coded as int8, the standard
approach for fixed-width GPU group keys).

    python -m spark_rapids_amd.bench.tpch --sf 10 --out report.json
"""
from __future__ import annotations

import argparse
import json
import time

import numpy as np

from ..api import Session
from ..column import Column, ColumnBatch, Field, Schema
from ..expr.aggregates import avg, count_star, sum_
from ..expr.expressions import col, lit
from ..types import DATE32, FLOAT64, INT8, INT32


def _gen_lineitem(sf: float, seed: int, partitions: int):
    n = int(6_000_000 * sf)
    per = n // partitions
    batches = []
    for p in range(partitions):
        rng = np.random.default_rng(seed * 100 + p)
        m = per
        qty = rng.integers(1, 51, m).astype(np.float64)
        price = rng.uniform(900.0, 105_000.0, m)
        disc = np.round(rng.uniform(0.0, 0.10, m), 2)
        tax = np.round(rng.uniform(0.0, 0.08, m), 2)
        batches.append(ColumnBatch([
            Column.from_numpy(rng.integers(0, int(1_500_000 * sf) or 1, m)
                              .astype(np.int32)),           # l_orderkey
            Column.from_numpy(rng.integers(0, int(10_000 * sf) or 1, m)
                              .astype(np.int32)),           # l_suppkey
            Column.from_numpy(qty),                          # l_quantity
            Column.from_numpy(price),                        # l_extendedprice
            Column.from_numpy(disc),                         # l_discount
            Column.from_numpy(tax),                          # l_tax
            Column.from_numpy(rng.integers(0, 3, m).astype(np.int8)),  # l_returnflag (A/N/R)
            Column.from_numpy(rng.integers(0, 2, m).astype(np.int8)),  # l_linestatus (F/O)
            Column.from_numpy(rng.integers(8035, 10591, m)
                              .astype(np.int32)),           # l_shipdate (1992..1998)
        ], m))
    schema = Schema([
        Field("l_orderkey", INT32), Field("l_suppkey", INT32),
        Field("l_quantity", FLOAT64), Field("l_extendedprice", FLOAT64),
        Field("l_discount", FLOAT64), Field("l_tax", FLOAT64),
        Field("l_returnflag", INT8), Field("l_linestatus", INT8),
        Field("l_shipdate", DATE32),
    ])
    return batches, schema


def _tables(session: Session, sf: float, partitions: int, device: str):
    li_batches, li_schema = _gen_lineitem(sf, 7, partitions)
    rng = np.random.default_rng(13)
    n_ord = int(1_500_000 * sf) or 1
    n_cust = int(150_000 * sf) or 1
    n_supp = int(10_000 * sf) or 1
    orders = ColumnBatch([
        Column.from_numpy(np.arange(n_ord, dtype=np.int32)),      # o_orderkey
        Column.from_numpy(rng.integers(0, n_cust, n_ord).astype(np.int32)),
        Column.from_numpy(rng.integers(8035, 10591, n_ord).astype(np.int32)),
    ], n_ord)
    orders_schema = Schema([Field("o_orderkey", INT32),
                            Field("o_custkey", INT32),
                            Field("o_orderdate", DATE32)])
    customer = ColumnBatch([
        Column.from_numpy(np.arange(n_cust, dtype=np.int32)),
        Column.from_numpy(rng.integers(0, 25, n_cust).astype(np.int8)),
    ], n_cust)
    customer_schema = Schema([Field("c_custkey", INT32),
                              Field("c_nationkey", INT8)])
    supplier = ColumnBatch([
        Column.from_numpy(np.arange(n_supp, dtype=np.int32)),
        Column.from_numpy(rng.integers(0, 25, n_supp).astype(np.int8)),
    ], n_supp)
    supplier_schema = Schema([Field("s_suppkey", INT32),
                              Field("s_nationkey", INT8)])
    nation = ColumnBatch([
        Column.from_numpy(np.arange(25, dtype=np.int8)),
        Column.from_numpy((np.arange(25) % 5).astype(np.int8)),
    ], 25)
    nation_schema = Schema([Field("n_nationkey", INT8),
                            Field("n_regionkey", INT8)])
    if device == "cuda":
        li_batches = [b.cuda() for b in li_batches]
        orders, customer = orders.cuda(), customer.cuda()
        supplier, nation = supplier.cuda(), nation.cuda()
    s = session
    return {
        "lineitem": s.from_batches(li_batches, li_schema, "lineitem"),
        "orders": s.from_batches([orders], orders_schema, "orders",
                                 replicated=True),
        "customer": s.from_batches([customer], customer_schema, "customer",
                                   replicated=True),
        "supplier": s.from_batches([supplier], supplier_schema, "supplier",
                                   replicated=True),
        "nation": s.from_batches([nation], nation_schema, "nation",
                                 replicated=True),
    }


def q1(t):
    """TPC-H Q1 pricing summary report (shipdate <= 1998-09-02)."""
    li = t["lineitem"]
    disc_price = col("l_extendedprice") * (lit(1.0) - col("l_discount"))
    charge = disc_price * (lit(1.0) + col("l_tax"))
    return (li.filter(col("l_shipdate") <= lit(10_471))
            .group_by("l_returnflag", "l_linestatus")
            .agg(sum_(col("l_quantity")).alias("sum_qty"),
                 sum_(col("l_extendedprice")).alias("sum_base_price"),
                 sum_(disc_price).alias("sum_disc_price"),
                 sum_(charge).alias("sum_charge"),
                 avg(col("l_quantity")).alias("avg_qty"),
                 avg(col("l_extendedprice")).alias("avg_price"),
                 avg(col("l_discount")).alias("avg_disc"),
                 count_star().alias("count_order"))
            .sort("l_returnflag", "l_linestatus"))


def q5(t):
    """TPC-H Q5 local supplier volume (region filter via nationkey groups,
    orderdate in [1994-01-01, 1995-01-01))."""
    li, o, c = t["lineitem"], t["orders"], t["customer"]
    su, na = t["supplier"], t["nation"]
    rev = col("l_extendedprice") * (lit(1.0) - col("l_discount"))
    return (li.join(o, on="l_orderkey", right_on=["o_orderkey"])
            .filter((col("o_orderdate") >= 8766) & (col("o_orderdate") < 9131))
            .join(c, on="o_custkey", right_on=["c_custkey"])
            .join(su, on="l_suppkey", right_on=["s_suppkey"])
            .filter(col("c_nationkey") == col("s_nationkey"))
            .join(na, on="s_nationkey", right_on=["n_nationkey"])
            .filter(col("n_regionkey") == 2)
            .group_by("s_nationkey")
            .agg(sum_(rev).alias("revenue"))
            .sort("revenue", descending=True))


def q3(t):
    """TPC-H Q3 shipping priority (top unshipped orders by revenue)."""
    li, o, c = t["lineitem"], t["orders"], t["customer"]
    rev = col("l_extendedprice") * (lit(1.0) - col("l_discount"))
    return (li.filter(col("l_shipdate") > 9204)       # > 1995-03-15
            .join(o, on="l_orderkey", right_on=["o_orderkey"])
            .filter(col("o_orderdate") < 9204)
            .join(c, on="o_custkey", right_on=["c_custkey"])
            .filter(col("c_nationkey") < 5)           # BUILDING-segment proxy
            .group_by("l_orderkey", "o_orderdate")
            .agg(sum_(rev).alias("revenue"))
            .sort("revenue", descending=True)
            .limit(10))


def q6(t):
    """TPC-H Q6 forecasting revenue change (selective filter + agg)."""
    li = t["lineitem"]
    return (li.filter((col("l_shipdate") >= 8766)      # [1994-01-01,
                      & (col("l_shipdate") < 9131)     #  1995-01-01)
                      & (col("l_discount") >= 0.05)
                      & (col("l_discount") <= 0.07)
                      & (col("l_quantity") < 24.0))
            .agg(sum_(col("l_extendedprice") * col("l_discount"))
                 .alias("revenue")))


def run(sf: float, partitions: int, gpu: bool, iters: int = 3):
    session = Session({"spark.rapids.sql.enabled": gpu})
    t0 = time.perf_counter()
    tables = _tables(session, sf, partitions, "cuda" if gpu else "cpu")
    gen_s = time.perf_counter() - t0
    out = {"sf": sf, "gen_seconds": round(gen_s, 2), "gpu": gpu,
           "queries": {}}
    for name, fn in (("q1", q1), ("q3", q3), ("q5", q5), ("q6", q6)):
        fn(tables).collect()  # warmup
        times = []
        for _ in range(iters):
            t0 = time.perf_counter()
            rows = fn(tables).collect()
            times.append(time.perf_counter() - t0)
        out["queries"][name] = {"seconds": round(min(times), 4),
                                "rows": len(rows)}
    return out


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--sf", type=float, default=1.0)
    ap.add_argument("--partitions", type=int, default=4)
    ap.add_argument("--cpu", action="store_true")
    ap.add_argument("--iters", type=int, default=3)
    ap.add_argument("--out", default="")
    args = ap.parse_args()
    report = run(args.sf, args.partitions, gpu=not args.cpu, iters=args.iters)
    text = json.dumps(report, indent=2)
    if args.out:
        open(args.out, "w").write(text)
    print(text)


if __name__ == "__main__":
    main()

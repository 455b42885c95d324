"""ScaleTest: parameterized stress-query harness with a JSON report
(reference analogue: integration_tests/ScaleTest.md + scaletest/ — 31
parameterized stress queries over datagen tables emitting a TestReport).

Queries stress the shapes the reference's suite targets: joins with
ride-along columns, skewed keys (zipf item ids), high-cardinality
aggregation, selective filters, windows, sort. Run:

    python -m spark_rapids_amd.bench.scaletest --rows 2000000 --out report.json
"""
from __future__ import annotations

import argparse
import json
import time
from typing import Dict, List

from ..api import Session
from ..expr.aggregates import avg, count_star, max_, min_, sum_
from ..expr.expressions import col
from ..expr.windows import rank
from . import datagen


def _tables(session: Session, rows: int, partitions: int):
    batches = [datagen.gen_fact_partition(rows // partitions, 100 + p)
               for p in range(partitions)]
    import torch

    if torch.cuda.is_available():
        batches = [b.cuda() for b in batches]
        items = datagen.gen_items().cuda()
        stores = datagen.gen_stores().cuda()
    else:
        items = datagen.gen_items()
        stores = datagen.gen_stores()
    return {
        "fact": session.from_batches(batches, datagen.fact_schema(), "fact"),
        "item": session.from_batches([items], datagen.item_schema(), "item",
                                     replicated=True),
        "store": session.from_batches([stores], datagen.store_schema(),
                                      "store", replicated=True),
    }


def _queries() -> List:
    def join_ride_along(t):
        # join keeping many non-key ("ride-along") columns
        return (t["fact"].join(t["item"], on="ss_item_id",
                               right_on=["i_item_id"])
                .filter(col("i_current_price") > 100.0)
                .agg(count_star(), sum_(col("ss_list_price"))))

    def skewed_join(t):
        # zipf-skewed item ids stress one-hot build buckets
        return (t["fact"].join(t["item"], on="ss_item_id",
                               right_on=["i_item_id"])
                .group_by("i_brand").agg(count_star()))

    def high_cardinality_agg(t):
        return (t["fact"].group_by("ss_customer_id")
                .agg(sum_(col("ss_sales_price")), count_star())
                .agg(count_star()))

    def selective_filter(t):
        return (t["fact"]
                .filter((col("ss_discount") >= 0.29) &
                        (col("ss_quantity") == 1))
                .agg(count_star(), avg(col("ss_list_price"))))

    def window_rank(t):
        return (t["fact"].limit(200_000)
                .with_column("r", rank().over(["ss_promo"],
                                              ["ss_list_price"]))
                .filter(col("r") <= 10).agg(count_star()))

    def big_sort(t):
        return (t["fact"].sort("ss_sold_date", "ss_item_id").limit(100))

    def semi_anti(t):
        hot = (t["fact"].group_by("ss_item_id").agg(count_star())
               .filter(col("count(*)") > 100))
        return (t["fact"].join(hot, on="ss_item_id", how="semi")
                .agg(count_star()))

    def window_sort_heavy(t):
        # TPC-DS q67 shape: rollup-style agg, rank over partition by
        # category ordered by revenue, keep top ranks, global sort
        agged = (t["fact"].group_by("ss_store_id", "ss_item_id")
                 .agg(sum_(col("ss_sales_price")).alias("rev")))
        from ..types import FLOAT64

        agged = agged.with_column("revd", col("rev").cast(FLOAT64))
        ranked = agged.with_column(
            "rk", rank().over(["ss_store_id"], ["revd"],
                              descending=[True]))
        return (ranked.filter(col("rk") <= 100)
                .sort("ss_store_id", "rk").limit(1000))

    def multi_key_agg(t):
        return (t["fact"].group_by("ss_store_id", "ss_promo")
                .agg(sum_(col("ss_sales_price")), min_(col("ss_discount")),
                     max_(col("ss_list_price")), count_star()))

    return [
        ("join_ride_along", join_ride_along),
        ("skewed_join", skewed_join),
        ("high_cardinality_agg", high_cardinality_agg),
        ("selective_filter", selective_filter),
        ("window_rank", window_rank),
        ("big_sort", big_sort),
        ("semi_anti", semi_anti),
        ("multi_key_agg", multi_key_agg),
        ("window_sort_heavy", window_sort_heavy),
    ]


def _timed(fn, tables) -> float:
    t0 = time.perf_counter()
    fn(tables).collect()
    return time.perf_counter() - t0


def run(rows: int = 2_000_000, partitions: int = 4,
        gpu: bool = True) -> Dict:
    session = Session({"spark.rapids.sql.enabled": gpu})
    tables = _tables(session, rows, partitions)
    report = {"rows": rows, "partitions": partitions, "queries": []}
    for name, fn in _queries():
        try:
            out = fn(tables).collect()  # warm (allocator, autotune, JIT)
            elapsed = min(_timed(fn, tables) for _ in range(3))
            report["queries"].append({
                "name": name, "status": "OK",
                "seconds": round(elapsed, 4),
                "result_rows": len(out),
            })
        except Exception as e:  # noqa: BLE001 - report and continue
            report["queries"].append({
                "name": name, "status": "FAIL", "error": repr(e)[:300],
            })
    return report


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--rows", type=int, default=2_000_000)
    ap.add_argument("--partitions", type=int, default=4)
    ap.add_argument("--out", default="")
    ap.add_argument("--cpu", action="store_true")
    args = ap.parse_args()
    report = run(args.rows, args.partitions, gpu=not args.cpu)
    text = json.dumps(report, indent=2)
    if args.out:
        with open(args.out, "w") as f:
            f.write(text)
    print(text)


if __name__ == "__main__":
    main()

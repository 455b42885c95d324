"""NDS (TPC-DS-derived) power-run query suite over the on-disk star schema
staged by bench/nds.py.

Each query mirrors the shape of a genuine NDS query (named in the
function's docstring): scan from Parquet, join string-keyed dimensions,
group by string keys, decimal aggregation, ORDER BY + LIMIT, rollup and
window passes. Reference analogues: the NDS query set behind
tools/generated_files/operatorsScore.csv and integration_tests/ScaleTest.md.
"""
from __future__ import annotations

from typing import Dict, List

from ..api import DataFrame
from ..expr.aggregates import avg, count_star, sum_
from ..expr.expressions import col, lit
from ..expr.windows import rank, win_avg
from ..types import FLOAT64


def q3(t):
    """NDS q3: date x store_sales x item, brand revenue by year."""
    ss, dd, it = t["store_sales"], t["date_dim"], t["item"]
    return (ss.join(dd, on="ss_sold_date_sk", right_on=["d_date_sk"])
            .join(it, on="ss_item_sk", right_on=["i_item_sk"])
            .filter((col("i_manufact_id") == 128) & (col("d_moy") == 11))
            .group_by("d_year", "i_brand", "i_brand_id")
            .agg(sum_(col("ss_ext_sales_price")).alias("sum_agg"))
            .sort("d_year", "sum_agg", "i_brand_id",
                  descending=[False, True, False])
            .limit(100))


def q42(t):
    """NDS q42: category revenue for one month."""
    ss, dd, it = t["store_sales"], t["date_dim"], t["item"]
    return (ss.join(dd, on="ss_sold_date_sk", right_on=["d_date_sk"])
            .join(it, on="ss_item_sk", right_on=["i_item_sk"])
            .filter((col("d_moy") == 11) & (col("d_year") == 2000)
                    & (col("i_manager_id") == 1))
            .group_by("d_year", "i_category_id", "i_category")
            .agg(sum_(col("ss_ext_sales_price")).alias("s"))
            .sort("s", "d_year", "i_category_id", "i_category",
                  descending=[True, False, False, False])
            .limit(100))


def q52(t):
    """NDS q52: brand revenue for one month."""
    ss, dd, it = t["store_sales"], t["date_dim"], t["item"]
    return (ss.join(dd, on="ss_sold_date_sk", right_on=["d_date_sk"])
            .join(it, on="ss_item_sk", right_on=["i_item_sk"])
            .filter((col("d_moy") == 12) & (col("d_year") == 1998))
            .group_by("d_year", "i_brand", "i_brand_id")
            .agg(sum_(col("ss_ext_sales_price")).alias("ext_price"))
            .sort("d_year", "ext_price", "i_brand_id",
                  descending=[False, True, False])
            .limit(100))


def q55(t):
    """NDS q55: manager's brand revenue."""
    ss, dd, it = t["store_sales"], t["date_dim"], t["item"]
    return (ss.join(dd, on="ss_sold_date_sk", right_on=["d_date_sk"])
            .join(it, on="ss_item_sk", right_on=["i_item_sk"])
            .filter((col("i_manager_id") == 28) & (col("d_moy") == 11))
            .group_by("i_brand", "i_brand_id")
            .agg(sum_(col("ss_ext_sales_price")).alias("ext_price"))
            .sort("ext_price", "i_brand_id", descending=[True, False])
            .limit(100))


def q7_shape(t):
    """NDS q7 shape: per-item averages over a promo slice, ordered by the
    102k-distinct string item id."""
    ss, it = t["store_sales"], t["item"]
    return (ss.filter(col("ss_promo_sk") < 150)
            .join(it, on="ss_item_sk", right_on=["i_item_sk"])
            .group_by("i_item_id")
            .agg(avg(col("ss_quantity")).alias("agg1"),
                 avg(col("ss_list_price").cast(FLOAT64)).alias("agg2"),
                 avg(col("ss_sales_price").cast(FLOAT64)).alias("agg3"))
            .sort("i_item_id")
            .limit(100))


def q96_shape(t):
    """NDS q96: selective count through store join."""
    ss, st, dd = t["store_sales"], t["store"], t["date_dim"]
    return (ss.join(dd, on="ss_sold_date_sk", right_on=["d_date_sk"])
            .join(st, on="ss_store_sk", right_on=["s_store_sk"])
            .filter((col("d_moy") == 7) & (col("ss_quantity") >= 40))
            .agg(count_star()))


def q_customer(t):
    """NDS q23/q4 shape: top customers by net revenue — a 1M-group
    group-by on the string business key c_customer_id (VERDICT Weak #5's
    CAS-build stressor)."""
    ss, c = t["store_sales"], t["customer"]
    rev = (col("ss_ext_sales_price").cast(FLOAT64)
           - col("ss_ext_discount_amt").cast(FLOAT64))
    return (ss.join(c, on="ss_customer_sk", right_on=["c_customer_sk"])
            .with_column("rev", rev)
            .group_by("c_customer_id")
            .agg(sum_(col("rev")).alias("revenue"), count_star())
            .sort("revenue", descending=True)
            .limit(100))


def q67_shape(t):
    """NDS q67 shape: rollup over (category, state) revenue, then rank
    within category, keep top 10 per category."""
    ss, st, it, dd = t["store_sales"], t["store"], t["item"], t["date_dim"]
    base = (ss.join(dd, on="ss_sold_date_sk", right_on=["d_date_sk"])
            .join(st, on="ss_store_sk", right_on=["s_store_sk"])
            .join(it, on="ss_item_sk", right_on=["i_item_sk"])
            .filter(col("d_year") == 2001)
            .rollup("i_category", "s_state")
            .agg(sum_(col("ss_ext_sales_price").cast(FLOAT64))
                 .alias("sumsales")))
    return (base
            .with_column("rk", rank().over(partition_by=["i_category"],
                                           order_by=["sumsales"],
                                           descending=[True]))
            .filter(col("rk") <= 10)
            .sort("i_category", "rk")
            .limit(200))


def q47_shape(t):
    """NDS q47 shape: monthly brand revenue with the moving average over
    the partition (window agg over string partition keys)."""
    ss, it, dd, st = t["store_sales"], t["item"], t["date_dim"], t["store"]
    monthly = (ss.join(dd, on="ss_sold_date_sk", right_on=["d_date_sk"])
               .join(it, on="ss_item_sk", right_on=["i_item_sk"])
               .join(st, on="ss_store_sk", right_on=["s_store_sk"])
               .filter(col("d_year") == 2000)
               .group_by("i_category", "i_brand", "s_state", "d_moy")
               .agg(sum_(col("ss_sales_price").cast(FLOAT64))
                    .alias("sum_sales")))
    return (monthly
            .with_column("avg_monthly",
                         win_avg(col("sum_sales")).over(
                             partition_by=["i_category", "i_brand",
                                           "s_state"]))
            .filter(col("sum_sales") > col("avg_monthly") * lit(1.1))
            .sort("i_category", "i_brand", "s_state", "d_moy")
            .limit(100))


def q36_shape(t):
    """NDS q36 shape: gross-margin ratio rollup over category/brand."""
    ss, it, dd = t["store_sales"], t["item"], t["date_dim"]
    return (ss.join(dd, on="ss_sold_date_sk", right_on=["d_date_sk"])
            .join(it, on="ss_item_sk", right_on=["i_item_sk"])
            .filter(col("d_year") == 2001)
            .rollup("i_category", "i_brand")
            .agg(sum_(col("ss_net_profit").cast(FLOAT64)).alias("profit"),
                 sum_(col("ss_ext_sales_price").cast(FLOAT64)).alias("sales"))
            .with_column("margin", col("profit") / col("sales"))
            .sort("margin")
            .limit(100))


def q89_shape(t):
    """NDS q89 shape: store/category monthly sales vs category average
    (windowed deviation filter), sorted by the deviation."""
    ss, it, dd, st = t["store_sales"], t["item"], t["date_dim"], t["store"]
    monthly = (ss.join(dd, on="ss_sold_date_sk", right_on=["d_date_sk"])
               .join(it, on="ss_item_sk", right_on=["i_item_sk"])
               .join(st, on="ss_store_sk", right_on=["s_store_sk"])
               .filter((col("d_year") == 1999)
                       & (col("i_category_id") <= 5))
               .group_by("i_category", "s_store_name", "d_moy")
               .agg(sum_(col("ss_sales_price").cast(FLOAT64))
                    .alias("sum_sales")))
    return (monthly
            .with_column("avg_m", win_avg(col("sum_sales")).over(
                partition_by=["i_category", "s_store_name"]))
            .with_column("dev", col("sum_sales") - col("avg_m"))
            .sort("dev", "s_store_name")
            .limit(100))


POWER_RUN: List = [
    ("q3", q3),
    ("q42", q42),
    ("q52", q52),
    ("q55", q55),
    ("q7", q7_shape),
    ("q96", q96_shape),
    ("q23", q_customer),
    ("q67", q67_shape),
    ("q47", q47_shape),
    ("q36", q36_shape),
    ("q89", q89_shape),
]


_DF_CACHE: Dict[int, List[tuple]] = {}


def run_power(tables: Dict[str, DataFrame], queries=None) -> List[tuple]:
    """Run the suite (all scans re-read from disk each call; the plan
    DataFrames are cached like prepared statements)."""
    key = (id(tables), tuple(queries) if queries else None)
    picked = [(n, f) for n, f in POWER_RUN
              if queries is None or n in queries]
    if key not in _DF_CACHE:
        _DF_CACHE[key] = [(name, fn(tables)) for name, fn in picked]
    return [(name, df.collect()) for name, df in _DF_CACHE[key]]

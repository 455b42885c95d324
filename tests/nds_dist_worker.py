"""Gloo 2-process worker: NDS star-schema queries over a SHARDED on-disk
parquet fact table with REPLICATED parquet dimensions (the flagship bench
topology of bench.py). The union of rank results must equal the
single-process result computed by the launching test.
"""
import json
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from spark_rapids_amd import Session, col, count_star, sum_
from spark_rapids_amd.types import FLOAT64


def open_tables(session, data_dir):
    t = {"store_sales": session.read_parquet(
        os.path.join(data_dir, "store_sales"))}
    for dim in ("date_dim", "item", "store", "customer"):
        t[dim] = session.read_parquet(os.path.join(data_dir, dim),
                                      replicated=True)
    return t


def queries(t):
    """No LIMIT: the union of per-rank rows must equal the global result."""
    ss, it, dd, st = t["store_sales"], t["item"], t["date_dim"], t["store"]
    return {
        "cat_rev": (ss.join(it, on="ss_item_sk", right_on=["i_item_sk"])
                    .group_by("i_category")
                    .agg(sum_(col("ss_ext_sales_price")), count_star())),
        "brand_year": (ss.join(dd, on="ss_sold_date_sk",
                               right_on=["d_date_sk"])
                       .join(it, on="ss_item_sk", right_on=["i_item_sk"])
                       .filter(col("d_moy") == 11)
                       .group_by("d_year", "i_brand")
                       .agg(sum_(col("ss_quantity")))),
        "state_rev": (ss.join(st, on="ss_store_sk", right_on=["s_store_sk"])
                      .group_by("s_state")
                      .agg(sum_(col("ss_net_profit").cast(FLOAT64)))),
    }


def to_jsonable(rows):
    """floats stay floats (tolerant compare); everything else stringified
    (Decimal is not JSON-serializable)."""
    return [[x if isinstance(x, float) else
             (None if x is None else str(x)) for x in r] for r in rows]


def normalize(rows):
    return sorted(rows, key=lambda r: str(r))


def rows_equal(a, b):
    """Multiset equality with float tolerance (distributed float sums
    reassociate)."""
    import math

    if len(a) != len(b):
        return False
    for ra, rb in zip(normalize(a), normalize(b)):
        if len(ra) != len(rb):
            return False
        for x, y in zip(ra, rb):
            if isinstance(x, float) or isinstance(y, float):
                if not math.isclose(float(x), float(y), rel_tol=1e-6,
                                    abs_tol=1e-6):
                    return False
            elif x != y:
                return False
    return True


def main():
    import torch.distributed as td

    data_dir, expected_file = sys.argv[1], sys.argv[2]
    td.init_process_group(backend="gloo")
    rank, world = td.get_rank(), td.get_world_size()
    s = Session({"spark.rapids.sql.enabled": False})
    t = open_tables(s, data_dir)
    expected = json.load(open(expected_file))
    for name, df in queries(t).items():
        mine = df.collect()
        gathered = [None] * world
        td.all_gather_object(gathered, mine)
        if rank == 0:
            union = to_jsonable([r for part in gathered for r in part])
            assert rows_equal(union, expected[name]), \
                (name, len(union), len(expected[name]),
                 normalize(union)[:2], normalize(expected[name])[:2])
    td.barrier()
    if rank == 0:
        print("DIST_OK nds parquet shard + replicated dims")
    td.destroy_process_group()


if __name__ == "__main__":
    main()

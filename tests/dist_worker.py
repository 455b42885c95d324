"""Worker for multi-process distributed tests (gloo, CPU). Launched by
test_distributed.py via torch.distributed.run; asserts that distributed
aggregation over the exchange layer matches a single-process run."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np
import torch
import torch.distributed as td

from spark_rapids_amd import Session, col, count_star, sum_, avg


def main():
    td.init_process_group(backend="gloo")
    rank = td.get_rank()
    world = td.get_world_size()

    # each rank owns a deterministic partition; the union is rows 0..N-1
    n_total = 10_000
    per = n_total // world
    lo, hi = rank * per, (rank + 1) * per
    rows = np.arange(lo, hi, dtype=np.int64)

    s = Session({"spark.rapids.sql.enabled": False})
    df = s.create_dataframe({
        "k": (rows % 97),
        "v": rows.astype(np.float64),
        "c": (rows % 5),
    }, num_partitions=3)

    # keyed aggregate: global result sharded by key across ranks
    out = df.group_by("k").agg(sum_(col("v")), count_star()).collect()
    # validate: every key's global sum; keys disjoint across ranks
    expect = {}
    all_rows = np.arange(0, n_total, dtype=np.int64)
    for k in range(97):
        m = all_rows % 97 == k
        expect[k] = (float(all_rows[m].sum()), int(m.sum()))
    for k, sv, cnt in out:
        ek, ec = expect[k]
        assert sv == ek and cnt == ec, (k, sv, cnt, expect[k])
    # keys disjoint + complete across ranks
    my_keys = sorted(k for k, _, _ in out)
    gathered = [None] * world
    td.all_gather_object(gathered, my_keys)
    if rank == 0:
        flat = [k for ks in gathered for k in ks]
        assert sorted(flat) == list(range(97)), "keys lost or duplicated"
        assert len(set(flat)) == len(flat)

    # keyless aggregate: every rank sees the global result
    tot = df.agg(sum_(col("v")), count_star(), avg(col("c"))).collect()
    assert tot[0][1] == n_total
    assert tot[0][0] == float(all_rows.sum())

    # filter + keyed agg + keyless rollup (the q4 shape)
    out2 = (df.filter(col("c") != 0).group_by("k")
            .agg(sum_(col("v"))).agg(count_star()).collect())
    assert out2[0][0] == 97, out2

    # sharded-right join: build side must be all-gathered (broadcast join)
    right_rows = np.arange(rank * 10, rank * 10 + 10, dtype=np.int64)
    right = s.create_dataframe({"k": right_rows % 97,
                                "r": right_rows * 100})
    joined = df.join(right, on="k").agg(count_star()).collect()
    # expected: every left row whose k is in union of right keys (0..19 %97)
    rk = set((np.arange(0, world * 10) % 97).tolist())
    exp_cnt = int(sum(1 for v in all_rows if (v % 97) in rk)) * \
        sum(1 for v in (np.arange(0, world * 10) % 97) if True) // (world * 10)
    # compute exactly: matches = sum over left rows of count of right dups
    from collections import Counter
    rc = Counter((np.arange(0, world * 10) % 97).tolist())
    exp = int(sum(rc[v % 97] for v in all_rows.tolist()))
    assert joined[0][0] == exp, (joined, exp)

    # replicated table must NOT be re-broadcast/duplicated
    rep = s.create_dataframe({"k": list(range(97)), "w": [1] * 97},
                             replicated=True)
    j2 = df.join(rep, on="k").agg(count_star()).collect()
    assert j2[0][0] == n_total, j2

    # shuffled-hash-join path: force it with a tiny broadcast threshold
    s2 = Session({"spark.rapids.sql.enabled": False,
                  "spark.rapids.sql.join.broadcastThreshold": 1})
    df2 = s2.create_dataframe({
        "k": (rows % 97), "v": rows.astype(np.float64)}, num_partitions=2)
    right2 = s2.create_dataframe({"k": right_rows % 97, "r": right_rows * 100})
    j3 = df2.join(right2, on="k").agg(count_star()).collect()
    assert j3[0][0] == exp, (j3, exp, "shuffled join path")
    left_join = df2.join(right2, on="k", how="left").agg(count_star()).collect()
    # left join rows = matched pairs + unmatched left rows
    unmatched = sum(1 for v in rows.tolist() if rc[v % 97] == 0)
    # per-rank left rows only; aggregate over ranks is global
    gexp = int(sum(rc[v % 97] if rc[v % 97] else 1 for v in all_rows.tolist()))
    assert left_join[0][0] == gexp, (left_join, gexp)

    # single-pass aggregates (collect_list / percentile / distinct): rows
    # are exchanged by key hash BEFORE aggregation — every rank must hold
    # each key's complete value set
    from spark_rapids_amd import collect_list, count_distinct, percentile

    outc = df.group_by("c").agg(collect_list(col("k")),
                                percentile(col("v"), 0.5)).collect()
    outd = dict((r[0], r[1]) for r in
                df.group_by("c").agg(count_distinct(col("k"))).collect())
    exp_by_c = {}
    for c in range(5):
        m = all_rows % 5 == c
        ks = (all_rows[m] % 97).tolist()
        exp_by_c[c] = (sorted(ks), len(set(ks)),
                       float(np.percentile(all_rows[m].astype(float), 50)))
    seen_cs = []
    for c, lst, p50 in outc:
        el, ec, ep = exp_by_c[c]
        assert sorted(lst) == el, (c, len(lst), len(el))
        assert outd[c] == ec and abs(p50 - ep) < 1e-9, (c, p50, exp_by_c[c])
        seen_cs.append(c)
    gath = [None] * world
    td.all_gather_object(gath, sorted(seen_cs))
    if rank == 0:
        flat = [c for cs in gath for c in cs]
        assert sorted(flat) == [0, 1, 2, 3, 4], flat

    # rollup across ranks: grand total row must count every rank's rows
    ro = df.rollup("c").agg(count_star()).collect()
    grand = [r for r in ro if r[1] == 1]
    gath2 = [None] * world
    td.all_gather_object(gath2, [tuple(r) for r in ro])
    if rank == 0:
        allro = [r for rs in gath2 for r in rs]
        g = [r for r in allro if r[1] == 1]
        assert sum(r[2] for r in g) == n_total, g

    # a rank with ZERO rows must still join every collective (keyed agg
    # exchange would otherwise deadlock)
    import spark_rapids_amd as _sr

    empty_on_1 = s.create_dataframe(
        {"k": [1, 2, 3] if rank == 0 else [],
         "v": [1.0, 2.0, 3.0] if rank == 0 else []},
        dtypes={"k": _sr.INT32, "v": _sr.FLOAT64})
    ek = sorted(empty_on_1.group_by("k").agg(sum_(col("v"))).collect())
    gathered_e = [None] * world
    td.all_gather_object(gathered_e, ek)
    if rank == 0:
        alle = sorted(r for rs in gathered_e for r in rs)
        assert alle == [(1, 1.0), (2, 2.0), (3, 3.0)], alle
    tot_e = empty_on_1.agg(count_star()).collect()
    assert tot_e == [(3,)], tot_e

    # shuffled join + sub-partitioned build in one plan: tiny broadcast
    # threshold forces the hash-exchange strategy, tiny subPartition
    # bytes forces the bucketed join of the exchanged build side
    s3 = Session({"spark.rapids.sql.enabled": False,
                  "spark.rapids.sql.join.broadcastThreshold": 1,
                  "spark.rapids.sql.join.subPartition.targetBytes": 512})
    df3 = s3.create_dataframe({"k": (rows % 97),
                               "v": rows.astype(np.float64)})
    right3 = s3.create_dataframe({"k": right_rows % 97,
                                  "r": right_rows * 100})
    j4 = df3.join(right3, on="k").agg(count_star()).collect()
    assert j4[0][0] == exp, (j4, exp, "shuffled+subpartitioned")

    # distributed global ORDER BY: rank r holds the r-th sorted range and
    # the rank-order concatenation is the full global sort
    sorted_rows = df.sort("v", descending=True).collect()
    my_vals = [r[1] for r in sorted_rows]
    assert my_vals == sorted(my_vals, reverse=True)
    gs = [None] * world
    td.all_gather_object(gs, my_vals)
    if rank == 0:
        flat = [v for vs in gs for v in vs]
        assert flat == sorted(flat, reverse=True), "ranges out of order"
        assert len(flat) == n_total

    # bounded-wave exchange: a tiny wave budget splits the shuffle into
    # many all-to-all rounds; results must be identical
    from spark_rapids_amd.shuffle import dist as _d

    baseline = sorted(df.group_by("k").agg(sum_(col("v"))).collect())
    _d.set_wave_bytes(2048)
    try:
        waved = sorted(df.group_by("k").agg(sum_(col("v"))).collect())
    finally:
        _d.set_wave_bytes(1 << 30)
    assert waved == baseline, (len(waved), len(baseline))

    # LIST columns survive the exchange (serializer nested walk): the
    # distributed sort range-exchanges every column, including lv
    lists_df = s.create_dataframe({
        "k": rows,
        "lv": [[int(v), int(v) + 1] if v % 3 else None
               for v in rows.tolist()],
    })
    srt = lists_df.sort("k").collect()
    for k, lv in srt:
        assert lv == ([k, k + 1] if k % 3 else None), (k, lv)
    t2 = torch.tensor([len(srt)], dtype=torch.int64)
    td.all_reduce(t2)
    assert int(t2.item()) == n_total

    # conditional hash join across ranks: per-rank sharded probe against
    # a replicated build, non-equi condition compacts pairs; the keyless
    # count aggregates globally so every rank reports the global count
    cright = s.create_dataframe({
        "k": np.arange(97, dtype=np.int64),
        "thr": np.arange(97, dtype=np.float64) * 50.0,
    }, replicated=True)
    cj = df.join(cright, on="k", condition=col("v") < col("thr")) \
        .agg(count_star()).collect()
    exp_cj = sum(1 for v in all_rows.tolist()
                 if float(v) < (v % 97) * 50.0)
    assert cj[0][0] == exp_cj, (cj, exp_cj)

    # nested-loop join across ranks (replicated right, semi pairs)
    nlr = s.create_dataframe({"b": [100.0, 5000.0]}, replicated=True)
    nl = df.join_nl(nlr, col("v") < col("b"), "semi") \
        .agg(count_star()).collect()
    exp_nl = sum(1 for v in all_rows.tolist() if float(v) < 5000.0)
    assert nl[0][0] == exp_nl, (nl, exp_nl)

    td.barrier()
    if rank == 0:
        print("DIST_OK")
    td.destroy_process_group()


if __name__ == "__main__":
    main()

"""Extended CPU-vs-GPU fuzz campaign: many seeds over the typed
special-value generators and the main query shapes (expressions, agg,
sort, join, window). Run on a GPU box:

  python tools/fuzz_campaign.py [n_seeds]

Prints one line per (shape, seed); exits non-zero on the first
divergence with the failing details.
"""
import sys

sys.path.insert(0, ".")

import numpy as np  # noqa: E402

import spark_rapids_amd as sr  # noqa: E402
from spark_rapids_amd import (col, count, count_star, lit, max_, min_,  # noqa: E402
                              row_number, sum_, win_sum)
from spark_rapids_amd.testing import (assert_gpu_and_cpu_are_equal,  # noqa: E402
                                      gen_column)
from spark_rapids_amd.types import (DType, FLOAT64, INT32, INT64,  # noqa: E402
                                    STRING)


def shape_exprs(seed):
    n = 3000
    data = {
        "i": gen_column(INT32, n, seed * 31 + 1),
        "l": gen_column(INT64, n, seed * 31 + 2),
        "f": gen_column(FLOAT64, n, seed * 31 + 3),
        "s": gen_column(STRING, n, seed * 31 + 4),
        "d": gen_column(DType.decimal(9, 2), n, seed * 31 + 5),
    }

    def q(s):
        df = s.create_dataframe({k: list(v) for k, v in data.items()},
                                dtypes={"d": DType.decimal(9, 2)})
        return (df.with_column("a", col("i").cast(sr.INT64) + col("l"))
                .with_column("b", col("f") * 2.0 - col("f"))
                .with_column("c", col("s").length())
                .with_column("e", col("d") + col("d"))
                .with_column("g", col("f") > col("f") * 0.5)
                .with_column("h", col("s").contains("a"))
                .filter(col("i").is_not_null() | col("f").is_null()))

    assert_gpu_and_cpu_are_equal(q)


def shape_agg(seed):
    n = 5000
    data = {
        "k": gen_column(INT32, n, seed * 37 + 1, null_frac=0.15),
        "f": gen_column(FLOAT64, n, seed * 37 + 2),
        "s": gen_column(STRING, n, seed * 37 + 3),
    }

    def q(s):
        df = s.create_dataframe({k: list(v) for k, v in data.items()})
        return (df.group_by("k")
                .agg(sum_(col("f")), count_star(), count(col("s")),
                     min_(col("s")), max_(col("s")), min_(col("f")),
                     max_(col("f"))).sort("k"))

    assert_gpu_and_cpu_are_equal(q, rel=1e-6)


def shape_join(seed):
    nl_, nr_ = 4000, 900
    data_l = {"k": gen_column(INT32, nl_, seed * 41 + 1, null_frac=0.1),
              "a": gen_column(FLOAT64, nl_, seed * 41 + 2)}
    data_r = {"k": gen_column(INT32, nr_, seed * 41 + 3, null_frac=0.1),
              "b": gen_column(FLOAT64, nr_, seed * 41 + 4)}

    def q(s):
        l = s.create_dataframe(data_l)
        r = s.create_dataframe(data_r)
        return l.join(r, on="k", how="left",
                      condition=col("a") < col("b"))

    assert_gpu_and_cpu_are_equal(q)


def shape_sort(seed):
    n = 6000
    data = {"f": gen_column(FLOAT64, n, seed * 43 + 1),
            "s": gen_column(STRING, n, seed * 43 + 2),
            "t": list(range(n))}

    def q(s):
        df = s.create_dataframe({k: list(v) for k, v in data.items()})
        return df.sort("f", "s", "t", descending=[True, False, False])

    assert_gpu_and_cpu_are_equal(q, ignore_order=False)


def shape_window(seed):
    n = 4000
    data = {
        "p": [abs(v) % 30 if v is not None else None
              for v in gen_column(INT32, n, seed * 47 + 1, null_frac=0.05)],
        "o": gen_column(FLOAT64, n, seed * 47 + 2, null_frac=0.05),
        "v": gen_column(FLOAT64, n, seed * 47 + 3),
        "t": list(range(n)),
    }

    def q(s):
        df = s.create_dataframe({k: list(v) for k, v in data.items()})
        df = df.with_column("rn", row_number().over(["p"], ["o", "t"]))
        return df.with_column("ws", win_sum(col("v")).over(
            ["p"], ["o", "t"], rows_between=(-3, 3)))

    assert_gpu_and_cpu_are_equal(q, rel=1e-6)


SHAPES = [("exprs", shape_exprs), ("agg", shape_agg),
          ("join", shape_join), ("sort", shape_sort),
          ("window", shape_window)]


def main():
    n_seeds = int(sys.argv[1]) if len(sys.argv) > 1 else 10
    ran = 0
    for seed in range(1000, 1000 + n_seeds):
        for name, fn in SHAPES:
            try:
                fn(seed)
                ran += 1
            except AssertionError as e:
                print(f"DIVERGENCE {name} seed={seed}: {e}")
                sys.exit(1)
        print(f"seed {seed}: {len(SHAPES)} shapes OK")
    print(f"campaign OK: {ran} shape-seed runs, 0 divergences")


if __name__ == "__main__":
    main()

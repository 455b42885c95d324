"""TPC-DS-shaped (NDS-like) star schema: string-keyed dimensions, decimal
fact columns, staged to Parquet on disk and scanned back through the GPU
parquet reader.

This is the round-2 answer to VERDICT.md "What's missing #1": the bench no
longer runs over an in-memory all-integer schema; it generates a scaled
store_sales fact plus item / store / customer / date_dim dimensions whose
join and group-by keys are strings (i_brand, i_category, s_state,
c_customer_id, i_item_id), writes them as Parquet (decimal as INT32/INT64
physical so the device decode kernels apply), and every benchmark step
scans from disk.

Reference analogues: datagen/src/main/scala/.../bigDataGen.scala (seeded,
skew-controlled generator) and the NDS table schemas behind
integration_tests/ScaleTest.md.

TPC-DS SF1 is ~2.88M store_sales rows; `sf_equivalent(rows)` reports the
fact-table scale factor a given row count corresponds to.
"""
from __future__ import annotations

import os
from typing import Dict, List, Optional

import numpy as np

N_ITEMS = 102_000
N_STORES = 1_002
N_CUSTOMERS = 1_000_000
N_DATES = 1_826          # 1998-01-01 .. 2002-12-31
BASE_YEAR = 1998
SF1_FACT_ROWS = 2_880_404  # TPC-DS store_sales rows at SF=1

_CATEGORIES = ["Books", "Children", "Electronics", "Home", "Jewelry",
               "Men", "Music", "Shoes", "Sports", "Women"]
_STATES = ["AL", "AK", "AZ", "AR", "CA", "CO", "CT", "DE", "FL", "GA",
           "HI", "ID", "IL", "IN", "IA", "KS", "KY", "LA", "ME", "MD",
           "MA", "MI", "MN", "MS", "MO", "MT", "NE", "NV", "NH", "NJ",
           "NM", "NY", "NC", "ND", "OH", "OK", "OR", "PA", "RI", "SC",
           "SD", "TN", "TX", "UT", "VT", "VA", "WA", "WV", "WI", "WY"]
_COUNTRIES = ["UNITED STATES", "CANADA", "MEXICO", "GERMANY", "FRANCE",
              "JAPAN", "BRAZIL", "INDIA", "CHINA", "AUSTRALIA",
              "UNITED KINGDOM", "ITALY", "SPAIN", "NETHERLANDS", "KOREA"]


def sf_equivalent(fact_rows: int) -> float:
    return fact_rows / SF1_FACT_ROWS


def _pa():
    import pyarrow

    return pyarrow


def _dec72(pa, ints: np.ndarray, mask: Optional[np.ndarray] = None):
    """decimal(7,2) arrow array from UNSCALED int64 cents (pa.array would
    treat plain ints as whole values, so build the buffers directly)."""
    n = len(ints)
    words = np.empty((n, 2), dtype=np.int64)
    words[:, 0] = ints
    words[:, 1] = ints.astype(np.int64) >> 63  # sign extension
    validity = None
    if mask is not None and mask.any():
        validity = pa.py_buffer(
            np.packbits(~mask, bitorder="little").tobytes())
    return pa.Array.from_buffers(
        pa.decimal128(7, 2), n, [validity, pa.py_buffer(words)])


def _id_strings(prefix: str, sks: np.ndarray, width: int = 16) -> np.ndarray:
    """Deterministic fixed-width business keys, e.g. AAAAAAAAAAAAA123."""
    pad = width - len(prefix)
    return np.char.add(prefix, np.char.zfill(sks.astype("U%d" % pad), pad))


# ---- dimensions ----------------------------------------------------------

def gen_date_dim():
    pa = _pa()
    sk = np.arange(N_DATES, dtype=np.int32)
    days = np.datetime64(f"{BASE_YEAR}-01-01") + sk.astype("timedelta64[D]")
    years = days.astype("datetime64[Y]").astype(int) + 1970
    months = days.astype("datetime64[M]").astype(int) % 12 + 1
    dom = (days - days.astype("datetime64[M]")).astype(int) + 1
    return pa.table({
        "d_date_sk": pa.array(sk, type=pa.int32()),
        "d_year": pa.array(years.astype(np.int32), type=pa.int32()),
        "d_moy": pa.array(months.astype(np.int32), type=pa.int32()),
        "d_dom": pa.array(dom.astype(np.int32), type=pa.int32()),
    })


def gen_item(seed: int = 7):
    pa = _pa()
    rng = np.random.default_rng(seed)
    sk = np.arange(N_ITEMS, dtype=np.int32)
    brand_id = rng.integers(1, 1001, N_ITEMS).astype(np.int32)
    cat_id = rng.integers(1, 11, N_ITEMS).astype(np.int32)
    cats = np.array(_CATEGORIES, dtype=object)[cat_id - 1]
    brands = np.char.add(
        np.char.add(np.array(_CATEGORIES)[cat_id - 1], "#brand"),
        (brand_id % 100).astype("U3"))
    price = np.round(rng.uniform(0.5, 300.0, N_ITEMS) * 100).astype(np.int64)
    return pa.table({
        "i_item_sk": pa.array(sk, type=pa.int32()),
        "i_item_id": pa.array(_id_strings("AAAAAAAA", sk), type=pa.string()),
        "i_brand_id": pa.array(brand_id, type=pa.int32()),
        "i_brand": pa.array(brands.astype(object), type=pa.string()),
        "i_category_id": pa.array(cat_id, type=pa.int32()),
        "i_category": pa.array(list(cats), type=pa.string()),
        "i_manufact_id": pa.array(
            rng.integers(1, 1001, N_ITEMS).astype(np.int32), type=pa.int32()),
        "i_manager_id": pa.array(
            rng.integers(1, 101, N_ITEMS).astype(np.int32), type=pa.int32()),
        "i_current_price": _dec72(pa, price),
    })


def gen_store(seed: int = 11):
    pa = _pa()
    rng = np.random.default_rng(seed)
    sk = np.arange(N_STORES, dtype=np.int32)
    state = np.array(_STATES, dtype=object)[
        rng.integers(0, len(_STATES), N_STORES)]
    return pa.table({
        "s_store_sk": pa.array(sk, type=pa.int32()),
        "s_store_id": pa.array(_id_strings("AAAAAAAA", sk), type=pa.string()),
        "s_store_name": pa.array(
            [f"store_{i % 997}" for i in sk], type=pa.string()),
        "s_state": pa.array(list(state), type=pa.string()),
    })


def gen_customer(seed: int = 13):
    pa = _pa()
    rng = np.random.default_rng(seed)
    sk = np.arange(N_CUSTOMERS, dtype=np.int32)
    country = np.array(_COUNTRIES, dtype=object)[
        rng.integers(0, len(_COUNTRIES), N_CUSTOMERS)]
    return pa.table({
        "c_customer_sk": pa.array(sk, type=pa.int32()),
        # 1M DISTINCT string business keys: the high-cardinality
        # string-group-by stressor (VERDICT.md Weak #5)
        "c_customer_id": pa.array(_id_strings("AAAAAAAA", sk),
                                  type=pa.string()),
        "c_birth_country": pa.array(list(country), type=pa.string()),
        "c_preferred_cust_flag": pa.array(
            list(np.where(rng.random(N_CUSTOMERS) < 0.5, "Y", "N")),
            type=pa.string()),
    })


# ---- fact ----------------------------------------------------------------

def gen_store_sales_partition(rows: int, seed: int):
    pa = _pa()
    rng = np.random.default_rng(seed)
    qty = rng.integers(1, 101, rows).astype(np.int32)
    # all-integer cent arithmetic (float rounding over 20M+ rows dominates
    # staging time otherwise)
    wholesale = rng.integers(100, 10001, rows)
    list_price = wholesale * rng.integers(100, 201, rows) // 100
    sales_price = list_price * rng.integers(30, 101, rows) // 100
    ext_sales = sales_price * qty // 10  # keep within decimal(7,2)
    ext_discount = ext_sales * rng.integers(0, 31, rows) // 100
    net_profit = ext_sales - (wholesale * qty // 10) - ext_discount
    null_mask = rng.integers(0, 50, rows) == 0  # 2% NULL sales price
    return pa.table({
        "ss_sold_date_sk": pa.array(
            rng.integers(0, N_DATES, rows).astype(np.int32), type=pa.int32()),
        "ss_item_sk": pa.array(
            (rng.zipf(1.3, rows) % N_ITEMS).astype(np.int32),
            type=pa.int32()),
        "ss_store_sk": pa.array(
            rng.integers(0, N_STORES, rows).astype(np.int32),
            type=pa.int32()),
        "ss_customer_sk": pa.array(
            rng.integers(0, N_CUSTOMERS, rows).astype(np.int32),
            type=pa.int32()),
        "ss_promo_sk": pa.array(
            rng.integers(0, 300, rows).astype(np.int32), type=pa.int32()),
        "ss_quantity": pa.array(qty, type=pa.int32()),
        "ss_wholesale_cost": _dec72(pa, wholesale),
        "ss_list_price": _dec72(pa, list_price),
        "ss_sales_price": _dec72(pa, sales_price, mask=null_mask),
        "ss_ext_discount_amt": _dec72(pa, ext_discount),
        "ss_ext_sales_price": _dec72(pa, ext_sales),
        "ss_net_profit": _dec72(pa, net_profit),
    })


# ---- staging -------------------------------------------------------------

# facts: PLAIN int/decimal pages (the device decode fast path; dictionary
# encoding 1M-distinct int keys is also pathologically slow to write)
_FACT_OPTS = dict(
    compression=None,            # measured path is GPU decode, not zstd
    use_dictionary=False,
    store_decimal_as_integer=True,   # decimal(7,2) -> INT32 physical
    data_page_size=8 << 20,
    # pyarrow otherwise caps pages at ~20k rows, shattering a 2.5M-row
    # chunk into 125 tiny pages whose header parse dominates the scan
    max_rows_per_page=1 << 20,
    row_group_size=1 << 23,      # one row group per staged file
)
# dims: dictionary-encoded strings (the GPU string decode path); the large
# dictionary limit keeps even c_customer_id (1M distinct) dictionary-coded
_DIM_OPTS = dict(
    compression=None,
    use_dictionary=True,
    store_decimal_as_integer=True,
    dictionary_pagesize_limit=1 << 26,
    data_page_size=8 << 20,
    max_rows_per_page=1 << 20,
)


def stage(data_dir: str, rows: int, rank: int, world: int,
          partitions: int = 8, force: bool = False) -> Dict[str, str]:
    """Write this rank's fact partitions + (rank 0) the dimensions under
    data_dir. Returns table-name -> path. Idempotent per (dir, rows)."""
    import pyarrow.parquet as pq

    fact_dir = os.path.join(data_dir, "store_sales")
    os.makedirs(fact_dir, exist_ok=True)
    stamp = os.path.join(data_dir, f".staged-r{rank}-{rows}x{world}")
    if force or not os.path.exists(stamp):
        per = rows // partitions
        for p in range(partitions):
            n = per if p < partitions - 1 else rows - per * (partitions - 1)
            t = gen_store_sales_partition(n, seed=(rank + 1) * 1000 + p)
            pq.write_table(
                t, os.path.join(fact_dir, f"part-{rank:03d}-{p:03d}.parquet"),
                **_FACT_OPTS)
        if rank == 0:
            for name, gen in (("date_dim", gen_date_dim),
                              ("item", gen_item), ("store", gen_store),
                              ("customer", gen_customer)):
                d = os.path.join(data_dir, name)
                os.makedirs(d, exist_ok=True)
                pq.write_table(gen(), os.path.join(d, "part-000.parquet"),
                               **_DIM_OPTS)
        open(stamp, "w").write("ok")
    return {name: os.path.join(data_dir, name)
            for name in ("store_sales", "date_dim", "item", "store",
                         "customer")}

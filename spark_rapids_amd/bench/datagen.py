"""Deterministic synthetic NDS/TPC-DS-like star schema generator
(reference analogue: datagen/ bigDataGen.scala — seeded, skew-controllable).
Generates a store_sales-style fact table plus item/store dimensions as
numpy arrays; scale is rows-per-partition so weak scaling is exact.
"""
from __future__ import annotations

from typing import Dict, List

import numpy as np

from ..column import Column, ColumnBatch, Field, Schema
from ..types import DATE32, FLOAT64, INT8, INT16, INT32, INT64, DType

DEC72 = DType.decimal(7, 2)

N_ITEMS = 102_000
N_STORES = 1_002
N_CUSTOMERS = 1_000_000
DATE_LO, DATE_HI = 10_000, 11_000  # ~1997..2000 in days-since-epoch


def fact_schema() -> Schema:
    return Schema([
        Field("ss_sold_date", DATE32),
        Field("ss_item_id", INT32),
        Field("ss_store_id", INT32),
        Field("ss_customer_id", INT32),
        Field("ss_promo", INT8),
        Field("ss_quantity", INT32),
        Field("ss_wholesale_cost", DEC72),
        Field("ss_list_price", DEC72),
        Field("ss_sales_price", DEC72),
        Field("ss_discount", FLOAT64),
    ])


def gen_fact_partition(rows: int, seed: int) -> ColumnBatch:
    rng = np.random.default_rng(seed)
    qty = rng.integers(1, 100, rows).astype(np.int32)
    wholesale = rng.uniform(1.0, 100.0, rows)
    list_price = wholesale * rng.uniform(1.0, 2.0, rows)
    sales_price = list_price * rng.uniform(0.3, 1.0, rows)
    discount = np.round(rng.uniform(0.0, 0.3, rows), 2)
    # 2% nulls on sales price (exercises null paths in aggregation)
    price_valid = rng.random(rows) >= 0.02
    cols = [
        Column.from_numpy(rng.integers(DATE_LO, DATE_HI, rows).astype(np.int32), DATE32),
        Column.from_numpy((rng.zipf(1.3, rows) % N_ITEMS).astype(np.int32), INT32),
        Column.from_numpy(rng.integers(0, N_STORES, rows).astype(np.int32), INT32),
        Column.from_numpy(rng.integers(0, N_CUSTOMERS, rows).astype(np.int32), INT32),
        Column.from_numpy(rng.integers(0, 4, rows).astype(np.int8), INT8),
        Column.from_numpy(qty, INT32),
        Column.from_numpy(np.round(wholesale * 100).astype(np.int64), DEC72),
        Column.from_numpy(np.round(list_price * 100).astype(np.int64), DEC72),
        Column.from_numpy(np.round(sales_price * 100).astype(np.int64), DEC72,
                          price_valid),
        Column.from_numpy(discount, FLOAT64),
    ]
    return ColumnBatch(cols, rows)


def item_schema() -> Schema:
    return Schema([
        Field("i_item_id", INT32),
        Field("i_category", INT8),
        Field("i_brand", INT16),
        Field("i_current_price", FLOAT64),
    ])


def gen_items(seed: int = 7) -> ColumnBatch:
    rng = np.random.default_rng(seed)
    n = N_ITEMS
    return ColumnBatch([
        Column.from_numpy(np.arange(n, dtype=np.int32), INT32),
        Column.from_numpy(rng.integers(0, 10, n).astype(np.int8), INT8),
        Column.from_numpy(rng.integers(0, 1000, n).astype(np.int16), INT16),
        Column.from_numpy(rng.uniform(1.0, 300.0, n), FLOAT64),
    ], n)


def store_schema() -> Schema:
    return Schema([
        Field("s_store_id", INT32),
        Field("s_state", INT8),
    ])


def gen_stores(seed: int = 11) -> ColumnBatch:
    rng = np.random.default_rng(seed)
    n = N_STORES
    return ColumnBatch([
        Column.from_numpy(np.arange(n, dtype=np.int32), INT32),
        Column.from_numpy(rng.integers(0, 50, n).astype(np.int8), INT8),
    ], n)

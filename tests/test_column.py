import numpy as np
import pytest

from spark_rapids_amd import Column, ColumnBatch, DType, INT32, INT64, FLOAT64, STRING
from spark_rapids_amd.column import make_validity, unpack_validity


def test_from_pylist_roundtrip_ints():
    c = Column.from_pylist([1, None, 3, None, 5], INT32)
    assert c.size == 5
    assert c.null_count == 2
    assert c.to_pylist() == [1, None, 3, None, 5]


def test_from_pylist_no_nulls_has_no_validity():
    c = Column.from_pylist([1, 2, 3], INT64)
    assert c.validity is None
    assert c.null_count == 0


def test_string_roundtrip():
    vals = ["hello", None, "", "wörld", None]
    c = Column.from_pylist(vals, STRING)
    assert c.to_pylist() == vals
    assert c.null_count == 2


def test_validity_bitmask_order():
    valid = np.array([True, False, True, True, False, False, True, True, True])
    m = make_validity(valid)
    assert unpack_validity(m, 9).tolist() == valid.tolist()
    # arrow LSB-first: first byte bits 0..7
    assert m[0].item() == 0b11001101


def test_nulls_column():
    c = Column.nulls(FLOAT64, 4)
    assert c.to_pylist() == [None] * 4


def test_batch_ragged_assert():
    a = Column.from_pylist([1, 2], INT32)
    b = Column.from_pylist([1], INT32)
    with pytest.raises(AssertionError):
        ColumnBatch([a, b])


def test_decimal_pylist():
    d = DType.decimal(10, 2)
    c = Column.from_pylist([12345, None], d)
    from decimal import Decimal
    assert c.to_pylist() == [Decimal("123.45"), None]

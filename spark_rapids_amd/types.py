"""Spark-semantics data type system for the MI355X columnar engine.

Mirrors the role of the reference's TypeSig/TypeChecks lattice
(reference: sql-plugin/src/main/scala/com/nvidia/spark/rapids/TypeChecks.scala:95-716)
but is a fresh design: a small closed enum of physical types plus per-operator
supported-type signatures used by the GPU-overrides tagging pass.
"""
from __future__ import annotations

import enum
from dataclasses import dataclass
from typing import Optional

import numpy as np


class TypeId(enum.Enum):
    BOOL = "boolean"
    INT8 = "tinyint"
    INT16 = "smallint"
    INT32 = "int"
    INT64 = "bigint"
    FLOAT32 = "float"
    FLOAT64 = "double"
    DECIMAL64 = "decimal64"   # int64 backing + scale
    DECIMAL128 = "decimal128"  # 2x int64 backing + scale
    DATE32 = "date"           # days since epoch, int32 backing
    TIMESTAMP = "timestamp"   # micros since epoch, int64 backing
    STRING = "string"         # arrow offsets + bytes
    NULL = "void"
    LIST = "array"
    STRUCT = "struct"
    MAP = "map"


_FIXED_WIDTH_BYTES = {
    TypeId.BOOL: 1,
    TypeId.INT8: 1,
    TypeId.INT16: 2,
    TypeId.INT32: 4,
    TypeId.INT64: 8,
    TypeId.FLOAT32: 4,
    TypeId.FLOAT64: 8,
    TypeId.DECIMAL64: 8,
    TypeId.DECIMAL128: 16,
    TypeId.DATE32: 4,
    TypeId.TIMESTAMP: 8,
}

_NUMPY_DTYPES = {
    TypeId.BOOL: np.uint8,
    TypeId.INT8: np.int8,
    TypeId.INT16: np.int16,
    TypeId.INT32: np.int32,
    TypeId.INT64: np.int64,
    TypeId.FLOAT32: np.float32,
    TypeId.FLOAT64: np.float64,
    TypeId.DECIMAL64: np.int64,
    TypeId.DATE32: np.int32,
    TypeId.TIMESTAMP: np.int64,
}

_INTEGRALS = {TypeId.INT8, TypeId.INT16, TypeId.INT32, TypeId.INT64}
_FLOATS = {TypeId.FLOAT32, TypeId.FLOAT64}


@dataclass(frozen=True)
class DType:
    """A data type instance: a TypeId plus parameters (decimal precision/scale,
    list element type, struct children)."""

    id: TypeId
    precision: int = 0
    scale: int = 0
    children: tuple = ()  # tuple[DType, ...] for LIST/STRUCT
    field_names: tuple = ()  # STRUCT member names (parallel to children)

    # ---- constructors -------------------------------------------------
    @staticmethod
    def bool_() -> "DType":
        return DType(TypeId.BOOL)

    @staticmethod
    def int8() -> "DType":
        return DType(TypeId.INT8)

    @staticmethod
    def int16() -> "DType":
        return DType(TypeId.INT16)

    @staticmethod
    def int32() -> "DType":
        return DType(TypeId.INT32)

    @staticmethod
    def int64() -> "DType":
        return DType(TypeId.INT64)

    @staticmethod
    def float32() -> "DType":
        return DType(TypeId.FLOAT32)

    @staticmethod
    def float64() -> "DType":
        return DType(TypeId.FLOAT64)

    @staticmethod
    def decimal(precision: int, scale: int) -> "DType":
        if precision <= 18:
            return DType(TypeId.DECIMAL64, precision, scale)
        return DType(TypeId.DECIMAL128, precision, scale)

    @staticmethod
    def date32() -> "DType":
        return DType(TypeId.DATE32)

    @staticmethod
    def timestamp() -> "DType":
        return DType(TypeId.TIMESTAMP)

    @staticmethod
    def string() -> "DType":
        return DType(TypeId.STRING)

    @staticmethod
    def list_(elem: "DType") -> "DType":
        return DType(TypeId.LIST, children=(elem,))

    @staticmethod
    def map_(key: "DType", value: "DType") -> "DType":
        """Spark MapType. Physical layout = LIST of STRUCT<key,value>
        entries (arrow map layout): offsets + one entry struct child."""
        return DType(TypeId.MAP, children=(key, value))

    @property
    def entry_dtype(self) -> "DType":
        """MAP only: the entry struct dtype (key, value)."""
        return DType.struct_([("key", self.children[0]),
                              ("value", self.children[1])])

    @staticmethod
    def struct(*fields: "DType") -> "DType":
        names = tuple(f"c{i}" for i in range(len(fields)))
        return DType(TypeId.STRUCT, children=tuple(fields),
                     field_names=names)

    @staticmethod
    def struct_(fields) -> "DType":
        """struct from (name, DType) pairs."""
        fields = list(fields)
        return DType(TypeId.STRUCT,
                     children=tuple(t for _, t in fields),
                     field_names=tuple(n for n, _ in fields))

    # ---- predicates ---------------------------------------------------
    @property
    def is_fixed_width(self) -> bool:
        return self.id in _FIXED_WIDTH_BYTES

    @property
    def itemsize(self) -> int:
        return _FIXED_WIDTH_BYTES[self.id]

    @property
    def is_integral(self) -> bool:
        return self.id in _INTEGRALS

    @property
    def is_floating(self) -> bool:
        return self.id in _FLOATS

    @property
    def is_numeric(self) -> bool:
        return (
            self.id in _INTEGRALS
            or self.id in _FLOATS
            or self.id in (TypeId.DECIMAL64, TypeId.DECIMAL128)
        )

    @property
    def is_decimal(self) -> bool:
        return self.id in (TypeId.DECIMAL64, TypeId.DECIMAL128)

    @property
    def is_timelike(self) -> bool:
        return self.id in (TypeId.DATE32, TypeId.TIMESTAMP)

    @property
    def is_nested(self) -> bool:
        return self.id in (TypeId.LIST, TypeId.STRUCT, TypeId.MAP)

    def numpy_dtype(self):
        return np.dtype(_NUMPY_DTYPES[self.id])

    def __str__(self) -> str:
        if self.is_decimal:
            return f"decimal({self.precision},{self.scale})"
        if self.id is TypeId.LIST:
            return f"array<{self.children[0]}>"
        if self.id is TypeId.MAP:
            return f"map<{self.children[0]},{self.children[1]}>"
        if self.id is TypeId.STRUCT:
            inner = ", ".join(f"{n}:{c}" for n, c in
                              zip(self.field_names, self.children))
            return f"struct<{inner}>"
        return self.id.value


BOOL = DType.bool_()
INT8 = DType.int8()
INT16 = DType.int16()
INT32 = DType.int32()
INT64 = DType.int64()
FLOAT32 = DType.float32()
FLOAT64 = DType.float64()
DATE32 = DType.date32()
TIMESTAMP = DType.timestamp()
STRING = DType.string()

# Numeric widening order used by binary-op type promotion (Spark semantics:
# result of int op float is the wider float, etc.)
_PROMOTION_ORDER = [
    TypeId.BOOL,
    TypeId.INT8,
    TypeId.INT16,
    TypeId.INT32,
    TypeId.INT64,
    TypeId.FLOAT32,
    TypeId.FLOAT64,
]


# integral -> decimal equivalents for mixed decimal arithmetic
# (reference analogue: Spark DecimalPrecision.scala integral promotion)
_INT_AS_DECIMAL = {
    TypeId.INT8: (3, 0),
    TypeId.INT16: (5, 0),
    TypeId.INT32: (10, 0),
    TypeId.INT64: (20, 0),
}


def as_decimal(dt: DType) -> DType:
    """The decimal type an operand takes in decimal arithmetic."""
    if dt.is_decimal:
        return dt
    if dt.id in _INT_AS_DECIMAL:
        p, sc = _INT_AS_DECIMAL[dt.id]
        return DType.decimal(p, sc)
    raise TypeError(f"{dt} cannot participate in decimal arithmetic")


def _adjust_decimal(p: int, s: int) -> DType:
    """Spark allowPrecisionLoss=true adjustment when precision > 38
    (DecimalType.adjustPrecisionScale)."""
    if p <= 38:
        return DType.decimal(p, s)
    int_digits = p - s
    min_scale = min(s, 6)
    adj_scale = max(38 - int_digits, min_scale)
    return DType.decimal(38, adj_scale)


def decimal_arith_type(op: str, lt: DType, rt: DType) -> DType:
    """Result type of decimal multiply / divide per Spark's
    DecimalPrecision rules (sql.decimalOperations.allowPrecisionLoss=true,
    the default)."""
    a, b = as_decimal(lt), as_decimal(rt)
    p1, s1, p2, s2 = a.precision, a.scale, b.precision, b.scale
    if op == "mul":
        return _adjust_decimal(p1 + p2 + 1, s1 + s2)
    if op == "div":
        s = max(6, s1 + p2 + 1)
        return _adjust_decimal(p1 - s1 + s2 + s, s)
    raise ValueError(f"decimal_arith_type: unsupported op {op}")


def promote(a: DType, b: DType) -> DType:
    """Common wider type for arithmetic between a and b (non-decimal path)."""
    if a == b:
        return a
    if a.is_decimal and b.is_decimal:
        # Spark DecimalPrecision widening (add/sub/compare common type):
        # scale = max(s1, s2); integral digits = max(p1-s1, p2-s2); routing
        # to DECIMAL128 when precision > 18 so the rescale of the
        # smaller-scale operand cannot overflow int64.
        s = max(a.scale, b.scale)
        p = max(a.precision - a.scale, b.precision - b.scale) + s
        return _adjust_decimal(p, s)
    if a.is_decimal or b.is_decimal:
        # decimal + integral -> decimal with enough precision; handled by caller
        d = a if a.is_decimal else b
        return d
    if a.is_timelike or b.is_timelike:
        # date/timestamp vs integral literal: compare in the timelike domain
        return a if a.is_timelike else b
    ia = _PROMOTION_ORDER.index(a.id)
    ib = _PROMOTION_ORDER.index(b.id)
    return DType(_PROMOTION_ORDER[max(ia, ib)])


class TypeSig:
    """A set of supported TypeIds for one operator slot, with optional extra
    predicates (e.g. decimal precision cap). Used by the overrides tagging pass
    to explain why an op cannot run on GPU (reference analogue:
    TypeChecks.scala TypeSig)."""

    def __init__(self, ids, max_decimal_precision: int = 38, allow_nested: bool = False):
        self.ids = frozenset(ids)
        self.max_decimal_precision = max_decimal_precision
        self.allow_nested = allow_nested

    @staticmethod
    def all_basic() -> "TypeSig":
        return TypeSig(
            {
                TypeId.BOOL,
                TypeId.INT8,
                TypeId.INT16,
                TypeId.INT32,
                TypeId.INT64,
                TypeId.FLOAT32,
                TypeId.FLOAT64,
                TypeId.DECIMAL64,
                TypeId.DECIMAL128,
                TypeId.DATE32,
                TypeId.TIMESTAMP,
                TypeId.STRING,
            }
        )

    @staticmethod
    def numeric() -> "TypeSig":
        return TypeSig(
            {
                TypeId.INT8,
                TypeId.INT16,
                TypeId.INT32,
                TypeId.INT64,
                TypeId.FLOAT32,
                TypeId.FLOAT64,
                TypeId.DECIMAL64,
                TypeId.DECIMAL128,
            }
        )

    @staticmethod
    def orderable() -> "TypeSig":
        return TypeSig.all_basic()

    def supports(self, dt: DType) -> Optional[str]:
        """Return None if supported, else a human-readable reason."""
        if dt.is_nested and not self.allow_nested:
            return f"nested type {dt} is not supported"
        if dt.id not in self.ids:
            return f"type {dt} is not supported"
        if dt.is_decimal and dt.precision > self.max_decimal_precision:
            return (
                f"decimal precision {dt.precision} exceeds max "
                f"{self.max_decimal_precision}"
            )
        return None

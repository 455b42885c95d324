"""Datetime format/parse/timezone expressions (reference analogues:
GpuDateFormatClass / GpuToTimestamp / GpuFromUTCTimestamp backed by
spark-rapids-jni's GpuTimeZoneDB and cudf's datetime kernels).

Format patterns: the fixed-width Spark subset yyyy, MM, dd, HH, mm, ss
plus literal separators — the patterns NDS/TPC-DS and most ETL use.
Unsupported patterns raise FormatUnsupported at plan time and the tagger
routes the expression to the CPU.
"""
from __future__ import annotations

from typing import List, Tuple

from ..column import Column, ColumnBatch, Schema
from ..types import DType, STRING, TIMESTAMP, TypeId
from .. import ops
from .expressions import CastExpr, Expression, _as_expr

DT_LIT, DT_YYYY, DT_MM, DT_DD, DT_HH, DT_MI, DT_SS = range(7)
_WIDTH = {DT_YYYY: 4, DT_MM: 2, DT_DD: 2, DT_HH: 2, DT_MI: 2, DT_SS: 2}


class FormatUnsupported(ValueError):
    pass


def compile_format(fmt: str) -> Tuple[List[Tuple[int, int]], int]:
    """Spark datetime pattern -> fixed-width token program + total width."""
    tokens: List[Tuple[int, int]] = []
    i = 0
    width = 0
    mapping = {"yyyy": DT_YYYY, "MM": DT_MM, "dd": DT_DD, "HH": DT_HH,
               "mm": DT_MI, "ss": DT_SS}
    while i < len(fmt):
        for pat, kind in mapping.items():
            if fmt.startswith(pat, i):
                tokens.append((kind, 0))
                width += _WIDTH[kind]
                i += len(pat)
                break
        else:
            c = fmt[i]
            if c.isalpha():
                raise FormatUnsupported(
                    f"datetime pattern {c!r} in {fmt!r} (supported: "
                    "yyyy MM dd HH mm ss + literals)")
            tokens.append((DT_LIT, ord(c)))
            width += 1
            i += 1
    return tokens, width


def _to_micros(e: Expression, schema) -> Expression:
    """date32 -> micros; timestamp passes through."""
    dt = e.dtype(schema)
    if dt.id is TypeId.DATE32:
        from .expressions import BinaryExpr, Literal
        from ..types import INT64

        days = CastExpr(e, INT64)
        return CastExpr(BinaryExpr("mul", days, Literal(86_400_000_000)),
                        TIMESTAMP)
    return e


class DateFormat(Expression):
    def __init__(self, child, fmt: str):
        self.child = _as_expr(child)
        self.fmt = fmt
        self.tokens, self.width = compile_format(fmt)

    @property
    def children(self):
        return (self.child,)

    def dtype(self, schema: Schema) -> DType:
        return STRING

    def eval(self, batch: ColumnBatch, schema: Schema) -> Column:
        c = _to_micros(self.child, schema).eval(batch, schema)
        return ops.date_format(c, self.tokens, self.width)

    def output_name(self) -> str:
        return f"date_format({self.child}, {self.fmt})"

    def __str__(self):
        return self.output_name()


class ToTimestamp(Expression):
    def __init__(self, child, fmt: str = "yyyy-MM-dd HH:mm:ss"):
        self.child = _as_expr(child)
        self.fmt = fmt
        self.tokens, self.width = compile_format(fmt)

    @property
    def children(self):
        return (self.child,)

    def dtype(self, schema: Schema) -> DType:
        return TIMESTAMP

    def nullable(self, schema: Schema) -> bool:
        return True

    def eval(self, batch: ColumnBatch, schema: Schema) -> Column:
        c = self.child.eval(batch, schema)
        return ops.ts_parse(c, self.tokens, self.width)

    def output_name(self) -> str:
        return f"to_timestamp({self.child}, {self.fmt})"

    def __str__(self):
        return self.output_name()


class TzConvert(Expression):
    """from_utc_timestamp / to_utc_timestamp against the TZif transition
    table (tools/tzdb.py; device binary search in k_tz_convert)."""

    def __init__(self, child, zone: str, to_utc: bool):
        from ..tools import tzdb

        self.child = _as_expr(child)
        self.zone = zone
        self.to_utc = to_utc
        tzdb.load(zone)  # validate at plan time

    @property
    def children(self):
        return (self.child,)

    def dtype(self, schema: Schema) -> DType:
        return TIMESTAMP

    def eval(self, batch: ColumnBatch, schema: Schema) -> Column:
        c = _to_micros(self.child, schema).eval(batch, schema)
        return ops.tz_convert(c, self.zone, self.to_utc)

    def output_name(self) -> str:
        fn = "to_utc_timestamp" if self.to_utc else "from_utc_timestamp"
        return f"{fn}({self.child}, {self.zone})"

    def __str__(self):
        return self.output_name()


def date_format(e, fmt: str) -> DateFormat:
    return DateFormat(e, fmt)


def to_timestamp(e, fmt: str = "yyyy-MM-dd HH:mm:ss") -> ToTimestamp:
    return ToTimestamp(e, fmt)


def from_utc_timestamp(e, zone: str) -> TzConvert:
    return TzConvert(e, zone, to_utc=False)


def to_utc_timestamp(e, zone: str) -> TzConvert:
    return TzConvert(e, zone, to_utc=True)


def from_unixtime(e, fmt: str = "yyyy-MM-dd HH:mm:ss") -> DateFormat:
    from .expressions import BinaryExpr, Literal
    from ..types import INT64

    us = BinaryExpr("mul", CastExpr(_as_expr(e), INT64),
                    Literal(1_000_000))
    return DateFormat(CastExpr(us, TIMESTAMP), fmt)

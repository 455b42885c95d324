"""CPU reference backend with Spark SQL semantics.

This is (a) the fallback execution path when an op is not GPU-enabled and
(b) the golden reference that GPU kernel numerics tests compare against
(reference analogue: CPU Spark itself in the integration-test harness,
integration_tests/src/main/python/asserts.py).

Spark-semantics notes implemented here:
- AND/OR use Kleene three-valued logic
- division / modulo by zero yields NULL (non-ANSI mode)
- integer arithmetic wraps (non-ANSI)
- float -> int cast truncates toward zero
- sum/min/max/avg ignore NULLs; all-NULL group aggregates to NULL
- murmur3 hash matches org.apache.spark.sql.catalyst.expressions.Murmur3Hash
"""
from __future__ import annotations

from typing import List, Optional, Sequence, Tuple

import numpy as np
import torch

from ..column import Column, ColumnBatch, make_validity
from ..types import DType, TypeId


# ---------------------------------------------------------------------------
# helpers
# ---------------------------------------------------------------------------

def _vals(col: Column):
    if col.dtype.id is TypeId.STRING:
        return np.array(col.to_pylist(), dtype=object)
    if col.dtype.id is TypeId.DECIMAL128:
        from ..column import dec128_unpack

        return np.array(dec128_unpack(
            col.data.cpu().numpy()[: 2 * col.size]), dtype=object)
    return col.data.cpu().numpy()[: col.size]


def _valid(col: Column) -> np.ndarray:
    return col.valid_array()


def _make(vals: np.ndarray, valid: Optional[np.ndarray], dtype: DType) -> Column:
    if dtype.id is TypeId.DECIMAL128:
        from ..column import dec128_pack

        packed = dec128_pack([int(v) for v in vals], 0)
        if valid is not None and not valid.all():
            return Column(dtype, len(vals), torch.from_numpy(packed.copy()),
                          make_validity(valid),
                          null_count=int(len(vals) - valid.sum()))
        return Column(dtype, len(vals), torch.from_numpy(packed.copy()),
                      None, null_count=0)
    if dtype.id is TypeId.STRING:
        out = []
        for i, v in enumerate(vals):
            if valid is not None and not valid[i]:
                out.append(None)
            else:
                out.append(v)
        return Column.from_pylist(out, dtype)
    vals = np.ascontiguousarray(vals, dtype=dtype.numpy_dtype())
    if valid is not None and not valid.all():
        return Column(dtype, len(vals), torch.from_numpy(vals.copy()),
                      make_validity(valid), null_count=int(len(vals) - valid.sum()))
    return Column(dtype, len(vals), torch.from_numpy(vals.copy()), None, null_count=0)


def _bool_col(vals: np.ndarray, valid: Optional[np.ndarray]) -> Column:
    return _make(vals.astype(np.uint8), valid, DType.bool_())


# ---------------------------------------------------------------------------
# binary ops
# ---------------------------------------------------------------------------

_CMP_OPS = {"eq", "ne", "lt", "le", "gt", "ge"}


def binary_op(op: str, lhs: Column, rhs: Column, out_dtype: DType) -> Column:
    a, av = _vals(lhs), _valid(lhs)
    b, bv = _vals(rhs), _valid(rhs)
    return _binary_impl(op, a, av, b, bv, lhs.dtype, out_dtype)


def decimal_mul_div(op: str, lhs: Column, rhs: Column,
                    out_dtype: DType) -> Column:
    """Exact decimal multiply/divide with python-int arithmetic.

    value(lhs)=a/10^s1, value(rhs)=b/10^s2; result at out_dtype.scale with
    HALF_UP rounding; NULL on divide-by-zero and on overflow of
    out_dtype.precision (Spark non-ANSI overflow semantics).
    """
    a, av = _vals(lhs), _valid(lhs)
    b, bv = _vals(rhs), _valid(rhs)
    s1, s2, st = lhs.dtype.scale, rhs.dtype.scale, out_dtype.scale
    valid = av & bv
    bound = 10 ** out_dtype.precision
    # vectorized fast paths (no int64 overflow possible by the type rules)
    if a.dtype != object and b.dtype != object:
        if op == "mul" and lhs.dtype.precision + rhs.dtype.precision + 1 <= 18:
            r = a.astype(np.int64) * b.astype(np.int64)
            delta = s1 + s2 - st
            if delta > 0:
                r = _round_half_up_div(r, 10 ** delta)
            return _make(r, valid if not valid.all() else None, out_dtype)
        m = st + s2 - s1
        if op == "div" and 0 <= m and lhs.dtype.precision + m <= 18 \
                and out_dtype.precision <= 18:
            valid = valid & (b != 0)
            num = a.astype(np.int64) * (10 ** m)
            den = np.where(b == 0, 1, b.astype(np.int64))
            sign = np.where((num < 0) != (den < 0), -1, 1)
            q = sign * ((2 * np.abs(num) + np.abs(den)) //
                        (2 * np.abs(den)))
            valid = valid & (np.abs(q) < bound)
            return _make(q, valid if not valid.all() else None, out_dtype)
    res = np.zeros(len(a), dtype=object)
    for i in range(len(a)):
        if not valid[i]:
            continue
        x, y = int(a[i]), int(b[i])
        if op == "mul":
            r = x * y
            delta = s1 + s2 - st
            if delta > 0:
                r = _div_half_up(r, 10 ** delta)
        else:
            if y == 0:
                valid[i] = False
                continue
            m = st + s2 - s1
            num, den = (x * 10 ** m, y) if m >= 0 else (x, y * 10 ** (-m))
            r = _div_half_up(num, den)
        if -bound < r < bound:
            res[i] = r
        else:
            valid[i] = False
    return _make(res, valid if not valid.all() else None, out_dtype)


def _div_half_up(num: int, den: int) -> int:
    """Signed integer division rounding HALF_UP (away from zero on .5)."""
    sign = -1 if (num < 0) != (den < 0) else 1
    num, den = abs(num), abs(den)
    return sign * ((2 * num + den) // (2 * den))


def binary_op_scalar(op: str, lhs: Column, scalar, out_dtype: DType) -> Column:
    a, av = _vals(lhs), _valid(lhs)
    if scalar is None:
        b = np.zeros(1, dtype=a.dtype if a.dtype != object else np.int64)
        b = np.broadcast_to(b, a.shape)
        bv = np.zeros(len(a), dtype=bool)
    else:
        if lhs.dtype.id is TypeId.STRING:
            b = np.array([scalar] * len(a), dtype=object)
        else:
            b = np.broadcast_to(np.asarray(scalar), a.shape)
        bv = np.ones(len(a), dtype=bool)
    return _binary_impl(op, a, av, b, bv, lhs.dtype, out_dtype)


def _binary_impl(op, a, av, b, bv, in_dtype: DType, out_dtype: DType) -> Column:
    n = len(a)
    if op == "and":
        at = a.astype(bool) if a.dtype != object else a
        bt = b.astype(bool)
        res = at & bt
        # Kleene: NULL unless (false present) or both valid
        valid = (av & bv) | (av & ~at.astype(bool)) | (bv & ~bt.astype(bool))
        return _bool_col(res.astype(np.uint8), valid)
    if op == "or":
        at = a.astype(bool)
        bt = b.astype(bool)
        res = at | bt
        valid = (av & bv) | (av & at) | (bv & bt)
        return _bool_col(res.astype(np.uint8), valid)
    if op == "eq_null_safe":
        eq = _safe_eq(a, b)
        res = np.where(av & bv, eq, av == bv)
        return _bool_col(res.astype(np.uint8), None)

    valid = av & bv
    if op == "concat":
        res = np.array([(x or "") + (y or "") for x, y in zip(a, b)],
                       dtype=object)
        return _make(res, valid if not valid.all() else None, out_dtype)
    if op in _CMP_OPS:
        if in_dtype.id is TypeId.STRING:
            # elementwise python compare on object arrays
            res = np.array([_str_cmp(op, x, y) for x, y in zip(a, b)], dtype=bool)
        elif a.dtype.kind == "f" or (hasattr(b, "dtype")
                                     and getattr(b, "dtype", None) is not None
                                     and getattr(b.dtype, "kind", "") == "f"):
            # Spark float ordering: NaN == NaN and NaN is GREATER than
            # every other value (matches the GPU kernels and Spark SQL)
            na = np.isnan(a.astype(np.float64))
            bb = np.asarray(b, dtype=np.float64)
            nb = np.isnan(bb) if bb.shape else np.full(n, np.isnan(bb))
            with np.errstate(invalid="ignore"):
                eq = (a == b) | (na & nb)
                lt = np.where(na, False, np.where(nb, True, a < b))
                gt = np.where(nb, False, np.where(na, True, a > b))
            res = {"eq": lambda: eq, "ne": lambda: ~eq,
                   "lt": lambda: lt, "le": lambda: lt | eq,
                   "gt": lambda: gt, "ge": lambda: gt | eq}[op]()
        else:
            with np.errstate(invalid="ignore"):
                res = {
                    "eq": lambda: _safe_eq(a, b),
                    "ne": lambda: ~_safe_eq(a, b),
                    "lt": lambda: a < b,
                    "le": lambda: a <= b,
                    "gt": lambda: a > b,
                    "ge": lambda: a >= b,
                }[op]()
        return _bool_col(res.astype(np.uint8), valid if not valid.all() else None)

    if out_dtype.id is TypeId.DECIMAL128:
        if op == "add":
            res = a + b
        elif op == "sub":
            res = a - b
        elif op == "min":
            res = np.array([x if x <= y else y for x, y in zip(a, b)],
                           dtype=object)
        elif op == "max":
            res = np.array([x if x >= y else y for x, y in zip(a, b)],
                           dtype=object)
        else:
            raise NotImplementedError(f"decimal128 op {op}")
        return _make(res, valid if not valid.all() else None, out_dtype)
    np_out = out_dtype.numpy_dtype()
    with np.errstate(divide="ignore", invalid="ignore", over="ignore"):
        if op == "add":
            res = (a.astype(np_out) + b.astype(np_out)).astype(np_out)
        elif op == "sub":
            res = (a.astype(np_out) - b.astype(np_out)).astype(np_out)
        elif op == "mul":
            res = (a.astype(np_out) * b.astype(np_out)).astype(np_out)
        elif op == "div":  # Spark `/` -> double (planner casts); 0 divisor -> NULL
            bf = b.astype(np.float64)
            res = (a.astype(np.float64) / np.where(bf == 0, 1, bf)).astype(np_out)
            valid = valid & (bf != 0)
        elif op == "int_div":
            bz = b == 0
            bb = np.where(bz, 1, b)
            q = np.trunc(a.astype(np.float64) / bb.astype(np.float64))
            res = q.astype(np_out)
            valid = valid & ~bz
        elif op == "mod":
            bz = b == 0
            bb = np.where(bz, 1, b)
            res = np.fmod(a, bb).astype(np_out)
            valid = valid & ~bz
        elif op == "pmod":
            bz = b == 0
            bb = np.where(bz, 1, b)
            r = np.fmod(a, bb)
            r = np.where((r != 0) & ((r < 0) != (bb < 0)), r + bb, r)
            res = r.astype(np_out)
            valid = valid & ~bz
        elif op == "pow":
            res = np.power(a.astype(np.float64), b.astype(np.float64)).astype(np_out)
        elif op == "bitand":
            res = (a & b).astype(np_out)
        elif op == "bitor":
            res = (a | b).astype(np_out)
        elif op == "bitxor":
            res = (a ^ b).astype(np_out)
        elif op == "shiftleft":
            res = (a.astype(np_out) << (b.astype(np.int32) & _shift_mask(np_out))).astype(np_out)
        elif op == "shiftright":
            res = (a.astype(np_out) >> (b.astype(np.int32) & _shift_mask(np_out))).astype(np_out)
        elif op == "min":
            res = np.minimum(a, b).astype(np_out)
        elif op == "max":
            res = np.maximum(a, b).astype(np_out)
        else:
            raise NotImplementedError(f"cpu binary op {op}")
    return _make(res, valid if not valid.all() else None, out_dtype)


def _shift_mask(np_out):
    return 63 if np.dtype(np_out).itemsize == 8 else 31


def _safe_eq(a, b):
    if a.dtype == object:
        return np.array([x == y for x, y in zip(a, b)], dtype=bool)
    return a == b


def _str_cmp(op, x, y):
    x = x if x is not None else ""
    y = y if y is not None else ""
    return {"eq": x == y, "ne": x != y, "lt": x < y, "le": x <= y,
            "gt": x > y, "ge": x >= y}[op]


# ---------------------------------------------------------------------------
# unary ops / cast
# ---------------------------------------------------------------------------

def unary_op(op: str, col: Column, out_dtype: DType) -> Column:
    if op == "initcap":
        out = [None if v is None else
               " ".join(w[:1].upper() + w[1:].lower() if w else w
                        for w in v.split(" "))
               for v in col.to_pylist()]
        return Column.from_pylist(out, DType.string())
    if op == "reverse":
        out = [None if v is None else v[::-1] for v in col.to_pylist()]
        return Column.from_pylist(out, DType.string())
    if op in ("trim", "ltrim", "rtrim"):
        fn = {"trim": str.strip, "ltrim": str.lstrip,
              "rtrim": str.rstrip}[op]
        out = [None if v is None else fn(v, " ")
               for v in col.to_pylist()]
        return Column.from_pylist(out, DType.string())
    a, av = _vals(col), _valid(col)
    np_out = out_dtype.numpy_dtype() if out_dtype.id is not TypeId.STRING else None
    valid = av
    with np.errstate(invalid="ignore", divide="ignore", over="ignore"):
        if op == "neg":
            res = (-a).astype(np_out)
        elif op == "abs":
            res = np.abs(a).astype(np_out)
        elif op == "not":
            res = (~a.astype(bool)).astype(np.uint8)
        elif op == "sqrt":
            res = np.sqrt(a.astype(np.float64)).astype(np_out)
        elif op == "exp":
            res = np.exp(a.astype(np.float64)).astype(np_out)
        elif op == "log":
            af = a.astype(np.float64)
            res = np.log(np.where(af <= 0, 1, af)).astype(np_out)
            valid = av & (af > 0)  # Spark: log(<=0) -> NULL
        elif op == "floor":
            res = np.floor(a).astype(np_out)
        elif op == "ceil":
            res = np.ceil(a).astype(np_out)
        elif op == "sin":
            res = np.sin(a.astype(np.float64)).astype(np_out)
        elif op == "cos":
            res = np.cos(a.astype(np.float64)).astype(np_out)
        elif op == "tan":
            res = np.tan(a.astype(np.float64)).astype(np_out)
        elif op == "is_nan":
            res = np.isnan(a.astype(np.float64)).astype(np.uint8)
            res = np.where(av, res, 0)
            return _bool_col(res, None)
        elif op == "length":  # string length in chars
            res = np.array([len(x) if x is not None else 0 for x in a], dtype=np.int32)
        elif op == "upper":
            return _make(np.array([x.upper() if x is not None else None for x in a],
                                  dtype=object), av, out_dtype)
        elif op == "lower":
            return _make(np.array([x.lower() if x is not None else None for x in a],
                                  dtype=object), av, out_dtype)
        elif op == "year":
            res = _dt_field(a, "year").astype(np_out)
        elif op == "month":
            res = _dt_field(a, "month").astype(np_out)
        elif op == "day":
            res = _dt_field(a, "day").astype(np_out)
        else:
            raise NotImplementedError(f"cpu unary op {op}")
    return _make(res, valid if not valid.all() else None, out_dtype)


def _dt_field(days: np.ndarray, field: str) -> np.ndarray:
    dt = days.astype("datetime64[D]")
    y = dt.astype("datetime64[Y]").astype(np.int64) + 1970
    if field == "year":
        return y
    m_idx = dt.astype("datetime64[M]").astype(np.int64)
    month = m_idx % 12 + 1
    if field == "month":
        return month
    day = (dt - dt.astype("datetime64[M]")).astype(np.int64) + 1
    return day


def cast(col: Column, to: DType) -> Column:
    a, av = _vals(col), _valid(col)
    src = col.dtype
    if to.id is TypeId.STRING:
        out = []
        for v, ok in zip(a, av):
            if not ok:
                out.append(None)
            elif src.id is TypeId.BOOL:
                out.append("true" if v else "false")
            elif src.is_decimal:
                out.append(_dec_str(int(v), src.scale))
            else:
                out.append(str(v))
        return Column.from_pylist(out, to)
    if src.id is TypeId.STRING:
        return _cast_string_exact(a, av, to)
    if src.id is TypeId.DECIMAL128 or to.id is TypeId.DECIMAL128:
        if src.is_decimal and to.is_decimal:
            vals, valid = _rescale_exact(a, av, src, to)
            return _make(vals, valid if not valid.all() else None, to)
        if src.id is TypeId.DECIMAL128 and to.is_floating:
            f = np.array([float(int(v)) / (10 ** src.scale) for v in a])
            return _make(f.astype(to.numpy_dtype()),
                         av if not av.all() else None, to)
        if to.id is TypeId.DECIMAL128:
            scaled = np.array([int(round(float(v) * (10 ** to.scale)))
                               for v in a], dtype=object)
            return _make(scaled, av if not av.all() else None, to)
        raise NotImplementedError(f"cast {src} -> {to}")
    if src.is_decimal and to.is_decimal:
        vals, valid = _rescale_exact(a, av, src, to)
        return _make(vals, valid if not valid.all() else None, to)
    if src.is_decimal:
        f = a.astype(np.float64) / (10 ** src.scale)
        res = f.astype(to.numpy_dtype())
        return _make(res, av if not av.all() else None, to)
    if to.is_decimal:
        scaled = np.round(a.astype(np.float64) * (10 ** to.scale)) if src.is_floating \
            else a.astype(np.int64) * (10 ** to.scale)
        return _make(scaled.astype(np.int64), av if not av.all() else None, to)
    if to.id is TypeId.BOOL:
        res = (a != 0).astype(np.uint8)
        return _make(res, av if not av.all() else None, to)
    if src.is_floating and to.is_integral:
        # Spark non-ANSI: NaN -> 0, saturate at integral bounds, trunc to zero
        info = np.iinfo(to.numpy_dtype())
        t = np.trunc(a.astype(np.float64))
        res = np.zeros(len(t), dtype=to.numpy_dtype())
        nan = np.isnan(t)
        big = ~nan & (t >= float(info.max))
        small = ~nan & (t <= float(info.min))
        mid = ~(nan | big | small)
        res[big] = info.max
        res[small] = info.min
        res[mid] = t[mid].astype(to.numpy_dtype())
        return _make(res, av if not av.all() else None, to)
    with np.errstate(invalid="ignore", over="ignore"):
        res = a.astype(to.numpy_dtype())
    return _make(res, av if not av.all() else None, to)


_INT_BOUNDS = {TypeId.INT8: 1 << 7, TypeId.INT16: 1 << 15,
               TypeId.INT32: 1 << 31, TypeId.INT64: 1 << 63}


def _cast_string_exact(a, av, to: DType) -> Column:
    """string -> numeric with Spark semantics, exact via python decimal
    (matches the device k_str_to_dec kernel): HALF_UP to the target scale
    for decimals, truncate-toward-zero for ints (no scientific notation
    for int targets), NULL on garbage/overflow."""
    import decimal as pydec

    ctx = pydec.Context(prec=50)
    n = len(a)
    valid = av.copy()
    if to.is_decimal:
        vals = [0] * n
        q = pydec.Decimal(1).scaleb(-to.scale)
        bound = 10 ** to.precision
        for i, v in enumerate(a):
            if not av[i]:
                continue
            try:
                d = ctx.create_decimal(str(v).strip())
                if not d.is_finite():
                    raise pydec.InvalidOperation
                u = int(d.quantize(q, rounding=pydec.ROUND_HALF_UP,
                                   context=ctx).scaleb(to.scale, ctx))
                if abs(u) >= bound:
                    raise pydec.InvalidOperation
                vals[i] = u
            except (pydec.InvalidOperation, ValueError, ArithmeticError):
                valid[i] = False
        arr = np.array(vals, dtype=object) \
            if to.id is TypeId.DECIMAL128 else np.array(vals, dtype=np.int64)
        return _make(arr, valid if not valid.all() else None, to)
    if to.is_integral:
        res = np.zeros(n, dtype=to.numpy_dtype())
        lim = _INT_BOUNDS[to.id]
        for i, v in enumerate(a):
            if not av[i]:
                continue
            sv = str(v).strip()
            try:
                if "e" in sv or "E" in sv:
                    raise ValueError  # Spark: no exponent in int literals
                d = ctx.create_decimal(sv)
                if not d.is_finite():
                    raise ValueError
                u = int(d.to_integral_value(rounding=pydec.ROUND_DOWN))
                if not (-lim <= u < lim):
                    raise ValueError
                res[i] = u
            except (pydec.InvalidOperation, ValueError, ArithmeticError):
                valid[i] = False
        return _make(res, valid if not valid.all() else None, to)
    if to.id is TypeId.BOOL:
        res = np.zeros(n, dtype=np.uint8)
        yes = {"true", "t", "yes", "y", "1"}
        no = {"false", "f", "no", "n", "0"}
        for i, v in enumerate(a):
            if not av[i]:
                continue
            sv = str(v).strip().lower()
            if sv in yes:
                res[i] = 1
            elif sv not in no:
                valid[i] = False
        return _make(res, valid if not valid.all() else None, to)
    # float target
    res = np.zeros(n, dtype=to.numpy_dtype())
    for i, v in enumerate(a):
        if not av[i]:
            continue
        try:
            res[i] = float(v)
        except (ValueError, TypeError):
            valid[i] = False
    return _make(res, valid if not valid.all() else None, to)


def _rescale_exact(a, av, src: DType, to: DType):
    """Exact decimal->decimal rescale in Python ints (no int64 overflow),
    HALF_UP on down-shift, NULL where the result exceeds the target
    precision or an int64 backing (Spark non-ANSI overflow -> null)."""
    shift = to.scale - src.scale
    bound = 10 ** to.precision
    up = 10 ** shift if shift >= 0 else None
    down = 10 ** -shift if shift < 0 else None
    valid = av.copy()
    out = []
    for i, v in enumerate(a):
        if not av[i]:
            out.append(0)
            continue
        x = int(v)
        if up is not None:
            x *= up
        else:
            sign = -1 if x < 0 else 1
            x = sign * ((2 * abs(x) + down) // (2 * down))
        if abs(x) >= bound or (to.id is TypeId.DECIMAL64
                               and abs(x) > 0x7FFFFFFFFFFFFFFF):
            valid[i] = False
            x = 0
        out.append(x)
    if to.id is TypeId.DECIMAL128:
        return np.array(out, dtype=object), valid
    return np.array(out, dtype=np.int64), valid


def _round_half_up_div(a: np.ndarray, d: int) -> np.ndarray:
    """HALF_UP (round half away from zero) on the magnitude: the floor-based
    formulation is wrong for negatives (-5/2 must give -3, not -4)."""
    sign = np.where(a < 0, -1, 1)
    aa = np.abs(a)
    return sign * ((2 * aa + d) // (2 * d))


def _dec_str(unscaled: int, scale: int) -> str:
    if scale == 0:
        return str(unscaled)
    sign = "-" if unscaled < 0 else ""
    s = str(abs(unscaled)).rjust(scale + 1, "0")
    return f"{sign}{s[:-scale]}.{s[-scale:]}"


def round_half_up(col: Column, scale: int) -> Column:
    a, av = _vals(col), _valid(col)
    p = 10.0 ** scale
    x = a.astype(np.float64) * p
    r = np.where(x >= 0, np.floor(x + 0.5), np.ceil(x - 0.5)) / p
    return _make(r, av if not av.all() else None, col.dtype)


def _like_to_regex(pattern: str) -> str:
    """SQL LIKE -> python regex with Spark's backslash escape: \\% and \\_
    match the literal char."""
    import re

    out = []
    i = 0
    while i < len(pattern):
        c = pattern[i]
        if c == "\\" and i + 1 < len(pattern):
            out.append(re.escape(pattern[i + 1]))
            i += 2
            continue
        if c == "%":
            out.append(".*")
        elif c == "_":
            out.append(".")
        else:
            out.append(re.escape(c))
        i += 1
    return "".join(out)


def str_predicate(op: str, col: Column, pattern: str) -> Column:
    a, av = _vals(col), _valid(col)
    if op == "rlike":
        import re

        # re.ASCII: Java regex (Spark's engine) treats \d \w \s as
        # ASCII-only by default, python re as unicode.
        # Remaining divergences (possessive quantifiers, \p classes) are
        # routed away by the tagger before this point.
        prog = re.compile(pattern, re.ASCII)
        res = np.array([bool(prog.search(x)) if x is not None else False
                        for x in a], dtype=np.uint8)
        return _make(res, av if not av.all() else None, DType.bool_())
    if op == "like":
        import re

        prog = re.compile(f"^{_like_to_regex(pattern)}$", re.DOTALL)
        res = np.array([bool(prog.match(x)) if x is not None else False
                        for x in a], dtype=np.uint8)
    else:
        fn = {"contains": lambda x: pattern in x,
              "starts_with": lambda x: x.startswith(pattern),
              "ends_with": lambda x: x.endswith(pattern)}[op]
        res = np.array([fn(x) if x is not None else False for x in a],
                       dtype=np.uint8)
    return _make(res, av if not av.all() else None, DType.bool_())


def str_split(col: Column, delimiter: str) -> Column:
    import re as _re

    rx = _re.compile(delimiter)
    out = []
    for v in col.to_pylist():
        if v is None:
            out.append(None)
            continue
        if v == "":
            out.append([""])
            continue
        parts = rx.split(v)
        while parts and parts[-1] == "":
            parts.pop()
        out.append(parts)
    return Column.from_pylist(out, DType.list_(DType.string()))


def array_size(col: Column) -> Column:
    valid = _valid(col)
    offs = col.offsets.numpy()
    sizes = (offs[1:] - offs[:-1]).astype(np.int32)
    return _make(sizes, valid if not valid.all() else None, DType.int32())


def element_at(col: Column, index: int) -> Column:
    out = []
    for v in col.to_pylist():
        if v is None:
            out.append(None)
        elif index > 0:
            out.append(v[index - 1] if index <= len(v) else None)
        else:
            out.append(v[index] if -index <= len(v) else None)
    return Column.from_pylist(out, col.dtype.children[0])


def array_contains(col: Column, value) -> Column:
    out = [None if v is None else (value in v) for v in col.to_pylist()]
    return Column.from_pylist(out, DType.bool_())


def map_get(col: Column, key) -> Column:
    out = []
    for v in col.to_pylist():  # dicts (last-win on duplicate keys)
        out.append(None if v is None else v.get(key))
    return Column.from_pylist(out, col.dtype.children[1])


def make_map(kcols, vcols) -> Column:
    from ..types import DType as _DT

    dtype = _DT.map_(kcols[0].dtype, vcols[0].dtype)
    kl = [c.to_pylist() for c in kcols]
    vl = [c.to_pylist() for c in vcols]
    rows = [list(zip(krow, vrow))
            for krow, vrow in zip(zip(*kl), zip(*vl))]
    return Column.from_pylist(rows, dtype)


def regexp_extract(col: Column, pattern: str, group: int) -> Column:
    import re as _re

    rx = _re.compile(pattern)
    out = []
    for v in col.to_pylist():
        if v is None:
            out.append(None)
            continue
        m = rx.search(v)
        if not m or group > rx.groups:
            out.append("")
        else:
            out.append(m.group(group) or "")
    return Column.from_pylist(out, DType.string())


def concat_ws(sep: str, cols) -> Column:
    lists = [c.to_pylist() for c in cols]
    out = [sep.join(v for v in row if v is not None)
           for row in zip(*lists)]
    return Column.from_pylist(out, DType.string())


def get_json_object(col: Column, path: str) -> Column:
    import json as _json

    assert path.startswith("$"), "json path must start with $"
    keys = [k for k in path[1:].lstrip(".").split(".") if k]
    out = []
    for v in col.to_pylist():
        if v is None:
            out.append(None)
            continue
        try:
            cur = _json.loads(v)
        except ValueError:
            out.append(None)
            continue
        for k in keys:
            if isinstance(cur, dict) and k in cur:
                cur = cur[k]
            else:
                cur = None
                break
        if cur is None:
            out.append(None)
        elif isinstance(cur, str):
            out.append(cur)
        elif isinstance(cur, bool):
            out.append("true" if cur else "false")
        elif isinstance(cur, (dict, list)):
            out.append(_json.dumps(cur, separators=(",", ":")))
        else:
            out.append(_json.dumps(cur))
    return Column.from_pylist(out, DType.string())


def regexp_extract_all(col: Column, pattern: str, group: int) -> Column:
    import re as _re

    rx = _re.compile(pattern)
    out = []
    for v in col.to_pylist():
        if v is None:
            out.append(None)
            continue
        vals = []
        for m in rx.finditer(v):
            g = m.group(group) if group <= rx.groups else None
            vals.append(g if g is not None else "")
        out.append(vals)
    return Column.from_pylist(out, DType.list_(DType.string()))


def regexp_replace(col: Column, pattern: str, replacement: str) -> Column:
    import re as _re

    rx = _re.compile(pattern)
    # java $g refs / \$ escapes -> python \g<g>
    pyrepl = ""
    i = 0
    while i < len(replacement):
        c = replacement[i]
        if c == "$" and i + 1 < len(replacement) and \
                replacement[i + 1].isdigit():
            pyrepl += f"\\g<{replacement[i + 1]}>"
            i += 2
        elif c == "\\" and i + 1 < len(replacement):
            ch = replacement[i + 1]
            # python templates: $ is literal; backslash must double
            pyrepl += ch if ch == "$" else (
                "\\\\" if ch == "\\" else "\\" + ch)
            i += 2
        else:
            pyrepl += c.replace("\\", "\\\\")
            i += 1
    out = [None if v is None else rx.sub(pyrepl, v)
           for v in col.to_pylist()]
    return Column.from_pylist(out, DType.string())


def substring(col: Column, pos: int, length: int = -1) -> Column:
    a, av = _vals(col), _valid(col)
    out = []
    for x, ok in zip(a, av):
        if not ok or x is None:
            out.append(None)
            continue
        n = len(x)
        begin = pos - 1 if pos > 0 else (n + pos if pos < 0 else 0)
        begin = max(begin, 0)
        end = n if length < 0 else min(begin + length, n)
        out.append(x[begin:end] if begin < n else "")
    return Column.from_pylist(out, DType.string())


def if_else(cond: Column, a: Column, b: Column) -> Column:
    """Rowwise cond ? a : b. A NULL condition selects b (Spark CASE WHEN)."""
    c = _vals(cond).astype(bool) & _valid(cond)
    if a.dtype.id is TypeId.STRING:
        av_, bv_ = _vals(a), _vals(b)
        aok, bok = _valid(a), _valid(b)
        out = [av_[i] if c[i] else bv_[i] for i in range(len(c))]
        ok = np.where(c, aok, bok)
        return _make(np.array(out, dtype=object), ok if not ok.all() else None, a.dtype)
    res = np.where(c, _vals(a), _vals(b))
    ok = np.where(c, _valid(a), _valid(b))
    return _make(res, ok if not ok.all() else None, a.dtype)


def is_null(col: Column) -> Column:
    valid = _valid(col)
    return _bool_col((~valid).astype(np.uint8), None)


# ---------------------------------------------------------------------------
# selection
# ---------------------------------------------------------------------------

def apply_boolean_mask(batch: ColumnBatch, mask: Column) -> ColumnBatch:
    m = _vals(mask).astype(bool) & _valid(mask)
    idx = np.nonzero(m)[0].astype(np.int32)
    return _gather_idx(batch, idx, np.ones(len(idx), dtype=bool))


def gather(batch: ColumnBatch, indices: Column, check_bounds: bool = False,
           negatives: bool = True) -> ColumnBatch:
    idx = _vals(indices).astype(np.int64)
    if not negatives and not check_bounds:
        return _gather_idx(batch, idx, np.ones(len(idx), dtype=bool))
    ok = idx >= 0
    if check_bounds:
        ok &= idx < batch.num_rows
    return _gather_idx(batch, np.where(ok, idx, 0).astype(np.int64), ok)


def _gather_idx(batch: ColumnBatch, idx: np.ndarray, row_ok: np.ndarray) -> ColumnBatch:
    cols = []
    for c in batch.columns:
        if c.dtype.is_nested:
            vals_py = c.to_pylist()
            out = [vals_py[i] if ok and 0 <= i < len(vals_py) else None
                   for i, ok in zip(idx, row_ok)]
            cols.append(Column.from_pylist(out, c.dtype))
            continue
        a, av = _vals(c), _valid(c)
        if len(a) == 0:
            vals = np.zeros(len(idx), dtype=object if c.dtype.id is TypeId.STRING
                            else c.dtype.numpy_dtype())
            valid = np.zeros(len(idx), dtype=bool)
        else:
            vals = a[idx]
            valid = av[idx] & row_ok
        cols.append(_make(vals, valid if not valid.all() else None, c.dtype))
    return ColumnBatch(cols, len(idx))


def concat_batches(batches: List[ColumnBatch]) -> ColumnBatch:
    ncols = batches[0].num_columns
    cols = []
    for i in range(ncols):
        dtype = batches[0].columns[i].dtype
        if dtype.is_nested:
            vals = []
            for b in batches:
                vals.extend(b.columns[i].to_pylist())
            cols.append(Column.from_pylist(vals, dtype))
            continue
        if dtype.id is TypeId.STRING:
            vals = []
            for b in batches:
                vals.extend(b.columns[i].to_pylist())
            cols.append(Column.from_pylist(vals, dtype))
        else:
            vals = np.concatenate([_vals(b.columns[i]) for b in batches])
            valid = np.concatenate([_valid(b.columns[i]) for b in batches])
            cols.append(_make(vals, valid if not valid.all() else None, dtype))
    return ColumnBatch(cols)


# ---------------------------------------------------------------------------
# Spark-compatible murmur3_x86_32
# ---------------------------------------------------------------------------

_U32 = 0xFFFFFFFF


def _rotl(x, r):
    return ((x << r) | (x >> (32 - r))) & _U32


def _mix_k1(k1):
    k1 = (k1 * 0xCC9E2D51) & _U32
    k1 = _rotl(k1, 15)
    return (k1 * 0x1B873593) & _U32


def _mix_h1(h1, k1):
    h1 = h1 ^ k1
    h1 = _rotl(h1, 13)
    return (h1 * 5 + 0xE6546B64) & _U32


def _fmix(h1, length):
    h1 ^= length
    h1 ^= h1 >> 16
    h1 = (h1 * 0x85EBCA6B) & _U32
    h1 ^= h1 >> 13
    h1 = (h1 * 0xC2B2AE35) & _U32
    h1 ^= h1 >> 16
    return h1


def _hash_int(v: np.ndarray, seed: np.ndarray) -> np.ndarray:
    k1 = _mix_k1(v.astype(np.int64) & _U32)
    h1 = _mix_h1(seed, k1)
    return _fmix(h1, 4)


def _hash_long(v: np.ndarray, seed: np.ndarray) -> np.ndarray:
    v = v.astype(np.int64)
    low = v & _U32
    high = (v >> 32) & _U32
    h1 = _mix_h1(seed, _mix_k1(low))
    h1 = _mix_h1(h1, _mix_k1(high))
    return _fmix(h1, 8)


def _hash_bytes_one(data: bytes, seed: int) -> int:
    h1 = seed
    n = len(data)
    i = 0
    # Spark hashUnsafeBytes: 4-byte little-endian blocks then per-byte tail
    while i + 4 <= n:
        k1 = int.from_bytes(data[i:i + 4], "little")
        h1 = _mix_h1(h1, _mix_k1(k1))
        i += 4
    while i < n:
        b = data[i]
        if b >= 128:
            b -= 256
        h1 = _mix_h1(h1, _mix_k1(b & _U32))
        i += 1
    return _fmix(h1, n)


def murmur3_hash(cols: List[Column], seed: int = 42) -> Column:
    n = cols[0].size
    h = np.full(n, seed, dtype=np.int64)
    for c in cols:
        a, av = _vals(c), _valid(c)
        if c.dtype.id is TypeId.STRING:
            nh = np.array([_hash_bytes_one((x or "").encode("utf-8"), int(s))
                           for x, s in zip(a, h)], dtype=np.int64)
        elif c.dtype.id is TypeId.DECIMAL128:
            lo = np.array([int(v) & 0xFFFFFFFFFFFFFFFF for v in a],
                          dtype=np.uint64).astype(np.int64)
            hi = np.array([(int(v) >> 64) & 0xFFFFFFFFFFFFFFFF for v in a],
                          dtype=np.uint64).astype(np.int64)
            nh = _hash_long(hi, _hash_long(lo, h))
        elif c.dtype.id in (TypeId.INT64, TypeId.TIMESTAMP, TypeId.DECIMAL64):
            nh = _hash_long(a, h)
        elif c.dtype.id is TypeId.FLOAT64:
            af = a.astype(np.float64).copy()
            af[af == 0.0] = 0.0  # -0.0 -> 0.0
            nh = _hash_long(af.view(np.int64), h)
        elif c.dtype.id is TypeId.FLOAT32:
            af = a.astype(np.float32).copy()
            af[af == 0.0] = 0.0
            nh = _hash_int(af.view(np.int32), h)
        else:  # int8/16/32, bool, date: hashed as int
            nh = _hash_int(a.astype(np.int32), h)
        h = np.where(av, nh, h)
    return _make(h.astype(np.int32), None, DType.int32())


# ---------------------------------------------------------------------------
# partition / groupby / join / sort
# ---------------------------------------------------------------------------

def hash_partition(batch: ColumnBatch, key_idx: List[int], num_parts: int):
    keys = [batch.columns[i] for i in key_idx]
    h = _vals(murmur3_hash(keys, 42)).astype(np.int64)
    part = ((h % num_parts) + num_parts) % num_parts  # pmod
    order = np.argsort(part, kind="stable").astype(np.int64)
    counts = np.bincount(part, minlength=num_parts)
    offsets = np.zeros(num_parts + 1, dtype=np.int64)
    np.cumsum(counts, out=offsets[1:])
    out = _gather_idx(batch, order, np.ones(len(order), dtype=bool))
    return out, offsets.tolist()


def reduce(op: str, col: Column):
    a, av = _vals(col), _valid(col)
    a = a[av]
    if op == "count":
        return int(av.sum())
    if op == "count_all":
        return col.size
    if len(a) == 0:
        return None
    if op == "sum":
        if col.dtype.is_floating:
            return float(np.sum(a.astype(np.float64)))
        return int(np.sum(a.astype(np.int64)))
    if op == "min":
        return a.min().item() if a.dtype != object else min(a)
    if op == "max":
        return a.max().item() if a.dtype != object else max(a)
    if op == "mean":
        return float(np.mean(a.astype(np.float64)))
    raise NotImplementedError(f"cpu reduce {op}")


def _group_codes(keys: List[Column]) -> Tuple[np.ndarray, np.ndarray]:
    """Return (codes, first_row_index_per_group)."""
    arrs = []
    for k in keys:
        a, av = _vals(k), _valid(k)
        if a.dtype == object:
            # encode strings / int128 objects (None distinct)
            uniq, inv = np.unique(
                np.array([repr(x) if x is not None else "\0\0NULL"
                          for x in a]), return_inverse=True)
            arrs.append(inv.astype(np.int64))
            arrs.append(av.astype(np.int64))
        else:
            code = a.view(np.int64) if a.dtype.itemsize == 8 else a.astype(np.int64)
            # normalize values under NULLs so all-null keys form one group,
            # and fold -0.0 / NaN payloads like the GPU grouping semantics
            if a.dtype.kind == "f":
                af = a.astype(np.float64).copy()
                af[af == 0.0] = 0.0
                af[np.isnan(af)] = np.nan
                code = af.view(np.int64)
            code = np.where(av, code, 0)
            arrs.append(code)
            arrs.append(av.astype(np.int64))
    stacked = np.stack(arrs, axis=1) if arrs else np.zeros((len(keys[0]._vals), 0))
    uniq, first_idx, codes = np.unique(stacked, axis=0, return_index=True,
                                       return_inverse=True)
    return codes.ravel(), first_idx


def group_by_aggregate(batch: ColumnBatch, key_idx: List[int],
                       aggs: List[Tuple[str, int, DType]]) -> ColumnBatch:
    keys = [batch.columns[i] for i in key_idx]
    n = batch.num_rows
    if not key_idx:
        # keyless (global) aggregate: ALWAYS one group, even over zero
        # rows — Spark returns a single row (count 0, sums null)
        codes = np.zeros(n, dtype=np.int64)
        first_idx = np.array([0] if n else [], dtype=np.int64)
        ngroups = 1
    else:
        codes, first_idx = _group_codes(keys)
        ngroups = len(first_idx)
    out_cols: List[Column] = []
    for k in keys:
        out_cols.append(_gather_idx(ColumnBatch([k]), first_idx,
                                    np.ones(ngroups, dtype=bool)).columns[0])
    for op, vidx, out_dtype in aggs:
        if op == "count_all":
            res = np.bincount(codes, minlength=ngroups).astype(np.int64)
            out_cols.append(_make(res, None, out_dtype))
            continue
        vc = batch.columns[vidx]
        a, av = _vals(vc), _valid(vc)
        if op == "count":
            res = np.bincount(codes[av], minlength=ngroups).astype(np.int64)
            out_cols.append(_make(res, None, out_dtype))
            continue
        if op in ("bit_and", "bit_or", "bit_xor"):
            ident = -1 if op == "bit_and" else 0
            res = np.full(ngroups, ident, dtype=np.int64)
            ufunc = {"bit_and": np.bitwise_and, "bit_or": np.bitwise_or,
                     "bit_xor": np.bitwise_xor}[op]
            ufunc.at(res, codes[av], a[av].astype(np.int64))
            cnt = np.bincount(codes[av], minlength=ngroups)
            gv = cnt > 0
            out_cols.append(_make(res.astype(out_dtype.numpy_dtype()),
                                  gv if not gv.all() else None, out_dtype))
            continue
        if op in ("first", "last"):
            idx = np.nonzero(av)[0]
            if op == "first":
                idx = idx[::-1]  # reversed assignment: earliest wins
            if a.dtype == object or vc.dtype.id in (TypeId.STRING,
                                                    TypeId.DECIMAL128):
                res = np.zeros(ngroups, dtype=object)
            else:
                res = np.zeros(ngroups, dtype=a.dtype)
            res[codes[idx]] = a[idx]
            cnt = np.bincount(codes[av], minlength=ngroups)
            gv = cnt > 0
            if vc.dtype.id is TypeId.STRING:
                out_cols.append(Column.from_pylist(
                    [v if ok else None for v, ok in zip(res, gv)], out_dtype))
            else:
                out_cols.append(_make(res, gv if not gv.all() else None,
                                      out_dtype))
            continue
        if op.startswith("hll:"):
            est = hll_groups(vc, codes, ngroups, int(op.split(":", 1)[1]))
            out_cols.append(_make(est, None, out_dtype))
            continue
        if op.startswith("percentile:"):
            pq = float(op.split(":", 1)[1])
            res = np.zeros(ngroups)
            gv = np.zeros(ngroups, dtype=bool)
            af = a[av].astype(np.float64)
            gc2 = codes[av]
            for g in range(ngroups):
                vals_g = af[gc2 == g]
                if len(vals_g):
                    res[g] = np.percentile(vals_g, 100.0 * pq)
                    gv[g] = True
            out_cols.append(_make(res, gv if not gv.all() else None,
                                  out_dtype))
            continue
        if op in ("collect_list", "collect_set"):
            lists: list = [[] for _ in range(ngroups)]
            for g, v, ok in zip(codes, vc.to_pylist(), av):
                if ok:
                    lists[g].append(v)
            if op == "collect_set":
                lists = [list(dict.fromkeys(l)) for l in lists]
            out_cols.append(Column.from_pylist(lists, out_dtype))
            continue
        if op in ("min", "max") and \
                batch.columns[vidx].dtype.id is TypeId.STRING:
            best: list = [None] * ngroups
            fn = min if op == "min" else max
            for g, v, ok in zip(codes, vc.to_pylist(), av):
                if ok:
                    best[g] = v if best[g] is None else fn(best[g], v)
            out_cols.append(Column.from_pylist(best, out_dtype))
            continue
        if batch.columns[vidx].dtype.id is TypeId.DECIMAL128 and \
                op not in ("sum", "count", "count_all"):
            raise NotImplementedError(
                f"{op} over decimal128 (float path would lose precision)")
        cnt = np.bincount(codes[av], minlength=ngroups)
        gvalid = cnt > 0
        gc = codes[av]
        if op == "sum" and out_dtype.id is TypeId.DECIMAL128:
            sums = [0] * ngroups
            has = [False] * ngroups
            for g, v, ok in zip(codes, a, av):
                if ok:
                    sums[g] += int(v)
                    has[g] = True
            gv = np.array(has)
            out_cols.append(_make(np.array(sums, dtype=object),
                                  gv if not gv.all() else None, out_dtype))
            continue
        if op == "sum" and not out_dtype.is_floating:
            # integral/decimal sum: accumulate in int64 (wraps like Spark non-ANSI)
            s = np.zeros(ngroups, dtype=np.int64)
            np.add.at(s, gc, a[av].astype(np.int64))
            out_cols.append(_make(s, gvalid if not gvalid.all() else None, out_dtype))
            continue
        af = a[av].astype(np.float64)
        if op in ("sum", "mean", "m2"):
            s = np.zeros(ngroups)
            with np.errstate(invalid="ignore"):  # inf + -inf -> nan (Spark)
                np.add.at(s, gc, af)
            if op == "sum":
                res = s
            elif op == "mean":
                res = s / np.where(cnt == 0, 1, cnt)
            else:  # m2: sum of squared deviations
                mean = s / np.where(cnt == 0, 1, cnt)
                d = af - mean[gc]
                res = np.zeros(ngroups)
                np.add.at(res, gc, d * d)
        elif op in ("min", "max"):
            init = np.inf if op == "min" else -np.inf
            res = np.full(ngroups, init)
            # Spark float ordering: NaN is the GREATEST value — min must
            # prefer any non-NaN (fmin drops NaN), max must return NaN
            # when present (maximum propagates it); all-NaN groups stay
            # NaN via fmin(nan,nan)=nan. Matches the GPU CAS ordering.
            ufunc = np.fmin if op == "min" else np.maximum
            ufunc.at(res, gc, af)
            if op == "min":
                nanm = np.isnan(af)
                if nanm.any():
                    # groups whose only values are NaN: fmin left init
                    only_nan = np.full(ngroups, True)
                    only_nan[gc[~nanm]] = False
                    seen = np.full(ngroups, False)
                    seen[gc] = True
                    res[only_nan & seen] = np.nan
        else:
            raise NotImplementedError(f"cpu groupby agg {op}")
        if out_dtype.is_floating:
            res = res.astype(out_dtype.numpy_dtype())
        else:
            res = res.astype(np.int64)
        out_cols.append(_make(res, gvalid if not gvalid.all() else None, out_dtype))
    return ColumnBatch(out_cols, ngroups)


def join_gather_maps(left: ColumnBatch, right: ColumnBatch,
                     left_keys: List[int], right_keys: List[int], how: str,
                     right_matched=None):
    import pandas as pd

    lk = {f"k{i}": _key_series(left.columns[c]) for i, c in enumerate(left_keys)}
    rk = {f"k{i}": _key_series(right.columns[c]) for i, c in enumerate(right_keys)}
    ldf = pd.DataFrame({**lk, "_l": np.arange(left.num_rows, dtype=np.int64)})
    rdf = pd.DataFrame({**rk, "_r": np.arange(right.num_rows, dtype=np.int64)})
    # Spark equi-join: NULL keys never match
    ldf_nn = ldf.dropna(subset=[f"k{i}" for i in range(len(left_keys))])
    rdf_nn = rdf.dropna(subset=[f"k{i}" for i in range(len(right_keys))])
    on = [f"k{i}" for i in range(len(left_keys))]
    if how == "inner":
        m = ldf_nn.merge(rdf_nn, on=on, how="inner")
        return (_make(m["_l"].to_numpy().astype(np.int32), None, DType.int32()),
                _make(m["_r"].to_numpy().astype(np.int32), None, DType.int32()))
    if how in ("left", "full"):
        m = ldf.merge(rdf_nn, on=on, how="left")
        lmap = m["_l"].to_numpy().astype(np.int32)
        rvals = m["_r"].to_numpy()
        rmap = np.where(np.isnan(rvals), -1, np.nan_to_num(rvals)).astype(np.int32)
        if how == "full" and right_matched is not None:
            hit = rmap[rmap >= 0]
            right_matched[hit] = True
        return (_make(lmap, None, DType.int32()),
                _make(rmap, None, DType.int32()))
    if how in ("semi", "anti"):
        keys_set = rdf_nn[on].drop_duplicates()
        m = ldf.merge(keys_set, on=on, how="left", indicator=True)
        hit = (m["_merge"] == "both").to_numpy()
        sel = hit if how == "semi" else ~hit
        return (_make(m["_l"].to_numpy()[sel].astype(np.int32), None, DType.int32()),
                None)
    raise NotImplementedError(f"cpu join {how}")


def _key_series(col: Column):
    import pandas as pd

    a, av = _vals(col), _valid(col)
    if a.dtype == object:
        return pd.Series(a, dtype=object).where(av, other=None)
    s = pd.Series(a)
    if not av.all():
        s = s.astype("float64").where(av, other=np.nan) if a.dtype.kind in "iuf" \
            else s.where(av, other=None)
    return s


def range_key(col: Column, desc: bool, nulls_last: bool) -> Column:
    """Monotone int64 sort-position proxy (see gpu_backend.range_key)."""
    a, av = _vals(col), _valid(col)
    n = len(a)
    if col.dtype.id is TypeId.STRING:
        key = np.zeros(n, dtype=np.int64)
        for i, v in enumerate(a):
            b = (v or "").encode("utf-8")[:8]
            u = int.from_bytes(b.ljust(8, b"\0"), "big")
            key[i] = (u ^ (1 << 63)) - (1 << 63)  # bias to signed order
    elif col.dtype.id is TypeId.DECIMAL128:
        key = np.array([int(x) >> 64 if x is not None else 0 for x in a],
                       dtype=np.int64)
    elif col.dtype.is_floating:
        f = a.astype(np.float64)
        bits = f.view(np.int64).copy()
        neg = bits < 0
        bits[neg] = np.int64(-0x8000000000000000) - bits[neg] - 1
        bits[np.isnan(f)] = np.int64(0x7FFFFFFFFFFFFFFE)  # NaN greatest
        key = bits
    else:
        key = a.astype(np.int64)
    if desc:
        key = ~key
    extreme = np.int64(0x7FFFFFFFFFFFFFFF) if nulls_last \
        else np.int64(-0x8000000000000000)
    key = np.where(av, key, extreme)
    return _make(key, None, DType.int64())


def sort_order(batch: ColumnBatch, key_idx: List[int],
               descending: List[bool], nulls_last: List[bool]) -> Column:
    n = batch.num_rows
    keys = []
    # np.lexsort: last key is primary -> iterate reversed
    for i, ci in enumerate(reversed(key_idx)):
        ri = len(key_idx) - 1 - i
        c = batch.columns[ci]
        a, av = _vals(c), _valid(c)
        desc = descending[ri]
        nl = nulls_last[ri]
        if a.dtype == object and c.dtype.id is TypeId.DECIMAL128:
            ints = [int(x) if x is not None else 0 for x in a]
            if desc:
                ints = [-v - 1 for v in ints]
            hi = np.array([v >> 64 for v in ints], dtype=np.int64)
            lo = np.array([v & ((1 << 64) - 1) for v in ints],
                          dtype=np.uint64).view(np.int64)
            # unsigned-order lo as secondary: bias to signed
            lo = (lo ^ np.int64(-0x8000000000000000))
            null_rank = np.where(av, 0, 1 if nl else -1)
            keys.append(lo)
            keys.append(hi)
            keys.append(null_rank)
            continue
        if a.dtype == object:
            uniq, codes = np.unique(
                np.array([x if x is not None else "" for x in a]), return_inverse=True)
            a = codes.ravel().astype(np.int64)
        else:
            a = a.copy()
        if a.dtype.kind == "f":
            key = a.astype(np.float64).copy()
            key[np.isnan(key)] = np.inf  # NaN greatest (then refined below)
            nan_rank = np.isnan(a.astype(np.float64)).astype(np.int64)
            if desc:
                key = -key
                nan_rank = -nan_rank
            key = np.where(av, key, 0.0)
            nan_rank = np.where(av, nan_rank, 0)
        else:
            key = a.astype(np.int64)
            nan_rank = np.zeros(len(a), dtype=np.int64)
            if desc:
                key = -key
            key = np.where(av, key, 0)
        null_rank = np.where(av, 0, 1 if nl else -1)
        keys.append(key)
        keys.append(nan_rank)
        keys.append(null_rank)
    order = np.lexsort(tuple(keys)) if keys else np.arange(n)
    return _make(order.astype(np.int32), None, DType.int32())


def str_pad(col: Column, width: int, fill: str, left: bool) -> Column:
    """lpad/rpad with Spark semantics: cycle the fill string, truncate to
    width codepoints; empty fill with short input -> truncate."""
    a, av = _vals(col), _valid(col)
    out = []
    for v, ok in zip(a, av):
        if not ok:
            out.append(None)
            continue
        if len(v) >= width or not fill:
            out.append(v[:width])
        else:
            pad = (fill * ((width - len(v)) // len(fill) + 1))[: width - len(v)]
            out.append(pad + v if left else v + pad)
    return Column.from_pylist(out, DType.string())


def str_locate(col: Column, substr: str, pos: int = 1) -> Column:
    a, av = _vals(col), _valid(col)
    out = []
    for v, ok in zip(a, av):
        if not ok:
            out.append(None)
        elif substr == "":
            out.append(pos if pos <= len(v) + 1 else 0)
        else:
            out.append(v.find(substr, max(pos - 1, 0)) + 1)
    return Column.from_pylist(out, DType.int32())


def tz_convert(col: Column, zone: str, to_utc: bool) -> Column:
    from ..tools import tzdb

    us = _vals(col).astype(np.int64)
    av = _valid(col)
    sec = np.floor_divide(us, 1_000_000)
    if not to_utc:
        out = us + tzdb.offset_at(zone, sec).astype(np.int64) * 1_000_000
    else:
        o0 = tzdb.offset_at(zone, sec).astype(np.int64)
        o1 = tzdb.offset_at(zone, sec - o0).astype(np.int64)
        out = us - o1 * 1_000_000
    return _make(out, av if not av.all() else None, DType.timestamp())


def _civil_from_days_np(days):
    z = days.astype(np.int64) + 719468
    era = np.floor_divide(z, 146097)
    doe = z - era * 146097
    yoe = (doe - doe // 1460 + doe // 36524 - doe // 146096) // 365
    y = yoe + era * 400
    doy = doe - (365 * yoe + yoe // 4 - yoe // 100)
    mp = (5 * doy + 2) // 153
    d = doy - (153 * mp + 2) // 5 + 1
    m = np.where(mp < 10, mp + 3, mp - 9)
    return y + (m <= 2), m, d


def date_format(col: Column, tokens, width: int) -> Column:
    us = _vals(col).astype(np.int64)
    av = _valid(col)
    sec = np.floor_divide(us, 1_000_000)
    days = np.floor_divide(sec, 86_400)
    tod = sec - days * 86_400
    y, m, d = _civil_from_days_np(days)
    hh, mi, ss = tod // 3600, tod // 60 % 60, tod % 60
    out = []
    from ..expr.datetime import (DT_DD, DT_HH, DT_LIT, DT_MI, DT_MM,
                                 DT_SS, DT_YYYY)

    for i in range(len(us)):
        if not av[i]:
            out.append(None)
            continue
        parts = []
        for kind, arg in tokens:
            if kind == DT_LIT:
                parts.append(chr(arg))
            elif kind == DT_YYYY:
                parts.append("%04d" % max(y[i], 0))
            elif kind == DT_MM:
                parts.append("%02d" % m[i])
            elif kind == DT_DD:
                parts.append("%02d" % d[i])
            elif kind == DT_HH:
                parts.append("%02d" % hh[i])
            elif kind == DT_MI:
                parts.append("%02d" % mi[i])
            elif kind == DT_SS:
                parts.append("%02d" % ss[i])
        out.append("".join(parts))
    return Column.from_pylist(out, DType.string())


def ts_parse(col: Column, tokens, width: int) -> Column:
    from datetime import datetime

    from ..expr.datetime import (DT_DD, DT_HH, DT_LIT, DT_MI, DT_MM,
                                 DT_SS, DT_YYYY)

    a, av = _vals(col), _valid(col)
    out = np.zeros(len(a), dtype=np.int64)
    valid = av.copy()
    for i, s in enumerate(a):
        if not av[i]:
            continue
        if s is None or len(s) != width:
            valid[i] = False
            continue
        p = 0
        f = {"y": 1970, "M": 1, "d": 1, "H": 0, "m": 0, "s": 0}
        ok = True
        for kind, arg in tokens:
            if kind == DT_LIT:
                if s[p] != chr(arg):
                    ok = False
                    break
                p += 1
            else:
                w = 4 if kind == DT_YYYY else 2
                seg = s[p:p + w]
                if not seg.isdigit():
                    ok = False
                    break
                v = int(seg)
                key = {DT_YYYY: "y", DT_MM: "M", DT_DD: "d", DT_HH: "H",
                       DT_MI: "m", DT_SS: "s"}[kind]
                f[key] = v
                p += w
        if ok:
            try:
                dt = datetime(f["y"], f["M"], f["d"], f["H"], f["m"],
                              f["s"])
                epoch = datetime(1970, 1, 1)
                out[i] = int((dt - epoch).total_seconds()) * 1_000_000
            except ValueError:
                ok = False
        valid[i] = ok
    return _make(out, valid if not valid.all() else None,
                 DType.timestamp())


def and_parent_validity(kid: Column, parent: Column) -> Column:
    """Null out child rows where the parent (struct) row is null."""
    pv = parent.valid_array()
    kv = kid.valid_array()
    both = pv & kv
    if both.all():
        return kid
    vals = kid.to_pylist()
    out = [v if ok else None for v, ok in zip(vals, both)]
    return Column.from_pylist(out, kid.dtype)


# ---- xxHash64 (canonical; CPU mirror of hash.hip) ------------------------

_XXP1 = 0x9E3779B185EBCA87
_XXP2 = 0xC2B2AE3D27D4EB4F
_XXP3 = 0x165667B19E3779F9
_XXP4 = 0x85EBCA77C2B2AE63
_XXP5 = 0x27D4EB2F165667C5
_M64 = (1 << 64) - 1


def _rotl64(x, r):
    return ((x << r) | (x >> (64 - r))) & _M64


def _xx_round(acc, inp):
    acc = (acc + inp * _XXP2) & _M64
    return (_rotl64(acc, 31) * _XXP1) & _M64


def _xx_avalanche(h):
    h ^= h >> 33
    h = (h * _XXP2) & _M64
    h ^= h >> 29
    h = (h * _XXP3) & _M64
    h ^= h >> 32
    return h


def xxh64_long(v: int, seed: int) -> int:
    h = (seed + _XXP5 + 8) & _M64
    h ^= _xx_round(0, v & _M64)
    h = (_rotl64(h, 27) * _XXP1 + _XXP4) & _M64
    return _xx_avalanche(h)


def xxh64_bytes(data: bytes, seed: int) -> int:
    import struct as st

    n = len(data)
    p = 0
    if n >= 32:
        v1 = (seed + _XXP1 + _XXP2) & _M64
        v2 = (seed + _XXP2) & _M64
        v3 = seed & _M64
        v4 = (seed - _XXP1) & _M64
        while p + 32 <= n:
            v1 = _xx_round(v1, st.unpack_from("<Q", data, p)[0])
            v2 = _xx_round(v2, st.unpack_from("<Q", data, p + 8)[0])
            v3 = _xx_round(v3, st.unpack_from("<Q", data, p + 16)[0])
            v4 = _xx_round(v4, st.unpack_from("<Q", data, p + 24)[0])
            p += 32
        h = (_rotl64(v1, 1) + _rotl64(v2, 7) + _rotl64(v3, 12)
             + _rotl64(v4, 18)) & _M64
        for v in (v1, v2, v3, v4):
            h = ((h ^ _xx_round(0, v)) * _XXP1 + _XXP4) & _M64
    else:
        h = (seed + _XXP5) & _M64
    h = (h + n) & _M64
    while p + 8 <= n:
        h ^= _xx_round(0, st.unpack_from("<Q", data, p)[0])
        h = (_rotl64(h, 27) * _XXP1 + _XXP4) & _M64
        p += 8
    if p + 4 <= n:
        h ^= (st.unpack_from("<I", data, p)[0] * _XXP1) & _M64
        h = (_rotl64(h, 23) * _XXP2 + _XXP3) & _M64
        p += 4
    while p < n:
        h ^= (data[p] * _XXP5) & _M64
        h = (_rotl64(h, 11) * _XXP1) & _M64
        p += 1
    return _xx_avalanche(h)


def _xxhash64_rows(cols, seed: int = 42) -> np.ndarray:
    import math
    import struct as st

    n = cols[0].size
    out = np.full(n, seed, dtype=np.uint64)
    for c in cols:
        av = _valid(c)
        if c.dtype.id is TypeId.STRING:
            vals = c.to_pylist()
            for i in range(n):
                if av[i]:
                    out[i] = xxh64_bytes(vals[i].encode(), int(out[i]))
        else:
            a = _vals(c)
            for i in range(n):
                if not av[i]:
                    continue
                v = a[i]
                if c.dtype.is_floating:
                    d = float(v)
                    if math.isnan(d):
                        d = math.nan
                    if d == 0.0:
                        d = 0.0
                    bits = st.unpack("<Q", st.pack("<d", d))[0]
                else:
                    bits = int(v) & _M64
                out[i] = xxh64_long(bits, int(out[i]))
    return out


def xxhash64(cols, seed: int = 42) -> Column:
    h = _xxhash64_rows(cols, seed).astype(np.int64)
    return _make(h, None, DType.int64())


def _hll_estimate_host(regs_np) -> np.ndarray:
    """Standard HLL estimate with the small-range linear-counting
    correction (no empirical bias tables — estimates differ slightly from
    Spark's HLL++ constants; documented divergence, like the reference's
    incompat-gated approximations)."""
    g, m = regs_np.shape
    alpha = 0.7213 / (1 + 1.079 / m)
    E = alpha * m * m / np.power(2.0, -regs_np.astype(np.float64)).sum(1)
    zeros = (regs_np == 0).sum(1)
    lc = m * np.log(np.maximum(m / np.maximum(zeros, 1), 1.0))
    small = (E <= 2.5 * m) & (zeros > 0)
    return np.rint(np.where(small, lc, E)).astype(np.int64)


def hll_groups(vc: Column, codes: np.ndarray, ngroups: int,
               p: int) -> np.ndarray:
    """HLL register build + estimate (CPU mirror of k_gb_hll)."""
    m = 1 << p
    regs = np.zeros((ngroups, m), dtype=np.uint8)
    av = _valid(vc)
    hashes = _xxhash64_rows([vc])
    for i in range(vc.size):
        if not av[i]:
            continue
        h = int(hashes[i])
        idx = h >> (64 - p)
        w = (h << p) & _M64
        rho = (64 - p + 1) if w == 0 else (64 - w.bit_length() + 1)
        g = codes[i]
        if regs[g, idx] < rho:
            regs[g, idx] = rho
    return _hll_estimate_host(regs)

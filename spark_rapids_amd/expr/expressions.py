"""Expression tree + evaluation.

The engine analogue of the reference's GpuExpression.columnarEval path
(reference: sql-plugin/src/main/scala/com/nvidia/spark/rapids/GpuExpressions.scala:139-334):
expressions evaluate batch-at-a-time to whole Columns; the same tree evaluates
on the CPU backend or on the GPU backend depending on where the batch lives.
"""
from __future__ import annotations

from typing import Optional, Sequence

from ..column import Column, ColumnBatch, Schema
from ..types import (BOOL, DType, FLOAT64, INT32, INT64, STRING, TypeId,
                     _adjust_decimal, as_decimal, decimal_arith_type, promote)
from .. import ops


class Expression:
    def dtype(self, schema: Schema) -> DType:
        raise NotImplementedError

    def nullable(self, schema: Schema) -> bool:
        return True

    @property
    def children(self) -> Sequence["Expression"]:
        return ()

    def eval(self, batch: ColumnBatch, schema: Schema) -> Column:
        raise NotImplementedError

    def output_name(self) -> str:
        return str(self)

    # ---- operator DSL --------------------------------------------------
    def _bin(self, op, other, swap=False):
        other = _as_expr(other)
        l, r = (other, self) if swap else (self, other)
        return BinaryExpr(op, l, r)

    def __add__(self, o):
        return self._bin("add", o)

    def __radd__(self, o):
        return self._bin("add", o, swap=True)

    def __sub__(self, o):
        return self._bin("sub", o)

    def __rsub__(self, o):
        return self._bin("sub", o, swap=True)

    def __mul__(self, o):
        return self._bin("mul", o)

    def __rmul__(self, o):
        return self._bin("mul", o, swap=True)

    def __truediv__(self, o):
        return self._bin("div", o)

    def __mod__(self, o):
        return self._bin("mod", o)

    def __lt__(self, o):
        return self._bin("lt", o)

    def __le__(self, o):
        return self._bin("le", o)

    def __gt__(self, o):
        return self._bin("gt", o)

    def __ge__(self, o):
        return self._bin("ge", o)

    def __eq__(self, o):  # noqa: PLE0302 - DSL
        return self._bin("eq", o)

    def __ne__(self, o):
        return self._bin("ne", o)

    def __and__(self, o):
        return self._bin("and", o)

    def __or__(self, o):
        return self._bin("or", o)

    def __invert__(self):
        return UnaryExpr("not", self)

    def __neg__(self):
        return UnaryExpr("neg", self)

    def __hash__(self):
        return id(self)

    def alias(self, name: str) -> "Alias":
        return Alias(self, name)

    def cast(self, to: DType) -> "CastExpr":
        return CastExpr(self, to)

    def is_null(self) -> "IsNull":
        return IsNull(self)

    def is_not_null(self) -> "Expression":
        return UnaryExpr("not", IsNull(self))

    # string DSL
    def contains(self, pattern: str) -> "StringPredicate":
        return StringPredicate("contains", self, pattern)

    def startswith(self, pattern: str) -> "StringPredicate":
        return StringPredicate("starts_with", self, pattern)

    def endswith(self, pattern: str) -> "StringPredicate":
        return StringPredicate("ends_with", self, pattern)

    def like(self, pattern: str) -> "StringPredicate":
        return StringPredicate("like", self, pattern)

    def rlike(self, pattern: str) -> "StringPredicate":
        """Java-regex find() semantics (RLike)."""
        return StringPredicate("rlike", self, pattern)

    def trim(self) -> "UnaryExpr":
        return UnaryExpr("trim", self)

    def ltrim(self) -> "UnaryExpr":
        return UnaryExpr("ltrim", self)

    def rtrim(self) -> "UnaryExpr":
        return UnaryExpr("rtrim", self)

    def concat(self, other) -> "BinaryExpr":
        """Spark concat(): NULL if either side is NULL."""
        return BinaryExpr("concat", self, _as_expr(other))

    def replace(self, search: str, replacement: str) -> "RegexpReplace":
        """Literal substring replace (StringReplace analogue), lowered to
        the regex engine with an escaped pattern."""
        esc = "".join("\\" + c if c in ".\\+*?()[]{}|^$" else c
                      for c in search)
        resc = replacement.replace("\\", "\\\\").replace("$", "\\$")
        return RegexpReplace(self, esc, resc)

    def split(self, delimiter: str) -> "StrSplit":
        """Split into a LIST of strings (java limit-0 semantics: trailing
        empty parts dropped)."""
        return StrSplit(self, delimiter)

    def element_at(self, index) -> "ElementAt":
        """1-based element of a LIST value (NULL beyond the length), or
        the value for a key of a MAP value (Spark element_at)."""
        return ElementAt(self, index)

    def size(self) -> "ArraySize":
        """Entry count of a LIST or MAP value (null -> null)."""
        return ArraySize(self)

    def array_contains(self, value) -> "ArrayContains":
        """True when the LIST value contains `value` (Spark
        array_contains; null list -> null)."""
        return ArrayContains(self, value)

    def regexp_extract(self, pattern: str, group: int = 1) -> "RegexpExtract":
        return RegexpExtract(self, pattern, group)

    def lpad(self, width: int, fill: str = " ") -> "PadExpr":
        return PadExpr(self, width, fill, left=True)

    def rpad(self, width: int, fill: str = " ") -> "PadExpr":
        return PadExpr(self, width, fill, left=False)

    def locate(self, substr: str, pos: int = 1) -> "LocateExpr":
        """1-based position of substr (0 when absent) — Spark locate/instr."""
        return LocateExpr(self, substr, pos)

    def get_json_object(self, path: str) -> "GetJsonObject":
        return GetJsonObject(self, path)

    def regexp_extract_all(self, pattern: str,
                           group: int = 1) -> "RegexpExtractAll":
        return RegexpExtractAll(self, pattern, group)

    def regexp_replace(self, pattern: str, replacement: str) -> "RegexpReplace":
        return RegexpReplace(self, pattern, replacement)

    def substr(self, pos: int, length: int = -1) -> "Substring":
        return Substring(self, pos, length)

    def length(self) -> "UnaryExpr":
        return UnaryExpr("length", self)

    def initcap(self) -> "UnaryExpr":
        """Capitalize each space-separated word (ASCII on GPU)."""
        return UnaryExpr("initcap", self)

    def reverse(self) -> "UnaryExpr":
        """Reverse codepoint order."""
        return UnaryExpr("reverse", self)

    def upper(self) -> "UnaryExpr":
        return UnaryExpr("upper", self)

    def lower(self) -> "UnaryExpr":
        return UnaryExpr("lower", self)


def _as_expr(v) -> Expression:
    if isinstance(v, Expression):
        return v
    return Literal(v)


class ColumnRef(Expression):
    def __init__(self, name: str):
        self.name = name

    def dtype(self, schema: Schema) -> DType:
        return schema.field(self.name).dtype

    def nullable(self, schema: Schema) -> bool:
        return schema.field(self.name).nullable

    def eval(self, batch: ColumnBatch, schema: Schema) -> Column:
        return batch.columns[schema.index(self.name)]

    def output_name(self) -> str:
        return self.name

    def __str__(self):
        return self.name


def _infer_literal_dtype(v) -> DType:
    if isinstance(v, bool):
        return BOOL
    if isinstance(v, int):
        return INT32 if -(2 ** 31) <= v < 2 ** 31 else INT64
    if isinstance(v, float):
        return FLOAT64
    if isinstance(v, str):
        return STRING
    if v is None:
        return DType(TypeId.NULL)
    raise TypeError(f"unsupported literal {v!r}")


class Literal(Expression):
    def __init__(self, value, dtype: Optional[DType] = None):
        self.value = value
        self._dtype = dtype or _infer_literal_dtype(value)

    def dtype(self, schema: Schema) -> DType:
        return self._dtype

    def nullable(self, schema: Schema) -> bool:
        return self.value is None

    def eval(self, batch: ColumnBatch, schema: Schema) -> Column:
        if self.value is None:
            return Column.nulls(self._dtype if self._dtype.id is not TypeId.NULL
                                else INT32, batch.num_rows, batch.device)
        return Column.full(self.value, self._dtype, batch.num_rows,
                           batch.device)

    def __str__(self):
        return repr(self.value)


def _decimal_exact(lt: DType, rt: DType) -> bool:
    """True when mul/div should use exact decimal arithmetic (at least one
    decimal operand, the other decimal or integral)."""
    return (lt.is_decimal or rt.is_decimal) \
        and (lt.is_decimal or lt.is_integral) \
        and (rt.is_decimal or rt.is_integral)


# ops whose result is boolean
_BOOL_OPS = {"eq", "ne", "lt", "le", "gt", "ge", "and", "or", "eq_null_safe"}
# ops that force double output (Spark `/`)
_DOUBLE_OPS = {"div", "pow"}


class BinaryExpr(Expression):
    def __init__(self, op: str, left: Expression, right: Expression):
        self.op = op
        self.left = left
        self.right = right

    @property
    def children(self):
        return (self.left, self.right)

    def _common(self, lt: DType, rt: DType) -> DType:
        if (lt.is_decimal or rt.is_decimal) and self.op in (
                "int_div", "mod", "pmod", "pow"):
            raise NotImplementedError(
                f"decimal {self.op} needs scale arithmetic (not implemented "
                "yet): cast to double first, e.g. col.cast(FLOAT64)")
        if (lt.is_decimal or rt.is_decimal) and self.op in ("mul", "div"):
            # mixed decimal/floating (the exact-decimal path handles
            # decimal/integral): Spark casts the decimal side to double
            return FLOAT64
        if lt.id is TypeId.NULL:
            return rt
        if rt.id is TypeId.NULL:
            return lt
        if lt == rt:
            return lt
        if self.op in _DOUBLE_OPS:
            return FLOAT64
        return promote(lt, rt)

    def _in_dtype(self, schema) -> DType:
        return self._common(self.left.dtype(schema), self.right.dtype(schema))

    def dtype(self, schema: Schema) -> DType:
        # NOTE: children's dtype() is called exactly once here — recursing
        # more than once makes deep expression chains exponential
        if self.op in _BOOL_OPS:
            return BOOL
        lt, rt = self.left.dtype(schema), self.right.dtype(schema)
        if self.op in ("mul", "div") and _decimal_exact(lt, rt):
            # Spark DecimalPrecision rules: scale s1+s2 (mul) /
            # max(6, s1+p2+1) (div), precision-loss adjustment at 38
            return decimal_arith_type(self.op, lt, rt)
        if self.op in _DOUBLE_OPS:
            return FLOAT64
        if self.op == "sub" and lt.is_timelike and rt.is_timelike:
            return INT32  # datediff domain
        it = self._common(lt, rt)
        if it.is_decimal and self.op in ("add", "sub"):
            # Spark add/sub result: max(p1-s1,p2-s2)+max(s1,s2)+1 at
            # scale max(s1,s2); promote() already produced the first part.
            return _adjust_decimal(it.precision + 1, it.scale)
        return it

    # ops where `scalar OP col` can run through the col-scalar kernel
    _COMMUTATIVE = {"add", "mul", "eq", "ne", "eq_null_safe", "and", "or",
                    "bitand", "bitor", "bitxor", "min", "max"}
    _SWAP_CMP = {"lt": "gt", "gt": "lt", "le": "ge", "ge": "le"}

    def eval(self, batch: ColumnBatch, schema: Schema) -> Column:
        if self.op in ("mul", "div"):
            lt, rt = self.left.dtype(schema), self.right.dtype(schema)
            if _decimal_exact(lt, rt):
                # operands keep their own scales; the backend computes the
                # exact product/quotient at the Spark result scale
                lcol = ops.cast(self.left.eval(batch, schema), as_decimal(lt))
                rcol = ops.cast(self.right.eval(batch, schema), as_decimal(rt))
                return ops.decimal_mul_div(
                    self.op, lcol, rcol, decimal_arith_type(self.op, lt, rt))
        common = self._in_dtype(schema)
        if self.op in _DOUBLE_OPS and not common.is_decimal:
            common = FLOAT64
        out = self.dtype(schema)
        if out.is_decimal and out.id is TypeId.DECIMAL128 \
                and common.id is TypeId.DECIMAL64:
            # result crosses into decimal128 (e.g. (18,0)+(18,0) -> (19,0)):
            # widen the operands so the 128-bit kernel runs end to end
            common = DType(TypeId.DECIMAL128, common.precision, common.scale)
        # scalar fast path: literal on either side
        if isinstance(self.right, Literal) and self.right.value is not None \
                and not common.is_decimal and self.op != "concat":
            lcol = ops.cast(self.left.eval(batch, schema), common)
            return ops.binary_op_scalar(self.op, lcol, _coerce_py(self.right.value, common), out)
        if isinstance(self.left, Literal) and self.left.value is not None \
                and not common.is_decimal and self.op != "concat":
            swapped = self.op if self.op in self._COMMUTATIVE \
                else self._SWAP_CMP.get(self.op)
            if swapped is not None:
                rcol = ops.cast(self.right.eval(batch, schema), common)
                return ops.binary_op_scalar(
                    swapped, rcol, _coerce_py(self.left.value, common), out)
            if self.op == "sub":
                # c - x == -(x - c): two scalar kernels, no literal column
                rcol = ops.cast(self.right.eval(batch, schema), common)
                d = ops.binary_op_scalar(
                    "sub", rcol, _coerce_py(self.left.value, common), out)
                return ops.unary_op("neg", d, out)
        lcol = ops.cast(self.left.eval(batch, schema), common)
        rcol = ops.cast(self.right.eval(batch, schema), common)
        return ops.binary_op(self.op, lcol, rcol, out)

    def __str__(self):
        return f"({self.left} {self.op} {self.right})"


def _coerce_py(v, dtype: DType):
    if dtype.is_floating:
        return float(v)
    if dtype.is_integral or dtype.is_decimal:
        return int(v)
    return v


_UNARY_OUT = {
    "not": lambda t: BOOL,
    "trim": lambda t: STRING,
    "initcap": lambda t: STRING,
    "reverse": lambda t: STRING,
    "ltrim": lambda t: STRING,
    "rtrim": lambda t: STRING,
    "is_nan": lambda t: BOOL,
    "neg": lambda t: t,
    "abs": lambda t: t,
    "sqrt": lambda t: FLOAT64,
    "exp": lambda t: FLOAT64,
    "log": lambda t: FLOAT64,
    "sin": lambda t: FLOAT64,
    "cos": lambda t: FLOAT64,
    "tan": lambda t: FLOAT64,
    "floor": lambda t: INT64 if not t.is_decimal else t,
    "ceil": lambda t: INT64 if not t.is_decimal else t,
    "length": lambda t: INT32,
    "upper": lambda t: STRING,
    "lower": lambda t: STRING,
    "year": lambda t: INT32,
    "month": lambda t: INT32,
    "day": lambda t: INT32,
}


class UnaryExpr(Expression):
    def __init__(self, op: str, child: Expression):
        self.op = op
        self.child = child

    @property
    def children(self):
        return (self.child,)

    def dtype(self, schema: Schema) -> DType:
        return _UNARY_OUT[self.op](self.child.dtype(schema))

    def eval(self, batch: ColumnBatch, schema: Schema) -> Column:
        c = self.child.eval(batch, schema)
        if self.op in ("sqrt", "exp", "log", "sin", "cos", "tan"):
            c = ops.cast(c, FLOAT64)
        return ops.unary_op(self.op, c, self.dtype(schema))

    def __str__(self):
        return f"{self.op}({self.child})"


class IsNull(Expression):
    def __init__(self, child: Expression):
        self.child = child

    @property
    def children(self):
        return (self.child,)

    def dtype(self, schema: Schema) -> DType:
        return BOOL

    def nullable(self, schema: Schema) -> bool:
        return False

    def eval(self, batch: ColumnBatch, schema: Schema) -> Column:
        return ops.is_null(self.child.eval(batch, schema))

    def __str__(self):
        return f"isnull({self.child})"


class CastExpr(Expression):
    def __init__(self, child: Expression, to: DType):
        self.child = child
        self.to = to

    @property
    def children(self):
        return (self.child,)

    def dtype(self, schema: Schema) -> DType:
        return self.to

    def eval(self, batch: ColumnBatch, schema: Schema) -> Column:
        return ops.cast(self.child.eval(batch, schema), self.to)

    def __str__(self):
        return f"cast({self.child} as {self.to})"


class Alias(Expression):
    def __init__(self, child: Expression, name: str):
        self.child = child
        self.name = name

    @property
    def children(self):
        return (self.child,)

    def dtype(self, schema: Schema) -> DType:
        return self.child.dtype(schema)

    def nullable(self, schema: Schema) -> bool:
        return self.child.nullable(schema)

    def eval(self, batch: ColumnBatch, schema: Schema) -> Column:
        return self.child.eval(batch, schema)

    def output_name(self) -> str:
        return self.name

    def __str__(self):
        return f"{self.child} AS {self.name}"


class CaseWhen(Expression):
    """CASE WHEN cond THEN v ... ELSE e END  (pairs of (cond, value))."""

    def __init__(self, branches, else_expr: Optional[Expression] = None):
        self.branches = [(c, _as_expr(v)) for c, v in branches]
        self.else_expr = _as_expr(else_expr) if else_expr is not None else None

    @property
    def children(self):
        out = []
        for c, v in self.branches:
            out.extend([c, v])
        if self.else_expr is not None:
            out.append(self.else_expr)
        return tuple(out)

    def dtype(self, schema: Schema) -> DType:
        t = self.branches[0][1].dtype(schema)
        for _, v in self.branches[1:]:
            vt = v.dtype(schema)
            if t.id is TypeId.NULL:
                t = vt
            elif vt.id is not TypeId.NULL:
                t = promote(t, vt)
        if self.else_expr is not None:
            et = self.else_expr.dtype(schema)
            if et.id is not TypeId.NULL:
                t = promote(t, et) if t.id is not TypeId.NULL else et
        return t

    def eval(self, batch: ColumnBatch, schema: Schema) -> Column:
        out_t = self.dtype(schema)
        # evaluate as nested if_else from the last branch backwards
        if self.else_expr is not None:
            acc = ops.cast(self.else_expr.eval(batch, schema), out_t)
        else:
            acc = Column.nulls(out_t, batch.num_rows, batch.device)
        backend = ops.backend_for(*batch.columns) if batch.columns else None
        for cond, val in reversed(self.branches):
            c = cond.eval(batch, schema)
            v = ops.cast(val.eval(batch, schema), out_t)
            acc = ops.backend_for(c, v, acc).if_else(c, v, acc)
        return acc

    def __str__(self):
        parts = " ".join(f"WHEN {c} THEN {v}" for c, v in self.branches)
        e = f" ELSE {self.else_expr}" if self.else_expr is not None else ""
        return f"CASE {parts}{e} END"


class StringPredicate(Expression):
    """contains / starts_with / ends_with / LIKE against a literal pattern
    (reference analogue: GpuContains/GpuStartsWith/GpuLike)."""

    def __init__(self, op: str, child: Expression, pattern: str):
        self.op = op
        self.child = child
        self.pattern = pattern

    @property
    def children(self):
        return (self.child,)

    def dtype(self, schema: Schema) -> DType:
        return BOOL

    def eval(self, batch: ColumnBatch, schema: Schema) -> Column:
        c = self.child.eval(batch, schema)
        return ops.backend_for(c).str_predicate(self.op, c, self.pattern)

    def __str__(self):
        return f"{self.op}({self.child}, {self.pattern!r})"


class Substring(Expression):
    """Spark substring(str, pos, len): 1-based pos in codepoints, negative
    pos counts from the end; len < 0 means to-the-end."""

    def __init__(self, child: Expression, pos: int, length: int = -1):
        self.child = child
        self.pos = pos
        self.length = length

    @property
    def children(self):
        return (self.child,)

    def dtype(self, schema: Schema) -> DType:
        return STRING

    def eval(self, batch: ColumnBatch, schema: Schema) -> Column:
        c = self.child.eval(batch, schema)
        return ops.backend_for(c).substring(c, self.pos, self.length)

    def __str__(self):
        return f"substring({self.child}, {self.pos}, {self.length})"


# ---------------------------------------------------------------------------
# public DSL
# ---------------------------------------------------------------------------

class StrSplit(Expression):
    """split(str, delim) -> array<string>. GPU kernel handles literal
    delimiters (k_str_split_* in strings.hip); regex delimiters run on
    the CPU via re.split (GpuStringSplit analogue)."""

    def __init__(self, child: Expression, delimiter: str):
        self.child = child
        self.delimiter = delimiter

    @property
    def children(self):
        return (self.child,)

    def dtype(self, schema: Schema) -> DType:
        return DType.list_(STRING)

    def eval(self, batch: ColumnBatch, schema: Schema) -> Column:
        return ops.str_split(self.child.eval(batch, schema), self.delimiter)

    def __str__(self):
        return f"split({self.child}, {self.delimiter!r})"


class ElementAt(Expression):
    """element_at(array, k) with 1-based k (negative k counts from the
    end); NULL when |k| exceeds the length. element_at(map, key) returns
    the value for key or NULL (GpuElementAt analogue)."""

    def __init__(self, child: Expression, index):
        self.child = child
        self.index = index

    @property
    def children(self):
        return (self.child,)

    def dtype(self, schema: Schema) -> DType:
        cdt = self.child.dtype(schema)
        return cdt.children[1] if cdt.id is TypeId.MAP else cdt.children[0]

    def eval(self, batch: ColumnBatch, schema: Schema) -> Column:
        c = self.child.eval(batch, schema)
        if c.dtype.id is TypeId.MAP:
            return ops.map_get(c, self.index)
        assert isinstance(self.index, int) and self.index != 0, \
            "element_at(array, k) is 1-based (Spark semantics)"
        return ops.element_at(c, self.index)

    def __str__(self):
        return f"element_at({self.child}, {self.index})"


class ArrayContains(Expression):
    """array_contains(array, value) -> bool (GpuArrayContains)."""

    def __init__(self, child: Expression, value):
        self.child = child
        self.value = value

    @property
    def children(self):
        return (self.child,)

    def dtype(self, schema: Schema) -> DType:
        return BOOL

    def eval(self, batch: ColumnBatch, schema: Schema) -> Column:
        return ops.array_contains(self.child.eval(batch, schema),
                                  self.value)

    def __str__(self):
        return f"array_contains({self.child}, {self.value!r})"


class ArraySize(Expression):
    """size(array) -> int32; null array -> null (GpuSize analogue)."""

    def __init__(self, child: Expression):
        self.child = child

    @property
    def children(self):
        return (self.child,)

    def dtype(self, schema: Schema) -> DType:
        return INT32

    def eval(self, batch: ColumnBatch, schema: Schema) -> Column:
        return ops.array_size(self.child.eval(batch, schema))

    def __str__(self):
        return f"size({self.child})"


class HostStringFn(Expression):
    """A scalar string function evaluated on the host (rows through a
    python callable); the overrides pass tags it off the GPU, so inside a
    GPU project it executes via CpuBridge. Used for the long tail of
    string builtins (repeat, substring_index, translate, ...) until they
    earn kernels."""

    def __init__(self, name: str, child: Expression, fn, out_dtype=None):
        self.name = name
        self.child = child
        self.fn = fn
        self._out = out_dtype or STRING

    @property
    def children(self):
        return (self.child,)

    def dtype(self, schema: Schema) -> DType:
        return self._out

    def eval(self, batch: ColumnBatch, schema: Schema) -> Column:
        c = self.child.eval(batch, schema)
        host = c if not c.is_cuda else c.cpu()
        out = [None if v is None else self.fn(v) for v in host.to_pylist()]
        res = Column.from_pylist(out, self._out)
        return res.cuda() if c.is_cuda else res

    def __str__(self):
        return f"{self.name}({self.child})"


def repeat_str(e, n: int) -> HostStringFn:
    return HostStringFn("repeat", _as_expr(e), lambda v: v * n)


def substring_index(e, delim: str, count: int) -> HostStringFn:
    """Spark substring_index: text before the count-th delimiter
    (negative count: after the count-th from the right)."""

    def fn(v, d=delim, c=count):
        parts = v.split(d)
        if c > 0:
            return d.join(parts[:c])
        if c < 0:
            return d.join(parts[c:])
        return ""

    return HostStringFn("substring_index", _as_expr(e), fn)


def translate(e, src: str, repl: str) -> HostStringFn:
    table = {ord(a): (repl[i] if i < len(repl) else None)
             for i, a in enumerate(src)}
    return HostStringFn("translate", _as_expr(e),
                        lambda v: v.translate(table))


def ascii_(e) -> HostStringFn:
    return HostStringFn("ascii", _as_expr(e),
                        lambda v: ord(v[0]) if v else 0, INT32)


class PadExpr(Expression):
    """lpad/rpad to a fixed width (GpuStringLPad/RPad analogue; CPU
    evaluation this round — tagged off the GPU by the overrides pass)."""

    def __init__(self, child: Expression, width: int, fill: str, left: bool):
        self.child = child
        self.width = width
        # empty fill is meaningful (Spark UTF8String.lpad: empty pad just
        # truncates to width); only None defaults to a space
        self.fill = " " if fill is None else fill
        self.left = left

    @property
    def children(self):
        return (self.child,)

    def dtype(self, schema: Schema) -> DType:
        return STRING

    def eval(self, batch: ColumnBatch, schema: Schema) -> Column:
        c = self.child.eval(batch, schema)
        return ops.str_pad(c, self.width, self.fill, self.left)

    def __str__(self):
        side = "lpad" if self.left else "rpad"
        return f"{side}({self.child}, {self.width}, {self.fill!r})"


class LocateExpr(Expression):
    """locate(substr, str, pos): 1-based find, 0 when absent (GpuLocate)."""

    def __init__(self, child: Expression, substr: str, pos: int = 1):
        self.child = child
        self.substr = substr
        self.pos = pos

    @property
    def children(self):
        return (self.child,)

    def dtype(self, schema: Schema) -> DType:
        return INT32

    def eval(self, batch: ColumnBatch, schema: Schema) -> Column:
        c = self.child.eval(batch, schema)
        return ops.str_locate(c, self.substr, self.pos)

    def __str__(self):
        return f"locate({self.substr!r}, {self.child}, {self.pos})"


class ConcatWs(Expression):
    """concat_ws(sep, c1, c2, ...): join non-null values with sep; rows
    with every value null give "" (never NULL — Spark concat_ws)."""

    def __init__(self, sep: str, *exprs: Expression):
        self.sep = sep
        self.exprs = [_as_expr(e) for e in exprs]

    @property
    def children(self):
        return tuple(self.exprs)

    def dtype(self, schema: Schema) -> DType:
        return STRING

    def nullable(self, schema: Schema) -> bool:
        return False

    def eval(self, batch: ColumnBatch, schema: Schema) -> Column:
        cols = [e.eval(batch, schema) for e in self.exprs]
        return ops.concat_ws(self.sep, cols)

    def __str__(self):
        args = ", ".join(str(e) for e in self.exprs)
        return f"concat_ws({self.sep!r}, {args})"


def concat_ws(sep: str, *exprs) -> ConcatWs:
    return ConcatWs(sep, *exprs)


class GetJsonObject(Expression):
    """get_json_object(col, '$.key'): top-level scalar extraction runs on
    the GPU via the JSON field kernel (csv.hip k_json_field); nested paths
    fall back to the CPU json parser (GpuGetJsonObject analogue)."""

    def __init__(self, child: Expression, path: str):
        self.child = child
        self.path = path

    @property
    def children(self):
        return (self.child,)

    def dtype(self, schema: Schema) -> DType:
        return STRING

    def eval(self, batch: ColumnBatch, schema: Schema) -> Column:
        return ops.get_json_object(self.child.eval(batch, schema),
                                   self.path)

    def __str__(self):
        return f"get_json_object({self.child}, {self.path!r})"


class CpuBridge(Expression):
    """Evaluate a CPU-only expression inside a GPU projection via a host
    round-trip, so one unsupported expression no longer demotes the whole
    project (reference analogue: GpuCpuBridgeExpression /
    GpuCpuBridgeOptimizer)."""

    def __init__(self, child: Expression):
        self.child = child

    @property
    def children(self):
        return (self.child,)

    def dtype(self, schema: Schema) -> DType:
        return self.child.dtype(schema)

    def nullable(self, schema: Schema) -> bool:
        return self.child.nullable(schema)

    def output_name(self) -> str:
        return self.child.output_name()

    def eval(self, batch: ColumnBatch, schema: Schema) -> Column:
        if batch.columns and batch.columns[0].is_cuda:
            out = self.child.eval(batch.cpu(), schema)
            return out.cuda()
        return self.child.eval(batch, schema)

    def __str__(self):
        return f"cpu_bridge({self.child})"


class RegexpExtract(Expression):
    """regexp_extract(str, pattern, idx): the capture group's text for the
    first match; "" when no match or non-participating group (Spark
    semantics). GPU: capture-group backtracking VM (regex.hip k_regex_extract);
    reference analogue: GpuRegExpExtract over the transpiled cudf regex."""

    def __init__(self, child: Expression, pattern: str, group: int = 1):
        self.child = child
        self.pattern = pattern
        self.group = group

    @property
    def children(self):
        return (self.child,)

    def dtype(self, schema: Schema) -> DType:
        return STRING

    def eval(self, batch: ColumnBatch, schema: Schema) -> Column:
        return ops.regexp_extract(self.child.eval(batch, schema),
                                  self.pattern, self.group)

    def __str__(self):
        return f"regexp_extract({self.child}, {self.pattern!r}, {self.group})"


class RegexpExtractAll(Expression):
    """regexp_extract_all: every match's group text as array<string>
    (GpuRegExpExtractAll analogue; shares the capture-group VM)."""

    def __init__(self, child: Expression, pattern: str, group: int = 1):
        self.child = child
        self.pattern = pattern
        self.group = group

    @property
    def children(self):
        return (self.child,)

    def dtype(self, schema: Schema) -> DType:
        return DType.list_(STRING)

    def eval(self, batch: ColumnBatch, schema: Schema) -> Column:
        return ops.regexp_extract_all(self.child.eval(batch, schema),
                                      self.pattern, self.group)

    def __str__(self):
        return (f"regexp_extract_all({self.child}, {self.pattern!r}, "
                f"{self.group})")


class RegexpReplace(Expression):
    """regexp_replace(str, pattern, replacement) with $g group references
    (GpuRegExpReplace analogue; java Matcher.appendReplacement semantics
    incl. empty-match advancement)."""

    def __init__(self, child: Expression, pattern: str, replacement: str):
        self.child = child
        self.pattern = pattern
        self.replacement = replacement

    @property
    def children(self):
        return (self.child,)

    def dtype(self, schema: Schema) -> DType:
        return STRING

    def eval(self, batch: ColumnBatch, schema: Schema) -> Column:
        return ops.regexp_replace(self.child.eval(batch, schema),
                                  self.pattern, self.replacement)

    def __str__(self):
        return (f"regexp_replace({self.child}, {self.pattern!r}, "
                f"{self.replacement!r})")


class Coalesce(Expression):
    def __init__(self, *exprs):
        self.exprs = [_as_expr(e) for e in exprs]

    @property
    def children(self):
        return tuple(self.exprs)

    def dtype(self, schema: Schema) -> DType:
        t = self.exprs[0].dtype(schema)
        for e in self.exprs[1:]:
            et = e.dtype(schema)
            if t.id is TypeId.NULL:
                t = et
            elif et.id is not TypeId.NULL and et != t:
                t = promote(t, et)
        return t

    def eval(self, batch: ColumnBatch, schema: Schema) -> Column:
        out_t = self.dtype(schema)
        acc = ops.cast(self.exprs[-1].eval(batch, schema), out_t)
        for e in reversed(self.exprs[:-1]):
            c = ops.cast(e.eval(batch, schema), out_t)
            acc = ops.backend_for(c, acc).if_else(_not_null(c), c, acc)
        return acc

    def __str__(self):
        return f"coalesce({', '.join(str(e) for e in self.exprs)})"


def _not_null(c: Column) -> Column:
    return ops.unary_op("not", ops.is_null(c), BOOL)


class Round(Expression):
    """Spark round(): HALF_UP at `scale` decimal places."""

    def __init__(self, child, scale: int = 0):
        self.child = _as_expr(child)
        self.scale = scale

    @property
    def children(self):
        return (self.child,)

    def dtype(self, schema: Schema) -> DType:
        t = self.child.dtype(schema)
        return t if t.is_decimal or t.is_integral else FLOAT64

    def eval(self, batch: ColumnBatch, schema: Schema) -> Column:
        t = self.child.dtype(schema)
        c = self.child.eval(batch, schema)
        if t.is_integral and self.scale >= 0:
            return c
        if t.is_decimal:
            shift = t.scale - self.scale
            if shift <= 0:
                return c
            rescaled = ops.cast(c, DType.decimal(t.precision, self.scale))
            return ops.cast(rescaled, t)
        c = ops.cast(c, FLOAT64)
        return ops.backend_for(c).round_half_up(c, self.scale)

    def __str__(self):
        return f"round({self.child}, {self.scale})"


def col(name: str) -> ColumnRef:
    return ColumnRef(name)


def lit(v, dtype: Optional[DType] = None) -> Literal:
    return Literal(v, dtype)


def when(cond: Expression, value) -> CaseWhen:
    return CaseWhen([(cond, value)])


def coalesce(*exprs) -> Coalesce:
    return Coalesce(*exprs)


def round_(e, scale: int = 0) -> Round:
    return Round(e, scale)


def _null_aware_fold(op: str, exprs):
    """greatest/least: Spark skips NULLs (result null only if all null)."""
    es = [_as_expr(e) for e in exprs]
    acc = es[0]
    for e in es[1:]:
        pick = BinaryExpr(op, acc, e)
        acc = CaseWhen([(IsNull(acc), e), (IsNull(e), acc)], pick)
    return acc


def greatest(*exprs) -> Expression:
    return _null_aware_fold("max", exprs)


def least(*exprs) -> Expression:
    return _null_aware_fold("min", exprs)


def isin(e, *values) -> Expression:
    e = _as_expr(e)
    acc = BinaryExpr("eq", e, Literal(values[0]))
    for v in values[1:]:
        acc = BinaryExpr("or", acc, BinaryExpr("eq", e, Literal(v)))
    return acc


def hour(ts) -> Expression:
    """hour of a timestamp (micros since epoch, UTC)."""
    e = _as_expr(ts)
    return BinaryExpr("pmod", BinaryExpr("int_div", CastExpr(e, INT64),
                                         Literal(3_600_000_000)),
                      Literal(24))


def minute(ts) -> Expression:
    e = _as_expr(ts)
    return BinaryExpr("pmod", BinaryExpr("int_div", CastExpr(e, INT64),
                                         Literal(60_000_000)), Literal(60))


def second(ts) -> Expression:
    e = _as_expr(ts)
    return BinaryExpr("pmod", BinaryExpr("int_div", CastExpr(e, INT64),
                                         Literal(1_000_000)), Literal(60))


def date_add(d, days) -> Expression:
    return BinaryExpr("add", _as_expr(d), _as_expr(days))


def date_sub(d, days) -> Expression:
    return BinaryExpr("sub", _as_expr(d), _as_expr(days))


def to_date(ts) -> Expression:
    """timestamp (micros) -> date32 days, floored for pre-epoch values."""
    e = _as_expr(ts)
    us = CastExpr(e, INT64)
    rem = BinaryExpr("pmod", us, Literal(86_400_000_000))
    days = BinaryExpr("int_div", BinaryExpr("sub", us, rem),
                      Literal(86_400_000_000))
    return CastExpr(days, DType.date32())


def unix_timestamp(ts) -> Expression:
    """timestamp -> whole seconds since epoch (floored)."""
    e = _as_expr(ts)
    us = CastExpr(e, INT64)
    rem = BinaryExpr("pmod", us, Literal(1_000_000))
    return BinaryExpr("int_div", BinaryExpr("sub", us, rem),
                      Literal(1_000_000))


def dayofweek(d) -> Expression:
    """Spark dayofweek: 1 = Sunday .. 7 = Saturday (1970-01-01 was a
    Thursday, day-number 4 in this scheme)."""
    e = _as_expr(d)
    days = CastExpr(CastExpr(e, DType.date32()), INT64)
    return BinaryExpr("add",
                      BinaryExpr("pmod", BinaryExpr("add", days,
                                                    Literal(4)),
                                 Literal(7)), Literal(1))


def quarter(d) -> Expression:
    """quarter 1..4 from the month."""
    e = _as_expr(d)
    m = UnaryExpr("month", e)
    return BinaryExpr("add",
                      BinaryExpr("int_div",
                                 BinaryExpr("sub", m, Literal(1)),
                                 Literal(3)), Literal(1))


def datediff(end, start) -> Expression:
    from ..types import DATE32

    return BinaryExpr("sub", CastExpr(_as_expr(end), DATE32),
                      CastExpr(_as_expr(start), DATE32))


class CreateNamedStruct(Expression):
    """named_struct(n1, e1, n2, e2, ...) -> STRUCT column (reference
    analogue: GpuCreateNamedStruct). The struct itself is never null."""

    def __init__(self, names, exprs):
        self.names = list(names)
        self.exprs = [_as_expr(e) for e in exprs]

    @property
    def children(self):
        return tuple(self.exprs)

    def dtype(self, schema: Schema) -> DType:
        return DType.struct_(
            [(n, e.dtype(schema)) for n, e in zip(self.names, self.exprs)])

    def nullable(self, schema: Schema) -> bool:
        return False

    def eval(self, batch: ColumnBatch, schema: Schema) -> Column:
        import torch

        kids = tuple(e.eval(batch, schema) for e in self.exprs)
        dev = kids[0].device if kids else batch.device
        return Column(self.dtype(schema), batch.num_rows,
                      torch.zeros(0, dtype=torch.uint8, device=dev),
                      None, None, 0, kids)

    def output_name(self) -> str:
        return "named_struct(" + ", ".join(self.names) + ")"

    def __str__(self):
        inner = ", ".join(f"{n}: {e}" for n, e in
                          zip(self.names, self.exprs))
        return f"named_struct({inner})"


class GetStructField(Expression):
    """struct.field access (reference analogue: GpuGetStructField)."""

    def __init__(self, child, name: str):
        self.child = _as_expr(child)
        self.name = name

    @property
    def children(self):
        return (self.child,)

    def _field_index(self, schema) -> int:
        st = self.child.dtype(schema)
        if st.id is not TypeId.STRUCT:
            raise TypeError(f"getField on non-struct {st}")
        if self.name not in st.field_names:
            raise KeyError(
                f"struct has no field {self.name!r} (has "
                f"{list(st.field_names)})")
        return st.field_names.index(self.name)

    def dtype(self, schema: Schema) -> DType:
        st = self.child.dtype(schema)
        return st.children[self._field_index(schema)]

    def nullable(self, schema: Schema) -> bool:
        return True

    def eval(self, batch: ColumnBatch, schema: Schema) -> Column:
        c = self.child.eval(batch, schema)
        kid = c.child[self._field_index(schema)]
        if c.validity is None:
            return kid
        return ops.backend_for(c).and_parent_validity(kid, c)

    def output_name(self) -> str:
        return f"{self.child}.{self.name}"

    def __str__(self):
        return self.output_name()


def named_struct(**fields) -> CreateNamedStruct:
    return CreateNamedStruct(list(fields), list(fields.values()))


def get_field(struct_expr, name: str) -> GetStructField:
    return GetStructField(struct_expr, name)


class CreateMap(Expression):
    """create_map(k1, v1, k2, v2, ...) -> MAP column (reference:
    GpuCreateMap). Entries keep source order; lookups are last-win."""

    def __init__(self, exprs):
        assert exprs and len(exprs) % 2 == 0, \
            "create_map needs key1, value1, key2, value2, ..."
        self.keys = [_as_expr(e) for e in exprs[0::2]]
        self.vals = [_as_expr(e) for e in exprs[1::2]]

    @property
    def children(self):
        return tuple(self.keys) + tuple(self.vals)

    def dtype(self, schema: Schema) -> DType:
        return DType.map_(self.keys[0].dtype(schema),
                          self.vals[0].dtype(schema))

    def nullable(self, schema: Schema) -> bool:
        return False

    def eval(self, batch: ColumnBatch, schema: Schema) -> Column:
        kc = [e.eval(batch, schema) for e in self.keys]
        vc = [e.eval(batch, schema) for e in self.vals]
        return ops.make_map(kc, vc)

    def output_name(self) -> str:
        return "map"

    def __str__(self):
        inner = ", ".join(f"{k}, {v}" for k, v in zip(self.keys, self.vals))
        return f"map({inner})"


class MapView(Expression):
    """map_keys / map_values / map_entries (zero-copy layout views;
    reference: GpuMapKeys/GpuMapValues/GpuMapEntries)."""

    def __init__(self, child, mode: str):
        self.child = _as_expr(child)
        self.mode = mode  # keys | values | entries

    @property
    def children(self):
        return (self.child,)

    def dtype(self, schema: Schema) -> DType:
        mt = self.child.dtype(schema)
        if mt.id is not TypeId.MAP:
            raise TypeError(f"map_{self.mode} on non-map {mt}")
        if self.mode == "keys":
            return DType.list_(mt.children[0])
        if self.mode == "values":
            return DType.list_(mt.children[1])
        return DType.list_(mt.entry_dtype)

    def eval(self, batch: ColumnBatch, schema: Schema) -> Column:
        c = self.child.eval(batch, schema)
        fn = {"keys": ops.map_keys, "values": ops.map_values,
              "entries": ops.map_entries}[self.mode]
        return fn(c)

    def output_name(self) -> str:
        return f"map_{self.mode}({self.child.output_name()})"

    def __str__(self):
        return f"map_{self.mode}({self.child})"


def create_map(*exprs) -> CreateMap:
    return CreateMap(list(exprs))


def map_keys(m) -> MapView:
    return MapView(m, "keys")


def map_values(m) -> MapView:
    return MapView(m, "values")


def map_entries(m) -> MapView:
    return MapView(m, "entries")

"""SQL frontend: a recursive-descent parser for the accelerated subset.

The reference accelerates Spark SQL; this engine owns the frontend, so a
compact SQL layer maps queries onto the same logical plans the DataFrame
API builds (and therefore through the same GPU overrides pass).

Supported: SELECT <exprs|*> FROM <table> [JOIN <table> ON a = b]...
[WHERE <cond>] [GROUP BY <cols>] [HAVING <cond>] [ORDER BY <cols> [ASC|DESC]]
[LIMIT n]. Expressions: + - * / %, comparisons, AND/OR/NOT, IS [NOT] NULL,
IN (...), BETWEEN, LIKE, CASE WHEN, CAST(x AS t), literals, and the
functions sum/avg/count/min/max/stddev/variance/coalesce/round/abs/sqrt/
upper/lower/length/substring/year/month/day.
"""
from __future__ import annotations

import re
from typing import List, Optional

from ..expr import aggregates as A
from ..expr.expressions import (BinaryExpr, CaseWhen, CastExpr, Coalesce,
                                ColumnRef, Expression, IsNull, Literal, Round,
                                StringPredicate, Substring, UnaryExpr)
from ..types import (BOOL, DATE32, DType, FLOAT32, FLOAT64, INT16, INT32,
                     INT64, INT8, STRING, TIMESTAMP)

_TOKEN_RE = re.compile(r"""
    \s*(?:
      (?P<num>\d+\.\d*(?:[eE][+-]?\d+)?|\.\d+|\d+(?:[eE][+-]?\d+)?)
    | (?P<str>'(?:[^'\\]|''|\\.)*')
    | (?P<name>[A-Za-z_][A-Za-z_0-9]*)
    | (?P<op><=|>=|<>|!=|=|<|>|\+|-|\*|/|%|\(|\)|,|\.)
    )""", re.VERBOSE)

_TYPES = {
    "boolean": BOOL, "tinyint": INT8, "smallint": INT16, "int": INT32,
    "integer": INT32, "bigint": INT64, "long": INT64, "float": FLOAT32,
    "double": FLOAT64, "date": DATE32, "timestamp": TIMESTAMP,
    "string": STRING,
}

_AGG_FUNCS = {"sum", "avg", "count", "min", "max", "stddev", "variance",
              "collect_list", "collect_set"}


class SqlError(ValueError):
    pass


_ESCAPES = {"0": "\0", "'": "'", '"': '"', "b": "\b", "n": "\n", "r": "\r",
            "t": "\t", "Z": "\x1a", "\\": "\\"}


def _unescape_sql_string(s: str) -> str:
    """Spark ParserUtils.unescapeSQLString: backslash escapes (\\n, \\t,
    \\uXXXX, ...) are processed; \\% and \\_ KEEP the backslash so LIKE
    sees them as its own escape; '' collapses to ' (ADVICE.md round 1)."""
    out = []
    i = 0
    while i < len(s):
        c = s[i]
        if c == "'" and i + 1 < len(s) and s[i + 1] == "'":
            out.append("'")
            i += 2
            continue
        if c == "\\" and i + 1 < len(s):
            nx = s[i + 1]
            if nx in ("%", "_"):
                out.append("\\" + nx)
            elif nx == "u" and i + 5 < len(s):
                try:
                    out.append(chr(int(s[i + 2:i + 6], 16)))
                    i += 6
                    continue
                except ValueError:
                    out.append(nx)
            elif nx in _ESCAPES:
                out.append(_ESCAPES[nx])
            else:
                out.append(nx)
            i += 2
            continue
        out.append(c)
        i += 1
    return "".join(out)


def tokenize(text: str) -> List[tuple]:
    out = []
    pos = 0
    while pos < len(text):
        m = _TOKEN_RE.match(text, pos)
        if not m or m.end() == pos:
            if text[pos:].strip() == "":
                break
            raise SqlError(f"cannot tokenize at: {text[pos:pos+20]!r}")
        pos = m.end()
        if m.group("num") is not None:
            out.append(("num", m.group("num")))
        elif m.group("str") is not None:
            out.append(("str", _unescape_sql_string(m.group("str")[1:-1])))
        elif m.group("name") is not None:
            out.append(("name", m.group("name")))
        else:
            out.append(("op", m.group("op")))
    out.append(("end", ""))
    return out


class Parser:
    def __init__(self, text: str, session):
        self.toks = tokenize(text)
        self.i = 0
        self.session = session

    # -- token helpers ---------------------------------------------------
    def peek(self, k=0):
        return self.toks[min(self.i + k, len(self.toks) - 1)]

    def next(self):
        t = self.toks[self.i]
        self.i += 1
        return t

    def kw(self, word) -> bool:
        t = self.peek()
        if t[0] == "name" and t[1].upper() == word.upper():
            self.i += 1
            return True
        return False

    def expect_kw(self, word):
        if not self.kw(word):
            raise SqlError(f"expected {word} at {self.peek()}")

    def op(self, sym) -> bool:
        t = self.peek()
        if t[0] == "op" and t[1] == sym:
            self.i += 1
            return True
        return False

    def expect_op(self, sym):
        if not self.op(sym):
            raise SqlError(f"expected {sym!r} at {self.peek()}")

    # -- grammar ---------------------------------------------------------
    def parse_query(self):
        self.explain_only = self.kw("EXPLAIN")
        self.expect_kw("SELECT")
        distinct = self.kw("DISTINCT")
        star = False
        items = []  # (expr_or_aggexpr, alias)
        if self.op("*"):
            star = True
        else:
            while True:
                e = self.parse_select_item()
                alias = None
                if self.kw("AS"):
                    alias = self.next()[1]
                elif self.peek()[0] == "name" and self.peek()[1].upper() not in (
                        "FROM", "WHERE", "GROUP", "ORDER", "LIMIT", "HAVING",
                        "JOIN", "LEFT", "INNER", "ON", "AND", "OR", "ASC",
                        "DESC", "BY"):
                    alias = self.next()[1]
                items.append((e, alias))
                if not self.op(","):
                    break
        self.expect_kw("FROM")
        df = self.parse_table()
        while True:
            how = None
            up = self.peek()[1].upper() if self.peek()[0] == "name" else ""
            if up == "JOIN":
                self.next()
                how = "inner"
            elif up == "INNER":
                self.next()
                self.expect_kw("JOIN")
                how = "inner"
            elif up == "LEFT":
                self.next()
                self.kw("OUTER")
                self.expect_kw("JOIN")
                how = "left"
            else:
                break
            right = self.parse_table()
            self.expect_kw("ON")
            lk, rk, cond = self.parse_join_keys()
            if not lk:
                raise ValueError(
                    "JOIN ON needs at least one equality conjunct "
                    "(a = b); pure non-equi joins: use CROSS JOIN + WHERE")
            df = df.join(right, on=lk, right_on=rk, how=how,
                         condition=cond)
        where = None
        if self.kw("WHERE"):
            where = self.parse_expr()
        group_cols = []
        group_mode = "plain"
        if self.kw("GROUP"):
            self.expect_kw("BY")
            if self.kw("ROLLUP") or self.kw("CUBE"):
                group_mode = "rollup" if \
                    self.toks[self.i - 1][1].upper() == "ROLLUP" else "cube"
                self.expect_op("(")
                while True:
                    group_cols.append(self.next()[1])
                    if not self.op(","):
                        break
                self.expect_op(")")
            else:
                while True:
                    group_cols.append(self.next()[1])
                    if not self.op(","):
                        break
        having = None
        if self.kw("HAVING"):
            having = self.parse_expr()
        order = []
        if self.kw("ORDER"):
            self.expect_kw("BY")
            while True:
                name = self.next()[1]
                desc = False
                if self.kw("DESC"):
                    desc = True
                else:
                    self.kw("ASC")
                order.append((name, desc))
                if not self.op(","):
                    break
        limit = None
        if self.kw("LIMIT"):
            limit = int(self.next()[1])
        if self.peek()[0] != "end":
            raise SqlError(f"unexpected trailing tokens at {self.peek()}")

        # assemble plan
        if where is not None:
            df = df.filter(where)
        if distinct and not star:
            # SELECT DISTINCT a, b ... -> project then dedupe
            exprs = [(e.alias(alias) if alias else e) for e, alias in items]
            df = df.select(*exprs).distinct()
            if order:
                df = df.sort(*[n for n, _ in order],
                             descending=[d for _, d in order])
            if limit is not None:
                df = df.limit(limit)
            if self.peek()[0] != "end":
                raise SqlError("trailing tokens")
            return df
        if distinct and star:
            df = df.distinct()
        from ..expr.windows import WindowExpr as _WE

        if any(isinstance(e, _WE) for e, _ in items):
            out_names = []
            for e, alias in items:
                name = alias or e.output_name()
                if isinstance(e, _WE):
                    df = df.with_column(name, e)
                out_names.append((e, name))
            sel = []
            for e, name in out_names:
                sel.append(ColumnRef(name) if isinstance(e, _WE)
                           else (e.alias(name)))
            df = df.select(*sel)
            if order:
                df = df.sort(*[n for n, _ in order],
                             descending=[d for _, d in order])
            if limit is not None:
                df = df.limit(limit)
            if self.peek()[0] != "end":
                raise SqlError("trailing tokens")
            return df
        has_aggs = any(isinstance(e, A.AggExpr) for e, _ in items)
        if group_cols or has_aggs:
            keys = group_cols
            aggs = []
            for e, alias in items:
                if isinstance(e, A.AggExpr):
                    aggs.append(e.alias(alias) if alias else e)
                elif isinstance(e, ColumnRef) and e.name in group_cols:
                    pass  # group key, included automatically
                elif star:
                    pass
                else:
                    raise SqlError(
                        f"non-aggregate select item {e} not in GROUP BY")
            if not keys:
                df = df.agg(*aggs)
            elif group_mode == "rollup":
                df = df.rollup(*keys).agg(*aggs)
            elif group_mode == "cube":
                df = df.cube(*keys).agg(*aggs)
            else:
                df = df.group_by(*keys).agg(*aggs)
        elif not star:
            exprs = [(e.alias(alias) if alias else e) for e, alias in items]
            # ORDER BY may reference pre-projection columns (Spark allows
            # both); sort first when a key is not in the select list
            if order and not all(n in [x.output_name() for x in exprs]
                                 for n, _ in order):
                df = df.sort(*[n for n, _ in order],
                             descending=[d for _, d in order])
                order = []
            df = df.select(*exprs)
        if having is not None:
            df = df.filter(having)
        if order:
            df = df.sort(*[n for n, _ in order],
                         descending=[d for _, d in order])
        if limit is not None:
            df = df.limit(limit)
        return df

    def parse_table(self):
        name = self.next()[1]
        df = self.session.table(name)
        # optional alias (ignored name binding; columns stay unqualified)
        t = self.peek()
        if t[0] == "name" and t[1].upper() not in (
                "JOIN", "LEFT", "INNER", "ON", "WHERE", "GROUP", "ORDER",
                "LIMIT", "HAVING"):
            self.next()
        return df

    def parse_join_keys(self):
        """AND-separated conjuncts: `a = b` pairs become equi keys, any
        other comparison becomes the non-equi join condition (conditional
        hash join)."""
        lk, rk = [], []
        cond = None
        while True:
            mark = self.i
            eq = None
            if self.peek()[0] == "name":
                try:
                    a = self.parse_qualified_name()
                    if self.op("="):
                        b = self.parse_qualified_name()
                        nxt = self.peek()
                        # a = b must end the conjunct (not `a = b + 1`)
                        if nxt[0] != "op" or nxt[1] in (")", ","):
                            eq = (a, b)
                except (ValueError, IndexError):
                    eq = None
            if eq is not None:
                lk.append(eq[0])
                rk.append(eq[1])
            else:
                self.i = mark
                c = self.parse_expr()
                cond = c if cond is None else (cond & c)
            if not self.kw("AND"):
                break
        return lk, rk, cond

    def parse_qualified_name(self) -> str:
        n = self.next()[1]
        if self.op("."):
            n = self.next()[1]  # drop qualifier; names are engine-global
        return n

    def parse_select_item(self):
        e = self.parse_expr()
        if self.kw("OVER"):
            e = self.parse_over(e)
        return e

    def parse_over(self, e):
        """<agg or ranking func> OVER (PARTITION BY .. ORDER BY ..
        [ROWS|RANGE BETWEEN .. AND ..])"""
        from ..expr import windows as W

        if isinstance(e, W.WindowFunc):
            func = e
        elif isinstance(e, A.AggExpr):
            if e.op not in ("sum", "count", "min", "max", "mean"):
                raise SqlError(f"{e.op} cannot be a window function")
            func = W.WindowFunc(e.op, e.child)
        else:
            raise SqlError("OVER needs an aggregate or ranking function")
        self.expect_op("(")
        part, order, desc = [], [], []
        rows_between = None
        range_between = None
        if self.kw("PARTITION"):
            self.expect_kw("BY")
            while True:
                part.append(self.next()[1])
                if not self.op(","):
                    break
        if self.kw("ORDER"):
            self.expect_kw("BY")
            while True:
                order.append(self.next()[1])
                if self.kw("DESC"):
                    desc.append(True)
                else:
                    self.kw("ASC")
                    desc.append(False)
                if not self.op(","):
                    break
        frame_kind = None
        if self.kw("ROWS"):
            frame_kind = "rows"
        elif self.kw("RANGE"):
            frame_kind = "range"
        if frame_kind:
            self.expect_kw("BETWEEN")
            lo = self._frame_bound(preceding=True)
            self.expect_kw("AND")
            hi = self._frame_bound(preceding=False)
            if frame_kind == "rows":
                rows_between = (int(lo) if lo is not None else None,
                                int(hi) if hi is not None else None)
                if rows_between[0] is None:
                    raise SqlError("ROWS UNBOUNDED PRECEDING frame is the "
                                   "default running frame; omit the clause")
            else:
                range_between = (lo, hi)
        self.expect_op(")")
        return W.WindowExpr(func, W.WindowSpec(
            part, order, desc or None, rows_between, range_between))

    def _frame_bound(self, preceding: bool):
        if self.kw("UNBOUNDED"):
            if not (self.kw("PRECEDING") or self.kw("FOLLOWING")):
                raise SqlError("expected PRECEDING/FOLLOWING")
            return None
        if self.kw("CURRENT"):
            self.expect_kw("ROW")
            return 0
        tok = self.next()[1]
        v = float(tok) if "." in tok else int(tok)
        if self.kw("PRECEDING"):
            return -v
        self.expect_kw("FOLLOWING")
        return v

    # expression precedence: OR < AND < NOT < cmp < add < mul < unary
    def parse_expr(self):
        e = self.parse_and()
        while self.kw("OR"):
            e = BinaryExpr("or", e, self.parse_and())
        return e

    def parse_and(self):
        e = self.parse_not()
        while self.kw("AND"):
            e = BinaryExpr("and", e, self.parse_not())
        return e

    def parse_not(self):
        if self.kw("NOT"):
            return UnaryExpr("not", self.parse_not())
        return self.parse_cmp()

    def parse_cmp(self):
        e = self.parse_add()
        t = self.peek()
        if t[0] == "op" and t[1] in ("=", "<>", "!=", "<", "<=", ">", ">="):
            self.next()
            opmap = {"=": "eq", "<>": "ne", "!=": "ne", "<": "lt",
                     "<=": "le", ">": "gt", ">=": "ge"}
            return BinaryExpr(opmap[t[1]], e, self.parse_add())
        if self.kw("IS"):
            neg = self.kw("NOT")
            self.expect_kw("NULL")
            x = IsNull(e)
            return UnaryExpr("not", x) if neg else x
        if self.kw("BETWEEN"):
            lo = self.parse_add()
            self.expect_kw("AND")
            hi = self.parse_add()
            return BinaryExpr("and", BinaryExpr("ge", e, lo),
                              BinaryExpr("le", e, hi))
        if self.kw("IN"):
            self.expect_op("(")
            vals = []
            while True:
                vals.append(self.parse_literal_value())
                if not self.op(","):
                    break
            self.expect_op(")")
            from ..expr.expressions import isin

            return isin(e, *vals)
        if self.kw("LIKE"):
            pat = self.next()
            if pat[0] != "str":
                raise SqlError("LIKE needs a string literal")
            return StringPredicate("like", e, pat[1])
        if self.kw("RLIKE") or self.kw("REGEXP"):
            pat = self.next()
            if pat[0] != "str":
                raise SqlError("RLIKE needs a string literal")
            return StringPredicate("rlike", e, pat[1])
        return e

    def parse_add(self):
        e = self.parse_mul()
        while True:
            if self.op("+"):
                e = BinaryExpr("add", e, self.parse_mul())
            elif self.op("-"):
                e = BinaryExpr("sub", e, self.parse_mul())
            else:
                return e

    def parse_mul(self):
        e = self.parse_unary()
        while True:
            if self.op("*"):
                e = BinaryExpr("mul", e, self.parse_unary())
            elif self.op("/"):
                e = BinaryExpr("div", e, self.parse_unary())
            elif self.op("%"):
                e = BinaryExpr("mod", e, self.parse_unary())
            else:
                return e

    def parse_unary(self):
        if self.op("-"):
            return UnaryExpr("neg", self.parse_unary())
        return self.parse_primary()

    def parse_literal_value(self):
        t = self.next()
        if t[0] == "num":
            return float(t[1]) if any(c in t[1] for c in ".eE") else int(t[1])
        if t[0] == "str":
            return t[1]
        if t[0] == "name" and t[1].upper() in ("TRUE", "FALSE"):
            return t[1].upper() == "TRUE"
        raise SqlError(f"expected literal at {t}")

    def parse_primary(self):
        t = self.peek()
        if t[0] == "op" and t[1] == "(":
            self.next()
            e = self.parse_expr()
            self.expect_op(")")
            return e
        if t[0] == "num":
            self.next()
            v = float(t[1]) if any(c in t[1] for c in ".eE") else int(t[1])
            return Literal(v)
        if t[0] == "str":
            self.next()
            return Literal(t[1])
        if t[0] == "name":
            up = t[1].upper()
            if up == "NULL":
                self.next()
                return Literal(None)
            if up in ("TRUE", "FALSE"):
                self.next()
                return Literal(up == "TRUE")
            if up == "CASE":
                return self.parse_case()
            if up == "CAST":
                self.next()
                self.expect_op("(")
                e = self.parse_expr()
                self.expect_kw("AS")
                tname = self.next()[1].lower()
                if tname not in _TYPES:
                    raise SqlError(f"unknown type {tname}")
                self.expect_op(")")
                return CastExpr(e, _TYPES[tname])
            if self.peek(1) == ("op", "("):
                return self.parse_func()
            self.next()
            return ColumnRef(self.parse_qualified_suffix(t[1]))
        raise SqlError(f"unexpected token {t}")

    def parse_qualified_suffix(self, first: str) -> str:
        if self.op("."):
            return self.next()[1]
        return first

    def parse_case(self):
        self.expect_kw("CASE")
        branches = []
        else_e = None
        while self.kw("WHEN"):
            c = self.parse_expr()
            self.expect_kw("THEN")
            v = self.parse_expr()
            branches.append((c, v))
        if self.kw("ELSE"):
            else_e = self.parse_expr()
        self.expect_kw("END")
        return CaseWhen(branches, else_e)

    @staticmethod
    def _int_arg(e) -> int:
        from ..expr.expressions import Literal, UnaryExpr

        if isinstance(e, Literal):
            return int(e.value)
        if isinstance(e, UnaryExpr) and e.op == "neg" and \
                isinstance(e.child, Literal):
            return -int(e.child.value)
        raise SqlError(f"expected an integer literal, got {e}")

    def parse_func(self):
        name = self.next()[1].lower()
        self.expect_op("(")
        if name == "count" and self.op("*"):
            self.expect_op(")")
            return A.count_star()
        distinct = False
        if name in _AGG_FUNCS and self.kw("DISTINCT"):
            distinct = True
        args = []
        if not self.op(")"):
            while True:
                args.append(self.parse_expr())
                if not self.op(","):
                    break
            self.expect_op(")")
        if name in ("row_number", "rank", "dense_rank"):
            from ..expr import windows as W

            return {"row_number": W.row_number, "rank": W.rank,
                    "dense_rank": W.dense_rank}[name]()
        if name in ("lag", "lead"):
            from ..expr import windows as W

            off = self._int_arg(args[1]) if len(args) > 1 else 1
            dflt = args[2].value if len(args) > 2 else None
            return {"lag": W.lag, "lead": W.lead}[name](args[0], off, dflt)
        if name == "regexp_extract":
            from ..expr.expressions import RegexpExtract

            idx = self._int_arg(args[2]) if len(args) > 2 else 1
            return RegexpExtract(args[0], args[1].value, idx)
        if name == "get_json_object":
            from ..expr.expressions import GetJsonObject

            return GetJsonObject(args[0], args[1].value)
        if name == "regexp_extract_all":
            from ..expr.expressions import RegexpExtractAll

            idx = self._int_arg(args[2]) if len(args) > 2 else 1
            return RegexpExtractAll(args[0], args[1].value, idx)
        if name == "regexp_replace":
            from ..expr.expressions import RegexpReplace

            return RegexpReplace(args[0], args[1].value, args[2].value)
        if name in ("percentile", "approx_percentile"):
            return A.percentile(args[0], float(args[1].value))
        if name in _AGG_FUNCS:
            ctor = {"sum": A.sum_, "avg": A.avg, "count": A.count,
                    "min": A.min_, "max": A.max_, "stddev": A.stddev,
                    "variance": A.variance,
                    "collect_list": A.collect_list,
                    "collect_set": A.collect_set}[name]
            agg = ctor(args[0])
            if distinct:
                agg = A.AggExpr(agg.op, agg.child, distinct=True)
            return agg
        if name in ("coalesce", "nvl", "ifnull"):
            return Coalesce(*args)
        if name == "nullif":
            from ..expr.expressions import CaseWhen, Literal

            return CaseWhen([(BinaryExpr("eq_null_safe", args[0], args[1]),
                              Literal(None))], args[0])
        if name in ("greatest", "least"):
            from ..expr.expressions import greatest as _g, least as _l

            return (_g if name == "greatest" else _l)(*args)
        if name == "round":
            scale = args[1].value if len(args) > 1 else 0
            return Round(args[0], int(scale))
        if name in ("abs", "sqrt", "exp", "log", "floor", "ceil", "upper",
                    "lower", "length", "year", "month", "day", "trim",
                    "ltrim", "rtrim", "initcap", "reverse"):
            return UnaryExpr(name, args[0])
        if name == "concat_ws":
            from ..expr.expressions import ConcatWs

            return ConcatWs(args[0].value, *args[1:])
        if name == "concat":
            out = args[0]
            for nxt in args[1:]:
                out = BinaryExpr("concat", out, nxt)
            return out
        if name == "split":
            from ..expr.expressions import StrSplit

            return StrSplit(args[0], args[1].value)
        if name == "element_at":
            from ..expr.expressions import ElementAt, Literal

            # int literal -> array index; any other literal -> map key
            if isinstance(args[1], Literal) and \
                    not isinstance(args[1].value, int):
                return ElementAt(args[0], args[1].value)
            return ElementAt(args[0], self._int_arg(args[1]))
        if name == "size":
            from ..expr.expressions import ArraySize

            return ArraySize(args[0])
        if name == "array_contains":
            from ..expr.expressions import ArrayContains, Literal

            return ArrayContains(args[0], args[1].value
                                 if isinstance(args[1], Literal)
                                 else args[1])
        if name in ("map_keys", "map_values", "map_entries"):
            from ..expr.expressions import MapView

            return MapView(args[0], name.split("_", 1)[1])
        if name == "map":
            from ..expr.expressions import CreateMap

            return CreateMap(args)
        if name == "repeat":
            from ..expr.expressions import repeat_str

            return repeat_str(args[0], self._int_arg(args[1]))
        if name == "substring_index":
            from ..expr.expressions import substring_index

            return substring_index(args[0], args[1].value,
                                   self._int_arg(args[2]))
        if name == "translate":
            from ..expr.expressions import translate

            return translate(args[0], args[1].value, args[2].value)
        if name == "ascii":
            from ..expr.expressions import ascii_

            return ascii_(args[0])
        if name in ("lpad", "rpad"):
            from ..expr.expressions import PadExpr

            fill = args[2].value if len(args) > 2 else " "
            return PadExpr(args[0], self._int_arg(args[1]), fill,
                           left=name == "lpad")
        if name in ("locate", "instr"):
            from ..expr.expressions import LocateExpr

            if name == "locate":  # locate(substr, str[, pos])
                pos = self._int_arg(args[2]) if len(args) > 2 else 1
                return LocateExpr(args[1], args[0].value, pos)
            return LocateExpr(args[0], args[1].value, 1)
        if name == "replace":
            return args[0].replace(args[1].value, args[2].value)
        if name == "substring" or name == "substr":
            pos = self._int_arg(args[1])
            ln = self._int_arg(args[2]) if len(args) > 2 else -1
            return Substring(args[0], pos, ln)
        raise SqlError(f"unknown function {name}")


def parse_sql(session, text: str):
    stripped = text.lstrip()
    up = stripped.upper()
    if up.startswith("CREATE "):
        # CREATE [OR REPLACE] [TEMP|TEMPORARY] VIEW <name> AS <query>
        import re as _re

        m = _re.match(r"CREATE\s+(?:OR\s+REPLACE\s+)?"
                      r"(?:TEMP(?:ORARY)?\s+)?VIEW\s+(\w+)\s+AS\s+(.*)",
                      stripped, _re.I | _re.S)
        if not m:
            raise SqlError("unsupported CREATE statement")
        name, body = m.group(1), m.group(2)
        df = parse_sql(session, body)
        session.register(name, df)
        return df
    p = Parser(text, session)
    df = p.parse_query()
    if getattr(p, "explain_only", False):
        return df.physical_plan().tree_string()
    return df

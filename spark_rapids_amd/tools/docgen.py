"""Generate docs from the live registries (reference analogue:
RapidsConf.help -> docs/configs.md and TypeChecks -> docs/supported_ops.md +
tools/generated_files CSVs). Run: python -m spark_rapids_amd.tools.docgen
"""
from __future__ import annotations

import os

from ..config import help_doc
from ..plan import overrides as ov
from ..types import TypeId

REPO = os.path.dirname(os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


# Spark-expression rows: (Spark expression, engine construct, placement,
# notes). Placement: GPU = device kernel; GPU* = device with documented
# condition; CPU = host fallback via tagger/CpuBridge.
_EXPR_ROWS = [
    # plumbing
    ("Alias", "Expression.alias", "GPU", ""),
    ("AttributeReference", "ColumnRef", "GPU", ""),
    ("Literal", "Literal", "GPU", "incl. decimal64/128, string, null"),
    ("SortOrder", "SortExec keys", "GPU", "asc/desc, nulls first/last"),
    # arithmetic
    ("Add", "BinaryExpr add", "GPU", "int wrap (non-ANSI); decimal exact with DecimalPrecision widening"),
    ("Subtract", "BinaryExpr sub", "GPU", ""),
    ("Multiply", "BinaryExpr mul", "GPU", "decimal: exact 128/256-bit kernels"),
    ("Divide", "BinaryExpr div", "GPU", "double result; decimal exact HALF_UP; x/0 -> NULL"),
    ("IntegralDivide", "BinaryExpr int_div", "GPU", ""),
    ("Remainder", "BinaryExpr mod", "GPU", "sign follows dividend; %0 -> NULL"),
    ("Pmod", "BinaryExpr pmod", "GPU", ""),
    ("UnaryMinus", "UnaryExpr neg", "GPU", ""),
    ("UnaryPositive", "identity", "GPU", ""),
    ("Abs", "UnaryExpr abs", "GPU", ""),
    ("Pow", "BinaryExpr pow", "GPU", ""),
    ("Sqrt", "UnaryExpr sqrt", "GPU", ""),
    ("Exp", "UnaryExpr exp", "GPU", ""),
    ("Log", "UnaryExpr log", "GPU", ""),
    ("Floor", "UnaryExpr floor", "GPU", ""),
    ("Ceil", "UnaryExpr ceil", "GPU", ""),
    ("Round", "Round expr", "GPU", "HALF_UP"),
    ("Sin", "UnaryExpr sin", "GPU", ""),
    ("Cos", "UnaryExpr cos", "GPU", ""),
    ("Tan", "UnaryExpr tan", "GPU", ""),
    ("Signum", "sign composition", "GPU", "via compare+case"),
    ("BitwiseAnd", "BinaryExpr bitand", "GPU", ""),
    ("BitwiseOr", "BinaryExpr bitor", "GPU", ""),
    ("BitwiseXor", "BinaryExpr bitxor", "GPU", ""),
    ("ShiftLeft", "BinaryExpr shiftleft", "GPU", ""),
    ("ShiftRight", "BinaryExpr shiftright", "GPU", ""),
    ("CheckOverflow", "decimal result bounds", "GPU", "null-on-overflow in every decimal kernel"),
    ("PromotePrecision", "DecimalPrecision widening in promote()", "GPU", ""),
    # predicates / conditional
    ("And", "BinaryExpr and", "GPU", "Kleene"),
    ("Or", "BinaryExpr or", "GPU", "Kleene"),
    ("Not", "UnaryExpr not", "GPU", ""),
    ("EqualTo", "BinaryExpr eq", "GPU", ""),
    ("EqualNullSafe", "BinaryExpr eq_null_safe", "GPU", "<=> never NULL"),
    ("LessThan", "BinaryExpr lt", "GPU", ""),
    ("LessThanOrEqual", "BinaryExpr le", "GPU", ""),
    ("GreaterThan", "BinaryExpr gt", "GPU", ""),
    ("GreaterThanOrEqual", "BinaryExpr ge", "GPU", ""),
    ("In / InSet", "isin", "GPU", "OR-expansion / hash-set semantics"),
    ("IsNull", "IsNull", "GPU", ""),
    ("IsNotNull", "not IsNull", "GPU", ""),
    ("IsNaN", "UnaryExpr is_nan", "GPU", ""),
    ("If", "CaseWhen 2-arm", "GPU", ""),
    ("CaseWhen", "CaseWhen", "GPU", ""),
    ("Coalesce", "Coalesce", "GPU", ""),
    ("Greatest", "max fold", "GPU", ""),
    ("Least", "min fold", "GPU", ""),
    ("NaNvl", "is_nan + if_else", "GPU", ""),
    # casts
    ("Cast int<->int", "CastExpr", "GPU", "wrap semantics"),
    ("Cast int<->float", "CastExpr", "GPU", "trunc-toward-zero, saturate"),
    ("Cast float->int", "CastExpr", "GPU", "NaN->0, saturate bounds"),
    ("Cast int->decimal", "CastExpr", "GPU", ""),
    ("Cast decimal->decimal", "CastExpr", "GPU", "exact 128-bit rescale, HALF_UP, null-on-overflow"),
    ("Cast decimal->float", "CastExpr", "GPU", ""),
    ("Cast float->decimal", "CastExpr", "GPU", "round-to-nearest"),
    ("Cast string->int8/16/32/64", "k_str_to_dec", "GPU", "UTF8String.toLong semantics: trim, truncate fraction, no exponent, overflow->NULL"),
    ("Cast string->decimal", "k_str_to_dec", "GPU", "exact u128 parse, HALF_UP at scale, exponent ok"),
    ("Cast string->float/double", "csv_parse f64", "GPU", ""),
    ("Cast string->bool", "literal table", "CPU", "'true'/'t'/'yes'/'y'/'1' etc"),
    ("Cast int->string", "i64_to_str", "GPU", ""),
    ("Cast decimal->string", "k_dec_to_str", "GPU", "fixed-scale trailing zeros"),
    ("Cast float->string", "python repr path", "CPU", "Java shortest-round-trip parity (reference gates this too)"),
    ("Cast bool->string", "host", "CPU", "true/false literals"),
    ("Cast date/timestamp<->int", "CastExpr", "GPU", "days / micros backing"),
    # strings
    ("Concat", "BinaryExpr concat", "GPU", "NULL if either NULL"),
    ("ConcatWs", "ConcatWs", "GPU", "never NULL"),
    ("Contains", "str_find", "GPU", ""),
    ("StartsWith", "str_find", "GPU", ""),
    ("EndsWith", "str_find", "GPU", ""),
    ("Like", "k_str_like", "GPU", "backslash escapes incl. \\% \\_"),
    ("RLike", "regex VM", "GPU*", "compiled subset; CPU redo on VM overflow; no backrefs/lookaround"),
    ("RegExpExtract", "regex VM captures", "GPU*", ""),
    ("RegExpExtractAll", "regex VM captures", "GPU*", ""),
    ("RegExpReplace", "regex VM captures", "GPU*", ""),
    ("StringSplit", "StrSplit", "GPU*", "literal delimiters on GPU; regex delimiters CPU"),
    ("StringReplace", "escaped-regex replace", "GPU", ""),
    ("StringTrim", "k_str_trim mode both", "GPU", "space-only (Spark trim)"),
    ("StringTrimLeft", "k_str_trim leading", "GPU", ""),
    ("StringTrimRight", "k_str_trim trailing", "GPU", ""),
    ("StringLPad", "k_str_pad", "GPU", "cycles fill, codepoint width"),
    ("StringRPad", "k_str_pad", "GPU", ""),
    ("StringLocate", "k_str_locate", "GPU", "1-based codepoint index"),
    ("StringInstr", "k_str_locate pos=1", "GPU", ""),
    ("Substring", "Substring", "GPU", "1-based, negative from end, codepoints"),
    ("SubstringIndex", "HostStringFn", "CPU", ""),
    ("InitCap", "k_str_case words", "GPU*", "ASCII; gated by incompatibleOps"),
    ("Lower", "k_str_case", "GPU*", "ASCII; incompat gate"),
    ("Upper", "k_str_case", "GPU*", "ASCII; incompat gate"),
    ("Length", "k_str_length", "GPU", "codepoints"),
    ("Reverse", "k_str_reverse", "GPU", "codepoint order"),
    ("StringRepeat", "HostStringFn", "CPU", ""),
    ("StringTranslate", "HostStringFn", "CPU", ""),
    ("GetJsonObject", "json key search kernel", "GPU*", "top-level keys; nested paths CPU"),
    ("Murmur3Hash", "k_murmur3", "GPU", "spark-exact, seeds chain"),
    # datetime
    ("Year", "UnaryExpr year", "GPU", "civil calendar kernel"),
    ("Month", "UnaryExpr month", "GPU", ""),
    ("DayOfMonth", "UnaryExpr day", "GPU", ""),
    ("Quarter", "quarter()", "GPU", "month composition"),
    ("DayOfWeek", "dayofweek()", "GPU", ""),
    ("Hour", "hour()", "GPU", ""),
    ("Minute", "minute()", "GPU", ""),
    ("Second", "second()", "GPU", ""),
    ("DateAdd", "date_add", "GPU", ""),
    ("DateSub", "date_sub", "GPU", ""),
    ("DateDiff", "date sub", "GPU", ""),
    ("ToDate", "to_date", "GPU", ""),
    ("UnixTimestamp", "unix_timestamp", "GPU", ""),
    ("FromUnixTime", "from_unixtime", "GPU", "k_date_format"),
    ("DateFormatClass", "date_format", "GPU*", "pattern subset yyyy MM dd HH mm ss + literals"),
    ("GetTimestamp/ToTimestamp", "to_timestamp", "GPU*", "k_ts_parse, fixed-width patterns, invalid->NULL"),
    ("FromUTCTimestamp", "from_utc_timestamp", "GPU", "k_tz_convert + TZif transition table (tools/tzdb.py)"),
    ("ToUTCTimestamp", "to_utc_timestamp", "GPU", "two-step wall-time resolve"),
    # aggregates
    ("Count", "count/count(*)", "GPU", ""),
    ("Sum", "sum", "GPU", "int64 wrap; decimal -> decimal128 accumulators"),
    ("Min", "min", "GPU", "incl. strings"),
    ("Max", "max", "GPU", "incl. strings"),
    ("Average", "avg", "GPU", "sum+count lowering"),
    ("StddevSamp", "stddev", "GPU", "sum/sumsq lowering"),
    ("VarianceSamp", "variance", "GPU", ""),
    ("First", "first", "GPU", ""),
    ("Last", "last", "GPU", ""),
    ("CollectList", "collect_list", "GPU", "single-pass exchange-then-agg"),
    ("CollectSet", "collect_set", "GPU*", "strings CPU"),
    ("Count(DISTINCT)", "rewrite", "GPU", "RewriteDistinctAggregates-style"),
    ("Sum(DISTINCT)", "rewrite", "GPU", ""),
    ("Percentile", "gb_percentile", "GPU", "exact sorted-groups kernel"),
    ("ApproximatePercentile", "exact rewrite", "GPU", "exact result (superset of t-digest accuracy)"),
    ("ApproxCountDistinct/HyperLogLogPlusPlus", "k_gb_hll + xxHash64", "GPU", "HLL++ sketch, Spark precision formula; estimates use standard HLL correction (no bias tables)"),
    ("XxHash64", "k_xxhash64_col/str", "GPU", "canonical XXH64, column-chained seeds"),
    ("BitAndAgg/BitOrAgg/BitXorAgg", "bit_and/or/xor", "GPU", ""),
    ("PivotFirst", "-", "CPU", "via host fallback"),
    # windows
    ("RowNumber", "row_number over", "GPU", ""),
    ("Rank", "rank over", "GPU", ""),
    ("DenseRank", "dense_rank over", "GPU", ""),
    ("Lag", "lag", "GPU", ""),
    ("Lead", "lead", "GPU", ""),
    ("NthValue", "nth_value", "GPU", ""),
    ("NTile", "ntile", "GPU", ""),
    ("WindowExpression ROWS frames", "win_sum/min/max/avg/count", "GPU", "running + bounded"),
    ("WindowExpression RANGE frames", "range_between", "GPU*", "ascending ranges; descending CPU"),
    # complex types
    ("CreateNamedStruct", "named_struct", "GPU", "struct columns, child-wise"),
    ("GetStructField", "get_field", "GPU", "parent null mask merged"),
    ("CreateMap", "create_map", "GPU", "entry interleave, last-win lookups"),
    ("MapKeys", "map_keys", "GPU", "zero-copy entries view"),
    ("MapValues", "map_values", "GPU", "zero-copy entries view"),
    ("MapEntries", "map_entries", "GPU", "LIST<STRUCT<key,value>> view"),
    ("ElementAt (map)", "col.element_at(key)", "GPU",
     "segment last-match reduction; nested values on CPU"),
    ("Explode", "explode", "GPU", ""),
    ("PosExplode", "posexplode", "GPU", ""),
    ("Explode_outer", "explode(outer=True)", "CPU", "padding path"),
    ("Size", "ArraySize", "GPU", ""),
    ("ElementAt (array)", "ElementAt", "GPU", "1-based, NULL overflow"),
    ("SortArray", "-", "CPU", ""),
    ("CreateArray", "from_pylist lists", "CPU", "construction host-side"),
    # misc
    ("Rand", "SampleHash deterministic draw", "GPU", "sample() lowering"),
    ("SparkPartitionID", "rank literal", "GPU", ""),
    ("MonotonicallyIncreasingID", "range + rank offset", "GPU", ""),
    ("ScalarSubquery", "pre-evaluated literal", "GPU", ""),
]

_EXECS = [
    ("Scan", "memory / parquet (GPU page decode or hybrid CPU) / csv / orc",
     "all basic types"),
    ("Filter", "boolean mask stream compaction", "all basic types"),
    ("Project", "expression evaluation", "all basic types"),
    ("Aggregate", "hash group-by, partial+merge, distributed exchange",
     "fixed-width + string + decimal128 keys on GPU"),
    ("Join", "hash equi-join inner/left/semi/anti/full, broadcast or "
     "shuffled build side", "fixed-width + string + decimal128 keys on GPU"),
    ("CrossJoin", "cartesian gather maps (+ filter for non-equi)", "all"),
    ("Generate", "explode / posexplode (+_outer) over LIST columns",
     "outer pads on CPU"),
    ("Sort", "stable LSD radix sort (incl. string + decimal128 keys); "
     "out-of-core range-partitioned spill buckets; distributed "
     "range-partitioned global ORDER BY; fused TopN", "all basic types"),
    ("Expand", "grouping-sets projections (rollup / cube)", "all"),
    ("Window", "ranking / running + bounded (ROWS and RANGE) + partition "
     "aggregates / lag / lead / ntile / nth_value",
     "GPU segmented scans + sparse-table min-max"),
    ("Limit", "row limit", "all"),
    ("Union", "concat", "all"),
    ("Sample", "deterministic murmur3 Bernoulli sample", "all"),
]


def supported_ops_doc() -> str:
    lines = [
        "# Supported operators and expressions",
        "",
        "Generated from the overrides registries "
        "(spark_rapids_amd/plan/overrides.py).",
        "",
        "## Execs", "",
        "| exec | implementation | GPU notes |", "|---|---|---|",
    ]
    for name, impl, notes in _EXECS:
        lines.append(f"| {name} | {impl} | {notes} |")
    lines += ["", "## Expressions (Spark expression -> engine mapping)", "",
              "placement: GPU = device kernels; GPU* = device within the",
              "documented condition (tagger falls back outside it); CPU =",
              "host fallback through the tagger/CpuBridge.", "",
              "| Spark expression | engine construct | placement | notes |",
              "|---|---|---|---|"]
    for nm, eng, pl, nt in _EXPR_ROWS:
        lines.append(f"| {nm} | {eng} | {pl} | {nt} |")
    lines += ["", f"({len(_EXPR_ROWS)} expression rows; "
              "GPU rows are covered by the CPU-vs-GPU equality suites "
              "under tests/)", ""]
    lines += ["", "## Binary expressions (GPU kernels)", ""]
    lines.append("`" + "`, `".join(sorted(ov._GPU_BINARY_OPS)) + "`")
    lines += ["", "## Unary expressions (GPU kernels)", ""]
    lines.append("`" + "`, `".join(sorted(ov._GPU_UNARY_OPS)) + "`")
    lines += ["", "## String expressions (GPU kernels)", ""]
    lines.append("compare: `" + "`, `".join(sorted(ov._GPU_STRING_OK)) + "`; "
                 "unary: `" + "`, `".join(sorted(ov._GPU_STRING_UNARY)) +
                 "`; plus `contains`, `starts_with`, `ends_with`, `like`, "
                 "`substring`, `split`, `concat_ws`, `element_at`/`size` "
                 "over arrays, `get_json_object` (top-level keys) — "
                 "strings.hip/csv.hip — and `rlike`, `regexp_extract`, "
                 "`regexp_extract_all`, `regexp_replace` (capture-group "
                 "regex VM, regex.hip, CPU fallback outside the subset)")
    lines += ["", "## Aggregate functions", "",
              "`sum`, `count`, `count(*)`, `min`, `max`, `avg`, `stddev`, "
              "`variance`, `first`, `last`, `count/sum(DISTINCT)`, "
              "`collect_list`, `collect_set`, `percentile`/"
              "`approx_percentile` (partial/merge lowering; collect and "
              "percentile run the single-pass path)", "",
              "## Window functions", "",
              "`row_number`, `rank`, `dense_rank`, `sum`, `count`, `min`, "
              "`max`, `avg` (running, bounded ROWS BETWEEN, whole "
              "partition, ROWS and RANGE frames), `lag`, `lead`, `ntile`, "
              "`nth_value`; SQL OVER clause", ""]
    return "\n".join(lines)


def main():
    docs = os.path.join(REPO, "docs")
    os.makedirs(docs, exist_ok=True)
    with open(os.path.join(docs, "configs.md"), "w") as f:
        f.write(help_doc())
    with open(os.path.join(docs, "supported_ops.md"), "w") as f:
        f.write(supported_ops_doc())
    print(f"wrote {docs}/configs.md and {docs}/supported_ops.md")


if __name__ == "__main__":
    main()

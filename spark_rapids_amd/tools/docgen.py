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

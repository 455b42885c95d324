"""spark_rapids_amd: MI355X-native columnar SQL engine with the capabilities
of the RAPIDS Accelerator for Apache Spark (see SURVEY.md for the blueprint).

Public surface: Session / DataFrame, the expression DSL (col, lit, when),
aggregates (sum_, avg, count, count_star, min_, max_), DType, and the
spark.rapids.* config registry.
"""
from .api import DataFrame, Session
from .column import Column, ColumnBatch, Field, Schema
from .config import RapidsConf, help_doc
from .metrics import reset_task_metrics, task_metrics
from .expr.aggregates import (approx_count_distinct, approx_percentile,
                              avg, bit_and, bit_or, bit_xor, collect_list,
                              collect_set, percentile,
                              count, count_distinct, count_star, first, last,
                              max_, min_, stddev, sum_distinct,
                              sum_, variance)
from .expr.expressions import (CaseWhen, ascii_, coalesce, col, concat_ws,
                               repeat_str, substring_index, translate,
                               date_add, date_sub,
                               datediff, dayofweek, greatest, hour, isin,
                               least, lit, quarter, to_date,
                               unix_timestamp,
                               minute, round_, second, when,
                               create_map, map_keys, map_values,
                               map_entries)
from .expr.windows import (dense_rank, lag, lead, nth_value, ntile,
                           rank, row_number, win_avg,
                           win_count, win_max, win_min, win_sum)
from .types import (BOOL, DATE32, FLOAT32, FLOAT64, INT8, INT16, INT32, INT64,
                    STRING, TIMESTAMP, DType)

__version__ = "0.1.0"

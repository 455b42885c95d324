"""Columnar op dispatch: device columns -> hand-written HIP kernels (hipdf),
host columns -> CPU reference backend.

The GPU path NEVER silently falls back to eager torch/CPU compute: if the
native extension is missing on a machine with a GPU, ops raise. This is the
boundary the reference crosses via JNI into libcudf
(reference: SURVEY.md section 2.8 kernel surface).
"""
from __future__ import annotations

from typing import List, Optional, Sequence, Tuple

from ..column import Column, ColumnBatch
from ..types import DType

from . import cpu_backend

_gpu = None
_gpu_err: Optional[str] = None


def _gpu_backend():
    global _gpu, _gpu_err
    if _gpu is None and _gpu_err is None:
        try:
            from . import gpu_backend as g
            _gpu = g
        except Exception as e:  # noqa: BLE001
            _gpu_err = f"hipdf native extension unavailable: {e!r}"
    if _gpu is None:
        raise RuntimeError(_gpu_err)
    return _gpu


def backend_for(*cols: Column):
    if any(c.is_cuda for c in cols):
        return _gpu_backend()
    return cpu_backend


def binary_op(op: str, lhs: Column, rhs: Column, out_dtype: DType) -> Column:
    return backend_for(lhs, rhs).binary_op(op, lhs, rhs, out_dtype)


def binary_op_scalar(op: str, lhs: Column, scalar, out_dtype: DType) -> Column:
    return backend_for(lhs).binary_op_scalar(op, lhs, scalar, out_dtype)


def str_split(col: Column, delimiter: str) -> Column:
    return backend_for(col).str_split(col, delimiter)


def array_size(col: Column) -> Column:
    return backend_for(col).array_size(col)


def array_contains(col: Column, value) -> Column:
    """array_contains(array, value) -> bool; null array -> null
    (GpuArrayContains analogue)."""
    return backend_for(col).array_contains(col, value)


def element_at(col: Column, index: int) -> Column:
    return backend_for(col).element_at(col, index)


def map_keys(col: Column) -> Column:
    """MAP -> LIST<key>: shares the offsets and the entry key child
    buffer (zero-copy, reference GpuMapKeys)."""
    return Column(DType.list_(col.dtype.children[0]), col.size, col.data,
                  col.validity, col.offsets, col._null_count,
                  col.child.child[0])


def map_values(col: Column) -> Column:
    return Column(DType.list_(col.dtype.children[1]), col.size, col.data,
                  col.validity, col.offsets, col._null_count,
                  col.child.child[1])


def map_entries(col: Column) -> Column:
    """MAP -> LIST<STRUCT<key,value>> (zero-copy view of the entries)."""
    return Column(DType.list_(col.dtype.entry_dtype), col.size, col.data,
                  col.validity, col.offsets, col._null_count, col.child)


def map_get(col: Column, key) -> Column:
    """element_at(map, key): the value for `key` per row, NULL when the
    key is absent or the map is null (GpuElementAt over maps)."""
    return backend_for(col).map_get(col, key)


def make_map(kcols: Sequence[Column], vcols: Sequence[Column]) -> Column:
    """create_map(k1,v1,k2,v2,...): one map per row with a fixed entry
    list; entries keep source order (no dedup; lookups are LAST_WIN —
    map_get returns the last match, matching dict() view semantics)."""
    return backend_for(*kcols, *vcols).make_map(list(kcols), list(vcols))


def regexp_extract(col: Column, pattern: str, group: int) -> Column:
    return backend_for(col).regexp_extract(col, pattern, group)


def concat_ws(sep: str, cols: Sequence[Column]) -> Column:
    return backend_for(*cols).concat_ws(sep, list(cols))


def get_json_object(col: Column, path: str) -> Column:
    return backend_for(col).get_json_object(col, path)


def regexp_extract_all(col: Column, pattern: str, group: int) -> Column:
    return backend_for(col).regexp_extract_all(col, pattern, group)


def regexp_replace(col: Column, pattern: str, replacement: str) -> Column:
    return backend_for(col).regexp_replace(col, pattern, replacement)


def decimal_mul_div(op: str, lhs: Column, rhs: Column,
                    out_dtype: DType) -> Column:
    """Exact decimal multiply/divide at the Spark result scale (operands
    keep their own scales; reference analogue: GpuMultiply/GpuDivide over
    cudf fixed-point with Spark's DecimalPrecision typing)."""
    return backend_for(lhs, rhs).decimal_mul_div(op, lhs, rhs, out_dtype)


def unary_op(op: str, col: Column, out_dtype: Optional[DType] = None) -> Column:
    return backend_for(col).unary_op(op, col, out_dtype or col.dtype)


def cast(col: Column, to: DType) -> Column:
    if col.dtype == to:
        return col
    return backend_for(col).cast(col, to)


def is_null(col: Column) -> Column:
    return backend_for(col).is_null(col)


def apply_boolean_mask(batch: ColumnBatch, mask: Column) -> ColumnBatch:
    """Filter: keep rows where mask is true (null mask = drop)."""
    return backend_for(*batch.columns, mask).apply_boolean_mask(batch, mask)


def gather(batch: ColumnBatch, indices: Column, check_bounds: bool = False,
           negatives: bool = True) -> ColumnBatch:
    """Take rows by int32 index column; negative index -> null row
    (OutOfBoundsPolicy.NULLIFY analogue for join gather maps).
    negatives=False: the caller guarantees no -1 indices (inner/cross
    maps), so non-null source columns stay mask-free — otherwise a
    validity mask re-materializes on every gathered column and defeats
    the downstream non-null fast paths."""
    return backend_for(*batch.columns, indices).gather(
        batch, indices, check_bounds, negatives)


def murmur3_hash(cols: Sequence[Column], seed: int = 42) -> Column:
    """Spark-compatible murmur3_x86_32 row hash over the columns."""
    return backend_for(*cols).murmur3_hash(list(cols), seed)


def hash_partition(batch: ColumnBatch, key_idx: Sequence[int], num_parts: int) -> Tuple[ColumnBatch, List[int]]:
    """Reorder rows so rows of one partition are contiguous; returns the
    reordered batch plus partition start offsets (len num_parts+1)."""
    return backend_for(*batch.columns).hash_partition(batch, list(key_idx), num_parts)


def reduce(op: str, col: Column):
    """Whole-column reduction -> python scalar or None."""
    return backend_for(col).reduce(op, col)


def group_by_aggregate(batch: ColumnBatch, key_idx: Sequence[int],
                       aggs: Sequence[Tuple[str, int, DType]]) -> ColumnBatch:
    """Hash group-by. aggs = [(op, value_col_idx, out_dtype)].
    Returns batch [keys..., agg results...]."""
    return backend_for(*batch.columns).group_by_aggregate(batch, list(key_idx), list(aggs))


def sort_order(batch: ColumnBatch, key_idx: Sequence[int],
               descending: Sequence[bool], nulls_last: Sequence[bool]) -> Column:
    """Return int32 permutation that sorts the batch by the keys."""
    return backend_for(*batch.columns).sort_order(batch, list(key_idx), list(descending), list(nulls_last))


def join_gather_maps(left: ColumnBatch, right: ColumnBatch,
                     left_keys: Sequence[int], right_keys: Sequence[int],
                     how: str, right_matched=None) -> Tuple[Column, Optional[Column]]:
    """Equi-join gather maps (left_map, right_map). For semi/anti only
    left_map is returned; full outer also records matched build rows into
    right_matched."""
    return backend_for(*left.columns, *right.columns).join_gather_maps(
        left, right, list(left_keys), list(right_keys), how, right_matched)


def concat_batches(batches: Sequence[ColumnBatch]) -> ColumnBatch:
    assert batches
    return backend_for(*batches[0].columns).concat_batches(list(batches))


def str_pad(col: Column, width: int, fill: str, left: bool) -> Column:
    return backend_for(col).str_pad(col, width, fill, left)


def str_locate(col: Column, substr: str, pos: int = 1) -> Column:
    return backend_for(col).str_locate(col, substr, pos)


def tz_convert(col: Column, zone: str, to_utc: bool) -> Column:
    return backend_for(col).tz_convert(col, zone, to_utc)


def date_format(col: Column, tokens, width: int) -> Column:
    return backend_for(col).date_format(col, tokens, width)


def ts_parse(col: Column, tokens, width: int) -> Column:
    return backend_for(col).ts_parse(col, tokens, width)

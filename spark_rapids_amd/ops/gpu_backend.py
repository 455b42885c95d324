"""GPU backend: drives the hipdf CDNA4 kernels over device Columns.

Buffers live in the PyTorch-ROCm caching allocator (the RMM-style pool);
every columnar computation below is a hand-written HIP kernel from
native/hipdf — torch is used only for allocation, D2H/H2D movement and
byte-level cat/zero (memcpy/memset-class work). If the native extension is
missing this module fails to import and ops.backend_for raises — there is
no silent eager fallback on GPU.
"""
from __future__ import annotations

import struct
from typing import List, Optional, Tuple

import numpy as np
import torch

from ..column import Column, ColumnBatch, mask_nbytes, torch_dtype
from ..types import DType, TypeId

import importlib


def _load_ext():
    try:
        return importlib.import_module("hipdf")
    except ImportError as e:
        raise ImportError(
            "hipdf native extension not built; run python native/hipdf/build.py"
        ) from e


ext = _load_ext()

# ---------------------------------------------------------------------------
# enums shared with the kernels
# ---------------------------------------------------------------------------
_HT = {
    TypeId.BOOL: 0, TypeId.INT8: 1, TypeId.INT16: 2, TypeId.INT32: 3,
    TypeId.DATE32: 3, TypeId.INT64: 4, TypeId.TIMESTAMP: 4,
    TypeId.DECIMAL64: 4, TypeId.FLOAT32: 5, TypeId.FLOAT64: 6,
}
_BIN_OPS = {
    "add": 0, "sub": 1, "mul": 2, "div": 3, "int_div": 4, "mod": 5,
    "pmod": 6, "pow": 7, "eq": 8, "ne": 9, "lt": 10, "le": 11, "gt": 12,
    "ge": 13, "eq_null_safe": 14, "and": 15, "or": 16, "bitand": 17,
    "bitor": 18, "bitxor": 19, "shiftleft": 20, "shiftright": 21,
    "min": 22, "max": 23, "round": 24,
}
_CMP_OPS = {"eq", "ne", "lt", "le", "gt", "ge", "eq_null_safe"}
_BOOL_OPS = {"and", "or"}
_NULL_PRODUCING = {"div", "int_div", "mod", "pmod"}
_UN_OPS = {
    "neg": 0, "abs": 1, "not": 2, "sqrt": 3, "exp": 4, "log": 5, "floor": 6,
    "ceil": 7, "sin": 8, "cos": 9, "tan": 10, "is_nan": 11, "year": 12,
    "month": 13, "day": 14,
}
_HK_INT, _HK_LONG, _HK_FLOAT, _HK_DOUBLE = 0, 1, 2, 3
_RED = {"sum": 0, "min": 1, "max": 2, "count": 3}
_GB = {"sum": 0, "min": 1, "max": 2, "count": 3, "count_all": 4,
       "first": 5, "last": 5, "bit_and": 6, "bit_or": 7, "bit_xor": 8}
_JOIN = {"inner": 0, "left": 1, "semi": 2, "anti": 3, "full": 4}





def _stream() -> int:
    # follow torch's CURRENT stream: the prefetch pool decodes files on
    # side streams (with torch.cuda.stream(...)) to overlap H2D + decode
    # with main-stream compute, so hipdf kernels must land on whatever
    # stream the calling thread has current
    return torch.cuda.current_stream().cuda_stream


def set_stream(handle: Optional[int]) -> None:
    # retained for API compatibility: _stream() now always follows
    # torch.cuda.current_stream(), which set_stream callers control via
    # `with torch.cuda.stream(...)`
    del handle


def _ptr(t: Optional[torch.Tensor]) -> int:
    return 0 if t is None else t.data_ptr()


def _ht(dt: DType) -> int:
    return _HT[dt.id]


def _alloc(n: int, dt: DType) -> torch.Tensor:
    words = 2 * n if dt.id is TypeId.DECIMAL128 else n
    if words == 0:
        return torch.empty(1, dtype=torch_dtype(dt), device="cuda")[:0]
    return torch.empty(words, dtype=torch_dtype(dt), device="cuda")


def _alloc_mask(n: int) -> torch.Tensor:
    return torch.empty(mask_nbytes(n), dtype=torch.uint8, device="cuda")


def _empty_col(dtype: DType) -> Column:
    if dtype.id is TypeId.STRING:
        return Column(dtype, 0, torch.zeros(0, dtype=torch.uint8, device="cuda"),
                      None, torch.zeros(1, dtype=torch.int32, device="cuda"), 0)
    if dtype.id is TypeId.STRUCT:
        return Column(dtype, 0,
                      torch.zeros(0, dtype=torch.uint8, device="cuda"),
                      None, None, 0,
                      tuple(_empty_col(c) for c in dtype.children))
    return Column(dtype, 0, torch.zeros(0, dtype=torch_dtype(dtype),
                                        device="cuda"), None, None, 0)


_I128_ARITH = {"add": 0, "sub": 1, "min": 22, "max": 23}


# ---------------------------------------------------------------------------
# elementwise
# ---------------------------------------------------------------------------

def binary_op(op: str, lhs: Column, rhs: Column, out_dtype: DType) -> Column:
    return _binary(op, lhs, rhs, None, out_dtype)


def binary_op_scalar(op: str, lhs: Column, scalar, out_dtype: DType) -> Column:
    return _binary(op, lhs, None, scalar, out_dtype)


_STR_CMP = {"eq": 0, "ne": 1, "lt": 2, "le": 3, "gt": 4, "ge": 5}


def decimal_mul_div(op: str, lhs: Column, rhs: Column,
                    out_dtype: DType) -> Column:
    """Exact decimal multiply/divide on device. dec64 operands ride
    k_dec64_mul_div (__int128 intermediates); any decimal128 operand
    routes to k_dec_mul_div_wide (256-bit product / u192-remainder long
    division). NULL on div-by-zero / overflow."""
    n = lhs.size
    s = _stream()
    s1, s2, st = lhs.dtype.scale, rhs.dtype.scale, out_dtype.scale
    shift = (s1 + s2 - st) if op == "mul" else (st + s2 - s1)
    is128 = out_dtype.id is TypeId.DECIMAL128
    width = 2 * n if is128 else n
    out = torch.empty(max(width, 1), dtype=torch.int64,
                      device="cuda")[:width]
    ov = _alloc_mask(n)
    a128 = lhs.dtype.id is TypeId.DECIMAL128
    b128 = rhs.dtype.id is TypeId.DECIMAL128
    if n and not (a128 or b128):
        ext.dec64_mul_div(1 if op == "div" else 0, lhs.data.data_ptr(),
                          rhs.data.data_ptr(), _ptr(lhs.validity),
                          _ptr(rhs.validity), out.data_ptr(), ov.data_ptr(),
                          1 if is128 else 0, shift, out_dtype.precision,
                          n, s)
    elif n:
        ext.dec_mul_div_wide(1 if op == "div" else 0, lhs.data.data_ptr(),
                             rhs.data.data_ptr(), _ptr(lhs.validity),
                             _ptr(rhs.validity), 1 if a128 else 0,
                             1 if b128 else 0, out.data_ptr(),
                             ov.data_ptr(), 1 if is128 else 0, shift,
                             out_dtype.precision, n, s)
    return Column(out_dtype, n, out, ov, null_count=None)


def _binary(op, lhs: Column, rhs: Optional[Column], scalar, out_dtype) -> Column:
    n = lhs.size
    s = _stream()
    scalar_rhs = rhs is None
    if lhs.dtype.id is TypeId.STRING:
        if op == "concat" and not scalar_rhs:
            return str_concat(lhs, rhs)
        if op not in _STR_CMP:
            raise NotImplementedError(f"string op {op} not on GPU")
        out = _alloc(n, out_dtype)
        if scalar_rhs:
            pat = _pattern_tensor(scalar)
            ext.str_cmp_scalar(_STR_CMP[op], lhs.offsets.data_ptr(),
                               lhs.data.data_ptr(), pat.data_ptr(),
                               pat.numel(), out.data_ptr(), n, s)
            v = lhs.validity.clone() if lhs.validity is not None else None
            return Column(out_dtype, n, out, v, null_count=lhs._null_count)
        ext.str_cmp(_STR_CMP[op], lhs.offsets.data_ptr(), lhs.data.data_ptr(),
                    rhs.offsets.data_ptr(), rhs.data.data_ptr(),
                    out.data_ptr(), n, s)
        # combine validities via the bool AND kernel on expanded masks
        return _with_and_validity(out, lhs, rhs, out_dtype, n, s)
    if lhs.dtype.id is TypeId.DECIMAL128:
        if scalar_rhs:
            raise NotImplementedError("decimal128 scalar ops not on GPU")
        ov = _alloc_mask(n)
        if op in _I128_ARITH and op in ("add", "sub", "min", "max"):
            out = _alloc(n, out_dtype)
            ext.i128_arith(_I128_ARITH[op], lhs.data.data_ptr(),
                           rhs.data.data_ptr(), _ptr(lhs.validity),
                           _ptr(rhs.validity), out.data_ptr(), ov.data_ptr(),
                           n, s)
            return Column(out_dtype, n, out, ov, null_count=None)
        if op in _STR_CMP:
            out = _alloc(n, out_dtype)
            ext.i128_cmp(_STR_CMP[op], lhs.data.data_ptr(),
                         rhs.data.data_ptr(), _ptr(lhs.validity),
                         _ptr(rhs.validity), out.data_ptr(), ov.data_ptr(),
                         n, s)
            has_valid = lhs.validity is not None or rhs.validity is not None
            return Column(out_dtype, n, out, ov if has_valid else None,
                          null_count=None if has_valid else 0)
        raise NotImplementedError(f"decimal128 op {op} not on GPU")
    t = _ht(lhs.dtype)
    sd, si = 0.0, 0
    if scalar_rhs and scalar is not None:
        if lhs.dtype.is_floating:
            sd = float(scalar)
        else:
            si = int(scalar)
    av = _ptr(lhs.validity)
    bv = 0 if scalar_rhs else _ptr(rhs.validity)
    has_in_valid = lhs.validity is not None or (not scalar_rhs and rhs.validity is not None)

    if op in _BOOL_OPS:
        out = _alloc(n, out_dtype)
        ov = _alloc_mask(n)  # Kleene output can always carry nulls
        ext.binary_bool(_BIN_OPS[op], lhs.data.data_ptr(),
                        0 if scalar_rhs else rhs.data.data_ptr(),
                        int(bool(si or sd)) if scalar_rhs else 0, scalar_rhs,
                        av, bv, out.data_ptr(), ov.data_ptr(), n, s)
        if not has_in_valid:
            return Column(out_dtype, n, out, None, null_count=0)
        return Column(out_dtype, n, out, ov, null_count=None)
    if op in _CMP_OPS:
        out = _alloc(n, out_dtype)
        need_mask = has_in_valid and op != "eq_null_safe"
        ov = _alloc_mask(n) if need_mask else None
        ext.binary_cmp(_BIN_OPS[op], t, lhs.data.data_ptr(),
                       0 if scalar_rhs else rhs.data.data_ptr(), sd, si,
                       scalar_rhs, av, bv, out.data_ptr(), _ptr(ov), n, s)
        return Column(out_dtype, n, out, ov,
                      null_count=None if need_mask else 0)
    out = _alloc(n, out_dtype)
    need_mask = has_in_valid or op in _NULL_PRODUCING
    ov = _alloc_mask(n) if need_mask else None
    ext.binary_arith(_BIN_OPS[op], t, lhs.data.data_ptr(),
                     0 if scalar_rhs else rhs.data.data_ptr(), sd, si,
                     scalar_rhs, av, bv, out.data_ptr(), _ptr(ov), n, s)
    return Column(out_dtype, n, out, ov, null_count=None if need_mask else 0)


def _pattern_tensor(pattern) -> torch.Tensor:
    b = str(pattern).encode("utf-8")
    if not b:
        return torch.zeros(1, dtype=torch.uint8, device="cuda")[:0]
    return torch.frombuffer(bytearray(b), dtype=torch.uint8).cuda()


def _and_masks(a: Optional[torch.Tensor], b: Optional[torch.Tensor]):
    if a is None:
        return b.clone() if b is not None else None
    if b is None:
        return a.clone()
    n = min(a.numel(), b.numel())
    return (a[:n].view(torch.int32) & b[:n].view(torch.int32)).view(torch.uint8)


def _with_and_validity(out, lhs, rhs, out_dtype, n, s):
    v = _and_masks(lhs.validity, rhs.validity)
    return Column(out_dtype, n, out, v,
                  null_count=None if v is not None else 0)


def str_predicate(op: str, col: Column, pattern: str) -> Column:
    n = col.size
    s = _stream()
    out = _alloc(n, DType.bool_())
    if op == "rlike":
        from .regex_compiler import compile_regex

        prog = compile_regex(pattern)  # tagger pre-checks; raise = bug
        flat = []
        for o, a0, a1 in prog.ops:
            flat.extend([o, a0, a1])
        prog_t = torch.tensor(flat, dtype=torch.int32).cuda()
        cls_blob = b"".join(prog.classes) or b"\x00" * 32
        cls_t = torch.frombuffer(bytearray(cls_blob),
                                 dtype=torch.uint8).cuda()
        overflow = torch.zeros(1, dtype=torch.int32, device="cuda")
        ext.regex_match(prog_t.data_ptr(), len(prog.ops), cls_t.data_ptr(),
                        col.offsets.data_ptr(), col.data.data_ptr(),
                        out.data_ptr(), overflow.data_ptr(), n, s)
        if int(overflow.item()) > 0:
            # pathological backtracking: redo the whole column on CPU
            from . import cpu_backend

            host = cpu_backend.str_predicate(op, col.cpu(), pattern)
            return host.cuda()
        v = col.validity.clone() if col.validity is not None else None
        return Column(DType.bool_(), n, out, v, null_count=col._null_count)
    pat = _pattern_tensor(pattern)
    if op == "like":
        ext.str_like(col.offsets.data_ptr(), col.data.data_ptr(),
                     pat.data_ptr(), pat.numel(), out.data_ptr(), n, s)
    else:
        mode = {"contains": 0, "starts_with": 1, "ends_with": 2}[op]
        ext.str_find(mode, col.offsets.data_ptr(), col.data.data_ptr(),
                     pat.data_ptr(), pat.numel(), out.data_ptr(), n, s)
    v = col.validity.clone() if col.validity is not None else None
    return Column(DType.bool_(), n, out, v, null_count=col._null_count)


def _rx_prog_tensors(pattern: str):
    from .regex_compiler import compile_regex

    prog = compile_regex(pattern)
    flat = []
    for o, a0, a1 in prog.ops:
        flat.extend([o, a0, a1])
    prog_t = torch.tensor(flat, dtype=torch.int32).cuda()
    cls_blob = b"".join(prog.classes) or b"\x00" * 32
    cls_t = torch.frombuffer(bytearray(cls_blob), dtype=torch.uint8).cuda()
    return prog, prog_t, cls_t


def regexp_extract(col: Column, pattern: str, group: int) -> Column:
    n = col.size
    s = _stream()
    if n == 0:
        return _empty_col(DType.string())
    prog, prog_t, cls_t = _rx_prog_tensors(pattern)
    if group > prog.ngroups:
        raise ValueError(f"regexp_extract group {group} > "
                         f"{prog.ngroups} capture groups")
    starts = torch.empty(n, dtype=torch.int32, device="cuda")
    lens = torch.empty(n, dtype=torch.int64, device="cuda")
    overflow = torch.zeros(1, dtype=torch.int32, device="cuda")
    ext.regex_extract(prog_t.data_ptr(), len(prog.ops), cls_t.data_ptr(),
                      col.offsets.data_ptr(), col.data.data_ptr(), group,
                      starts.data_ptr(), lens.data_ptr(),
                      overflow.data_ptr(), n, s)
    if int(overflow.item()) > 0:
        from . import cpu_backend

        return cpu_backend.regexp_extract(col.cpu(), pattern, group).cuda()
    return _strings_from_spans(col, starts, lens, n, s)


def _strings_from_spans(col: Column, starts, lens, n, s) -> Column:
    """Assemble a string column from absolute (start, len) spans over the
    source bytes (substr_copy compaction)."""
    scanned, total = _exclusive_scan_i64(lens)
    out_bytes = torch.empty(max(total, 1), dtype=torch.uint8,
                            device="cuda")[:total]
    if total:
        ext.substr_copy(col.data.data_ptr(), starts.data_ptr(),
                        lens.data_ptr(), scanned.data_ptr(),
                        out_bytes.data_ptr(), n, s)
    offs = torch.empty(n + 1, dtype=torch.int32, device="cuda")
    ext.narrow_i64_i32(scanned.data_ptr(), offs.data_ptr(), n, s)
    offs[n] = total
    v = col.validity.clone() if col.validity is not None else None
    return Column(DType.string(), n, out_bytes, v, offs,
                  null_count=col._null_count)


def _compile_replacement(replacement: str, ngroups: int):
    """Java appendReplacement template -> (ops int32 triples, literal
    bytes): $g group refs, backslash escapes."""
    ops_l = []
    lit = bytearray()
    cur = bytearray()

    def flush():
        nonlocal cur
        if cur:
            ops_l.append((0, len(lit), len(cur)))
            lit.extend(cur)
            cur = bytearray()

    i = 0
    while i < len(replacement):
        c = replacement[i]
        if c == "$" and i + 1 < len(replacement) and \
                replacement[i + 1].isdigit():
            g = int(replacement[i + 1])
            if g > ngroups:
                raise ValueError(f"replacement group ${g} out of range")
            flush()
            ops_l.append((1, g, 0))
            i += 2
        elif c == "\\" and i + 1 < len(replacement):
            cur.extend(replacement[i + 1].encode("utf-8"))
            i += 2
        else:
            cur.extend(c.encode("utf-8"))
            i += 1
    flush()
    return ops_l, bytes(lit)


def concat_ws(sep: str, cols) -> Column:
    n = cols[0].size
    s = _stream()
    if n == 0:
        return _empty_col(DType.string())
    blob = b"".join(struct.pack("<qqq", c.offsets.data_ptr(),
                                c.data.data_ptr(), _ptr(c.validity))
                    for c in cols)
    desc = torch.frombuffer(bytearray(blob), dtype=torch.uint8).cuda()
    sep_t = _pattern_tensor(sep or "\x00")
    seplen = len(sep.encode("utf-8"))
    lens = torch.empty(n, dtype=torch.int64, device="cuda")
    ext.str_concat_ws(desc.data_ptr(), len(cols), sep_t.data_ptr(), seplen,
                      0, lens.data_ptr(), 0, 0, n, s)
    scanned, total = _exclusive_scan_i64(lens)
    out = torch.empty(max(total, 1), dtype=torch.uint8,
                      device="cuda")[:total]
    if total:
        ext.str_concat_ws(desc.data_ptr(), len(cols), sep_t.data_ptr(),
                          seplen, scanned.data_ptr(), lens.data_ptr(),
                          out.data_ptr(), 1, n, s)
    offs = torch.empty(n + 1, dtype=torch.int32, device="cuda")
    ext.narrow_i64_i32(scanned.data_ptr(), offs.data_ptr(), n, s)
    offs[n] = total
    return Column(DType.string(), n, out, None, offs, 0)


def get_json_object(col: Column, path: str) -> Column:
    """Top-level scalar key extraction ($.key) via k_json_field over the
    string column's own row spans."""
    keys = [k for k in path[1:].lstrip(".").split(".") if k]
    if len(keys) != 1:
        raise NotImplementedError("gpu get_json_object: nested path")
    n = col.size
    s = _stream()
    if n == 0:
        return _empty_col(DType.string())
    nameb = keys[0].encode("utf-8")
    name_t = torch.frombuffer(bytearray(nameb), dtype=torch.uint8).cuda()
    ss = torch.empty(n, dtype=torch.int32, device="cuda")
    sl = torch.empty(n, dtype=torch.int64, device="cuda")
    valid_u8 = torch.empty(n, dtype=torch.uint8, device="cuda")
    unsupported = torch.zeros(1, dtype=torch.int32, device="cuda")
    row_start = col.offsets[:n]
    row_end = col.offsets[1:]
    ext.json_field(col.data.data_ptr(), row_start.data_ptr(),
                   row_end.data_ptr(), name_t.data_ptr(), len(nameb), 4,
                   0, 0, ss.data_ptr(), sl.data_ptr(), valid_u8.data_ptr(),
                   unsupported.data_ptr(), n, s)
    if int(unsupported.item()) > 0:
        from . import cpu_backend

        return cpu_backend.get_json_object(col.cpu(), path).cuda()
    out = _strings_from_spans(col, ss, sl, n, s)
    from ..column import mask_nbytes

    v64 = torch.empty(n, dtype=torch.int64, device="cuda")
    ext.cast(0, 4, valid_u8.data_ptr(), v64.data_ptr(), n, s)
    mask = torch.empty(mask_nbytes(n), dtype=torch.uint8, device="cuda")
    ext.mask_from_nonzero(v64.data_ptr(), mask.data_ptr(), n, s)
    if col.validity is not None:
        mask = _and_masks(col.validity, mask)
    return Column(DType.string(), n, out.data, mask, out.offsets,
                  null_count=None)


def regexp_extract_all(col: Column, pattern: str, group: int) -> Column:
    """All matches' group text per row as LIST<STRING>."""
    n = col.size
    s = _stream()
    lt = DType.list_(DType.string())
    if n == 0:
        return Column.from_pylist([], lt).cuda()
    prog, prog_t, cls_t = _rx_prog_tensors(pattern)
    if group > prog.ngroups:
        raise ValueError(f"group {group} > {prog.ngroups}")
    counts = torch.empty(n, dtype=torch.int64, device="cuda")
    overflow = torch.zeros(1, dtype=torch.int32, device="cuda")
    ext.regex_extract_all(prog_t.data_ptr(), len(prog.ops),
                          cls_t.data_ptr(), col.offsets.data_ptr(),
                          col.data.data_ptr(), group, 0, counts.data_ptr(),
                          0, 0, 0, overflow.data_ptr(), n, s)
    if int(overflow.item()) > 0:
        from . import cpu_backend

        return cpu_backend.regexp_extract_all(col.cpu(), pattern,
                                              group).cuda()
    part_off, total = _exclusive_scan_i64(counts)
    ss = torch.empty(max(total, 1), dtype=torch.int32, device="cuda")[:total]
    sl = torch.empty(max(total, 1), dtype=torch.int64, device="cuda")[:total]
    if total:
        ext.regex_extract_all(prog_t.data_ptr(), len(prog.ops),
                              cls_t.data_ptr(), col.offsets.data_ptr(),
                              col.data.data_ptr(), group,
                              part_off.data_ptr(), counts.data_ptr(),
                              ss.data_ptr(), sl.data_ptr(), 1,
                              overflow.data_ptr(), n, s)
    scanned, nbytes = _exclusive_scan_i64(sl) if total else (sl, 0)
    out_bytes = torch.empty(max(nbytes, 1), dtype=torch.uint8,
                            device="cuda")[:nbytes]
    if nbytes:
        ext.substr_copy(col.data.data_ptr(), ss.data_ptr(), sl.data_ptr(),
                        scanned.data_ptr(), out_bytes.data_ptr(), total, s)
    coffs = torch.empty(total + 1, dtype=torch.int32, device="cuda")
    if total:
        ext.narrow_i64_i32(scanned.data_ptr(), coffs.data_ptr(), total, s)
    coffs[total] = nbytes
    child = Column(DType.string(), total, out_bytes, None, coffs, 0)
    loffs = torch.empty(n + 1, dtype=torch.int32, device="cuda")
    ext.narrow_i64_i32(part_off.data_ptr(), loffs.data_ptr(), n, s)
    loffs[n] = total
    v = col.validity.clone() if col.validity is not None else None
    return Column(lt, n, torch.zeros(0, dtype=torch.uint8, device="cuda"),
                  v, loffs, col._null_count, child)


def regexp_replace(col: Column, pattern: str, replacement: str) -> Column:
    n = col.size
    s = _stream()
    if n == 0:
        return _empty_col(DType.string())
    prog, prog_t, cls_t = _rx_prog_tensors(pattern)
    ops_l, lit = _compile_replacement(replacement, prog.ngroups)
    flat = []
    for k, a, b in ops_l:
        flat.extend([k, a, b])
    repl_t = torch.tensor(flat or [0, 0, 0], dtype=torch.int32).cuda()
    lit_t = torch.frombuffer(bytearray(lit or b"\x00"),
                             dtype=torch.uint8).cuda()
    lens = torch.empty(n, dtype=torch.int64, device="cuda")
    overflow = torch.zeros(1, dtype=torch.int32, device="cuda")
    ext.regex_replace(prog_t.data_ptr(), len(prog.ops), cls_t.data_ptr(),
                      col.offsets.data_ptr(), col.data.data_ptr(),
                      repl_t.data_ptr(), len(ops_l), lit_t.data_ptr(),
                      0, lens.data_ptr(), 0, 0, overflow.data_ptr(), n, s)
    if int(overflow.item()) > 0:
        from . import cpu_backend

        return cpu_backend.regexp_replace(col.cpu(), pattern,
                                          replacement).cuda()
    scanned, total = _exclusive_scan_i64(lens)
    out_bytes = torch.empty(max(total, 1), dtype=torch.uint8,
                            device="cuda")[:total]
    if total:
        ext.regex_replace(prog_t.data_ptr(), len(prog.ops), cls_t.data_ptr(),
                          col.offsets.data_ptr(), col.data.data_ptr(),
                          repl_t.data_ptr(), len(ops_l), lit_t.data_ptr(),
                          scanned.data_ptr(), lens.data_ptr(),
                          out_bytes.data_ptr(), 1, overflow.data_ptr(), n, s)
    offs = torch.empty(n + 1, dtype=torch.int32, device="cuda")
    ext.narrow_i64_i32(scanned.data_ptr(), offs.data_ptr(), n, s)
    offs[n] = total
    v = col.validity.clone() if col.validity is not None else None
    return Column(DType.string(), n, out_bytes, v, offs,
                  null_count=col._null_count)


def str_trim(col: Column, mode: str) -> Column:
    """trim/ltrim/rtrim (ascii space, Spark default)."""
    n = col.size
    s = _stream()
    if n == 0:
        return _empty_col(DType.string())
    m = {"both": 0, "leading": 1, "trailing": 2, "cast": 3}[mode]
    bstart = torch.empty(n, dtype=torch.int32, device="cuda")
    blen = torch.empty(n, dtype=torch.int64, device="cuda")
    ext.str_trim_ranges(m, col.offsets.data_ptr(), col.data.data_ptr(),
                        bstart.data_ptr(), blen.data_ptr(), n, s)
    return _strings_from_spans(col, bstart, blen, n, s)


def str_concat(a: Column, b: Column) -> Column:
    """concat(a, b): NULL if either side is NULL (Spark concat)."""
    n = a.size
    s = _stream()
    if n == 0:
        return _empty_col(DType.string())
    lens = torch.empty(n, dtype=torch.int64, device="cuda")
    ext.str_concat2(a.offsets.data_ptr(), a.data.data_ptr(),
                    b.offsets.data_ptr(), b.data.data_ptr(), 0,
                    lens.data_ptr(), 0, 0, n, s)
    scanned, total = _exclusive_scan_i64(lens)
    out_bytes = torch.empty(max(total, 1), dtype=torch.uint8,
                            device="cuda")[:total]
    if total:
        ext.str_concat2(a.offsets.data_ptr(), a.data.data_ptr(),
                        b.offsets.data_ptr(), b.data.data_ptr(),
                        scanned.data_ptr(), lens.data_ptr(),
                        out_bytes.data_ptr(), 1, n, s)
    offs = torch.empty(n + 1, dtype=torch.int32, device="cuda")
    ext.narrow_i64_i32(scanned.data_ptr(), offs.data_ptr(), n, s)
    offs[n] = total
    v = _and_masks(a.validity, b.validity)
    return Column(DType.string(), n, out_bytes, v, offs,
                  null_count=None if v is not None else 0)


def str_split(col: Column, delimiter: str) -> Column:
    """Literal-delimiter split into LIST<STRING> (see StrSplit)."""
    n = col.size
    s = _stream()
    lt = DType.list_(DType.string())
    if n == 0:
        return Column.from_pylist([], lt).cuda()
    d = _pattern_tensor(delimiter)
    counts = torch.empty(n, dtype=torch.int64, device="cuda")
    ext.str_split_count(col.offsets.data_ptr(), col.data.data_ptr(),
                        d.data_ptr(), d.numel(), counts.data_ptr(), n, s)
    part_off, total = _exclusive_scan_i64(counts)
    ss = torch.empty(max(total, 1), dtype=torch.int32, device="cuda")[:total]
    sl = torch.empty(max(total, 1), dtype=torch.int64, device="cuda")[:total]
    if total:
        ext.str_split_fill(col.offsets.data_ptr(), col.data.data_ptr(),
                           d.data_ptr(), d.numel(), part_off.data_ptr(),
                           counts.data_ptr(), ss.data_ptr(), sl.data_ptr(),
                           n, s)
    # child strings from (start, len) spans
    scanned, nbytes = _exclusive_scan_i64(sl) if total else (sl, 0)
    out_bytes = torch.empty(max(nbytes, 1), dtype=torch.uint8,
                            device="cuda")[:nbytes]
    if nbytes:
        ext.substr_copy(col.data.data_ptr(), ss.data_ptr(), sl.data_ptr(),
                        scanned.data_ptr(), out_bytes.data_ptr(), total, s)
    coffs = torch.empty(total + 1, dtype=torch.int32, device="cuda")
    if total:
        ext.narrow_i64_i32(scanned.data_ptr(), coffs.data_ptr(), total, s)
    coffs[total] = nbytes
    child = Column(DType.string(), total, out_bytes, None, coffs, 0)
    loffs = torch.empty(n + 1, dtype=torch.int32, device="cuda")
    ext.narrow_i64_i32(part_off.data_ptr(), loffs.data_ptr(), n, s)
    loffs[n] = total
    v = col.validity.clone() if col.validity is not None else None
    return Column(lt, n, torch.zeros(0, dtype=torch.uint8, device="cuda"),
                  v, loffs, col._null_count, child)


def element_at(col: Column, index: int) -> Column:
    """Gather child elements at per-row offset +- index with bounds
    nullification (reuses the negative-index gather)."""
    n = col.size
    s = _stream()
    elem_dt = col.dtype.children[0]
    if n == 0:
        return _empty_col(elem_dt)
    starts = Column(DType.int32(), n, col.offsets[:n], None, null_count=0)
    ends = Column(DType.int32(), n, col.offsets[1:], None, null_count=0)
    if index > 0:
        idx = binary_op_scalar("add", starts, index - 1, DType.int32())
        ok = binary_op("lt", idx, ends, DType.bool_())
    else:
        idx = binary_op_scalar("add", ends, index, DType.int32())
        ok = binary_op("ge", idx, starts, DType.bool_())
    if col.validity is not None:
        # null list -> null element: fold into the gather index (-1 rows
        # nullify) so string elements need no typed if_else
        nn = unary_op("not", is_null(col), DType.bool_())
        ok = binary_op("and", ok, nn, DType.bool_())
    neg1 = Column.full(-1, DType.int32(), n, "cuda")
    srcm = if_else(ok, idx, neg1)
    return _gather_col(col.child, srcm.data, n, maybe_negative=True)


def array_contains(col: Column, value) -> Column:
    """Entry-level compare + per-segment any() via scatter amax."""
    n = col.size
    if n == 0:
        return _empty_col(DType.bool_())
    elem = col.child
    ne = elem.size
    hit = torch.zeros(n, dtype=torch.uint8, device="cuda")
    if ne:
        if elem.dtype.id is TypeId.STRING:
            sw = str_predicate("starts_with", elem, value)
            ln = unary_op("length", elem, DType.int32())
            lm = binary_op_scalar("eq", ln, len(value), DType.bool_())
            match = binary_op("and", sw, lm, DType.bool_())
        else:
            match = binary_op_scalar("eq", elem, value, DType.bool_())
        mt = match.data[:ne].bool()
        if elem.validity is not None:
            ev = torch.empty(ne, dtype=torch.uint8, device="cuda")
            ext.mask_expand(elem.validity.data_ptr(), ev.data_ptr(),
                            False, ne, _stream())
            mt &= ev.bool()
        offs = col.offsets.long()
        counts = offs[1:n + 1] - offs[:n]
        seg = torch.repeat_interleave(
            torch.arange(n, device="cuda", dtype=torch.int64), counts)
        if mt.any():
            hit.scatter_reduce_(0, seg[mt],
                                torch.ones(int(mt.sum()),
                                           dtype=torch.uint8,
                                           device="cuda"),
                                reduce="amax", include_self=True)
    v = col.validity.clone() if col.validity is not None else None
    return Column(DType.bool_(), n, hit, v, null_count=col._null_count)


def map_get(col: Column, key) -> Column:
    """element_at(map, key): entry-level key compare + per-segment
    last-match reduction (torch scatter_reduce amax over the entry iota),
    then a nullable gather of the value child."""
    n = col.size
    vdt = col.dtype.children[1]
    if n == 0:
        return _empty_col(vdt)
    entry = col.child
    kcol, vcol = entry.child
    ne = kcol.size
    if ne == 0:
        out = _gather_col(vcol, torch.full((n,), -1, dtype=torch.int32,
                                           device="cuda"), n,
                          maybe_negative=True)
        return out
    if kcol.dtype.id is TypeId.STRING:
        sw = str_predicate("starts_with", kcol, key)
        ln = unary_op("length", kcol, DType.int32())
        lm = binary_op_scalar("eq", ln, len(key), DType.bool_())
        match = binary_op("and", sw, lm, DType.bool_())
    else:
        match = binary_op_scalar("eq", kcol, key, DType.bool_())
    mt = match.data[:ne].bool()
    offs = col.offsets.long()
    counts = offs[1:n + 1] - offs[:n]
    seg = torch.repeat_interleave(
        torch.arange(n, device="cuda", dtype=torch.int64), counts)
    eidx = torch.arange(ne, device="cuda", dtype=torch.int64)
    res = torch.full((n,), -1, dtype=torch.int64, device="cuda")
    if mt.any():
        res.scatter_reduce_(0, seg[mt], eidx[mt], reduce="amax",
                            include_self=True)
    if col.validity is not None:
        rv = torch.empty(n, dtype=torch.uint8, device="cuda")
        ext.mask_expand(col.validity.data_ptr(), rv.data_ptr(), False, n,
                        _stream())
        res = torch.where(rv.bool(), res,
                          torch.full_like(res, -1))
    return _gather_col(vcol, res.to(torch.int32), n, maybe_negative=True)


def make_map(kcols, vcols) -> Column:
    """create_map: interleave the per-pair key/value columns into entry
    children (block concat + permutation gather) with row-major offsets."""
    npairs = len(kcols)
    n = kcols[0].size
    dtype = DType.map_(kcols[0].dtype, vcols[0].dtype)
    perm_np = (np.arange(n * npairs, dtype=np.int32) % npairs) * n \
        + np.arange(n * npairs, dtype=np.int32) // npairs
    perm = torch.from_numpy(perm_np).cuda()
    def inter(cols):
        if len(cols) == 1:
            return cols[0]
        blk = concat_batches([ColumnBatch([c], n) for c in cols]).columns[0]
        return _gather_col(blk, perm, n * npairs, maybe_negative=False)
    kc = inter(kcols)
    vc = inter(vcols)
    entry = Column(dtype.entry_dtype, n * npairs,
                   torch.zeros(0, dtype=torch.uint8, device="cuda"),
                   None, None, 0, (kc, vc))
    offsets = torch.arange(0, (n + 1) * npairs, npairs,
                           dtype=torch.int32, device="cuda")
    return Column(dtype, n, torch.zeros(0, dtype=torch.uint8,
                                        device="cuda"),
                  None, offsets, 0, entry)


def array_size(col: Column) -> Column:
    n = col.size
    s = _stream()
    if n == 0:
        return _empty_col(DType.int32())
    out = torch.empty(n, dtype=torch.int32, device="cuda")
    # sizes = offsets[i+1] - offsets[i] via the int32 sub kernel on views
    a = Column(DType.int32(), n, col.offsets[1:], None, null_count=0)
    b = Column(DType.int32(), n, col.offsets[:n], None, null_count=0)
    diff = _binary("sub", a, b, None, DType.int32())
    v = col.validity.clone() if col.validity is not None else None
    return Column(DType.int32(), n, diff.data, v,
                  null_count=col._null_count)


def substring(col: Column, pos: int, length: int = -1) -> Column:
    n = col.size
    s = _stream()
    if n == 0:
        return _empty_col(DType.string())
    bstart = torch.empty(n, dtype=torch.int32, device="cuda")
    blen = torch.empty(n, dtype=torch.int64, device="cuda")
    ext.substr_ranges(col.offsets.data_ptr(), col.data.data_ptr(), pos,
                      length, bstart.data_ptr(), blen.data_ptr(), n, s)
    scanned, total = _exclusive_scan_i64(blen)
    out_bytes = torch.empty(max(total, 1), dtype=torch.uint8,
                            device="cuda")[:total]
    if total:
        ext.substr_copy(col.data.data_ptr(), bstart.data_ptr(),
                        blen.data_ptr(), scanned.data_ptr(),
                        out_bytes.data_ptr(), n, s)
    offs = torch.empty(n + 1, dtype=torch.int32, device="cuda")
    ext.narrow_i64_i32(scanned.data_ptr(), offs.data_ptr(), n, s)
    offs[n] = total
    v = col.validity.clone() if col.validity is not None else None
    return Column(DType.string(), n, out_bytes, v, offs,
                  null_count=col._null_count)


def round_half_up(col: Column, scale: int) -> Column:
    return binary_op_scalar("round", col, float(10 ** scale), col.dtype)


def unary_op(op: str, col: Column, out_dtype: DType) -> Column:
    n = col.size
    s = _stream()
    if col.dtype.id is TypeId.STRING:
        v = col.validity.clone() if col.validity is not None else None
        if op == "length":
            out = _alloc(n, out_dtype)
            ext.str_length(col.offsets.data_ptr(), col.data.data_ptr(),
                           out.data_ptr(), n, s)
            return Column(out_dtype, n, out, v, null_count=col._null_count)
        if op in ("trim", "ltrim", "rtrim"):
            return str_trim(col, {"trim": "both", "ltrim": "leading",
                                  "rtrim": "trailing"}[op])
        if op in ("initcap", "reverse"):
            nb = int(col.data.numel())
            ob = torch.empty(max(nb, 1), dtype=torch.uint8,
                             device="cuda")[:nb]
            if n:
                fn = ext.str_initcap if op == "initcap" else ext.str_reverse
                fn(col.offsets.data_ptr(), col.data.data_ptr(),
                   ob.data_ptr(), n, s)
            return Column(out_dtype, n, ob, v, col.offsets.clone(),
                          null_count=col._null_count)
        if op in ("upper", "lower"):
            nb = int(col.data.numel())
            ob = torch.empty(max(nb, 1), dtype=torch.uint8,
                             device="cuda")[:nb]
            if nb:
                ext.str_case(op == "upper", col.data.data_ptr(),
                             ob.data_ptr(), nb, s)
            return Column(out_dtype, n, ob, v, col.offsets.clone(),
                          null_count=col._null_count)
        raise NotImplementedError(f"string unary {op} not on GPU")
    out = _alloc(n, out_dtype)
    if op in ("not", "is_nan", "year", "month", "day"):
        ext.unary(_UN_OPS[op], _ht(col.dtype), col.data.data_ptr(),
                  _ptr(col.validity), out.data_ptr(), 0, n, s)
        if op == "is_nan":
            return Column(out_dtype, n, out, None, null_count=0)
        v = col.validity.clone() if col.validity is not None else None
        return Column(out_dtype, n, out, v, null_count=col._null_count)
    need_mask = col.validity is not None or op == "log"
    ov = _alloc_mask(n) if need_mask else None
    ext.unary(_UN_OPS[op], _ht(col.dtype), col.data.data_ptr(),
              _ptr(col.validity), out.data_ptr(), _ptr(ov), n, s)
    return Column(out_dtype, n, out, ov, null_count=None if need_mask else 0)


def cast(col: Column, to: DType) -> Column:
    n = col.size
    s = _stream()
    v = col.validity.clone() if col.validity is not None else None
    if col.dtype.id is TypeId.STRING:
        # must run before the decimal branch: string -> decimal parses as
        # f64 first (truncating through int64 would drop the fraction)
        return _cast_string_to(col, to, v)
    if to.id is TypeId.STRING:
        return _cast_to_string(col, v)
    if col.dtype.is_decimal or to.is_decimal:
        return _cast_decimal(col, to, v)
    out = _alloc(n, to)
    ext.cast(_ht(col.dtype), _ht(to), col.data.data_ptr(), out.data_ptr(), n, s)
    return Column(to, n, out, v, null_count=col._null_count)


_STR_TO_KIND = {TypeId.INT8: 2, TypeId.INT16: 3, TypeId.INT32: 4,
                TypeId.INT64: 5}


def _cast_string_to(col: Column, to: DType, v) -> Column:
    """string -> numeric cast (CastStrings analogue). Integral and decimal
    targets parse EXACTLY on device (k_str_to_dec: u128 digit accumulate,
    HALF_UP for decimals, truncate-toward-zero for ints, null on
    overflow/garbage); float targets ride the CSV f64 parser."""
    n = col.size
    s = _stream()
    if to.is_decimal or to.is_integral:
        if to.id is TypeId.DECIMAL128:
            kind, scale, prec = 1, to.scale, to.precision
            width = 2 * n
        elif to.id is TypeId.DECIMAL64:
            kind, scale, prec = 0, to.scale, to.precision
            width = n
        else:
            kind, scale, prec = _STR_TO_KIND[to.id], 0, 39
            width = n
        out64 = torch.empty(max(width, 1), dtype=torch.int64,
                            device="cuda")[:width]
        ov = _alloc_mask(n)
        if n:
            ext.str_to_dec(col.offsets.data_ptr(), col.data.data_ptr(),
                           _ptr(v), kind, scale, prec, out64.data_ptr(),
                           ov.data_ptr(), n, s)
        if to.is_decimal or to.id is TypeId.INT64:
            return Column(to, n, out64, ov, null_count=None)
        narrow = _alloc(n, to)
        if n:
            ext.cast(4, _ht(to), out64.data_ptr(), narrow.data_ptr(), n, s)
        return Column(to, n, narrow, ov, null_count=None)
    if not to.is_floating:
        raise NotImplementedError(f"gpu cast string -> {to}")
    trimmed = str_trim(col, "cast")  # full ASCII whitespace (Spark casts)
    f64 = torch.empty(max(n, 1), dtype=torch.float64, device="cuda")[:n]
    valid_u8 = torch.empty(max(n, 1), dtype=torch.uint8, device="cuda")[:n]
    unsupported = torch.zeros(1, dtype=torch.int32, device="cuda")
    if n:
        ext.csv_parse(trimmed.data.data_ptr(),
                      trimmed.offsets[:n].data_ptr(),
                      trimmed.offsets[1:].data_ptr(), 0, 0, 1,
                      0, f64.data_ptr(), 0, 0, valid_u8.data_ptr(),
                      unsupported.data_ptr(), n, s)
    from ..column import mask_nbytes

    v64 = torch.empty(max(n, 1), dtype=torch.int64, device="cuda")[:n]
    if n:
        ext.cast(0, 4, valid_u8.data_ptr(), v64.data_ptr(), n, s)
    pmask = torch.empty(mask_nbytes(n), dtype=torch.uint8, device="cuda")
    if n:
        ext.mask_from_nonzero(v64.data_ptr(), pmask.data_ptr(), n, s)
    mask = _and_masks(v, pmask) if v is not None else pmask
    wide = Column(DType.float64(), n, f64, mask, null_count=None)
    if to.id is TypeId.FLOAT64:
        return wide
    return cast(wide, to)


def _cast_to_string(col: Column, v) -> Column:
    """integral/decimal/date -> string on device; floats fall back (Java
    shortest-roundtrip float formatting parity is CPU-side, mirroring the
    reference's castFloatToString incompat gate)."""
    n = col.size
    s = _stream()
    if col.dtype.is_decimal:
        is128 = 1 if col.dtype.id is TypeId.DECIMAL128 else 0
        lens = torch.empty(max(n, 1), dtype=torch.int64, device="cuda")[:n]
        if n:
            ext.dec_to_str(col.data.data_ptr(), is128, col.dtype.scale,
                           0, lens.data_ptr(), 0, 0, n, s)
        scanned, total = _exclusive_scan_i64(lens) if n else (lens, 0)
        out = torch.empty(max(total, 1), dtype=torch.uint8,
                          device="cuda")[:total]
        if total:
            ext.dec_to_str(col.data.data_ptr(), is128, col.dtype.scale,
                           scanned.data_ptr(), lens.data_ptr(),
                           out.data_ptr(), 1, n, s)
        offs = torch.empty(n + 1, dtype=torch.int32, device="cuda")
        if n:
            ext.narrow_i64_i32(scanned.data_ptr(), offs.data_ptr(), n, s)
        offs[n] = total
        return Column(DType.string(), n, out, v, offs,
                      null_count=None if v is not None else 0)
    if not col.dtype.is_integral and col.dtype.id is not TypeId.BOOL:
        raise NotImplementedError(f"gpu cast {col.dtype} -> string")
    i64 = cast(Column(col.dtype, n, col.data, None, null_count=0),
               DType.int64()) if col.dtype.id is not TypeId.INT64         else Column(DType.int64(), n, col.data, None, null_count=0)
    lens = torch.empty(max(n, 1), dtype=torch.int64, device="cuda")[:n]
    if n:
        ext.i64_to_str(i64.data.data_ptr(), 0, lens.data_ptr(), 0, 0, n, s)
    scanned, total = _exclusive_scan_i64(lens) if n else (lens, 0)
    out = torch.empty(max(total, 1), dtype=torch.uint8,
                      device="cuda")[:total]
    if total:
        ext.i64_to_str(i64.data.data_ptr(), scanned.data_ptr(),
                       lens.data_ptr(), out.data_ptr(), 1, n, s)
    offs = torch.empty(n + 1, dtype=torch.int32, device="cuda")
    if n:
        ext.narrow_i64_i32(scanned.data_ptr(), offs.data_ptr(), n, s)
    offs[n] = total
    return Column(DType.string(), n, out, v, offs,
                  null_count=None if v is not None else 0)


def _cast_decimal(col: Column, to: DType, v) -> Column:
    n = col.size
    s = _stream()
    if col.dtype.is_decimal and to.is_decimal:
        if to.scale == col.dtype.scale and to.id is col.dtype.id \
                and to.precision >= col.dtype.precision:
            # pure widening retype: no rescale, no overflow possible
            return Column(to, n, col.data.clone(), v,
                          null_count=col._null_count)
        # Widen to int128 FIRST, then rescale exactly in 128-bit with
        # HALF_UP down-shift and null-on-overflow (Spark non-ANSI).
        # Rescaling inside int64 before widening silently overflows
        # (decimal(18,0) + decimal(18,10)); the same
        # kernel checks the target precision for narrowing user casts.
        if col.dtype.id is TypeId.DECIMAL64:
            wide = torch.empty(max(2 * n, 1), dtype=torch.int64,
                               device="cuda")[:2 * n]
            if n:
                ext.i64_to_i128(col.data.data_ptr(), wide.data_ptr(), n, s)
            src128 = wide
        else:
            src128 = col.data
        shift = to.scale - col.dtype.scale
        out_is_64 = to.id is TypeId.DECIMAL64
        width = n if out_is_64 else 2 * n
        out = torch.empty(max(width, 1), dtype=torch.int64,
                          device="cuda")[:width]
        ov = _alloc_mask(n)
        if n:
            ext.i128_rescale(src128.data_ptr(), _ptr(v), out.data_ptr(),
                             ov.data_ptr(), shift, to.precision,
                             1 if out_is_64 else 0, n, s)
        return Column(to, n, out, ov, null_count=None)
    if col.dtype.id is TypeId.DECIMAL128 or to.id is TypeId.DECIMAL128:
        if col.dtype.id is TypeId.DECIMAL128 and to.is_floating:
            dbl = torch.empty(max(n, 1), dtype=torch.float64,
                              device="cuda")[:n]
            if n:
                ext.i128_to_f64(col.data.data_ptr(), dbl.data_ptr(), n, s)
            c = Column(DType.float64(), n, dbl, v, null_count=col._null_count)
            scaled = binary_op_scalar("div", c, float(10 ** col.dtype.scale),
                                      DType.float64())
            scaled = Column(DType.float64(), n, scaled.data, v,
                            null_count=col._null_count)
            return cast(scaled, to) if to.id is not TypeId.FLOAT64 else scaled
        if col.dtype.is_integral and to.id is TypeId.DECIMAL128:
            # integral -> decimal128: treat as decimal(scale 0) and rescale
            i64 = col if col.dtype.id is TypeId.INT64 \
                else cast(Column(col.dtype, n, col.data, None, null_count=0),
                          DType.int64())
            as_dec = Column(DType.decimal(18, 0), n, i64.data, v,
                            null_count=col._null_count)
            return _cast_decimal(as_dec, to, v)
        raise NotImplementedError(f"gpu cast {col.dtype} -> {to}")
    if col.dtype.is_decimal:
        # decimal -> float/int: via double divide
        dbl = _alloc(n, DType.float64())
        ext.cast(_ht(col.dtype), 6, col.data.data_ptr(), dbl.data_ptr(), n, s)
        c = Column(DType.float64(), n, dbl, v, null_count=col._null_count)
        scaled = binary_op_scalar("div", c, float(10 ** col.dtype.scale),
                                  DType.float64())
        scaled = Column(DType.float64(), n, scaled.data, v,
                        null_count=col._null_count)
        return cast(scaled, to) if to.id is not TypeId.FLOAT64 else scaled
    # numeric -> decimal: scale up in int64 (float sources rounded)
    i64 = cast(col, DType.int64()) if not col.dtype.is_floating else None
    if i64 is not None:
        out = _alloc(n, to)
        ext.decimal_rescale(i64.data.data_ptr(), out.data_ptr(),
                            10 ** to.scale, True, n, s)
        return Column(to, n, out, v, null_count=col._null_count)
    dbl = cast(col, DType.float64())
    scaled = binary_op_scalar("mul", dbl, float(10 ** to.scale),
                              DType.float64())
    out = _alloc(n, to)
    # round-to-nearest (not the truncating int cast): 40.5074 * 10^4 is
    # 405073.9999.. in binary and must become 405074
    ext.f64_to_i64_rint(scaled.data.data_ptr(), out.data_ptr(), n, s)
    return Column(to, n, out, v, null_count=col._null_count)


def is_null(col: Column) -> Column:
    n = col.size
    out = torch.zeros(max(n, 1), dtype=torch.uint8, device="cuda")[:n]
    if col.validity is not None and n:
        ext.mask_expand(col.validity.data_ptr(), out.data_ptr(), True, n,
                        _stream())
    return Column(DType.bool_(), n, out, None, null_count=0)


def if_else(cond: Column, a: Column, b: Column) -> Column:
    n = cond.size
    s = _stream()
    # cond-as-selector: rows where cond true+valid take a, else b
    out = _alloc(n, a.dtype) if a.dtype.id is not TypeId.STRING else None
    if out is None:
        raise NotImplementedError("if_else on strings not on GPU yet")
    ov = _alloc_mask(n)
    ext.if_else(_ht(a.dtype), cond.data.data_ptr(), _ptr(cond.validity),
                a.data.data_ptr(), _ptr(a.validity), b.data.data_ptr(),
                _ptr(b.validity), out.data_ptr(), ov.data_ptr(), n, s)
    has_valid = a.validity is not None or b.validity is not None
    return Column(a.dtype, n, out, ov if has_valid else None,
                  null_count=None if has_valid else 0)


# ---------------------------------------------------------------------------
# scan helper
# ---------------------------------------------------------------------------

def _exclusive_scan_i64(vals: torch.Tensor) -> Tuple[torch.Tensor, int]:
    """Returns (exclusive scan tensor, total)."""
    n = vals.numel()
    s = _stream()
    if n == 0:
        return vals, 0
    out = torch.empty(n, dtype=torch.int64, device="cuda")
    nb = ext.scan_num_blocks(n)
    sums = torch.empty(nb, dtype=torch.int64, device="cuda")
    ext.scan_block(vals.data_ptr(), out.data_ptr(), sums.data_ptr(), n, s)
    if nb > 1:
        scanned_sums, total = _exclusive_scan_i64(sums)
        ext.scan_add_offsets(out.data_ptr(), scanned_sums.data_ptr(), n, s)
        return out, total
    return out, int(sums[0].item())


# ---------------------------------------------------------------------------
# selection
# ---------------------------------------------------------------------------

def mask_to_sel(mask: Column, n: int) -> torch.Tensor:
    """Selection vector (int32 row indices) of rows where mask is true."""
    s = _stream()
    nb = ext.sel_num_blocks(n)
    counts = torch.empty(nb, dtype=torch.int64, device="cuda")
    ext.mask_count(mask.data.data_ptr(), _ptr(mask.validity),
                   counts.data_ptr(), n, s)
    offsets, total = _exclusive_scan_i64(counts)
    idx = torch.empty(max(total, 1), dtype=torch.int32, device="cuda")[:total]
    if total:
        ext.mask_scatter(mask.data.data_ptr(), _ptr(mask.validity),
                         offsets.data_ptr(), idx.data_ptr(), n, s)
    return idx


def apply_boolean_mask(batch: ColumnBatch, mask: Column) -> ColumnBatch:
    idx = mask_to_sel(mask, batch.num_rows)
    return _gather_by_idx(batch, idx, idx.numel(), maybe_negative=False)


def gather(batch: ColumnBatch, indices: Column, check_bounds: bool = False,
           negatives: bool = True) -> ColumnBatch:
    return _gather_by_idx(batch, indices.data, indices.size,
                          maybe_negative=negatives)


def _gather_col(c: Column, idx: torch.Tensor, n_out: int,
                maybe_negative: bool) -> Column:
    s = _stream()
    if n_out == 0:
        return _empty_col(c.dtype)
    if c.dtype.id is TypeId.STRUCT:
        kids = tuple(_gather_col(k, idx, n_out, maybe_negative)
                     for k in c.child)
        ov = None
        if c.validity is not None or maybe_negative:
            ov = _alloc_mask(n_out)
            ext.gather_validity(_ptr(c.validity), c.validity is not None,
                                idx.data_ptr(), ov.data_ptr(), n_out, s)
        return Column(c.dtype, n_out,
                      torch.zeros(0, dtype=torch.uint8, device="cuda"),
                      ov, None, None if ov is not None else 0, kids)
    if c.dtype.id is TypeId.STRING:
        lens = torch.empty(n_out, dtype=torch.int64, device="cuda")
        ext.gather_str_lens(c.offsets.data_ptr(), idx.data_ptr(),
                            lens.data_ptr(), n_out, s)
        scanned, total = _exclusive_scan_i64(lens)
        out_bytes = torch.empty(max(total, 1), dtype=torch.uint8,
                                device="cuda")[:total]
        if total:
            ext.gather_str_bytes(c.data.data_ptr(), c.offsets.data_ptr(),
                                 idx.data_ptr(), scanned.data_ptr(),
                                 out_bytes.data_ptr(), n_out, total, s)
        offs = torch.empty(n_out + 1, dtype=torch.int32, device="cuda")
        ext.narrow_i64_i32(scanned.data_ptr(), offs.data_ptr(), n_out, s)
        offs[n_out] = total
        ov = None
        if c.validity is not None or maybe_negative:
            ov = _alloc_mask(n_out)
            ext.gather_validity(_ptr(c.validity), c.validity is not None,
                                idx.data_ptr(), ov.data_ptr(), n_out, s)
        return Column(c.dtype, n_out, out_bytes, ov, offs, null_count=None if ov is not None else 0)
    out = _alloc(n_out, c.dtype)
    ext.gather_fixed(c.dtype.itemsize, c.data.data_ptr(), idx.data_ptr(),
                     out.data_ptr(), n_out, s)
    ov = None
    nc = 0
    if c.validity is not None or maybe_negative:
        ov = _alloc_mask(n_out)
        ext.gather_validity(_ptr(c.validity), c.validity is not None,
                            idx.data_ptr(), ov.data_ptr(), n_out, s)
        nc = None
    return Column(c.dtype, n_out, out, ov, null_count=nc)


def _gather_by_idx(batch: ColumnBatch, idx: torch.Tensor, n_out: int,
                   maybe_negative: bool) -> ColumnBatch:
    s = _stream()
    cols: List[Optional[Column]] = [None] * len(batch.columns)
    fixed: List[tuple] = []  # (position, column)
    for ci, c in enumerate(batch.columns):
        if c.dtype.id in (TypeId.STRING, TypeId.STRUCT) or n_out == 0:
            cols[ci] = _gather_col(c, idx, n_out, maybe_negative)
        else:
            fixed.append((ci, c))
    if fixed:
        blobs = []
        outs = []
        for ci, c in fixed:
            out = _alloc(n_out, c.dtype)
            need_valid = c.validity is not None or maybe_negative
            ov = _alloc_mask(n_out) if need_valid else None
            outs.append((ci, c, out, ov))
            blobs.append(struct.pack(
                "<iiqqqq", c.dtype.itemsize,
                1 if c.validity is not None else 0, c.data.data_ptr(),
                _ptr(c.validity), out.data_ptr(), _ptr(ov)))
        desc = torch.frombuffer(bytearray(b"".join(blobs)),
                                dtype=torch.uint8).cuda()
        ext.gather_table(desc.data_ptr(), len(fixed), idx.data_ptr(), n_out, s)
        for ci, c, out, ov in outs:
            cols[ci] = Column(c.dtype, n_out, out, ov,
                              null_count=None if ov is not None else 0)
    return ColumnBatch(cols, n_out)


def concat_batches(batches: List[ColumnBatch]) -> ColumnBatch:
    s = _stream()
    total = sum(b.num_rows for b in batches)
    ncols = batches[0].num_columns
    out_cols = []
    for ci in range(ncols):
        ins = [b.columns[ci] for b in batches]
        if ins[0].dtype.id is TypeId.LIST:
            raise NotImplementedError(
                "concat of LIST columns on GPU (explode or collect before "
                "unioning, or keep the union on the CPU)")
        dtype = ins[0].dtype
        if dtype.id is TypeId.STRUCT:
            any_valid = any(c.validity is not None for c in ins)
            out_valid = None
            if any_valid:
                out_valid = torch.zeros(mask_nbytes(total),
                                        dtype=torch.uint8, device="cuda")
                off = 0
                for c in ins:
                    if c.size:
                        ext.copy_valid_range(_ptr(c.validity),
                                             c.validity is not None, off,
                                             out_valid.data_ptr(), c.size,
                                             s)
                    off += c.size
            kids = tuple(
                concat_batches([ColumnBatch([c.child[k]], c.size)
                                for c in ins]).columns[0]
                for k in range(len(dtype.children)))
            out_cols.append(Column(
                dtype, total, torch.zeros(0, dtype=torch.uint8,
                                          device="cuda"),
                out_valid, None, None if any_valid else 0, kids))
            continue
        any_valid = any(c.validity is not None for c in ins)
        out_valid = None
        if any_valid:
            out_valid = torch.zeros(mask_nbytes(total), dtype=torch.uint8,
                                    device="cuda")
            off = 0
            for c in ins:
                if c.size:
                    ext.copy_valid_range(_ptr(c.validity),
                                         c.validity is not None, off,
                                         out_valid.data_ptr(), c.size, s)
                off += c.size
        if dtype.id is TypeId.STRING:
            data = torch.cat([c.data for c in ins])
            offs = torch.empty(total + 1, dtype=torch.int32, device="cuda")
            offs[0] = 0
            row_off, byte_off = 0, 0
            for c in ins:
                if c.size:
                    seg = offs[row_off + 1: row_off + 1 + c.size]
                    seg.copy_(c.offsets[1:c.size + 1])
                    if byte_off:
                        ext.binary_arith(_BIN_OPS["add"], 3, seg.data_ptr(),
                                         0, 0.0, byte_off, True, 0, 0,
                                         seg.data_ptr(), 0, c.size, s)
                row_off += c.size
                byte_off += int(c.data.numel())
            out_cols.append(Column(dtype, total, data, out_valid, offs,
                                   null_count=None if any_valid else 0))
        else:
            data = torch.cat([c.data for c in ins]) if total else \
                torch.zeros(0, dtype=torch_dtype(dtype), device="cuda")
            out_cols.append(Column(dtype, total, data, out_valid,
                                   null_count=None if any_valid else 0))
    return ColumnBatch(out_cols, total)


# ---------------------------------------------------------------------------
# hashing / partition
# ---------------------------------------------------------------------------

_HASH_KIND = {
    TypeId.BOOL: _HK_INT, TypeId.INT8: _HK_INT, TypeId.INT16: _HK_INT,
    TypeId.INT32: _HK_INT, TypeId.DATE32: _HK_INT,
    TypeId.INT64: _HK_LONG, TypeId.TIMESTAMP: _HK_LONG,
    TypeId.DECIMAL64: _HK_LONG,
    TypeId.FLOAT32: _HK_FLOAT, TypeId.FLOAT64: _HK_DOUBLE,
}


def _murmur3_tensor(cols: List[Column], seed: int,
                    sel: Optional[torch.Tensor] = None,
                    n_out: Optional[int] = None) -> torch.Tensor:
    n = cols[0].size if sel is None else (n_out if n_out is not None
                                          else sel.numel())
    s = _stream()
    seeds = torch.full((max(n, 1),), seed, dtype=torch.int32, device="cuda")[:n]
    selp = 0 if sel is None else sel.data_ptr()
    for c in cols:
        if n == 0:
            break
        if c.dtype.id is TypeId.STRING:
            ext.murmur3_str(c.offsets.data_ptr(), c.data.data_ptr(),
                            _ptr(c.validity), selp, seeds.data_ptr(), n, s)
        elif c.dtype.id is TypeId.DECIMAL128:
            ext.murmur3_col(5, 4, c.data.data_ptr(), _ptr(c.validity), selp,
                            seeds.data_ptr(), n, s)
        else:
            ext.murmur3_col(_HASH_KIND[c.dtype.id], _ht(c.dtype),
                            c.data.data_ptr(), _ptr(c.validity), selp,
                            seeds.data_ptr(), n, s)
    return seeds


def murmur3_hash(cols: List[Column], seed: int = 42) -> Column:
    h = _murmur3_tensor(cols, seed)
    return Column(DType.int32(), cols[0].size, h, None, null_count=0)


def hash_partition(batch: ColumnBatch, key_idx: List[int], num_parts: int):
    n = batch.num_rows
    s = _stream()
    if n == 0:
        return batch, [0] * (num_parts + 1)
    h = _murmur3_tensor([batch.columns[i] for i in key_idx], 42)
    part = torch.empty(n, dtype=torch.int32, device="cuda")
    ext.pmod_part(h.data_ptr(), num_parts, part.data_ptr(), n, s)
    perm, offsets = _scatter_by_part(part, n, num_parts)
    out = _gather_by_idx(batch, perm, n, maybe_negative=False)
    return out, offsets


def _scatter_by_part(part: torch.Tensor, n: int, num_parts: int):
    s = _stream()
    nb = ext.part_num_blocks(n)
    counts = torch.empty(num_parts * nb, dtype=torch.int64, device="cuda")
    ext.part_hist(part.data_ptr(), num_parts, counts.data_ptr(), n, s)
    scanned, _total = _exclusive_scan_i64(counts)
    perm = torch.empty(n, dtype=torch.int32, device="cuda")
    ext.part_scatter(part.data_ptr(), num_parts, scanned.data_ptr(),
                     perm.data_ptr(), n, s)
    # partition start offsets: exclusive scan of per-part totals
    per_part = counts.view(num_parts, nb).sum(dim=1).cpu().numpy()
    offs = np.zeros(num_parts + 1, dtype=np.int64)
    np.cumsum(per_part, out=offs[1:])
    return perm, offs.tolist()


# ---------------------------------------------------------------------------
# reduce
# ---------------------------------------------------------------------------

def reduce(op: str, col: Column):
    n = col.size
    if op == "count_all":
        return n
    if n == 0:
        return 0 if op == "count" else None
    s = _stream()
    is_f = col.dtype.is_floating
    if op == "mean":
        total = reduce("sum", col)
        cnt = reduce("count", col)
        return None if cnt == 0 else float(total) / cnt if total is not None else None
    init = {"sum": 0, "min": None, "max": None, "count": 0}[op]
    if is_f:
        acc = torch.full((1,), {"sum": 0.0, "min": float("inf"),
                                "max": float("-inf"), "count": 0.0}[op],
                         dtype=torch.float64, device="cuda")
    else:
        acc = torch.full((1,), {"sum": 0, "min": 2 ** 63 - 1,
                                "max": -2 ** 63, "count": 0}[op],
                         dtype=torch.int64, device="cuda")
    cnt = torch.zeros(1, dtype=torch.int64, device="cuda")
    if op == "count" and not col.dtype.is_fixed_width:
        # count over STRING/nested: only the validity matters
        nn8 = torch.ones(max(n, 1), dtype=torch.uint8, device="cuda")[:n]
        col = Column(DType.bool_(), n, nn8, col.validity,
                     null_count=col._null_count)
    ext.reduce(_RED.get(op, 0), _ht(col.dtype), col.data.data_ptr(),
               _ptr(col.validity), acc.data_ptr(), cnt.data_ptr(), n, s)
    c = int(cnt.item())
    if op == "count":
        return c
    if c == 0:
        return None
    v = acc.item()
    if not is_f and col.dtype.id not in (TypeId.DECIMAL64,):
        v = int(v)
    return v


# ---------------------------------------------------------------------------
# group-by aggregate
# ---------------------------------------------------------------------------

def _key_desc(cols: List[Column]) -> torch.Tensor:
    """Pack KeyCol structs (see native/hipdf/kernels/keys.h) into device mem:
    {int type; int is_string; void* data; u64* valid; void* aux} = 32 B."""
    blobs = []
    for c in cols:
        if c.dtype.id is TypeId.STRING:
            blobs.append(struct.pack("<iiqqq", 0, 1, c.offsets.data_ptr(),
                                     _ptr(c.validity), c.data.data_ptr()))
        elif c.dtype.id is TypeId.DECIMAL128:
            blobs.append(struct.pack("<iiqqq", 7, 0, c.data.data_ptr(),
                                     _ptr(c.validity), 0))
        else:
            blobs.append(struct.pack("<iiqqq", _ht(c.dtype), 0,
                                     c.data.data_ptr(), _ptr(c.validity), 0))
    host = torch.frombuffer(bytearray(b"".join(blobs)), dtype=torch.uint8)
    return host.cuda()


def _next_pow2(x: int) -> int:
    p = 1
    while p < x:
        p <<= 1
    return p


def group_by_aggregate(batch: ColumnBatch, key_idx: List[int],
                       aggs: List[Tuple[str, int, DType]],
                       sel: Optional[torch.Tensor] = None) -> ColumnBatch:
    """Hash group-by; with `sel` (int32 row indices) only the selected rows
    participate — the filter-into-aggregate fusion path (no materialized
    gather of the filtered batch)."""
    n = batch.num_rows if sel is None else sel.numel()
    s = _stream()
    keys = [batch.columns[i] for i in key_idx]
    if n == 0 and key_idx:
        out = [_empty_col(k.dtype) for k in keys]
        out += [_empty_col(dt) for _, _, dt in aggs]
        return ColumnBatch(out, 0)
    selp = 0 if sel is None else sel.data_ptr()
    dense = None
    if not key_idx:
        # global aggregate: single group
        row_gid = torch.zeros(n, dtype=torch.int32, device="cuda")
        ngroups = 1
        leaders = torch.zeros(1, dtype=torch.int32, device="cuda")
    elif (dense := _try_dense_keys(keys, aggs, selp, n)) is not None:
        # dense integer-key fast path: gid = radix index over the small
        # value ranges — no hash table, no probe, no leader gather
        # (k_dense_gid); empty ids are compacted after aggregation
        row_gid, ngroups, dense_info = dense
        leaders = None
    else:
        h = _murmur3_tensor(keys, 42, sel, n)
        cap = max(1024, _next_pow2(2 * n))
        desc = _key_desc(keys)
        slot_row = torch.full((cap,), -1, dtype=torch.int32, device="cuda")
        row_slot = torch.empty(n, dtype=torch.int32, device="cuda")
        claimed = torch.empty(n, dtype=torch.int32, device="cuda")
        ngroups_t = torch.zeros(1, dtype=torch.int32, device="cuda")
        ext.gb_build(h.data_ptr(), desc.data_ptr(), len(keys), selp,
                     slot_row.data_ptr(), row_slot.data_ptr(),
                     claimed.data_ptr(), ngroups_t.data_ptr(), cap, n, s)
        slot_gid = torch.empty(cap, dtype=torch.int32, device="cuda")
        leaders = torch.empty(n, dtype=torch.int32, device="cuda")
        ext.gb_number(claimed.data_ptr(), slot_row.data_ptr(),
                      slot_gid.data_ptr(), ngroups_t.data_ptr(),
                      leaders.data_ptr(), n, s)
        ngroups = int(ngroups_t.item())
        row_gid = torch.empty(n, dtype=torch.int32, device="cuda")
        ext.gb_rowgid(row_slot.data_ptr(), slot_gid.data_ptr(),
                      row_gid.data_ptr(), n, s)
    if leaders is not None:
        leaders = leaders[:ngroups]
    # leaders are ORIGINAL row ids (pre-selection), so gather keys from the
    # full-length key columns (hash path; the dense path reconstructs keys
    # arithmetically after compaction)
    key_batch = _gather_by_idx(ColumnBatch(keys, keys[0].size), leaders,
                               ngroups, maybe_negative=False) \
        if keys and leaders is not None else None
    out_cols = list(key_batch.columns) if key_batch is not None else []

    # fused multi-aggregate: one kernel pass accumulates every agg;
    # decimal128 sums go through dedicated carry-correct i128 kernels
    # replica count for the non-LDS atomic path: spreads a skew-hot
    # group's atomics over nrep accumulator copies (zipf keys would
    # otherwise serialize millions of atomics on one address)
    nrep = 1 if ngroups * max(1, len(aggs)) <= 4096 \
        else max(1, min(64, (1 << 21) // ngroups))
    allocs = []
    blobs = []
    for op, vidx, out_dtype in aggs:
        vc = batch.columns[vidx] if vidx >= 0 else None
        if op == "sum" and out_dtype.id is TypeId.DECIMAL128:
            acc = torch.zeros(2 * max(ngroups, 1), dtype=torch.int64,
                              device="cuda")
            cnt = torch.zeros(max(ngroups, 1), dtype=torch.int64,
                              device="cuda")
            if n:
                in64 = 0 if vc.dtype.id is TypeId.DECIMAL128 else 1
                if ngroups <= 2048:
                    # LDS-staged: one global flush per block instead of
                    # per-row atomics on a few hot accumulators
                    ext.gb_sum_i128_lds(
                        in64, vc.data.data_ptr(), _ptr(vc.validity),
                        row_gid.data_ptr(), selp, acc.data_ptr(),
                        cnt.data_ptr(), ngroups, n, s)
                elif in64:
                    ext.gb_sum_i64_to_i128(
                        vc.data.data_ptr(), _ptr(vc.validity),
                        row_gid.data_ptr(), selp, acc.data_ptr(),
                        cnt.data_ptr(), n, s)
                else:
                    ext.gb_sum_i128(vc.data.data_ptr(), _ptr(vc.validity),
                                    row_gid.data_ptr(), selp, acc.data_ptr(),
                                    cnt.data_ptr(), n, s)
            allocs.append(("sum_d128", out_dtype, False, acc, cnt, 0))
            continue
        if op in ("collect_list", "collect_set"):
            col = _gb_collect(vc, row_gid, selp, n, ngroups, out_dtype,
                              op == "collect_set", s)
            allocs.append(("collect", out_dtype, False, col, None, 0))
            continue
        if op.startswith("percentile:"):
            col = _gb_percentile(vc, row_gid, selp, n, ngroups,
                                 float(op.split(":", 1)[1]), s)
            allocs.append(("collect", out_dtype, False, col, None, 0))
            continue
        if op.startswith("hll:"):
            col = _gb_hll(vc, row_gid, selp, n, ngroups,
                          int(op.split(":", 1)[1]), s)
            allocs.append(("collect", out_dtype, False, col, None, 0))
            continue
        if op in ("min", "max") and vc is not None \
                and vc.dtype.id is TypeId.STRING:
            col = _gb_minmax_str(vc, row_gid, sel, n, ngroups,
                                 op == "max", s)
            allocs.append(("collect", out_dtype, False, col, None, 0))
            continue
        if op == "count" and vc is not None and not vc.dtype.is_fixed_width:
            # count over STRING/nested only needs the validity: swap in a
            # u8 non-null indicator column so no typed accumulator is hit
            nn8 = torch.ones(max(vc.size, 1), dtype=torch.uint8,
                             device="cuda")[:vc.size]
            vc = Column(DType.bool_(), vc.size, nn8, vc.validity,
                        null_count=vc._null_count)
        acc_is_double = out_dtype.is_floating or (
            vc is not None and vc.dtype.is_floating)
        acc = torch.empty(max(nrep * ngroups, 1),
                          dtype=torch.float64 if acc_is_double else torch.int64,
                          device="cuda")
        cnt = torch.zeros(max(nrep * ngroups, 1), dtype=torch.int64,
                          device="cuda")
        t = _ht(vc.dtype) if vc is not None else 4
        if op not in ("count", "count_all"):
            ext.gb_acc_init(_GB[op], acc.data_ptr(), acc_is_double,
                            nrep * ngroups, s)
        # sum / integer min-max over a non-null column: group validity
        # is implied by group existence, so the per-row count atomic is
        # pure traffic (roofline: k_gb_agg_multi writes were ~2x the
        # accumulator data). FLOAT min/max keep the count: the replica
        # fold needs it to tell an empty replica's ±inf init from a real
        # ±inf when NaN values are present (Spark: NaN is greatest).
        skip_cnt = 1 if (vc is not None and vc.validity is None and n > 0
                         and (op == "sum"
                              or (op in ("min", "max")
                                  and not vc.dtype.is_floating))) else 0
        allocs.append((op, out_dtype, acc_is_double, acc, cnt,
                       skip_cnt))
        blobs.append(struct.pack(
            "<iiiiqqqq", _GB[op], t, acc_is_double, skip_cnt,
            _ptr(vc.data if vc is not None else None),
            _ptr(vc.validity if vc is not None else None),
            acc.data_ptr(), cnt.data_ptr()))
    if blobs:
        desc = torch.frombuffer(bytearray(b"".join(blobs)),
                                dtype=torch.uint8).cuda()
        ext.gb_agg_multi(desc.data_ptr(), len(blobs), row_gid.data_ptr(),
                         selp, ngroups, nrep, n, s)
        if nrep > 1:
            for op, _, acc_is_double, acc, cnt, _skip in allocs:
                if op in ("sum_d128", "collect"):
                    continue
                ext.gb_reduce_reps(_GB.get(op, 0), acc.data_ptr(),
                                   1 if acc_is_double else 0,
                                   cnt.data_ptr(), ngroups, nrep, s)
    for op, out_dtype, acc_is_double, acc, cnt, skip_cnt in allocs:
        if op == "collect":
            out_cols.append(acc)
            continue
        if op == "sum_d128":
            ov = _alloc_mask(ngroups)
            ext.mask_from_nonzero(cnt.data_ptr(), ov.data_ptr(), ngroups, s)
            out_cols.append(Column(out_dtype, ngroups, acc[:2 * ngroups],
                                   ov, null_count=None))
            continue
        if op in ("count", "count_all"):
            out_cols.append(Column(out_dtype, ngroups, cnt[:ngroups].clone(),
                                   None, null_count=0))
            continue
        acc_dt = DType.float64() if acc_is_double else DType.int64()
        data = acc[:ngroups]
        if torch_dtype(out_dtype) != data.dtype:
            out_data = _alloc(ngroups, out_dtype)
            ext.cast(_ht(acc_dt), _ht(out_dtype), data.data_ptr(),
                     out_data.data_ptr(), ngroups, s)
        else:
            out_data = data
        if skip_cnt:
            # every observed group has >= 1 contributing row
            out_cols.append(Column(out_dtype, ngroups, out_data, None,
                                   null_count=0))
        else:
            ov = _alloc_mask(ngroups)
            ext.mask_from_nonzero(cnt.data_ptr(), ov.data_ptr(), ngroups, s)
            out_cols.append(Column(out_dtype, ngroups, out_data, ov,
                                   null_count=None))
    if dense is not None:
        return _compact_dense(out_cols, allocs, keys, dense_info, ngroups,
                              row_gid, n, s)
    return ColumnBatch(out_cols, ngroups)


_DENSE_OK = {TypeId.BOOL, TypeId.INT8, TypeId.INT16, TypeId.INT32,
             TypeId.INT64, TypeId.DATE32}


def _try_dense_keys(keys, aggs, selp, n):
    """Dense path applicability: all keys are non-null small-range integers
    and the range product stays tiny relative to n."""
    if any(k.validity is not None for k in keys):
        return None
    if any(k.dtype.id not in _DENSE_OK for k in keys):
        return None
    if any(op in ("collect_list", "collect_set") for op, _, _ in aggs):
        return None  # LIST columns cannot be re-gathered in compaction
    mins, ranges = [], []
    for k in keys:
        if k._minmax is None:
            # cached on the column: stored tables are re-scanned every
            # query, and columns are immutable by convention
            k._minmax = (reduce("min", k), reduce("max", k))
        kmin, kmax = k._minmax
        if kmin is None:
            return None
        rng = int(kmax) - int(kmin) + 1
        if rng <= 0:
            return None
        mins.append(int(kmin))
        ranges.append(rng)
    prod = 1
    for r in ranges:
        prod *= r
        if prod > (1 << 24):
            return None
    if prod > max(1 << 16, 4 * n):
        return None
    strides = [0] * len(keys)
    acc = 1
    for i in range(len(keys) - 1, -1, -1):
        strides[i] = acc
        acc *= ranges[i]
    s = _stream()
    blob = b"".join(
        struct.pack("<iiqqqqq", _ht(k.dtype), 0, k.data.data_ptr(), 0,
                    mins[i], ranges[i], strides[i])
        for i, k in enumerate(keys))
    desc = torch.frombuffer(bytearray(blob), dtype=torch.uint8).cuda()
    row_gid = torch.empty(n, dtype=torch.int32, device="cuda")
    ext.dense_gid(desc.data_ptr(), len(keys), selp, row_gid.data_ptr(), n, s)
    return row_gid, prod, (mins, ranges, strides)


def _compact_dense(out_cols, allocs, keys, dense_info, ngroups, row_gid,
                   n, s):
    """Drop empty dense group ids and reconstruct the key columns from the
    surviving ids (kmin + (g // stride) % range per key)."""
    mins, ranges, strides = dense_info
    counts = None
    for (op, _, _, acc, cnt, _sk) in allocs:
        if op == "count_all":
            counts = cnt
            break
    if counts is None:
        counts = torch.zeros(max(ngroups, 1), dtype=torch.int64,
                             device="cuda")
        if n:
            ext.gb_collect_count(0, row_gid.data_ptr(), 0,
                                 counts.data_ptr(), n, s)
    ccol = Column(DType.int64(), ngroups, counts[:ngroups], None,
                  null_count=0)
    live = binary_op_scalar("gt", ccol, 0, DType.bool_())
    gsel = mask_to_sel(live, ngroups)
    ncomp = gsel.numel()
    gcol = Column(DType.int32(), ncomp, gsel, None, null_count=0)
    g64 = cast(gcol, DType.int64())
    key_cols = []
    for i, k in enumerate(keys):
        code = g64
        if strides[i] != 1:
            code = binary_op_scalar("int_div", code, strides[i],
                                    DType.int64())
        if i > 0:  # leading key needs no modulo (gid < stride_{i-1}*range)
            code = binary_op_scalar("mod", code, ranges[i], DType.int64())
        if mins[i]:
            code = binary_op_scalar("add", code, mins[i], DType.int64())
        key_cols.append(cast(code, k.dtype))
    agg_batch = _gather_by_idx(ColumnBatch(out_cols, ngroups), gsel, ncomp,
                               maybe_negative=False)
    return ColumnBatch(key_cols + list(agg_batch.columns), ncomp)


def _gb_collect_str(vc: Column, row_gid: torch.Tensor, selp, n: int,
                    ngroups: int, out_dtype, s) -> Column:
    """collect_list over strings: stable-sort rows by group id, then one
    string gather in group order; offsets come from the per-group valid
    counts."""
    if selp:
        raise NotImplementedError("string collect under filter fusion")
    # drop null values first (collect skips nulls) while keeping gids
    gid_col = Column(DType.int32(), n, row_gid[:n], None, null_count=0)
    if vc.validity is not None:
        nn = unary_op("not", is_null(vc), DType.bool_())
        kept = apply_boolean_mask(ColumnBatch([gid_col, vc], n), nn)
        gid_col, vc = kept.columns[0], kept.columns[1]
        n = kept.num_rows
    pair = ColumnBatch([gid_col, vc], n)
    perm = sort_order(pair, [0], [False], [False])
    child = _gather_col(vc, perm.data, n, maybe_negative=False)
    counts = torch.zeros(max(ngroups, 1), dtype=torch.int64, device="cuda")
    if n:
        ext.gb_collect_count(0, gid_col.data.data_ptr(), 0,
                             counts.data_ptr(), n, s)
    scanned, total = _exclusive_scan_i64(counts[:ngroups]) if ngroups \
        else (counts, 0)
    offs = torch.empty(ngroups + 1, dtype=torch.int32, device="cuda")
    if ngroups:
        ext.narrow_i64_i32(scanned.data_ptr(), offs.data_ptr(), ngroups, s)
    offs[ngroups] = total
    return Column(out_dtype, ngroups, torch.zeros(0, dtype=torch.uint8,
                                                  device="cuda"),
                  None, offs, 0, child)


def _gb_percentile(vc: Column, row_gid: torch.Tensor, selp, n: int,
                   ngroups: int, p: float, s) -> Column:
    """Exact percentile (linear interpolation): sort (gid, value) with
    nulls last, then interpolate inside each group's valid window
    (k_gb_percentile). Reference analogue: GpuPercentile."""
    if selp:
        raise NotImplementedError("percentile under filter fusion")
    vf = cast(vc, DType.float64()) if vc.dtype.id is not TypeId.FLOAT64 \
        else vc
    gid_col = Column(DType.int32(), n, row_gid[:n], None, null_count=0)
    pair = ColumnBatch([gid_col, vf], n)
    perm = sort_order(pair, [0, 1], [False, False], [True, True])
    total = torch.zeros(max(ngroups, 1), dtype=torch.int64, device="cuda")
    vcnt = torch.zeros(max(ngroups, 1), dtype=torch.int64, device="cuda")
    if n:
        ext.gb_collect_count(0, row_gid.data_ptr(), 0, total.data_ptr(),
                             n, s)
        ext.gb_collect_count(_ptr(vf.validity), row_gid.data_ptr(), 0,
                             vcnt.data_ptr(), n, s)
    starts, _ = _exclusive_scan_i64(total[:ngroups]) if ngroups \
        else (total, 0)
    out = torch.empty(max(ngroups, 1), dtype=torch.float64,
                      device="cuda")[:ngroups]
    if ngroups:
        ext.gb_percentile(vf.data.data_ptr(), perm.data.data_ptr(),
                          starts.data_ptr(), vcnt.data_ptr(), p,
                          out.data_ptr(), ngroups, s)
    ov = _alloc_mask(ngroups)
    ext.mask_from_nonzero(vcnt.data_ptr(), ov.data_ptr(), ngroups, s)
    return Column(DType.float64(), ngroups, out, ov, null_count=None)


def _gb_minmax_str(vc: Column, row_gid: torch.Tensor, sel, n: int,
                   ngroups: int, is_max: bool, s) -> Column:
    """min/max over STRING values per group, sort-based: stable radix
    sort of (gid, value) (string keys via the chunked big-endian LSD
    machinery), then the head row of each gid segment is the extreme.
    Nulls sort last so any non-null wins; an all-null group's head is a
    null row and gathers as NULL (Spark ignore-null semantics)."""
    gid_col = Column(DType.int32(), n, row_gid[:n], None, null_count=0)
    if sel is not None:
        vals = _gather_col(vc, sel, n, maybe_negative=False)
    else:
        vals = vc
    pair = ColumnBatch([gid_col, vals], n)
    perm = sort_order(pair, [0, 1], [False, is_max], [False, True])
    pg = perm.data[:n].long()
    sg = row_gid[:n].long()[pg] if sel is None else \
        row_gid[:n].long()[pg]
    head = torch.ones(n, dtype=torch.bool, device="cuda")
    if n > 1:
        head[1:] = sg[1:] != sg[:-1]
    hp = head.nonzero(as_tuple=True)[0]
    orig = pg[hp] if sel is None else sel[:n].long()[pg[hp]]
    ridx = torch.full((ngroups,), -1, dtype=torch.int32, device="cuda")
    ridx[sg[hp]] = orig.to(torch.int32)
    return _gather_col(vc, ridx, ngroups, maybe_negative=True)


def _gb_collect(vc: Column, row_gid: torch.Tensor, selp, n: int,
                ngroups: int, out_dtype, is_set: bool, s) -> Column:
    """collect_list/collect_set: count-per-group + scan + atomic-cursor
    scatter (k_gb_collect_* in groupby.hip). collect_set dedupes first by
    grouping the (gid, value) pairs with the generic hash groupby."""
    elem_dt = out_dtype.children[0]
    if elem_dt.id is TypeId.STRING and not is_set:
        return _gb_collect_str(vc, row_gid, selp, n, ngroups, out_dtype, s)
    if is_set:
        if selp:
            raise NotImplementedError(
                "collect_set under filter fusion: the single-pass exec "
                "path materializes filtered batches, so sel is never set")
        # dedupe the (gid, value) pairs with the generic hash groupby, then
        # collect the unique pairs using gid as a direct group index
        gid_col = Column(DType.int32(), n, row_gid[:n], None, null_count=0)
        pair = ColumnBatch([gid_col, vc], n)
        uniq = group_by_aggregate(pair, [0, 1], [])
        ug, uv = uniq.columns[0], uniq.columns[1]
        return _gb_collect(uv, ug.data[:uniq.num_rows], None,
                           uniq.num_rows, ngroups, out_dtype, False, s)
    counts = torch.zeros(max(ngroups, 1), dtype=torch.int64, device="cuda")
    if n:
        ext.gb_collect_count(_ptr(vc.validity), row_gid.data_ptr(),
                             selp or 0, counts.data_ptr(), n, s)
    scanned, total = _exclusive_scan_i64(counts[:ngroups]) if ngroups         else (counts, 0)
    esize = 16 if elem_dt.id is TypeId.DECIMAL128 else         torch.empty(0, dtype=torch_dtype(elem_dt)).element_size()
    nelem = 2 * total if elem_dt.id is TypeId.DECIMAL128 else total
    out_data = torch.empty(max(nelem, 1), dtype=torch_dtype(elem_dt),
                           device="cuda")[:nelem]
    cursor = torch.zeros(max(ngroups, 1), dtype=torch.int64, device="cuda")
    if n and total:
        ext.gb_collect_fill(esize, vc.data.data_ptr(), _ptr(vc.validity),
                            row_gid.data_ptr(), selp or 0,
                            scanned.data_ptr(), cursor.data_ptr(),
                            out_data.data_ptr(), n, s)
    offs = torch.empty(ngroups + 1, dtype=torch.int32, device="cuda")
    if ngroups:
        ext.narrow_i64_i32(scanned.data_ptr(), offs.data_ptr(), ngroups, s)
    offs[ngroups] = total
    child = Column(elem_dt, total, out_data, None, null_count=0)
    return Column(out_dtype, ngroups, torch.zeros(0, dtype=torch.uint8,
                                                  device="cuda"),
                  None, offs, 0, child)


# ---------------------------------------------------------------------------
# join
# ---------------------------------------------------------------------------

def join_gather_maps(left: ColumnBatch, right: ColumnBatch,
                     left_keys: List[int], right_keys: List[int], how: str,
                     right_matched: Optional[torch.Tensor] = None):
    s = _stream()
    nl, nr = left.num_rows, right.num_rows
    lk = [left.columns[i] for i in left_keys]
    rk = [right.columns[i] for i in right_keys]
    cap = max(1024, _next_pow2(2 * max(nr, 1)))
    rh = _murmur3_tensor(rk, 42)
    rdesc = _key_desc(rk)
    ldesc = _key_desc(lk)
    head = torch.full((cap,), -1, dtype=torch.int32, device="cuda")
    nxt = torch.empty(max(nr, 1), dtype=torch.int32, device="cuda")
    ext.join_build(rh.data_ptr(), rdesc.data_ptr(), len(rk), head.data_ptr(),
                   nxt.data_ptr(), cap, nr, s)
    lh = _murmur3_tensor(lk, 42)
    counts = torch.empty(nl, dtype=torch.int64, device="cuda")
    ext.join_count(_JOIN[how], lh.data_ptr(), ldesc.data_ptr(),
                   rdesc.data_ptr(), len(lk), head.data_ptr(), nxt.data_ptr(),
                   cap, counts.data_ptr(), nl, s)
    offsets, total = _exclusive_scan_i64(counts)
    lmap = torch.empty(max(total, 1), dtype=torch.int32, device="cuda")[:total]
    rmap = torch.empty(max(total, 1), dtype=torch.int32, device="cuda")[:total] \
        if how in ("inner", "left", "full") else None
    if total:
        ext.join_fill(_JOIN[how], lh.data_ptr(), ldesc.data_ptr(),
                      rdesc.data_ptr(), len(lk), head.data_ptr(),
                      nxt.data_ptr(), cap, offsets.data_ptr(),
                      lmap.data_ptr(), _ptr(rmap), _ptr(right_matched), nl, s)
    lcol = Column(DType.int32(), total, lmap, None, null_count=0)
    rcol = Column(DType.int32(), total, rmap, None, null_count=0) \
        if rmap is not None else None
    return lcol, rcol


# ---------------------------------------------------------------------------
# sort: stable LSD radix over order-preserving u64 keys (native/hipdf sort.hip)
# ---------------------------------------------------------------------------

def range_key(col: Column, desc: bool, nulls_last: bool) -> Column:
    """Monotone int64 proxy of the sort position of each row (signed order;
    ties allowed). Used by external sort to range-partition rows into
    bounded buckets before the in-core sort of each bucket."""
    n = col.size
    s = _stream()
    keys = torch.empty(max(n, 1), dtype=torch.int64, device="cuda")[:n]
    if n:
        if col.dtype.id is TypeId.STRING:
            # first 8 bytes as the proxy (ties share a bucket)
            ext.make_sort_keys_str(col.offsets.data_ptr(),
                                   col.data.data_ptr(), _ptr(col.validity),
                                   0, desc, 0, keys.data_ptr(), n, s)
        elif col.dtype.id is TypeId.DECIMAL128:
            ext.make_sort_keys_i128(col.data.data_ptr(),
                                    _ptr(col.validity), 0, desc, 1,
                                    keys.data_ptr(), n, s)
        else:
            ext.make_sort_keys(_ht(col.dtype), col.data.data_ptr(),
                               _ptr(col.validity), 0, desc, nulls_last,
                               False, keys.data_ptr(), n, s)
    kc = Column(DType.int64(), n, keys, None, null_count=0)
    # make_sort_keys output orders UNSIGNED; bias the top bit for signed use
    kc = binary_op_scalar("bitxor", kc, -(1 << 63), DType.int64())
    if col.validity is not None:
        extreme = (1 << 63) - 1 if nulls_last else -(1 << 63)
        nm = is_null(col)
        kc = if_else(nm, Column.full(extreme, DType.int64(), n, "cuda"), kc)
    return Column(DType.int64(), n, kc.data, None, null_count=0)


def sort_order(batch: ColumnBatch, key_idx: List[int], descending: List[bool],
               nulls_last: List[bool]) -> Column:
    n = batch.num_rows
    s = _stream()
    if n == 0:
        return _empty_col(DType.int32())
    perm: Optional[torch.Tensor] = None
    keys_a = torch.empty(n, dtype=torch.int64, device="cuda")
    keys_b = torch.empty(n, dtype=torch.int64, device="cuda")
    perm_a = torch.empty(n, dtype=torch.int32, device="cuda")
    perm_b = torch.empty(n, dtype=torch.int32, device="cuda")
    nb = ext.sort_num_blocks(n)
    counts = torch.empty(256 * nb, dtype=torch.int64, device="cuda")
    # least-significant key first: stability carries earlier orders forward
    for ci, desc, nl in reversed(list(zip(key_idx, descending, nulls_last))):
        col = batch.columns[ci]
        wide = col.dtype.id in (TypeId.STRING, TypeId.DECIMAL128)
        t = 4 if wide else _ht(col.dtype)
        has_valid = col.validity is not None
        vwidth = ext.sort_key_width(t)
        embedded_null = vwidth < 8  # null byte fits above the value bytes

        cur_keys, alt_keys = keys_a, keys_b
        cur_perm, alt_perm = perm, perm_a if perm is not perm_a else perm_b

        def _passes(npasses, shift0=0):
            nonlocal cur_keys, alt_keys, cur_perm, alt_perm
            for p in range(npasses):
                sh = 8 * (shift0 + p)
                ext.radix_count(cur_keys.data_ptr(), sh, counts.data_ptr(), n, s)
                offsets, _ = _exclusive_scan_i64(counts)
                ext.radix_scatter(cur_keys.data_ptr(),
                                  0 if cur_perm is None else cur_perm.data_ptr(),
                                  sh, offsets.data_ptr(), alt_keys.data_ptr(),
                                  alt_perm.data_ptr(), n, s)
                cur_keys, alt_keys = alt_keys, cur_keys
                nxt = perm_b if alt_perm is perm_a else perm_a
                cur_perm, alt_perm = alt_perm, nxt

        if col.dtype.id is TypeId.STRING:
            # LSD over 8-byte big-endian chunks, last chunk first; the
            # stable radix carries earlier chunk orders forward
            lens = Column(DType.int32(), n, col.offsets[1:], None,
                          null_count=0)
            starts0 = Column(DType.int32(), n, col.offsets[:n], None,
                             null_count=0)
            ldiff = _binary("sub", lens, starts0, None, DType.int32())
            maxlen = reduce("max", ldiff) or 0
            nchunks = max(1, (int(maxlen) + 7) // 8)
            for chunk in range(nchunks - 1, -1, -1):
                ext.make_sort_keys_str(
                    col.offsets.data_ptr(), col.data.data_ptr(),
                    _ptr(col.validity),
                    0 if cur_perm is None else cur_perm.data_ptr(), desc,
                    chunk, cur_keys.data_ptr(), n, s)
                _passes(8)
        elif col.dtype.id is TypeId.DECIMAL128:
            for word in (0, 1):  # lo (unsigned) then hi (sign-biased)
                ext.make_sort_keys_i128(
                    col.data.data_ptr(), _ptr(col.validity),
                    0 if cur_perm is None else cur_perm.data_ptr(), desc,
                    word, cur_keys.data_ptr(), n, s)
                _passes(8)
        else:
            ext.make_sort_keys(t, col.data.data_ptr(), _ptr(col.validity),
                               0 if cur_perm is None else cur_perm.data_ptr(),
                               desc, nl, False, cur_keys.data_ptr(), n, s)
            _passes(vwidth + (1 if (has_valid and embedded_null) else 0))
        wide_key = col.dtype.id in (TypeId.STRING, TypeId.DECIMAL128) \
            or not embedded_null
        if has_valid and wide_key:
            # no embedded null byte: one extra null-ordering pass
            ext.make_sort_keys(4, col.data.data_ptr(), _ptr(col.validity),
                               0 if cur_perm is None else cur_perm.data_ptr(),
                               desc, nl, True, cur_keys.data_ptr(), n, s)
            _passes(1)
        perm = cur_perm
        keys_a, keys_b = cur_keys, alt_keys
    if perm is None:
        perm = torch.arange(n, dtype=torch.int32, device="cuda")
    return Column(DType.int32(), n, perm.clone(), None, null_count=0)


def str_pad(col: Column, width: int, fill: str, left: bool) -> Column:
    n = col.size
    s = _stream()
    v = col.validity.clone() if col.validity is not None else None
    fb = fill.encode("utf-8")
    fill_t = _pattern_tensor(fill) if fb else torch.zeros(
        1, dtype=torch.uint8, device="cuda")
    lens = torch.empty(max(n, 1), dtype=torch.int64, device="cuda")[:n]
    if n:
        ext.str_pad(1 if left else 0, col.offsets.data_ptr(),
                    col.data.data_ptr(), fill_t.data_ptr(), len(fb),
                    len(fill), width, 0, lens.data_ptr(), 0, 0, n, s)
    scanned, total = _exclusive_scan_i64(lens) if n else (lens, 0)
    out = torch.empty(max(total, 1), dtype=torch.uint8,
                      device="cuda")[:total]
    if total:
        ext.str_pad(1 if left else 0, col.offsets.data_ptr(),
                    col.data.data_ptr(), fill_t.data_ptr(), len(fb),
                    len(fill), width, scanned.data_ptr(), lens.data_ptr(),
                    out.data_ptr(), 1, n, s)
    offs = torch.empty(n + 1, dtype=torch.int32, device="cuda")
    if n:
        ext.narrow_i64_i32(scanned.data_ptr(), offs.data_ptr(), n, s)
    offs[n] = total
    return Column(DType.string(), n, out, v, offs,
                  null_count=col._null_count)


def str_locate(col: Column, substr: str, pos: int = 1) -> Column:
    n = col.size
    s = _stream()
    v = col.validity.clone() if col.validity is not None else None
    nb = substr.encode("utf-8")
    needle = _pattern_tensor(substr) if nb else torch.zeros(
        1, dtype=torch.uint8, device="cuda")
    out = _alloc(n, DType.int32())
    if n:
        ext.str_locate(col.offsets.data_ptr(), col.data.data_ptr(),
                       needle.data_ptr(), len(nb), pos, out.data_ptr(),
                       n, s)
    return Column(DType.int32(), n, out, v, null_count=col._null_count)


_TZ_DEV_CACHE: dict = {}


def _tz_device(zone: str):
    hit = _TZ_DEV_CACHE.get(zone)
    if hit is None:
        from ..tools import tzdb

        trans, offs = tzdb.load(zone)
        hit = (torch.from_numpy(trans.copy()).cuda(),
               torch.from_numpy(offs.copy()).cuda())
        _TZ_DEV_CACHE[zone] = hit
    return hit


def tz_convert(col: Column, zone: str, to_utc: bool) -> Column:
    n = col.size
    s = _stream()
    v = col.validity.clone() if col.validity is not None else None
    trans, offs = _tz_device(zone)
    out = _alloc(n, DType.timestamp())
    if n:
        ext.tz_convert(col.data.data_ptr(), trans.data_ptr(),
                       offs.data_ptr(), trans.numel(), 1 if to_utc else 0,
                       out.data_ptr(), n, s)
    return Column(DType.timestamp(), n, out, v,
                  null_count=col._null_count)


def _token_tensor(tokens):
    flat = []
    for kind, arg in tokens:
        flat.extend([kind, arg])
    return torch.tensor(flat, dtype=torch.int32).cuda()


def date_format(col: Column, tokens, width: int) -> Column:
    n = col.size
    s = _stream()
    v = col.validity.clone() if col.validity is not None else None
    tok = _token_tensor(tokens)
    out = torch.empty(max(n * width, 1), dtype=torch.uint8,
                      device="cuda")[:n * width]
    if n:
        ext.date_format(col.data.data_ptr(), tok.data_ptr(), len(tokens),
                        width, out.data_ptr(), n, s)
    offs = torch.arange(0, (n + 1) * width, width, dtype=torch.int32,
                        device="cuda")
    return Column(DType.string(), n, out, v, offs,
                  null_count=col._null_count)


def ts_parse(col: Column, tokens, width: int) -> Column:
    n = col.size
    s = _stream()
    tok = _token_tensor(tokens)
    out = _alloc(n, DType.timestamp())
    ov = _alloc_mask(n)
    if n:
        ext.ts_parse(col.offsets.data_ptr(), col.data.data_ptr(),
                     _ptr(col.validity), tok.data_ptr(), len(tokens),
                     width, out.data_ptr(), ov.data_ptr(), n, s)
    return Column(DType.timestamp(), n, out, ov, null_count=None)


def and_parent_validity(kid: Column, parent: Column) -> Column:
    """Null out child rows where the parent (struct) row is null."""
    if parent.validity is None:
        return kid
    n = kid.size
    s = _stream()
    mask = _and_masks(kid.validity, parent.validity) \
        if kid.validity is not None else parent.validity.clone()
    return Column(kid.dtype, n, kid.data, mask, kid.offsets,
                  null_count=None, child=kid.child)


def _xxhash64_tensor(cols: List[Column], seed: int = 42,
                     sel: Optional[torch.Tensor] = None,
                     n_out: Optional[int] = None) -> torch.Tensor:
    """Column-chained xxHash64 (Hash.xxhash64 analogue; fixed-width
    values widened to 8 bytes, floats normalized, strings over bytes)."""
    n = cols[0].size if sel is None else (n_out if n_out is not None
                                          else sel.numel())
    s = _stream()
    seeds = torch.full((max(n, 1),), seed, dtype=torch.int64,
                       device="cuda")[:n]
    selp = 0 if sel is None else sel.data_ptr()
    for c in cols:
        if n == 0:
            break
        if c.dtype.id is TypeId.STRING:
            ext.xxhash64_str(c.offsets.data_ptr(), c.data.data_ptr(),
                             _ptr(c.validity), selp, seeds.data_ptr(), n, s)
        else:
            ext.xxhash64_col(_HASH_KIND[c.dtype.id], _ht(c.dtype),
                             c.data.data_ptr(), _ptr(c.validity), selp,
                             seeds.data_ptr(), n, s)
    return seeds


def xxhash64(cols: List[Column], seed: int = 42) -> Column:
    t = _xxhash64_tensor(cols, seed)
    return Column(DType.int64(), cols[0].size, t, None, null_count=0)


def _gb_hll(vc: Column, row_gid: torch.Tensor, selp, n: int, ngroups: int,
            p: int, s) -> Column:
    import numpy as np

    sel_t = None
    if selp:
        raise NotImplementedError("approx_count_distinct under filter "
                                  "fusion")
    from .cpu_backend import _hll_estimate_host

    hashes = _xxhash64_tensor([vc], 42)
    m = 1 << p
    regs = torch.zeros(max(ngroups * m, 1), dtype=torch.uint8,
                       device="cuda")
    if n:
        ext.gb_hll(hashes.data_ptr(), _ptr(vc.validity),
                   row_gid.data_ptr(), 0, regs.data_ptr(), p, n, s)
    regs_np = regs.cpu().numpy().reshape(ngroups, m)
    est = _hll_estimate_host(regs_np)
    return Column.from_numpy(est, DType.int64()).cuda()

"""Arrow-layout columnar data model, host and device.

The device representation is the zero-copy contract of the whole engine
(reference analogue: GpuColumnVector over a cudf column,
sql-plugin/src/main/java/com/nvidia/spark/rapids/GpuColumnVector.java):

- fixed-width column: typed data buffer of `size` elements
- string column:      int32 offsets buffer (size+1) + uint8 bytes buffer
- validity:           optional Arrow bitmask, LSB-first, uint8 buffer of
                      ceil(size/8) bytes (bit set = valid), 64B padded

Buffers are torch tensors so they live in the PyTorch-ROCm caching allocator
(the RMM-style pool on 288 GB HBM3E) and can move over RCCL. All GPU compute
on them is done by hand-written HIP kernels in native/hipdf — torch is the
allocator/transport substrate, not the compute path.
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import List, Optional, Sequence

import numpy as np
import torch

from .types import DType, TypeId

_TORCH_DTYPES = {
    TypeId.BOOL: torch.uint8,
    TypeId.INT8: torch.int8,
    TypeId.INT16: torch.int16,
    TypeId.INT32: torch.int32,
    TypeId.INT64: torch.int64,
    TypeId.FLOAT32: torch.float32,
    TypeId.FLOAT64: torch.float64,
    TypeId.DECIMAL64: torch.int64,
    TypeId.DATE32: torch.int32,
    TypeId.TIMESTAMP: torch.int64,
}


def torch_dtype(dt: DType) -> torch.dtype:
    if dt.id is TypeId.DECIMAL128:
        return torch.int64  # 2 words per row, interleaved (lo, hi) LE
    return _TORCH_DTYPES[dt.id]


def dec128_pack(values, scale: int) -> np.ndarray:
    """Python ints/floats -> interleaved (lo, hi) int64 pairs."""
    import decimal

    out = np.zeros(2 * len(values), dtype=np.uint64)
    for i, v in enumerate(values):
        if v is None:
            continue
        if isinstance(v, float):
            u = int(round(v * (10 ** scale)))
        elif isinstance(v, decimal.Decimal):
            u = int(v.scaleb(scale))
        else:
            u = int(v)
        u &= (1 << 128) - 1
        out[2 * i] = u & 0xFFFFFFFFFFFFFFFF
        out[2 * i + 1] = (u >> 64) & 0xFFFFFFFFFFFFFFFF
    return out.view(np.int64)


def dec128_unpack(arr: np.ndarray):
    """Interleaved pairs -> python ints (signed 128-bit)."""
    u = arr.view(np.uint64)
    out = []
    for i in range(len(u) // 2):
        v = int(u[2 * i]) | (int(u[2 * i + 1]) << 64)
        if v >= 1 << 127:
            v -= 1 << 128
        out.append(v)
    return out


def mask_nbytes(size: int) -> int:
    """Bitmask bytes for `size` rows, padded to 64 B like Arrow."""
    return ((size + 7) // 8 + 63) // 64 * 64


def make_validity(valid: np.ndarray, device="cpu") -> torch.Tensor:
    """Pack a boolean numpy array into an Arrow LSB-first bitmask tensor."""
    packed = np.packbits(valid.astype(np.uint8), bitorder="little")
    buf = np.zeros(mask_nbytes(len(valid)), dtype=np.uint8)
    buf[: len(packed)] = packed
    t = torch.from_numpy(buf)
    return t.to(device) if device != "cpu" else t


def unpack_validity(mask: torch.Tensor, size: int) -> np.ndarray:
    arr = mask.cpu().numpy()
    bits = np.unpackbits(arr, bitorder="little")[:size]
    return bits.astype(bool)


class Column:
    """One column of data; immutable by convention."""

    __slots__ = ("dtype", "size", "data", "validity", "offsets",
                 "_null_count", "child", "_minmax")

    def __init__(
        self,
        dtype: DType,
        size: int,
        data: torch.Tensor,
        validity: Optional[torch.Tensor] = None,
        offsets: Optional[torch.Tensor] = None,
        null_count: Optional[int] = None,
        child: Optional["Column"] = None,
    ):
        self.dtype = dtype
        self.size = size
        self.data = data
        self.validity = validity
        self.offsets = offsets
        self._null_count = null_count
        self.child = child  # LIST element column / tuple for STRUCT
        self._minmax = None  # cached (min, max) for dense-key group-by
        if dtype.id is TypeId.STRING:
            assert offsets is not None and offsets.numel() == size + 1
        if dtype.id in (TypeId.LIST, TypeId.MAP):
            assert offsets is not None and child is not None
        if dtype.id is TypeId.STRUCT:
            assert isinstance(child, tuple) and \
                len(child) == len(dtype.children)

    # ---- properties ---------------------------------------------------
    @property
    def device(self) -> str:
        return "cuda" if self.data.is_cuda else "cpu"

    @property
    def is_cuda(self) -> bool:
        return self.data.is_cuda

    @property
    def has_nulls(self) -> bool:
        return self.null_count > 0

    @property
    def null_count(self) -> int:
        if self._null_count is None:
            if self.validity is None:
                self._null_count = 0
            else:
                nbits = int(unpack_validity(self.validity, self.size).sum())
                self._null_count = self.size - nbits
        return self._null_count

    @property
    def nbytes(self) -> int:
        n = self.data.numel() * self.data.element_size()
        if self.validity is not None:
            n += self.validity.numel()
        if self.offsets is not None:
            n += self.offsets.numel() * 4
        if isinstance(self.child, tuple):
            n += sum(c.nbytes for c in self.child)
        elif self.child is not None:
            n += self.child.nbytes
        return n

    # ---- movement -----------------------------------------------------
    def to(self, device: str, non_blocking: bool = False) -> "Column":
        if device == self.device:
            return self
        child = self.child
        if isinstance(child, tuple):
            child = tuple(c.to(device, non_blocking=non_blocking)
                          for c in child)
        elif child is not None:
            child = child.to(device, non_blocking=non_blocking)
        return Column(
            self.dtype,
            self.size,
            self.data.to(device, non_blocking=non_blocking),
            None if self.validity is None else self.validity.to(device, non_blocking=non_blocking),
            None if self.offsets is None else self.offsets.to(device, non_blocking=non_blocking),
            self._null_count,
            child,
        )

    def cuda(self) -> "Column":
        return self.to("cuda")

    def cpu(self) -> "Column":
        return self.to("cpu")

    # ---- construction -------------------------------------------------
    @staticmethod
    def from_numpy(arr: np.ndarray, dtype: Optional[DType] = None,
                   valid: Optional[np.ndarray] = None, device: str = "cpu") -> "Column":
        if dtype is None:
            dtype = _infer_dtype(arr.dtype)
        np_dt = dtype.numpy_dtype()
        arr = np.ascontiguousarray(arr, dtype=np_dt)
        data = torch.from_numpy(arr)
        if device != "cpu":
            data = data.to(device)
        validity = None
        nc = 0
        if valid is not None:
            nc = int(len(valid) - valid.sum())
            if nc > 0:
                validity = make_validity(valid, device)
            else:
                validity = None
        return Column(dtype, len(arr), data, validity, null_count=nc if validity is not None else 0)

    @staticmethod
    def from_pylist(values: Sequence, dtype: DType, device: str = "cpu") -> "Column":
        n = len(values)
        if dtype.id is TypeId.STRUCT:
            valid = np.array([v is not None for v in values], dtype=bool)
            kids = []
            for name, cdt in zip(dtype.field_names, dtype.children):
                kid_vals = [None if v is None else
                            (v.get(name) if isinstance(v, dict)
                             else v[dtype.field_names.index(name)])
                            for v in values]
                kids.append(Column.from_pylist(kid_vals, cdt))
            col = Column(dtype, n, torch.zeros(0, dtype=torch.uint8),
                         make_validity(valid) if not valid.all() else None,
                         None, None, tuple(kids))
            return col.to(device) if device != "cpu" else col
        if dtype.id in (TypeId.LIST, TypeId.MAP):
            valid = np.array([v is not None for v in values], dtype=bool)
            flat: list = []
            offsets = np.zeros(n + 1, dtype=np.int32)
            for i, v in enumerate(values):
                if v is not None:
                    flat.extend(v.items() if isinstance(v, dict) else v)
                offsets[i + 1] = len(flat)
            elem_dt = dtype.entry_dtype if dtype.id is TypeId.MAP \
                else dtype.children[0]
            child = Column.from_pylist(flat, elem_dt)
            col = Column(dtype, n, torch.zeros(0, dtype=torch.uint8),
                         make_validity(valid) if not valid.all() else None,
                         torch.from_numpy(offsets), None, child)
            return col.to(device) if device != "cpu" else col
        if dtype.id is TypeId.STRING:
            valid = np.array([v is not None for v in values], dtype=bool)
            parts = [(v if v is not None else "").encode("utf-8") for v in values]
            offsets = np.zeros(n + 1, dtype=np.int32)
            np.cumsum([len(p) for p in parts], out=offsets[1:])
            data = np.frombuffer(b"".join(parts), dtype=np.uint8).copy() if parts else np.zeros(0, np.uint8)
            col = Column(
                dtype, n,
                torch.from_numpy(data),
                make_validity(valid) if not valid.all() else None,
                torch.from_numpy(offsets),
            )
            return col.to(device) if device != "cpu" else col
        valid = np.array([v is not None for v in values], dtype=bool)
        if dtype.id is TypeId.DECIMAL128:
            packed = dec128_pack(values, dtype.scale)
            data = torch.from_numpy(packed.copy())
            validity = make_validity(valid) if not valid.all() else None
            col = Column(dtype, n, data, validity,
                         null_count=int(n - valid.sum()) if validity is not None else 0)
            return col.to(device) if device != "cpu" else col
        if dtype.id is TypeId.DECIMAL64:
            # same input convention as dec128_pack: Decimal/float inputs
            # are VALUES (scaled here), plain ints are raw unscaled units
            import decimal as _dec

            conv = []
            for v in values:
                if v is None:
                    conv.append(0)
                elif isinstance(v, float):
                    conv.append(int(round(v * 10 ** dtype.scale)))
                elif isinstance(v, _dec.Decimal):
                    conv.append(int(v.scaleb(dtype.scale)))
                else:
                    conv.append(int(v))
            dense = np.array(conv, dtype=np.int64)
            return Column.from_numpy(dense, dtype,
                                     None if valid.all() else valid, device)
        np_dt = dtype.numpy_dtype()
        fill = 0
        dense = np.array([v if v is not None else fill for v in values], dtype=np_dt)
        if dtype.id is TypeId.BOOL:
            dense = dense.astype(np.uint8)
        return Column.from_numpy(dense, dtype, None if valid.all() else valid, device)

    @staticmethod
    def full(value, dtype: DType, size: int, device: str = "cpu") -> "Column":
        """Broadcast a non-null scalar to a column without a python list."""
        assert value is not None
        if dtype.id is TypeId.STRING:
            b = str(value).encode("utf-8")
            data = torch.from_numpy(
                np.frombuffer(b * size, dtype=np.uint8).copy()) if size else \
                torch.zeros(0, dtype=torch.uint8)
            offsets = torch.arange(0, (size + 1) * len(b) or 1, len(b) or 1,
                                   dtype=torch.int32)[: size + 1] if b else \
                torch.zeros(size + 1, dtype=torch.int32)
            col = Column(dtype, size, data, None, offsets, 0)
            return col.to(device) if device != "cpu" else col
        if dtype.id is TypeId.BOOL:
            value = int(bool(value))
        if dtype.id is TypeId.DECIMAL128:
            one = torch.from_numpy(dec128_pack([value], dtype.scale).copy())
            data = one.repeat(size).to(device) if device != "cpu" \
                else one.repeat(size)
            return Column(dtype, size, data, None, null_count=0)
        if dtype.id is TypeId.DECIMAL64:
            import decimal as _dec

            if isinstance(value, _dec.Decimal):
                value = int(value.scaleb(dtype.scale)
                            .to_integral_value(_dec.ROUND_HALF_UP))
            elif isinstance(value, float):
                value = int(round(value * (10 ** dtype.scale)))
        data = torch.full((size,), value, dtype=torch_dtype(dtype),
                          device=device)
        return Column(dtype, size, data, None, null_count=0)

    @staticmethod
    def nulls(dtype: DType, size: int, device: str = "cpu") -> "Column":
        if dtype.id is TypeId.STRING:
            data = torch.zeros(0, dtype=torch.uint8, device=device)
            offsets = torch.zeros(size + 1, dtype=torch.int32, device=device)
            validity = torch.zeros(mask_nbytes(size), dtype=torch.uint8,
                                   device=device)
            return Column(dtype, size, data, validity, offsets,
                          null_count=size)
        nwords = 2 * size if dtype.id is TypeId.DECIMAL128 else size
        data = torch.zeros(nwords, dtype=torch_dtype(dtype), device=device)
        validity = torch.zeros(mask_nbytes(size), dtype=torch.uint8, device=device)
        offsets = None
        if dtype.id is TypeId.STRING:
            offsets = torch.zeros(size + 1, dtype=torch.int32, device=device)
            data = torch.zeros(0, dtype=torch.uint8, device=device)
        return Column(dtype, size, data, validity, offsets, null_count=size)

    # ---- host conversion ----------------------------------------------
    def valid_array(self) -> np.ndarray:
        if self.validity is None:
            return np.ones(self.size, dtype=bool)
        return unpack_validity(self.validity, self.size)

    def to_numpy(self) -> np.ndarray:
        assert self.dtype.id is not TypeId.STRING
        return self.data.cpu().numpy()[: self.size]

    def to_pylist(self) -> list:
        valid = self.valid_array()
        if self.dtype.id is TypeId.STRUCT:
            kid_lists = [c.to_pylist() for c in self.child]
            names = self.dtype.field_names
            return [dict(zip(names, row)) if valid[i] else None
                    for i, row in enumerate(zip(*kid_lists))] if kid_lists \
                else [{} if v else None for v in valid]
        if self.dtype.id is TypeId.LIST:
            offs = self.offsets.cpu().numpy()
            elems = self.child.to_pylist()
            return [list(elems[offs[i]:offs[i + 1]]) if valid[i] else None
                    for i in range(self.size)]
        if self.dtype.id is TypeId.MAP:
            offs = self.offsets.cpu().numpy()
            elems = self.child.to_pylist()  # entry dicts {key, value}
            return [{e["key"]: e["value"]
                     for e in elems[offs[i]:offs[i + 1]]}
                    if valid[i] else None for i in range(self.size)]
        if self.dtype.id is TypeId.STRING:
            offs = self.offsets.cpu().numpy()
            raw = self.data.cpu().numpy().tobytes()
            out = []
            for i in range(self.size):
                if not valid[i]:
                    out.append(None)
                else:
                    out.append(raw[offs[i]: offs[i + 1]].decode("utf-8"))
            return out
        if self.dtype.id is TypeId.DECIMAL128:
            import decimal

            ints = dec128_unpack(self.data.cpu().numpy()[: 2 * self.size])
            scale = self.dtype.scale
            ctx = decimal.Context(prec=50)  # decimal128 needs > default 28
            return [(decimal.Decimal(v).scaleb(-scale, ctx) if scale else v)
                    if ok else None for v, ok in zip(ints, valid)]
        arr = self.to_numpy()
        if self.dtype.id is TypeId.BOOL:
            arr = arr.astype(bool)
        if self.dtype.is_decimal:
            import decimal

            scale = self.dtype.scale
            return [
                (decimal.Decimal(int(v)).scaleb(-scale) if scale else int(v))
                if ok else None
                for v, ok in zip(arr, valid)
            ]
        return [v.item() if ok else None for v, ok in zip(arr, valid)]

    def __repr__(self):
        return (f"Column({self.dtype}, size={self.size}, device={self.device}, "
                f"nulls={self.null_count})")


def _infer_dtype(np_dtype) -> DType:
    m = {
        np.dtype(np.bool_): TypeId.BOOL,
        np.dtype(np.int8): TypeId.INT8,
        np.dtype(np.int16): TypeId.INT16,
        np.dtype(np.int32): TypeId.INT32,
        np.dtype(np.int64): TypeId.INT64,
        np.dtype(np.float32): TypeId.FLOAT32,
        np.dtype(np.float64): TypeId.FLOAT64,
        np.dtype(np.uint8): TypeId.BOOL,
    }
    return DType(m[np.dtype(np_dtype)])


@dataclass
class Field:
    name: str
    dtype: DType
    nullable: bool = True


class Schema:
    def __init__(self, fields: List[Field]):
        self.fields = list(fields)

    @property
    def names(self):
        return [f.name for f in self.fields]

    def __len__(self):
        return len(self.fields)

    def __iter__(self):
        return iter(self.fields)

    def index(self, name: str) -> int:
        return self.names.index(name)

    def field(self, name: str) -> Field:
        return self.fields[self.index(name)]

    def __repr__(self):
        inner = ", ".join(f"{f.name}:{f.dtype}" for f in self.fields)
        return f"Schema({inner})"


class ColumnBatch:
    """A batch of columns — the unit flowing between physical operators
    (reference analogue: ColumnarBatch of GpuColumnVector)."""

    __slots__ = ("columns", "num_rows")

    def __init__(self, columns: List[Column], num_rows: Optional[int] = None):
        self.columns = list(columns)
        if num_rows is None:
            num_rows = columns[0].size if columns else 0
        self.num_rows = num_rows
        for c in self.columns:
            assert c.size == self.num_rows, "ragged batch"

    @property
    def num_columns(self) -> int:
        return len(self.columns)

    @property
    def device(self) -> str:
        return self.columns[0].device if self.columns else "cpu"

    @property
    def is_cuda(self) -> bool:
        return bool(self.columns) and self.columns[0].is_cuda

    @property
    def nbytes(self) -> int:
        return sum(c.nbytes for c in self.columns)

    def column(self, i: int) -> Column:
        return self.columns[i]

    def to(self, device: str) -> "ColumnBatch":
        if self.device == device:
            return self
        return ColumnBatch([c.to(device) for c in self.columns], self.num_rows)

    def cuda(self) -> "ColumnBatch":
        return self.to("cuda")

    def cpu(self) -> "ColumnBatch":
        return self.to("cpu")

    def select(self, indices: Sequence[int]) -> "ColumnBatch":
        return ColumnBatch([self.columns[i] for i in indices], self.num_rows)

    def __repr__(self):
        return f"ColumnBatch({self.num_columns} cols x {self.num_rows} rows, {self.device})"

"""GPU-decode Parquet reader: host footer/page-header parse + CPU
decompression, then hipdf kernels decode PLAIN / *_DICTIONARY pages and
RLE/bit-packed definition levels on the MI355X.

Reference analogue: GpuParquetScan + libcudf's parquet decode kernels
(SURVEY.md §2.3/§2.8); the reference's CPU-decompress option
(CpuCompressionConfig, GpuParquetScan.scala:1549) is the model for the
decompress-on-host/decode-on-device split used here.

Scope: flat schemas plus one-level LIST (repetition levels decoded on
device), STRUCT (dotted leaves, struct validity from def levels) and MAP
(key/value leaves through the LIST path); INT32/INT64/FLOAT/DOUBLE/
BOOLEAN (+date32/timestamp/decimal on those physical types, including
width-widening INT32-physical decimals) PLAIN, dictionary or
DELTA_BINARY_PACKED; BYTE_ARRAY strings PLAIN, dictionary,
DELTA_BYTE_ARRAY or DELTA_LENGTH_BYTE_ARRAY. Anything else raises
NotImplementedError and the caller falls back to the CPU (hybrid) reader
per file.
"""
from __future__ import annotations

import struct as pystruct
from typing import List, Optional

import numpy as np
import torch

from ..column import Column, ColumnBatch, mask_nbytes
from ..types import DType, TypeId
from . import thrift_compact as tc

PLAIN, PLAIN_DICTIONARY, RLE, RLE_DICTIONARY = 0, 2, 3, 8
DELTA_BINARY_PACKED = 5
DELTA_LENGTH_BYTE_ARRAY, DELTA_BYTE_ARRAY = 6, 7
DATA_PAGE, DICTIONARY_PAGE, DATA_PAGE_V2 = 0, 2, 3

_PHYS_NP = {
    "INT32": np.dtype("<i4"),
    "INT64": np.dtype("<i8"),
    "FLOAT": np.dtype("<f4"),
    "DOUBLE": np.dtype("<f8"),
}


_PINNED = __import__("threading").local()

# (path, mtime) -> (mmap, base_addr, nbytes, registered) — the mapping is
# hipHostRegister'ed once so chunk uploads are direct DMA from the page
# cache (no pinned bounce memcpy); kept alive for the process lifetime
_MMAP_CACHE: dict = {}
_MMAP_LOCK = __import__("threading").Lock()


def _file_mmap(path: str):
    import mmap as _mmap
    import os as _os

    key = (path, _os.path.getmtime(path))
    with _MMAP_LOCK:
        hit = _MMAP_CACHE.get(key)
        if hit is not None:
            return hit
        f = open(path, "rb")
        mm = _mmap.mmap(f.fileno(), 0, prot=_mmap.PROT_READ)
        f.close()
        arr = np.frombuffer(memoryview(mm), dtype=np.uint8)
        base = arr.ctypes.data
        # NOTE: hipHostRegister on the file-backed mapping looked
        # attractive (direct DMA from page cache) but the GPU faults on
        # DMA from MAP_PRIVATE file pages on this stack — uploads bounce
        # through the per-thread pinned buffer instead
        if len(_MMAP_CACHE) > 512:
            _MMAP_CACHE.clear()
        hit = (mm, memoryview(mm), base, arr.nbytes, False)
        _MMAP_CACHE[key] = hit
        return hit


def _upload_parts(parts, ext, stream, torch_dtype=torch.uint8,
                  pad: int = 0) -> torch.Tensor:
    """Upload several host byte views into one device buffer. Views that
    live inside a hipHostRegister'ed mapping go straight to DMA
    (hipMemcpyAsync from pinned memory); others bounce through the
    per-thread pinned buffer."""
    arrs = [a if isinstance(a, np.ndarray) else
            np.frombuffer(a, dtype=np.uint8) for a in parts]
    total = sum(a.nbytes for a in arrs) + pad
    if total == 0:
        return torch.zeros(0, dtype=torch_dtype, device="cuda")
    dev = torch.empty(total, dtype=torch.uint8, device="cuda")
    off = 0
    for a in arrs:
        n = a.nbytes
        if n == 0:
            continue
        rc = ext.memcpy_h2d(dev.data_ptr() + off, a.ctypes.data, n, stream)
        if rc != 0:
            raise RuntimeError(f"hipMemcpyAsync failed rc={rc}")
        off += n
    if pad:
        dev[total - pad:] = 0
    itemsize = torch.empty(0, dtype=torch_dtype).element_size()
    return dev.view(torch_dtype) if itemsize == 1 else         dev[: (total // itemsize) * itemsize].view(torch_dtype)


def _upload(data, torch_dtype=torch.uint8) -> torch.Tensor:
    """bytes/ndarray -> device tensor through a reusable pinned staging
    buffer (one host memcpy + async DMA, instead of frombuffer().copy()
    + a pageable H2D which stages internally anyway). Per-thread buffer:
    the multi-file prefetch pool uploads concurrently."""
    if isinstance(data, np.ndarray):
        arr = data.view(np.uint8).reshape(-1)
    else:
        arr = np.frombuffer(data, dtype=np.uint8)
    nbytes = arr.nbytes
    if nbytes == 0:
        return torch.zeros(0, dtype=torch_dtype, device="cuda")
    buf = getattr(_PINNED, "buf", None)
    if buf is None or buf.numel() < nbytes:
        cap = max(nbytes, 1 << 20)
        cap = 1 << (cap - 1).bit_length()
        buf = torch.empty(cap, dtype=torch.uint8, pin_memory=True)
        _PINNED.buf = buf
    import time as _t

    from .parquet import PHASE_STATS

    t0 = _t.perf_counter()
    buf.numpy()[:nbytes] = arr
    dev = torch.empty(nbytes, dtype=torch.uint8, device="cuda")
    # blocking copy: the pinned buffer is reused by the next upload on
    # this thread, so the DMA must complete before returning
    dev.copy_(buf[:nbytes])
    PHASE_STATS["upload_s"] += _t.perf_counter() - t0
    PHASE_STATS["upload_bytes"] += nbytes
    itemsize = torch.empty(0, dtype=torch_dtype).element_size()
    return dev.view(torch_dtype) if itemsize == 1 else \
        dev[: (nbytes // itemsize) * itemsize].view(torch_dtype)


def _codec(name: str):
    import pyarrow as pa

    name = name.lower()
    if name in ("uncompressed", "none"):
        return None
    return pa.Codec(name)


def _decompress(codec, payload: bytes, usize: int) -> bytes:
    if codec is None:
        return payload
    out = codec.decompress(payload, usize)
    return out.to_pybytes() if hasattr(out, "to_pybytes") else bytes(out)


class _Page:
    __slots__ = ("n", "levels", "values", "encoding", "all_valid", "rep")

    def __init__(self, n, levels, values, encoding, all_valid, rep=b""):
        self.n = n
        self.levels = levels
        self.values = values
        self.encoding = encoding
        self.all_valid = all_valid
        self.rep = rep


def _rle_all_valid(levels: bytes, n: int, max_def: int) -> bool:
    """Host-side check that a def-level RLE stream encodes `n` copies of
    max_def (no nulls). All-valid pages are written as a single short RLE
    run, so this reads a handful of bytes and lets the decoder skip the
    level decode + validity-mask pipeline entirely."""
    pos = 0
    seen = 0
    nb = len(levels)
    byte_per_val = 1  # def levels for flat schemas: bit width 1..7 -> 1B
    while seen < n:
        if pos >= nb:
            return False
        h = 0
        shift = 0
        while pos < nb:
            b = levels[pos]
            pos += 1
            h |= (b & 0x7F) << shift
            if not (b & 0x80):
                break
            shift += 7
        if h & 1:
            return False  # bit-packed run: has a mix -> not trivially valid
        count = h >> 1
        if pos >= nb or levels[pos] != max_def:
            return False
        pos += byte_per_val
        seen += count
    return True


class _ChunkDecoder:
    """Decodes one column chunk (one column of one row group) on the GPU."""

    def __init__(self, raw: bytes, phys: str, dtype: DType, max_def: int,
                 codec, num_values: int, ext, stream: int,
                 max_rep: int = 0, list_dtype: Optional[DType] = None,
                 opt_list: bool = True, opt_elem: bool = True,
                 capture_def: bool = False):
        self.raw = raw
        self.phys = phys
        self.dtype = dtype  # ELEMENT dtype when max_rep > 0
        self.max_def = max_def
        self.codec = codec
        self.num_values = num_values
        self.ext = ext
        self.s = stream
        # nested (LIST) decode state: one repetition level, arrow-style
        # 3-level list groups (reference: GpuParquetScan nested schemas)
        self.max_rep = max_rep
        self.list_dtype = list_dtype
        self.opt_list = opt_list
        self.opt_elem = opt_elem
        # when set, the chunk's decoded def levels are stashed on
        # self.captured_def (None = trivially all max_def) so a STRUCT
        # parent can derive its own validity from any leaf's levels
        self.capture_def = capture_def
        self.captured_def: Optional[torch.Tensor] = None
        self.dict_fixed: Optional[torch.Tensor] = None  # device dict values
        self.dict_str: Optional[tuple] = None  # (offsets cuda, bytes cuda)

    def decode(self) -> Column:
        """Chunk-granular decode: page headers are parsed host-side first;
        the hot paths then run ONE upload + ONE kernel sequence for the
        whole column chunk instead of a launch storm per page (the round-1
        profile showed per-page rle_hybrid_decode + a 6-kernel validity
        chain per page as 60%+ of NDS scan kernel time)."""
        import time as _t

        from .parquet import PHASE_STATS

        t0 = _t.perf_counter()
        pages = self._parse_pages()
        PHASE_STATS["parse_s"] += _t.perf_counter() - t0
        if not pages:
            return Column.nulls(self.list_dtype if self.max_rep
                                else self.dtype, 0, "cuda")
        if self.max_rep:
            return self._decode_list(pages)
        t0 = _t.perf_counter()
        fast = self._decode_chunk(pages)
        PHASE_STATS["fast_s"] += _t.perf_counter() - t0
        if fast is not None:
            return fast
        page_cols = []
        cap = []
        for pg in pages:
            levels = None
            if not pg.all_valid and self.max_def > 0:
                levels = self._decode_levels(pg.levels, pg.n)
            cap.append(levels)
            page_cols.append(self._materialize(pg.values, pg.n, levels,
                                               pg.encoding))
        if self.capture_def and any(c is not None for c in cap):
            full = [c if c is not None else
                    torch.full((pg.n,), self.max_def, dtype=torch.int32,
                               device="cuda")
                    for c, pg in zip(cap, pages)]
            self.captured_def = torch.cat(full) if len(full) > 1 else full[0]
        if len(page_cols) == 1:
            return page_cols[0]
        from ..ops import gpu_backend

        return gpu_backend.concat_batches(
            [ColumnBatch([c]) for c in page_cols]).columns[0]

    def _decode_level_stream(self, raw: bytes, n: int, bw: int
                             ) -> torch.Tensor:
        dev = torch.from_numpy(np.frombuffer(raw, np.uint8).copy()).cuda()
        out = torch.empty(n, dtype=torch.int32, device="cuda")
        self.ext.rle_hybrid_decode(dev.data_ptr(), dev.numel(), bw,
                                   out.data_ptr(), n, self.s)
        return out

    def _decode_list(self, pages) -> Column:
        """Device decode of a one-level LIST column (arrow 3-level list
        groups). Per page: RLE-decode repetition+definition levels, derive
        rows (rep==0), per-row entry counts, list validity and the entry
        def subset; the element values then reuse the flat _materialize
        path (reference analogue: the nested-schema decode of
        GpuParquetScan/libcudf; rep depth 1 here, deeper nesting falls
        back to the hybrid reader)."""
        from ..ops import gpu_backend as gb

        opt_list = 1 if self.opt_list else 0
        entry_thr = opt_list + 1  # def >= this -> a list entry exists
        counts_parts = []
        validrow_parts = []
        elem_cols = []
        for pg in pages:
            n = pg.n
            rep = self._decode_level_stream(pg.rep, n, 1)
            deft = self._decode_level_stream(
                pg.levels, n, max(1, self.max_def.bit_length()))
            row_flag = rep == 0
            first_def = deft[row_flag]
            elem_mask = deft >= entry_thr
            n_entries = int(elem_mask.sum().item())
            row_id = torch.cumsum(row_flag.to(torch.int64), 0) - 1
            nrows = first_def.numel()
            counts = torch.zeros(nrows, dtype=torch.int64, device="cuda")
            if n_entries:
                counts.index_add_(
                    0, row_id[elem_mask],
                    torch.ones(n_entries, dtype=torch.int64,
                               device="cuda"))
            counts_parts.append(counts)
            validrow_parts.append(first_def > 0 if opt_list
                                  else torch.ones(nrows, dtype=torch.bool,
                                                  device="cuda"))
            if n_entries == 0:
                elem_cols.append(None)
                continue
            entry_defs = deft[elem_mask].contiguous() if self.opt_elem \
                else None
            elem_cols.append(self._materialize(pg.values, n_entries,
                                               entry_defs, pg.encoding))
        counts = torch.cat(counts_parts) if len(counts_parts) > 1 \
            else counts_parts[0]
        n_rows = counts.numel()
        offs64 = torch.zeros(n_rows + 1, dtype=torch.int64, device="cuda")
        torch.cumsum(counts, 0, out=offs64[1:])
        offsets = offs64.to(torch.int32)
        pieces = [c for c in elem_cols if c is not None]
        if not pieces:
            elem = gb._empty_col(self.dtype)
        elif len(pieces) == 1:
            elem = pieces[0]
        else:
            elem = gb.concat_batches(
                [ColumnBatch([c], c.size) for c in pieces]).columns[0]
        validity = None
        null_count = 0
        if opt_list:
            valid_rows = torch.cat(validrow_parts) \
                if len(validrow_parts) > 1 else validrow_parts[0]
            if not bool(valid_rows.all()):
                rv = valid_rows.to(torch.int32).contiguous()
                mask = torch.empty(mask_nbytes(n_rows), dtype=torch.uint8,
                                   device="cuda")
                self.ext.levels_to_mask(rv.data_ptr(), 1, mask.data_ptr(),
                                        n_rows, self.s)
                validity = mask
                null_count = None
        return Column(self.list_dtype, n_rows,
                      torch.zeros(0, dtype=torch.uint8, device="cuda"),
                      validity, offsets, null_count, elem)

    def _parse_pages(self):
        """Parse headers + decompress payloads for every page of the chunk;
        load the dictionary; mark pages whose def-levels are trivially
        all-valid (a single RLE run of max_def — the overwhelmingly common
        case, checked host-side in a few bytes)."""
        pos = 0
        decoded = 0
        pages = []
        while decoded < self.num_values and pos < len(self.raw):
            ph = tc.parse_page_header(self.raw, pos)
            pos += ph.header_size
            payload = self.raw[pos: pos + ph.compressed_page_size]
            pos += ph.compressed_page_size
            if ph.type == DICTIONARY_PAGE:
                data = _decompress(self.codec, payload,
                                   ph.uncompressed_page_size)
                self._load_dict(data, ph.dictionary_page.num_values)
            elif ph.type == DATA_PAGE:
                data = _decompress(self.codec, payload,
                                   ph.uncompressed_page_size)
                hdr = ph.data_page
                n = hdr.num_values
                levels = b""
                rep = b""
                vpos = 0
                if self.max_rep > 0:
                    (rl,) = pystruct.unpack_from("<I", data, 0)
                    rep = data[4:4 + rl]
                    vpos = 4 + rl
                if self.max_def > 0:
                    (lvl_len,) = pystruct.unpack_from("<I", data, vpos)
                    levels = data[vpos + 4:vpos + 4 + lvl_len]
                    vpos += 4 + lvl_len
                pages.append(_Page(n, levels, data[vpos:], hdr.encoding,
                                   self.max_rep == 0
                                   and (self.max_def == 0
                                        or _rle_all_valid(levels, n,
                                                          self.max_def)),
                                   rep))
                decoded += n
            elif ph.type == DATA_PAGE_V2:
                hdr = ph.data_page_v2
                n = hdr.num_values
                lvl_len = (hdr.def_levels_byte_length
                           + hdr.rep_levels_byte_length)
                levels = payload[hdr.rep_levels_byte_length:lvl_len]
                vals = payload[lvl_len:]
                if hdr.is_compressed and self.codec is not None:
                    vals = _decompress(self.codec, vals,
                                       ph.uncompressed_page_size - lvl_len)
                all_valid = self.max_rep == 0 and (
                    self.max_def == 0 or hdr.num_nulls == 0)
                # v2 levels carry no 4-byte length prefix and are a pure
                # RLE stream like v1's
                pages.append(_Page(n, levels, vals, hdr.encoding, all_valid,
                                   payload[:hdr.rep_levels_byte_length]))
                decoded += n
            else:
                raise NotImplementedError(f"page type {ph.type}")
        return pages

    # -- chunk-granular fast paths --------------------------------------

    def _decode_chunk(self, pages) -> Optional[Column]:
        encs = {p.encoding for p in pages}
        total = sum(p.n for p in pages)
        all_valid = all(p.all_valid for p in pages)
        if encs == {PLAIN} and self.phys in _PHYS_NP:
            return self._chunk_plain_fixed(pages, total, all_valid)
        if encs <= {PLAIN_DICTIONARY, RLE_DICTIONARY} and all_valid:
            return self._chunk_dict(pages, total)
        if encs == {PLAIN} and self.phys == "BYTE_ARRAY" and all_valid:
            # concat value sections: one plain byte-array decode for the
            # whole chunk
            blob = b"".join(p.values for p in pages)
            return self._materialize(blob, total, None, PLAIN)
        return None

    def _rle_streams_to_i32(self, streams, total) -> torch.Tensor:
        """Decode several RLE/bit-packed hybrid streams into one int32
        tensor. Fast path: host walk of the run headers (one varint per
        run, C++) + fully parallel k_rle_expand — a 2.5M-value stream is
        decoded by the whole chip instead of one workgroup. Falls back to
        the per-stream workgroup kernel when a walk fails."""
        runs_list = []
        parts = []
        src = out = 0
        ok = True
        for data, n, bw in streams:
            arr = np.frombuffer(data, dtype=np.uint8)
            max_runs = len(arr) + 2  # every run consumes >= 1 header byte
            runs = np.empty((max_runs, 4), dtype=np.int64)
            cnt = self.ext.rle_walk_host(arr.ctypes.data, len(arr), bw, n,
                                         src, out, runs.ctypes.data,
                                         max_runs)
            if cnt < 0:
                ok = False
                break
            runs_list.append(runs[:cnt])
            parts.append(data)
            src += len(arr)
            out += n
        out_t = torch.empty(max(total, 1), dtype=torch.int32,
                            device="cuda")[:total]
        if ok and total:
            # +8 zero bytes: the bit-packed unpack reads an 8-byte window
            base = _upload_parts(parts, self.ext, self.s, pad=8)
            all_runs = np.concatenate(runs_list) if len(runs_list) > 1 \
                else runs_list[0]
            rt = _upload_parts([np.ascontiguousarray(all_runs)
                                .view(np.uint8).reshape(-1)],
                               self.ext, self.s).view(torch.int64)
            self.ext.rle_expand(base.data_ptr(), rt.data_ptr(),
                                len(all_runs), 0, out_t.data_ptr(), total,
                                self.s)
            return out_t
        # fallback: per-stream workgroup decode
        base = _upload_parts([d for d, _, _ in streams], self.ext, self.s)
        descs = np.empty((len(streams), 5), dtype=np.int64)
        src = out = 0
        for i, (data, n, bw) in enumerate(streams):
            descs[i] = (src, len(data), out, n, bw)
            src += len(data)
            out += n
        dt = _upload_parts([descs.view(np.uint8).reshape(-1)],
                           self.ext, self.s).view(torch.int64)
        if total:
            self.ext.rle_hybrid_batch(base.data_ptr(), dt.data_ptr(),
                                      len(streams), out_t.data_ptr(),
                                      self.s)
        return out_t

    def _chunk_validity(self, pages, total):
        """(mask, valid_idx, n_valid) for the whole chunk: decode all
        pages' def-level RLE streams at once, then run the validity
        chain once over the contiguous levels array."""
        bw = max(1, self.max_def.bit_length())
        levels = self._rle_streams_to_i32(
            [(p.levels, p.n, bw) for p in pages], total)
        if self.capture_def:
            self.captured_def = levels
        return self._valid_parts(levels, total)

    def _chunk_plain_fixed(self, pages, total, all_valid) -> Column:
        from ..column import torch_dtype

        np_dt = _PHYS_NP[self.phys]
        dense = _upload_parts([p.values for p in pages], self.ext, self.s,
                              torch.from_numpy(np.empty(0, dtype=np_dt)).dtype)
        tdt = torch_dtype(self.dtype)
        if dense.dtype != tdt:
            dense = self._cast_raw(dense, tdt)
        if all_valid:
            return Column(self.dtype, total, dense[:total], None,
                          null_count=0)
        mask, valid_idx, n_valid = self._chunk_validity(pages, total)
        if mask is None:
            return Column(self.dtype, total, dense[:total], None,
                          null_count=0)
        out = torch.zeros(total, dtype=tdt, device="cuda")
        if n_valid:
            self.ext.scatter_fixed(self.dtype.itemsize, dense.data_ptr(),
                                   valid_idx.data_ptr(), out.data_ptr(),
                                   n_valid, self.s)
        return Column(self.dtype, total, out, mask, null_count=None)

    def _chunk_dict(self, pages, total) -> Optional[Column]:
        if self.dict_fixed is None and self.dict_str is None:
            return None
        # decode every page's index stream (first value byte = bit width)
        ridx = self._rle_streams_to_i32(
            [(p.values[1:], p.n, p.values[0] if len(p.values) else 0)
             for p in pages], total)
        return self._gather_dict(ridx, total, None)

    # -- dictionary ------------------------------------------------------
    def _load_dict(self, data: bytes, count: int):
        if self.phys in _PHYS_NP:
            vals = np.frombuffer(data, dtype=_PHYS_NP[self.phys], count=count)
            self.dict_fixed = torch.from_numpy(vals.copy()).cuda()
        elif self.phys == "BYTE_ARRAY":
            # device-side parse of the length-prefixed records: the 1M-entry
            # customer dictionary took ~300ms in a python loop here
            self.dict_str = self._byte_array_dense(data, count)
        else:
            raise NotImplementedError(f"dict for {self.phys}")

    def _string_dense_to_rows(self, offs_dense, out_bytes, n, mask,
                              valid_idx, n_valid) -> Column:
        """Dense (valid-only) string buffers -> row-level string column,
        scattering through a null mask when present."""
        from ..ops import gpu_backend as gb

        dense_col = Column(DType.string(), n_valid, out_bytes, None,
                           offs_dense, 0)
        if mask is None:
            return dense_col
        # scatter dense -> rows with nulls: build ridx then string-gather
        ridx = torch.full((n,), -1, dtype=torch.int32, device="cuda")
        dense_iota = torch.empty(max(n_valid, 1), dtype=torch.int32,
                                 device="cuda")[:n_valid]
        if n_valid:
            self.ext.iota_i32(dense_iota.data_ptr(), n_valid, self.s)
            self.ext.scatter_fixed(4, dense_iota.data_ptr(),
                                   valid_idx.data_ptr(), ridx.data_ptr(),
                                   n_valid, self.s)
        out = gb._gather_col(dense_col, ridx, n, maybe_negative=True)
        return Column(self.dtype, n, out.data, mask, out.offsets,
                      null_count=None)

    def _delta_byte_array_dense(self, values: bytes, count: int,
                                encoding: int):
        """Host decode of DELTA_LENGTH_BYTE_ARRAY / DELTA_BYTE_ARRAY
        string pages -> (offsets int64 [count+1], dense payload bytes).
        The length/prefix vectors are serial varint+bitpack chains
        (pq_delta_walk in C++); reconstruction for DELTA_BYTE_ARRAY is a
        serial prefix-chain memcpy pass (delta_ba_concat). Reference
        analogue: GpuParquetScan's cudf delta decoders."""
        if count == 0:
            return np.zeros(1, dtype=np.int64), b""
        arr = np.frombuffer(values, dtype=np.uint8)
        if encoding == DELTA_LENGTH_BYTE_ARRAY:
            lens = np.empty(count, dtype=np.int64)
            consumed = self.ext.pq_delta_walk_host(
                arr.ctypes.data, len(arr), lens.ctypes.data, count)
            if consumed < 0 or lens.min() < 0:
                raise NotImplementedError("corrupt DELTA_LENGTH stream")
            offs = np.zeros(count + 1, dtype=np.int64)
            np.cumsum(lens, out=offs[1:])
            total = int(offs[-1])
            if consumed + total > len(arr):
                raise NotImplementedError("DELTA_LENGTH payload overrun")
            return offs, np.ascontiguousarray(
                arr[consumed:consumed + total])
        pre = np.empty(count, dtype=np.int64)
        c1 = self.ext.pq_delta_walk_host(arr.ctypes.data, len(arr),
                                         pre.ctypes.data, count)
        if c1 < 0:
            raise NotImplementedError("corrupt DELTA_BYTE_ARRAY prefixes")
        suf = np.empty(count, dtype=np.int64)
        c2 = self.ext.pq_delta_walk_host(arr.ctypes.data + c1,
                                         len(arr) - c1, suf.ctypes.data,
                                         count)
        if c2 < 0:
            raise NotImplementedError("corrupt DELTA_BYTE_ARRAY suffixes")
        offs = np.zeros(count + 1, dtype=np.int64)
        np.cumsum(pre + suf, out=offs[1:])
        total = int(offs[-1])
        outbuf = np.empty(max(total, 1), dtype=np.uint8)[:total]
        rc = self.ext.delta_ba_concat_host(
            pre.ctypes.data, suf.ctypes.data, arr.ctypes.data + c1 + c2,
            len(arr) - c1 - c2, count, outbuf.ctypes.data, offs.ctypes.data)
        if rc < 0:
            raise NotImplementedError("corrupt DELTA_BYTE_ARRAY payload")
        return offs, outbuf

    def _byte_array_dense(self, data: bytes, count: int):
        """Parse parquet length-prefixed BYTE_ARRAY records -> (offsets
        int32 [count+1] cuda, compact bytes cuda). The serial record walk
        runs on the host in C++ (see byte_array_offsets_host); the GPU
        does the parallel payload compaction."""
        if count == 0:
            return (torch.zeros(1, dtype=torch.int32, device="cuda"),
                    torch.zeros(0, dtype=torch.uint8, device="cuda"))
        arr = np.frombuffer(data, dtype=np.uint8)
        starts_h = np.empty(count, dtype=np.int32)
        lens_h = np.empty(count, dtype=np.int64)
        total = self.ext.byte_array_offsets_host(
            arr.ctypes.data, len(data), count, starts_h.ctypes.data,
            lens_h.ctypes.data)
        if total < 0:
            raise NotImplementedError("corrupt byte-array records")
        offs_h = np.empty(count + 1, dtype=np.int64)
        offs_h[0] = 0
        np.cumsum(lens_h, out=offs_h[1:])
        page = _upload_parts([arr], self.ext, self.s)
        starts = torch.from_numpy(starts_h).cuda()
        lens = torch.from_numpy(lens_h).cuda()
        scanned = torch.from_numpy(offs_h[:-1]).cuda()
        out_bytes = torch.empty(max(total, 1), dtype=torch.uint8,
                                device="cuda")[:total]
        if total:
            self.ext.substr_copy(page.data_ptr(), starts.data_ptr(),
                                 lens.data_ptr(), scanned.data_ptr(),
                                 out_bytes.data_ptr(), count, self.s)
        offs = torch.from_numpy(offs_h.astype(np.int32)).cuda()
        return offs, out_bytes

    # -- data pages ------------------------------------------------------
    def _decode_levels(self, rle: bytes, n: int) -> torch.Tensor:
        return self._decode_level_stream(rle, n,
                                         max(1, self.max_def.bit_length()))

    def _valid_parts(self, levels: Optional[torch.Tensor], n: int):
        """(validity_mask or None, valid_idx tensor, n_valid)."""
        from ..ops import gpu_backend as gb

        if levels is None:
            return None, None, n
        mask = torch.empty(mask_nbytes(n), dtype=torch.uint8, device="cuda")
        self.ext.levels_to_mask(levels.data_ptr(), self.max_def,
                                mask.data_ptr(), n, self.s)
        # positions of valid rows (stream-compact the defs)
        defs_u8 = torch.empty(n, dtype=torch.uint8, device="cuda")
        self.ext.mask_expand(mask.data_ptr(), defs_u8.data_ptr(), False, n,
                             self.s)
        nb = self.ext.sel_num_blocks(n)
        counts = torch.empty(nb, dtype=torch.int64, device="cuda")
        self.ext.mask_count(defs_u8.data_ptr(), 0, counts.data_ptr(), n, self.s)
        offsets, n_valid = gb._exclusive_scan_i64(counts)
        idx = torch.empty(max(n_valid, 1), dtype=torch.int32,
                          device="cuda")[:n_valid]
        if n_valid:
            self.ext.mask_scatter(defs_u8.data_ptr(), 0, offsets.data_ptr(),
                                  idx.data_ptr(), n, self.s)
        return mask, idx, n_valid

    def _materialize(self, values: bytes, n: int, levels, encoding) -> Column:
        from ..column import torch_dtype

        mask, valid_idx, n_valid = self._valid_parts(levels, n)
        nulls = mask is not None
        if encoding == PLAIN and self.phys == "BOOLEAN":
            # plain booleans are LSB-first bit-packed — same layout as the
            # validity bitmask, so the mask-expand kernel decodes them
            nbytes = (n_valid + 7) // 8
            bits = torch.from_numpy(np.frombuffer(
                values, dtype=np.uint8, count=nbytes).copy()).cuda()
            padded = torch.zeros(mask_nbytes(n_valid), dtype=torch.uint8,
                                 device="cuda")
            padded[:nbytes] = bits
            dense = torch.empty(max(n_valid, 1), dtype=torch.uint8,
                                device="cuda")[:n_valid]
            if n_valid:
                self.ext.mask_expand(padded.data_ptr(), dense.data_ptr(),
                                     False, n_valid, self.s)
            if not nulls:
                return Column(self.dtype, n, dense, None, null_count=0)
            out = torch.zeros(n, dtype=torch.uint8, device="cuda")
            if n_valid:
                self.ext.scatter_fixed(1, dense.data_ptr(),
                                       valid_idx.data_ptr(), out.data_ptr(),
                                       n_valid, self.s)
            return Column(self.dtype, n, out, mask, null_count=None)
        if encoding == PLAIN and self.phys == "BYTE_ARRAY":
            offs_dense, out_bytes = self._byte_array_dense(values, n_valid)
            return self._string_dense_to_rows(offs_dense, out_bytes, n,
                                              mask, valid_idx, n_valid)
        if encoding in (DELTA_LENGTH_BYTE_ARRAY, DELTA_BYTE_ARRAY) \
                and self.phys == "BYTE_ARRAY":
            offs_h, payload = self._delta_byte_array_dense(
                values, n_valid, encoding)
            offs_dense = torch.from_numpy(
                offs_h.astype(np.int32)).cuda()
            out_bytes = _upload_parts([payload], self.ext, self.s) \
                if len(payload) else torch.zeros(0, dtype=torch.uint8,
                                                 device="cuda")
            return self._string_dense_to_rows(offs_dense, out_bytes, n,
                                              mask, valid_idx, n_valid)
        if encoding == DELTA_BINARY_PACKED and self.phys in ("INT32",
                                                             "INT64"):
            page = torch.from_numpy(np.frombuffer(
                values, dtype=np.uint8).copy()).cuda()
            vals = torch.empty(max(n_valid, 1), dtype=torch.int64,
                               device="cuda")[:n_valid]
            if n_valid:
                self.ext.pq_delta_i64(page.data_ptr(), page.numel(),
                                      n_valid, vals.data_ptr(), self.s)
            tdt = torch_dtype(self.dtype)
            if tdt != torch.int64:
                from ..ops import gpu_backend as gb

                wide = Column(DType.int64(), n_valid, vals, None,
                              null_count=0)
                dense = gb.cast(wide, self.dtype).data
            else:
                dense = vals
            if not nulls:
                return Column(self.dtype, n, dense, None, null_count=0)
            out = torch.zeros(n, dtype=tdt, device="cuda")
            if n_valid:
                self.ext.scatter_fixed(self.dtype.itemsize, dense.data_ptr(),
                                       valid_idx.data_ptr(), out.data_ptr(),
                                       n_valid, self.s)
            return Column(self.dtype, n, out, mask, null_count=None)
        if encoding == PLAIN:
            if self.phys not in _PHYS_NP:
                raise NotImplementedError(f"PLAIN {self.phys}")
            np_dt = _PHYS_NP[self.phys]
            dense_np = np.frombuffer(values, dtype=np_dt, count=n_valid)
            dense = torch.from_numpy(dense_np.copy()).cuda()
            tdt = torch_dtype(self.dtype)
            if dense.dtype != tdt:
                dense = self._cast_raw(dense, tdt)
            if not nulls:
                return Column(self.dtype, n, dense, None, null_count=0)
            out = torch.zeros(n, dtype=tdt, device="cuda")
            if n_valid:
                self.ext.scatter_fixed(self.dtype.itemsize, dense.data_ptr(),
                                       valid_idx.data_ptr(), out.data_ptr(),
                                       n_valid, self.s)
            return Column(self.dtype, n, out, mask, null_count=None)
        if encoding in (PLAIN_DICTIONARY, RLE_DICTIONARY):
            bit_width = values[0]
            rle = torch.from_numpy(np.frombuffer(
                values, dtype=np.uint8, offset=1).copy()).cuda()
            dense_idx = torch.empty(max(n_valid, 1), dtype=torch.int32,
                                    device="cuda")[:n_valid]
            if n_valid:
                self.ext.rle_hybrid_decode(rle.data_ptr(), rle.numel(),
                                           int(bit_width),
                                           dense_idx.data_ptr(), n_valid,
                                           self.s)
            # row-level dictionary index, -1 for null rows
            if nulls:
                ridx = torch.full((n,), -1, dtype=torch.int32, device="cuda")
                if n_valid:
                    self.ext.scatter_fixed(4, dense_idx.data_ptr(),
                                           valid_idx.data_ptr(),
                                           ridx.data_ptr(), n_valid, self.s)
            else:
                ridx = dense_idx
            return self._gather_dict(ridx, n, mask)
        raise NotImplementedError(f"encoding {encoding}")

    _RAW_HT = {torch.uint8: 0, torch.int8: 1, torch.int16: 2,
               torch.int32: 3, torch.int64: 4, torch.float32: 5,
               torch.float64: 6}

    def _cast_raw(self, dense: torch.Tensor, tdt) -> torch.Tensor:
        # same-width reinterpret (e.g. int32 -> date32 backing)
        if dense.element_size() == torch.tensor([], dtype=tdt).element_size():
            return dense.view(tdt)
        # width-changing RAW widen (e.g. INT32-physical decimal into its
        # int64 backing: parquet stores the unscaled value, so this must
        # NOT rescale like a value-level decimal cast would)
        n = dense.numel()
        out = torch.empty(max(n, 1), dtype=tdt, device="cuda")[:n]
        if n:
            self.ext.cast(self._RAW_HT[dense.dtype], self._RAW_HT[tdt],
                          dense.data_ptr(), out.data_ptr(), n, self.s)
        return out

    def _gather_dict(self, ridx: torch.Tensor, n: int, mask) -> Column:
        from ..ops import gpu_backend as gb

        idx_col = Column(DType.int32(), n, ridx, None, null_count=0)
        if self.dict_str is not None:
            offsets, blob = self.dict_str
            nd = offsets.numel() - 1
            dcol = Column(DType.string(), nd, blob, None, offsets, 0)
            out = gb._gather_col(dcol, ridx, n, maybe_negative=mask is not None)
            return Column(self.dtype, n, out.data, mask, out.offsets,
                          null_count=None if mask is not None else 0)
        if self.dict_fixed is None:
            raise NotImplementedError("data page before dictionary page")
        tdt = torch_dtype_of(self.dtype)
        dvals = self.dict_fixed
        if dvals.dtype != tdt:
            dvals = self._cast_raw(dvals, tdt)
        dcol = Column(self.dtype, dvals.numel(), dvals, None, null_count=0)
        out = gb._gather_col(dcol, ridx, n, maybe_negative=mask is not None)
        return Column(self.dtype, n, out.data, mask, null_count=None
                      if mask is not None else 0)


def torch_dtype_of(dt: DType):
    from ..column import torch_dtype

    return torch_dtype(dt)


# footer/metadata cache keyed by (path, mtime): the reference's FileCache
# caches parquet footers for exactly this reason — every query of a power
# run re-opens the same files (SURVEY.md §2.3 "File cache" row)
_META_CACHE: dict = {}


def _file_meta(path: str):
    import os

    import pyarrow.parquet as pq

    key = (path, os.path.getmtime(path))
    hit = _META_CACHE.get(key)
    if hit is None:
        pf = pq.ParquetFile(path)
        hit = (pf.metadata, pf.schema_arrow, pf.schema)
        if len(_META_CACHE) > 8192:
            _META_CACHE.clear()
        _META_CACHE[key] = hit
    return hit


def read_parquet_gpu(path: str, columns: List[str],
                     keep_rgs=None) -> ColumnBatch:
    import time as _time

    from ..ops.gpu_backend import ext, _stream
    from .parquet import PHASE_STATS, arrow_to_dtype

    t0 = _time.perf_counter()
    md, arrow_schema, pq_schema = _file_meta(path)
    PHASE_STATS["meta_s"] += _time.perf_counter() - t0
    import mmap as _mmap
    name_to_idx = {md.row_group(0).column(j).path_in_schema: j
                   for j in range(md.num_columns)} if md.num_row_groups else {}
    s = _stream()

    rg_batches = []
    if True:
        # zero-copy chunk access through the per-file cached mapping,
        # hipHostRegister'ed once so uploads are direct DMA
        _, mv, _, _, _ = _file_mmap(path)
        for rg in range(md.num_row_groups):
            if keep_rgs is not None and rg not in keep_rgs:
                continue  # pruned by row-group min/max statistics
            rgmd = md.row_group(rg)
            cols = []

            def _mk_dec(j, leaf_dtype, **kw):
                cmd = rgmd.column(j)
                if pq_schema.column(j).max_repetition_level > 0 \
                        and not kw.get("max_rep"):
                    raise NotImplementedError("nested parquet column")
                start = cmd.dictionary_page_offset \
                    if cmd.dictionary_page_offset is not None \
                    else cmd.data_page_offset
                t1 = _time.perf_counter()
                raw = mv[start:start + cmd.total_compressed_size]
                PHASE_STATS["io_s"] += _time.perf_counter() - t1
                return _ChunkDecoder(
                    raw, cmd.physical_type, leaf_dtype,
                    pq_schema.column(j).max_definition_level,
                    _codec(cmd.compression), cmd.num_values, ext, s, **kw)

            for name in columns:
                field = arrow_schema.field(name)
                dtype = arrow_to_dtype(field.type)
                j = name_to_idx.get(name)
                t2 = _time.perf_counter()
                if j is not None:
                    cols.append(_mk_dec(j, dtype).decode())
                elif dtype.id is TypeId.LIST \
                        and not dtype.children[0].is_nested:
                    cands = [jj for p, jj in name_to_idx.items()
                             if p.startswith(name + ".")]
                    if len(cands) != 1:
                        raise NotImplementedError(
                            f"nested parquet column {name}")
                    dec = _mk_dec(
                        cands[0], dtype.children[0], max_rep=1,
                        list_dtype=dtype, opt_list=field.nullable,
                        opt_elem=field.type.value_field.nullable)
                    cols.append(dec.decode())
                elif dtype.id is TypeId.MAP \
                        and not any(c.is_nested for c in dtype.children):
                    # MAP = two rep-1 leaves (key, value) sharing the
                    # entry offsets; decode each through the LIST path
                    # and zip the entry children into the entry struct
                    leaves = {p: jj for p, jj in name_to_idx.items()
                              if p.startswith(name + ".")}
                    jk = next((jj for p, jj in leaves.items()
                               if p.endswith(".key")), None)
                    jv = next((jj for p, jj in leaves.items()
                               if p.endswith(".value")), None)
                    if jk is None or jv is None:
                        raise NotImplementedError(
                            f"map leaf layout for {name}")
                    kt, vt = dtype.children
                    kl = _mk_dec(jk, kt, max_rep=1,
                                 list_dtype=DType.list_(kt),
                                 opt_list=field.nullable,
                                 opt_elem=False).decode()
                    vl = _mk_dec(jv, vt, max_rep=1,
                                 list_dtype=DType.list_(vt),
                                 opt_list=field.nullable,
                                 opt_elem=field.type.item_field.nullable
                                 ).decode()
                    entry = Column(
                        dtype.entry_dtype, kl.child.size,
                        torch.zeros(0, dtype=torch.uint8, device="cuda"),
                        None, None, 0, (kl.child, vl.child))
                    cols.append(Column(
                        dtype, kl.size,
                        torch.zeros(0, dtype=torch.uint8, device="cuda"),
                        kl.validity, kl.offsets, kl._null_count, entry))
                elif dtype.id is TypeId.STRUCT \
                        and not any(c.is_nested for c in dtype.children):
                    # device STRUCT decode: each leaf is a flat chunk
                    # (dotted path); the struct's own validity comes from
                    # the first leaf's def levels (struct-null rows have
                    # def 0 on EVERY leaf)
                    kids = []
                    sv = None
                    for ci, (cname, cdt) in enumerate(
                            zip(dtype.field_names, dtype.children)):
                        jj = name_to_idx.get(f"{name}.{cname}")
                        if jj is None:
                            raise NotImplementedError(
                                f"struct leaf {name}.{cname}")
                        dec = _mk_dec(jj, cdt,
                                      capture_def=(ci == 0
                                                   and field.nullable))
                        kids.append(dec.decode())
                        if ci == 0:
                            sv = dec.captured_def
                    n_rows = rgmd.num_rows
                    validity = None
                    null_count = 0
                    if sv is not None:
                        rv = (sv >= 1).to(torch.int32).contiguous()
                        if not bool((rv == 1).all()):
                            mask = torch.empty(mask_nbytes(n_rows),
                                               dtype=torch.uint8,
                                               device="cuda")
                            ext.levels_to_mask(rv.data_ptr(), 1,
                                               mask.data_ptr(), n_rows, s)
                            validity = mask
                            null_count = None
                    cols.append(Column(
                        dtype, n_rows,
                        torch.zeros(0, dtype=torch.uint8, device="cuda"),
                        validity, None, null_count, tuple(kids)))
                else:
                    raise NotImplementedError(
                        f"nested parquet column {name}")
                PHASE_STATS["decode_s"] += _time.perf_counter() - t2
            rg_batches.append(ColumnBatch(cols, rgmd.num_rows))
    if len(rg_batches) == 1:
        return rg_batches[0]
    from ..ops import gpu_backend as gb

    return gb.concat_batches(rg_batches)

"""Minimal Thrift compact-protocol parser for Parquet page headers.

Parquet page headers are thrift-compact structs (parquet-format
PageHeader). The reference reads them natively in libcudf's parquet reader;
here a small host-side parser feeds the hipdf GPU decode kernels. Only the
subset of the protocol Parquet uses is implemented (structs, i32/i64 zigzag
varints, binary, bool, nested struct skip).
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import Optional, Tuple

CT_STOP = 0
CT_TRUE = 1
CT_FALSE = 2
CT_BYTE = 3
CT_I16 = 4
CT_I32 = 5
CT_I64 = 6
CT_DOUBLE = 7
CT_BINARY = 8
CT_LIST = 9
CT_SET = 10
CT_MAP = 11
CT_STRUCT = 12


def _varint(buf: bytes, pos: int) -> Tuple[int, int]:
    out = 0
    shift = 0
    while True:
        b = buf[pos]
        pos += 1
        out |= (b & 0x7F) << shift
        if not b & 0x80:
            return out, pos
        shift += 7


def _zigzag(v: int) -> int:
    return (v >> 1) ^ -(v & 1)


def _skip(buf: bytes, pos: int, ctype: int) -> int:
    if ctype in (CT_TRUE, CT_FALSE):
        return pos
    if ctype == CT_BYTE:
        return pos + 1
    if ctype in (CT_I16, CT_I32, CT_I64):
        _, pos = _varint(buf, pos)
        return pos
    if ctype == CT_DOUBLE:
        return pos + 8
    if ctype == CT_BINARY:
        n, pos = _varint(buf, pos)
        return pos + n
    if ctype == CT_STRUCT:
        return _skip_struct(buf, pos)
    if ctype in (CT_LIST, CT_SET):
        h = buf[pos]
        pos += 1
        size = h >> 4
        et = h & 0x0F
        if size == 15:
            size, pos = _varint(buf, pos)
        for _ in range(size):
            pos = _skip(buf, pos, et)
        return pos
    raise NotImplementedError(f"thrift skip type {ctype}")


def _skip_struct(buf: bytes, pos: int) -> int:
    last = 0
    while True:
        b = buf[pos]
        pos += 1
        if b == CT_STOP:
            return pos
        delta = b >> 4
        ctype = b & 0x0F
        if delta == 0:
            fid_z, pos = _varint(buf, pos)
            last = _zigzag(fid_z)
        else:
            last += delta
        pos = _skip(buf, pos, ctype)


class _StructReader:
    """Iterate (field_id, ctype) of a compact struct, with typed getters."""

    def __init__(self, buf: bytes, pos: int):
        self.buf = buf
        self.pos = pos
        self.last = 0

    def fields(self):
        while True:
            b = self.buf[self.pos]
            self.pos += 1
            if b == CT_STOP:
                return
            delta = b >> 4
            ctype = b & 0x0F
            if delta == 0:
                fid_z, self.pos = _varint(self.buf, self.pos)
                self.last = _zigzag(fid_z)
            else:
                self.last += delta
            yield self.last, ctype

    def read_i(self) -> int:
        v, self.pos = _varint(self.buf, self.pos)
        return _zigzag(v)

    def skip(self, ctype: int):
        self.pos = _skip(self.buf, self.pos, ctype)


@dataclass
class DataPageHeader:
    num_values: int = 0
    encoding: int = 0
    def_level_encoding: int = 0
    rep_level_encoding: int = 0


@dataclass
class DataPageHeaderV2:
    num_values: int = 0
    num_nulls: int = 0
    num_rows: int = 0
    encoding: int = 0
    def_levels_byte_length: int = 0
    rep_levels_byte_length: int = 0
    is_compressed: bool = True


@dataclass
class DictionaryPageHeader:
    num_values: int = 0
    encoding: int = 0


@dataclass
class PageHeader:
    type: int = -1          # 0 DATA_PAGE, 2 DICTIONARY_PAGE, 3 DATA_PAGE_V2
    uncompressed_page_size: int = 0
    compressed_page_size: int = 0
    data_page: Optional[DataPageHeader] = None
    data_page_v2: Optional[DataPageHeaderV2] = None
    dictionary_page: Optional[DictionaryPageHeader] = None
    header_size: int = 0


def parse_page_header(buf: bytes, pos: int = 0) -> PageHeader:
    start = pos
    r = _StructReader(buf, pos)
    ph = PageHeader()
    for fid, ctype in r.fields():
        if fid == 1:
            ph.type = r.read_i()
        elif fid == 2:
            ph.uncompressed_page_size = r.read_i()
        elif fid == 3:
            ph.compressed_page_size = r.read_i()
        elif fid == 5 and ctype == CT_STRUCT:
            dp = DataPageHeader()
            rr = _StructReader(r.buf, r.pos)
            for f2, t2 in rr.fields():
                if f2 == 1:
                    dp.num_values = rr.read_i()
                elif f2 == 2:
                    dp.encoding = rr.read_i()
                elif f2 == 3:
                    dp.def_level_encoding = rr.read_i()
                elif f2 == 4:
                    dp.rep_level_encoding = rr.read_i()
                else:
                    rr.skip(t2)
            r.pos = rr.pos
            ph.data_page = dp
        elif fid == 7 and ctype == CT_STRUCT:
            dp = DictionaryPageHeader()
            rr = _StructReader(r.buf, r.pos)
            for f2, t2 in rr.fields():
                if f2 == 1:
                    dp.num_values = rr.read_i()
                elif f2 == 2:
                    dp.encoding = rr.read_i()
                elif t2 in (CT_TRUE, CT_FALSE):
                    pass
                else:
                    rr.skip(t2)
            r.pos = rr.pos
            ph.dictionary_page = dp
        elif fid == 8 and ctype == CT_STRUCT:
            dp = DataPageHeaderV2()
            rr = _StructReader(r.buf, r.pos)
            for f2, t2 in rr.fields():
                if f2 == 1:
                    dp.num_values = rr.read_i()
                elif f2 == 2:
                    dp.num_nulls = rr.read_i()
                elif f2 == 3:
                    dp.num_rows = rr.read_i()
                elif f2 == 4:
                    dp.encoding = rr.read_i()
                elif f2 == 5:
                    dp.def_levels_byte_length = rr.read_i()
                elif f2 == 6:
                    dp.rep_levels_byte_length = rr.read_i()
                elif f2 == 7:
                    dp.is_compressed = t2 == CT_TRUE
                else:
                    rr.skip(t2)
            r.pos = rr.pos
            ph.data_page_v2 = dp
        else:
            r.skip(ctype)
    ph.header_size = r.pos - start
    return ph


# ---------------------------------------------------------------------------
# compact-protocol WRITER (parquet file authoring: PageHeader + footer
# FileMetaData). Symmetric subset of the parser above.
# ---------------------------------------------------------------------------

def _w_varint(out: bytearray, v: int):
    while True:
        b = v & 0x7F
        v >>= 7
        if v:
            out.append(b | 0x80)
        else:
            out.append(b)
            return


def _w_zigzag(out: bytearray, v: int):
    _w_varint(out, (v << 1) ^ (v >> 63) if v < 0 else v << 1)


class StructWriter:
    """Compact-protocol struct writer: call the typed field methods in
    ascending field-id order, then bytes()."""

    def __init__(self):
        self.buf = bytearray()
        self.last_fid = 0

    def _hdr(self, fid: int, ctype: int):
        delta = fid - self.last_fid
        if 0 < delta < 16:
            self.buf.append((delta << 4) | ctype)
        else:
            self.buf.append(ctype)
            _w_zigzag(self.buf, fid)
        self.last_fid = fid

    def f_i32(self, fid: int, v: int):
        self._hdr(fid, CT_I32)
        _w_zigzag(self.buf, v)

    def f_i64(self, fid: int, v: int):
        self._hdr(fid, CT_I64)
        _w_zigzag(self.buf, v)

    def f_bool(self, fid: int, v: bool):
        self._hdr(fid, CT_TRUE if v else CT_FALSE)

    def f_binary(self, fid: int, b: bytes):
        self._hdr(fid, CT_BINARY)
        _w_varint(self.buf, len(b))
        self.buf.extend(b)

    def f_struct(self, fid: int, body: bytes):
        self._hdr(fid, CT_STRUCT)
        self.buf.extend(body)

    def f_list_struct(self, fid: int, items):
        self._hdr(fid, CT_LIST)
        n = len(items)
        if n < 15:
            self.buf.append((n << 4) | CT_STRUCT)
        else:
            self.buf.append(0xF0 | CT_STRUCT)
            _w_varint(self.buf, n)
        for it in items:
            self.buf.extend(it)

    def f_list_binary(self, fid: int, items):
        self._hdr(fid, CT_LIST)
        n = len(items)
        if n < 15:
            self.buf.append((n << 4) | CT_BINARY)
        else:
            self.buf.append(0xF0 | CT_BINARY)
            _w_varint(self.buf, n)
        for b in items:
            _w_varint(self.buf, len(b))
            self.buf.extend(b)

    def f_list_i32(self, fid: int, items):
        self._hdr(fid, CT_LIST)
        n = len(items)
        if n < 15:
            self.buf.append((n << 4) | CT_I32)
        else:
            self.buf.append(0xF0 | CT_I32)
            _w_varint(self.buf, n)
        for v in items:
            _w_zigzag(self.buf, v)

    def bytes(self) -> bytes:
        self.buf.append(CT_STOP)
        return bytes(self.buf)

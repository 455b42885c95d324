"""Avro Object Container File reader/writer (flat records).

Reference analogue: GpuAvroScan (sql-plugin avro module). Host-side
decode this round (CPU-ONLY row in docs/STATUS.md): the container format
is varint/zigzag-heavy and row-oriented, so the decode is a python/numpy
pass that lands in regular host ColumnBatches; a device field-walker in
the CSV/JSON style is a later round. Supports the null and deflate
codecs, records of null/boolean/int/long/float/double/string/bytes and
the standard ["null", T] nullable unions.
"""
from __future__ import annotations

import json
import os
import struct
import zlib
from typing import List, Optional, Tuple

from ..column import Column, ColumnBatch, Field, Schema
from ..types import BOOL, DType, FLOAT32, FLOAT64, INT32, INT64, STRING

_MAGIC = b"Obj\x01"

_AVRO_TO_DTYPE = {
    "boolean": BOOL,
    "int": INT32,
    "long": INT64,
    "float": FLOAT32,
    "double": FLOAT64,
    "string": STRING,
    "bytes": STRING,
}

_DTYPE_TO_AVRO = {
    "boolean": "boolean", "int": "int", "bigint": "long",
    "float": "float", "double": "double", "string": "string",
}


def _zigzag_enc(v: int) -> bytes:
    u = (v << 1) ^ (v >> 63)
    out = bytearray()
    while True:
        b = u & 0x7F
        u >>= 7
        if u:
            out.append(b | 0x80)
        else:
            out.append(b)
            return bytes(out)


def _zigzag_dec(b: bytes, p: int) -> Tuple[int, int]:
    u = 0
    sh = 0
    while True:
        v = b[p]
        p += 1
        u |= (v & 0x7F) << sh
        if not v & 0x80:
            break
        sh += 7
    return (u >> 1) ^ -(u & 1), p


def _field_type(t):
    """-> (dtype, nullable). Accepts "long" or ["null", "long"]."""
    if isinstance(t, list):
        nn = [x for x in t if x != "null"]
        if len(nn) != 1 or not isinstance(nn[0], str):
            raise NotImplementedError(f"avro union {t}")
        return _AVRO_TO_DTYPE[nn[0]], True
    if isinstance(t, str) and t in _AVRO_TO_DTYPE:
        return _AVRO_TO_DTYPE[t], False
    raise NotImplementedError(f"avro type {t}")


def read_avro(path: str) -> Tuple[Schema, ColumnBatch]:
    with open(path, "rb") as f:
        raw = f.read()
    assert raw[:4] == _MAGIC, "not an avro container file"
    p = 4
    # file metadata map: blocks of (count, then count * (key, value))
    meta = {}
    while True:
        cnt, p = _zigzag_dec(raw, p)
        if cnt == 0:
            break
        if cnt < 0:  # block with byte size prefix
            _, p = _zigzag_dec(raw, p)
            cnt = -cnt
        for _ in range(cnt):
            klen, p = _zigzag_dec(raw, p)
            key = raw[p:p + klen].decode()
            p += klen
            vlen, p = _zigzag_dec(raw, p)
            meta[key] = raw[p:p + vlen]
            p += vlen
    sync = raw[p:p + 16]
    p += 16
    codec = meta.get("avro.codec", b"null").decode()
    schema_json = json.loads(meta["avro.schema"])
    if schema_json.get("type") != "record":
        raise NotImplementedError("avro: non-record schema")
    fields = []
    ftypes = []
    for fd in schema_json["fields"]:
        dt, nullable = _field_type(fd["type"])
        fields.append(Field(fd["name"], dt, nullable))
        ftypes.append((fd["type"], dt, nullable))
    schema = Schema(fields)

    cols: List[List] = [[] for _ in fields]
    while p < len(raw):
        nrec, p = _zigzag_dec(raw, p)
        nbytes, p = _zigzag_dec(raw, p)
        block = raw[p:p + nbytes]
        p += nbytes
        assert raw[p:p + 16] == sync, "avro: sync marker mismatch"
        p += 16
        if codec == "deflate":
            block = zlib.decompress(block, -15)
        elif codec != "null":
            raise NotImplementedError(f"avro codec {codec}")
        q = 0
        for _ in range(nrec):
            for j, (ft, dt, nullable) in enumerate(ftypes):
                v, q = _read_value(block, q, ft)
                cols[j].append(v)
    batch = ColumnBatch(
        [Column.from_pylist(vals, f.dtype)
         for vals, f in zip(cols, fields)],
        len(cols[0]) if cols else 0)
    return schema, batch


def _read_value(b: bytes, p: int, ft):
    if isinstance(ft, list):
        branch, p = _zigzag_dec(b, p)
        t = ft[branch]
        if t == "null":
            return None, p
        return _read_value(b, p, t)
    if ft == "boolean":
        return bool(b[p]), p + 1
    if ft in ("int", "long"):
        return _zigzag_dec(b, p)
    if ft == "float":
        return struct.unpack_from("<f", b, p)[0], p + 4
    if ft == "double":
        return struct.unpack_from("<d", b, p)[0], p + 8
    if ft in ("string", "bytes"):
        ln, p = _zigzag_dec(b, p)
        s = b[p:p + ln]
        return (s.decode() if ft == "string" else s), p + ln
    raise NotImplementedError(f"avro type {ft}")


def write_avro(batch: ColumnBatch, schema: Schema, path: str,
               codec: str = "deflate"):
    fields_json = []
    for f in schema.fields:
        t = _DTYPE_TO_AVRO.get(str(f.dtype))
        if t is None:
            raise NotImplementedError(f"avro write: {f.dtype}")
        fields_json.append({"name": f.name, "type": ["null", t]})
    sj = json.dumps({"type": "record", "name": "row",
                     "fields": fields_json}).encode()
    sync = b"hipdfhipdfhipdf!"  # any 16 bytes
    out = bytearray(_MAGIC)
    out += _zigzag_enc(2)
    for k, v in (("avro.schema", sj), ("avro.codec", codec.encode())):
        out += _zigzag_enc(len(k)) + k.encode()
        out += _zigzag_enc(len(v)) + v
    out += _zigzag_enc(0)
    out += sync
    body = bytearray()
    pylists = [c.to_pylist() for c in batch.columns]
    for i in range(batch.num_rows):
        for vals, f in zip(pylists, schema.fields):
            v = vals[i]
            if v is None:
                body += _zigzag_enc(0)
                continue
            body += _zigzag_enc(1)
            t = _DTYPE_TO_AVRO[str(f.dtype)]
            if t == "boolean":
                body += bytes([1 if v else 0])
            elif t in ("int", "long"):
                body += _zigzag_enc(int(v))
            elif t == "float":
                body += struct.pack("<f", v)
            elif t == "double":
                body += struct.pack("<d", float(v))
            else:
                eb = v.encode() if isinstance(v, str) else bytes(v)
                body += _zigzag_enc(len(eb)) + eb
    payload = zlib.compress(bytes(body), 6)[2:-4] if codec == "deflate" \
        else bytes(body)
    out += _zigzag_enc(batch.num_rows)
    out += _zigzag_enc(len(payload))
    out += payload
    out += sync
    with open(path, "wb") as f:
        f.write(out)


class AvroTable:
    def __init__(self, path: str):
        from .formats import _expand

        self.files = _expand(path)
        self.schema, self._first = read_avro(self.files[0])

    def partitions(self):
        yield self._first
        for f in self.files[1:]:
            yield read_avro(f)[1]


# ---- generic (nested) record layer ---------------------------------------
# Iceberg manifests are deeply nested avro records (records in records,
# arrays, maps, fixed); this generic reader/writer works on python dicts
# and backs io/iceberg.py (reference analogue: the iceberg-core manifest
# reading the reference reaches through its bridge classes).

def _read_generic(b, p, ft, named):
    if isinstance(ft, str) and ft in named:
        ft = named[ft]
    if isinstance(ft, list):
        branch, p = _zigzag_dec(b, p)
        t = ft[branch]
        if t == "null":
            return None, p
        return _read_generic(b, p, t, named)
    if isinstance(ft, dict):
        t = ft["type"]
        if t == "record":
            named[ft["name"]] = ft
            out = {}
            for fd in ft["fields"]:
                out[fd["name"]], p = _read_generic(b, p, fd["type"], named)
            return out, p
        if t == "array":
            items = []
            while True:
                cnt, p = _zigzag_dec(b, p)
                if cnt == 0:
                    break
                if cnt < 0:
                    _, p = _zigzag_dec(b, p)  # byte size, unused
                    cnt = -cnt
                for _ in range(cnt):
                    v, p = _read_generic(b, p, ft["items"], named)
                    items.append(v)
            return items, p
        if t == "map":
            out = {}
            while True:
                cnt, p = _zigzag_dec(b, p)
                if cnt == 0:
                    break
                if cnt < 0:
                    _, p = _zigzag_dec(b, p)
                    cnt = -cnt
                for _ in range(cnt):
                    klen, p = _zigzag_dec(b, p)
                    k = b[p:p + klen].decode()
                    p += klen
                    out[k], p = _read_generic(b, p, ft["values"], named)
            return out, p
        if t == "fixed":
            named[ft["name"]] = ft
            n = ft["size"]
            return bytes(b[p:p + n]), p + n
        if t == "enum":
            named[ft["name"]] = ft
            idx, p = _zigzag_dec(b, p)
            return ft["symbols"][idx], p
        # logical types wrap a primitive
        return _read_generic(b, p, t, named)
    return _read_value(b, p, ft)


def read_avro_records(path: str):
    """-> (schema_json, list[dict]) for arbitrary nested record schemas."""
    with open(path, "rb") as f:
        raw = f.read()
    assert raw[:4] == _MAGIC, "not an avro container file"
    p = 4
    meta = {}
    while True:
        cnt, p = _zigzag_dec(raw, p)
        if cnt == 0:
            break
        if cnt < 0:
            _, p = _zigzag_dec(raw, p)
            cnt = -cnt
        for _ in range(cnt):
            klen, p = _zigzag_dec(raw, p)
            key = raw[p:p + klen].decode()
            p += klen
            vlen, p = _zigzag_dec(raw, p)
            meta[key] = raw[p:p + vlen]
            p += vlen
    sync = raw[p:p + 16]
    p += 16
    codec = meta.get("avro.codec", b"null").decode()
    schema_json = json.loads(meta["avro.schema"])
    records = []
    while p < len(raw):
        nrec, p = _zigzag_dec(raw, p)
        nbytes, p = _zigzag_dec(raw, p)
        block = raw[p:p + nbytes]
        p += nbytes
        assert raw[p:p + 16] == sync, "avro: sync marker mismatch"
        p += 16
        if codec == "deflate":
            block = zlib.decompress(block, -15)
        elif codec != "null":
            raise NotImplementedError(f"avro codec {codec}")
        q = 0
        for _ in range(nrec):
            v, q = _read_generic(block, q, schema_json, {})
            records.append(v)
    return schema_json, records


def _write_generic(out: bytearray, v, ft, named):
    if isinstance(ft, str) and ft in named:
        ft = named[ft]
    if isinstance(ft, list):
        if v is None and "null" in ft:
            out += _zigzag_enc(ft.index("null"))
            return
        branch = next(i for i, t in enumerate(ft) if t != "null")
        out += _zigzag_enc(branch)
        _write_generic(out, v, ft[branch], named)
        return
    if isinstance(ft, dict):
        t = ft["type"]
        if t == "record":
            named[ft["name"]] = ft
            for fd in ft["fields"]:
                _write_generic(out, v.get(fd["name"]), fd["type"], named)
            return
        if t == "array":
            if v:
                out += _zigzag_enc(len(v))
                for item in v:
                    _write_generic(out, item, ft["items"], named)
            out += _zigzag_enc(0)
            return
        if t == "map":
            if v:
                out += _zigzag_enc(len(v))
                for k, mv in v.items():
                    kb = k.encode()
                    out += _zigzag_enc(len(kb)) + kb
                    _write_generic(out, mv, ft["values"], named)
            out += _zigzag_enc(0)
            return
        if t == "fixed":
            named[ft["name"]] = ft
            out += v
            return
        _write_generic(out, v, t, named)
        return
    if ft == "boolean":
        out.append(1 if v else 0)
    elif ft in ("int", "long"):
        out += _zigzag_enc(int(v))
    elif ft == "float":
        out += struct.pack("<f", v)
    elif ft == "double":
        out += struct.pack("<d", v)
    elif ft == "string":
        vb = v.encode()
        out += _zigzag_enc(len(vb)) + vb
    elif ft == "bytes":
        out += _zigzag_enc(len(v)) + v
    elif ft == "null":
        pass
    else:
        raise NotImplementedError(f"avro write type {ft}")


def write_avro_records(schema_json, records, path: str):
    """Write arbitrary nested records (python dicts) as an avro container
    file (null codec)."""
    body = bytearray()
    for r in records:
        _write_generic(body, r, schema_json, {})
    sync = os.urandom(16) if hasattr(os, "urandom") else b"\x01" * 16
    out = bytearray(_MAGIC)
    meta = {"avro.schema": json.dumps(schema_json).encode(),
            "avro.codec": b"null"}
    out += _zigzag_enc(len(meta))
    for k, v in meta.items():
        kb = k.encode()
        out += _zigzag_enc(len(kb)) + kb
        out += _zigzag_enc(len(v)) + v
    out += _zigzag_enc(0)
    out += sync
    out += _zigzag_enc(len(records))
    out += _zigzag_enc(len(body))
    out += body
    out += sync
    with open(path, "wb") as f:
        f.write(out)

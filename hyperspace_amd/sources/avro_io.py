"""Minimal Avro Object Container File reader/writer.

The reference's default source accepts avro among its supported formats
(util/HyperspaceConf.scala:110-115, via Spark's avro FileFormat).  No
avro library ships in this environment, so this implements the container
format directly (spec: avro.apache.org/docs/current/specification —
"Obj\\x01" magic, metadata map with schema JSON + codec, 16-byte sync
marker, then blocks of zigzag-varint-framed records).

Scope: flat record schemas of the primitive types the engine models
(boolean/int/long/float/double/string/bytes) and their nullable
["null", T] unions; codecs "null" and "deflate".  This is a
compatibility ingest path — indexes built from avro sources are native
Parquet, so queries served from an index still run the device pipeline.
"""

from __future__ import annotations

import io
import json
import os
import struct
import zlib
from typing import Any, Dict, List, Optional, Tuple

import numpy as np

MAGIC = b"Obj\x01"


# ---------------------------------------------------------------------------
# varint / zigzag primitives
# ---------------------------------------------------------------------------

def _read_long(buf: memoryview, pos: int) -> Tuple[int, int]:
    out = 0
    shift = 0
    while True:
        b = buf[pos]
        pos += 1
        out |= (b & 0x7F) << shift
        if not b & 0x80:
            break
        shift += 7
    return (out >> 1) ^ -(out & 1), pos


def _write_long(n: int) -> bytes:
    n = (n << 1) ^ (n >> 63)
    out = bytearray()
    while True:
        b = n & 0x7F
        n >>= 7
        if n:
            out.append(b | 0x80)
        else:
            out.append(b)
            return bytes(out)


# ---------------------------------------------------------------------------
# schema model
# ---------------------------------------------------------------------------

class _Field:
    __slots__ = ("name", "type", "nullable", "null_first")

    def __init__(self, name: str, ftype: str, nullable: bool,
                 null_first: bool = True):
        self.name = name
        self.type = ftype  # avro primitive name
        self.nullable = nullable
        self.null_first = null_first  # union branch order


_PRIMITIVES = {"boolean", "int", "long", "float", "double", "string",
               "bytes"}


def _parse_schema(schema_json: str) -> List[_Field]:
    sch = json.loads(schema_json)
    if sch.get("type") != "record":
        raise ValueError("only record schemas are supported")
    fields = []
    for f in sch["fields"]:
        t = f["type"]
        nullable = False
        null_first = True
        if isinstance(t, list):  # union
            branches = [b for b in t if b != "null"]
            if len(branches) != 1 or len(t) > 2:
                raise ValueError(f"unsupported union {t}")
            nullable = "null" in t
            null_first = t[0] == "null"
            t = branches[0]
        if isinstance(t, dict):
            t = t.get("type")
        if t not in _PRIMITIVES:
            raise ValueError(f"unsupported avro type {t}")
        fields.append(_Field(f["name"], t, nullable, null_first))
    return fields


# ---------------------------------------------------------------------------
# reader
# ---------------------------------------------------------------------------

def read_avro(path: str):
    """Decode an avro container file into a pyarrow.Table."""
    import pyarrow as pa
    with open(path, "rb") as f:
        data = f.read()
    if data[:4] != MAGIC:
        raise ValueError(f"not an avro container file: {path}")
    buf = memoryview(data)
    pos = 4
    meta: Dict[str, bytes] = {}
    while True:
        count, pos = _read_long(buf, pos)
        if count == 0:
            break
        if count < 0:  # negative block count: byte size follows
            count = -count
            _, pos = _read_long(buf, pos)
        for _ in range(count):
            klen, pos = _read_long(buf, pos)
            key = bytes(buf[pos:pos + klen]).decode()
            pos += klen
            vlen, pos = _read_long(buf, pos)
            meta[key] = bytes(buf[pos:pos + vlen])
            pos += vlen
    codec = meta.get("avro.codec", b"null").decode()
    if codec not in ("null", "deflate"):
        raise ValueError(f"unsupported avro codec {codec}")
    fields = _parse_schema(meta["avro.schema"].decode())
    sync = bytes(buf[pos:pos + 16])
    pos += 16

    cols: Dict[str, list] = {f.name: [] for f in fields}
    while pos < len(buf):
        count, pos = _read_long(buf, pos)
        nbytes, pos = _read_long(buf, pos)
        block = buf[pos:pos + nbytes]
        pos += nbytes
        if bytes(buf[pos:pos + 16]) != sync:
            raise ValueError("sync marker mismatch (corrupt avro block)")
        pos += 16
        if codec == "deflate":
            block = memoryview(zlib.decompress(bytes(block), -15))
        _decode_block(block, count, fields, cols)

    arrays = {}
    for f in fields:
        vals = cols[f.name]
        if f.type == "string":
            arrays[f.name] = pa.array(vals, type=pa.string())
        elif f.type == "bytes":
            arrays[f.name] = pa.array(vals, type=pa.binary())
        elif f.type == "boolean":
            arrays[f.name] = pa.array(vals, type=pa.bool_())
        elif f.type == "int":
            arrays[f.name] = pa.array(vals, type=pa.int32())
        elif f.type == "long":
            arrays[f.name] = pa.array(vals, type=pa.int64())
        elif f.type == "float":
            arrays[f.name] = pa.array(vals, type=pa.float32())
        else:
            arrays[f.name] = pa.array(vals, type=pa.float64())
    return pa.table(arrays)


def _decode_block(buf: memoryview, count: int, fields: List[_Field],
                  cols: Dict[str, list]) -> None:
    pos = 0
    for _ in range(count):
        for f in fields:
            if f.nullable:
                branch, pos = _read_long(buf, pos)
                is_null = (branch == 0) == f.null_first
                if is_null:
                    cols[f.name].append(None)
                    continue
            v, pos = _decode_value(buf, pos, f.type)
            cols[f.name].append(v)


def _decode_value(buf: memoryview, pos: int, ftype: str):
    if ftype in ("int", "long"):
        return _read_long(buf, pos)
    if ftype == "boolean":
        return buf[pos] != 0, pos + 1
    if ftype == "float":
        return struct.unpack_from("<f", buf, pos)[0], pos + 4
    if ftype == "double":
        return struct.unpack_from("<d", buf, pos)[0], pos + 8
    # string / bytes
    n, pos = _read_long(buf, pos)
    raw = bytes(buf[pos:pos + n])
    return (raw.decode() if ftype == "string" else raw), pos + n


# ---------------------------------------------------------------------------
# generic (nested) record reader/writer — full avro type system.
# Needed for real Iceberg metadata: manifest lists and manifests are avro
# container files of nested records with unions, maps and arrays
# (iceberg spec "Manifests"/"Manifest Lists"; reference consumes them via
# the iceberg library in index/sources/iceberg/IcebergRelation.scala).
# ---------------------------------------------------------------------------


def _resolve_schema(schema: Any, names: Dict[str, Any]) -> Any:
    """Normalize a schema node; register/resolve named types."""
    if isinstance(schema, str):
        if schema in _PRIMITIVES or schema == "null":
            return schema
        if schema in names:
            return names[schema]
        raise ValueError(f"unknown avro type name {schema}")
    if isinstance(schema, list):
        return [_resolve_schema(b, names) for b in schema]
    t = schema.get("type")
    if t in ("record", "enum", "fixed"):
        name = schema.get("name")
        if name:
            names[name] = schema
            full = (schema.get("namespace", "") + "." + name).lstrip(".")
            names[full] = schema
        if t == "record":
            for f in schema.get("fields", []):
                f["type"] = _resolve_schema(f["type"], names)
        return schema
    if t == "array":
        schema["items"] = _resolve_schema(schema["items"], names)
        return schema
    if t == "map":
        schema["values"] = _resolve_schema(schema["values"], names)
        return schema
    if isinstance(t, (dict, list)):
        return _resolve_schema(t, names)
    return t  # primitive spelled as {"type": "long"}


def _decode_any(buf: memoryview, pos: int, schema: Any):
    if isinstance(schema, str):
        if schema == "null":
            return None, pos
        return _decode_value(buf, pos, schema)
    if isinstance(schema, list):  # union: branch index then value
        branch, pos = _read_long(buf, pos)
        return _decode_any(buf, pos, schema[branch])
    t = schema.get("type")
    if t == "record":
        out = {}
        for f in schema.get("fields", []):
            out[f["name"]], pos = _decode_any(buf, pos, f["type"])
        return out, pos
    if t == "enum":
        idx, pos = _read_long(buf, pos)
        return schema["symbols"][idx], pos
    if t == "fixed":
        n = schema["size"]
        return bytes(buf[pos:pos + n]), pos + n
    if t == "array":
        out = []
        while True:
            count, pos = _read_long(buf, pos)
            if count == 0:
                break
            if count < 0:
                count = -count
                _, pos = _read_long(buf, pos)  # block byte size
            for _ in range(count):
                v, pos = _decode_any(buf, pos, schema["items"])
                out.append(v)
        return out, pos
    if t == "map":
        out = {}
        while True:
            count, pos = _read_long(buf, pos)
            if count == 0:
                break
            if count < 0:
                count = -count
                _, pos = _read_long(buf, pos)
            for _ in range(count):
                k, pos = _decode_value(buf, pos, "string")
                out[k], pos = _decode_any(buf, pos, schema["values"])
        return out, pos
    return _decode_value(buf, pos, t)


def read_avro_records(path: str) -> List[Any]:
    """Decode an avro container file of arbitrary (nested) records into
    a list of python values (dicts for records)."""
    with open(path, "rb") as f:
        data = f.read()
    if data[:4] != MAGIC:
        raise ValueError(f"not an avro container file: {path}")
    buf = memoryview(data)
    pos = 4
    meta: Dict[str, bytes] = {}
    while True:
        count, pos = _read_long(buf, pos)
        if count == 0:
            break
        if count < 0:
            count = -count
            _, pos = _read_long(buf, pos)
        for _ in range(count):
            klen, pos = _read_long(buf, pos)
            key = bytes(buf[pos:pos + klen]).decode()
            pos += klen
            vlen, pos = _read_long(buf, pos)
            meta[key] = bytes(buf[pos:pos + vlen])
            pos += vlen
    codec = meta.get("avro.codec", b"null").decode()
    if codec not in ("null", "deflate"):
        raise ValueError(f"unsupported avro codec {codec}")
    schema = _resolve_schema(json.loads(meta["avro.schema"].decode()), {})
    sync = bytes(buf[pos:pos + 16])
    pos += 16
    out: List[Any] = []
    while pos < len(buf):
        count, pos = _read_long(buf, pos)
        nbytes, pos = _read_long(buf, pos)
        block = buf[pos:pos + nbytes]
        pos += nbytes
        if bytes(buf[pos:pos + 16]) != sync:
            raise ValueError("sync marker mismatch (corrupt avro block)")
        pos += 16
        if codec == "deflate":
            block = memoryview(zlib.decompress(bytes(block), -15))
        bpos = 0
        for _ in range(count):
            v, bpos = _decode_any(block, bpos, schema)
            out.append(v)
    return out


def _encode_any(out: io.BytesIO, value: Any, schema: Any) -> None:
    if isinstance(schema, str):
        if schema == "null":
            return
        _encode_value(out, value, schema)
        return
    if isinstance(schema, list):  # union: pick first matching branch
        for i, b in enumerate(schema):
            bt = b if isinstance(b, str) else b.get("type")
            if value is None and bt == "null":
                out.write(_write_long(i))
                return
            if value is not None and bt != "null":
                out.write(_write_long(i))
                _encode_any(out, value, b)
                return
        raise ValueError(f"no union branch for {value!r} in {schema}")
    t = schema.get("type")
    if t == "record":
        for f in schema.get("fields", []):
            _encode_any(out, value.get(f["name"]), f["type"])
        return
    if t == "enum":
        out.write(_write_long(schema["symbols"].index(value)))
        return
    if t == "fixed":
        out.write(value)
        return
    if t == "array":
        if value:
            out.write(_write_long(len(value)))
            for v in value:
                _encode_any(out, v, schema["items"])
        out.write(_write_long(0))
        return
    if t == "map":
        if value:
            out.write(_write_long(len(value)))
            for k, v in value.items():
                _encode_value(out, k, "string")
                _encode_any(out, v, schema["values"])
        out.write(_write_long(0))
        return
    _encode_value(out, value, t)


def _encode_value(out: io.BytesIO, v: Any, t: str) -> None:
    if t in ("int", "long"):
        out.write(_write_long(int(v)))
    elif t == "boolean":
        out.write(b"\x01" if v else b"\x00")
    elif t == "float":
        out.write(struct.pack("<f", float(v)))
    elif t == "double":
        out.write(struct.pack("<d", float(v)))
    elif t == "string":
        raw = str(v).encode()
        out.write(_write_long(len(raw)))
        out.write(raw)
    elif t == "bytes":
        out.write(_write_long(len(v)))
        out.write(v)
    else:
        raise ValueError(f"unsupported avro type {t}")


def write_avro_records(path: str, schema: Dict[str, Any],
                       records: List[Any]) -> None:
    """Write nested records as one avro container file (codec null)."""
    resolved = _resolve_schema(json.loads(json.dumps(schema)), {})
    body = io.BytesIO()
    for r in records:
        _encode_any(body, r, resolved)
    payload = body.getvalue()
    sync = os.urandom(16)
    with open(path, "wb") as f:
        f.write(MAGIC)
        meta = {"avro.schema": json.dumps(schema).encode(),
                "avro.codec": b"null"}
        f.write(_write_long(len(meta)))
        for k, v in meta.items():
            kb = k.encode()
            f.write(_write_long(len(kb)))
            f.write(kb)
            f.write(_write_long(len(v)))
            f.write(v)
        f.write(_write_long(0))
        f.write(sync)
        if records:
            f.write(_write_long(len(records)))
            f.write(_write_long(len(payload)))
            f.write(payload)
            f.write(sync)


# ---------------------------------------------------------------------------
# writer (tests + tooling; codec "null")
# ---------------------------------------------------------------------------

_NP_TO_AVRO = {"int64": "long", "int32": "int", "float64": "double",
               "float32": "float", "bool": "boolean"}


def write_avro(columns: "Dict[str, Any]", path: str,
               masks: "Optional[Dict[str, np.ndarray]]" = None) -> None:
    """Write columns (numpy arrays or lists of str) as one avro container
    file with the null codec.  ``masks``: optional validity arrays."""
    names = list(columns)
    n = len(next(iter(columns.values()))) if names else 0
    fields_json = []
    ftypes = {}
    for name in names:
        col = columns[name]
        if isinstance(col, np.ndarray):
            t = _NP_TO_AVRO[str(col.dtype)]
        else:
            t = "string"
        ftypes[name] = t
        nullable = masks is not None and name in masks
        fields_json.append(
            {"name": name, "type": ["null", t] if nullable else t})
    schema = {"type": "record", "name": "row", "fields": fields_json}

    body = io.BytesIO()
    for i in range(n):
        for name in names:
            t = ftypes[name]
            valid = masks is None or name not in masks or \
                bool(masks[name][i])
            if masks is not None and name in masks:
                body.write(_write_long(1 if valid else 0))
                if not valid:
                    continue
            v = columns[name][i]
            if t in ("int", "long"):
                body.write(_write_long(int(v)))
            elif t == "boolean":
                body.write(b"\x01" if v else b"\x00")
            elif t == "float":
                body.write(struct.pack("<f", float(v)))
            elif t == "double":
                body.write(struct.pack("<d", float(v)))
            else:
                raw = str(v).encode()
                body.write(_write_long(len(raw)))
                body.write(raw)
    payload = body.getvalue()

    sync = os.urandom(16)
    with open(path, "wb") as f:
        f.write(MAGIC)
        meta = {"avro.schema": json.dumps(schema).encode(),
                "avro.codec": b"null"}
        f.write(_write_long(len(meta)))
        for k, v in meta.items():
            kb = k.encode()
            f.write(_write_long(len(kb)))
            f.write(kb)
            f.write(_write_long(len(v)))
            f.write(v)
        f.write(_write_long(0))
        f.write(sync)
        if n:
            f.write(_write_long(n))
            f.write(_write_long(len(payload)))
            f.write(payload)
            f.write(sync)

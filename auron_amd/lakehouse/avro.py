"""Minimal Apache Avro object-container-file codec (reader + writer).

No avro library ships in this environment, and Iceberg's manifest lists /
manifest files are Avro OCF — so this implements the subset the Iceberg
spec uses: records, string/bytes/fixed, int/long (zigzag varint),
float/double, boolean, null, arrays, maps, unions, null/deflate codecs.
Schemas are the JSON forms embedded in the file header.
"""
from __future__ import annotations

import json
import struct
import zlib
from typing import Any, Dict, List, Tuple

MAGIC = b"Obj\x01"


# ------------------------------------------------------------- primitives
def zigzag_encode(v: int) -> int:
    return (v << 1) ^ (v >> 63)


def zigzag_decode(v: int) -> int:
    return (v >> 1) ^ -(v & 1)


def write_long(out: bytearray, v: int):
    u = zigzag_encode(v) & (2 ** 64 - 1)
    while True:
        b = u & 0x7F
        u >>= 7
        if u:
            out.append(b | 0x80)
        else:
            out.append(b)
            return


def read_long(buf: bytes, pos: int) -> Tuple[int, int]:
    u = 0
    shift = 0
    while True:
        b = buf[pos]
        pos += 1
        u |= (b & 0x7F) << shift
        if not (b & 0x80):
            break
        shift += 7
    return zigzag_decode(u), pos


def write_bytes(out: bytearray, b: bytes):
    write_long(out, len(b))
    out += b


def read_bytes(buf: bytes, pos: int) -> Tuple[bytes, int]:
    n, pos = read_long(buf, pos)
    return bytes(buf[pos:pos + n]), pos + n


# ---------------------------------------------------------------- schema
def _norm(schema, named: Dict[str, Any]):
    """Resolve named-type references and normalize shorthand."""
    if isinstance(schema, str):
        if schema in named:
            return named[schema]
        return {"type": schema}
    if isinstance(schema, list):
        return {"type": "union", "branches": [_norm(s, named) for s in schema]}
    t = schema.get("type")
    if t in ("record", "enum", "fixed") and schema.get("name"):
        named[schema["name"]] = schema
    if t == "record":
        for f in schema["fields"]:
            f["type"] = _norm(f["type"], named)
    elif t == "array":
        schema["items"] = _norm(schema["items"], named)
    elif t == "map":
        schema["values"] = _norm(schema["values"], named)
    elif isinstance(t, (dict, list)):
        return _norm(t, named)
    return schema


def parse_schema(text: str):
    return _norm(json.loads(text), {})


# ---------------------------------------------------------------- decode
def decode(schema, buf: bytes, pos: int) -> Tuple[Any, int]:
    t = schema["type"] if isinstance(schema, dict) else schema
    if t == "union":
        idx, pos = read_long(buf, pos)
        return decode(schema["branches"][idx], buf, pos)
    if t == "null":
        return None, pos
    if t == "boolean":
        return buf[pos] != 0, pos + 1
    if t in ("int", "long"):
        return read_long(buf, pos)
    if t == "float":
        return struct.unpack_from("<f", buf, pos)[0], pos + 4
    if t == "double":
        return struct.unpack_from("<d", buf, pos)[0], pos + 8
    if t == "bytes":
        return read_bytes(buf, pos)
    if t == "string":
        b, pos = read_bytes(buf, pos)
        return b.decode("utf-8"), pos
    if t == "fixed":
        n = schema["size"]
        return bytes(buf[pos:pos + n]), pos + n
    if t == "enum":
        i, pos = read_long(buf, pos)
        return schema["symbols"][i], pos
    if t == "record":
        out = {}
        for f in schema["fields"]:
            out[f["name"]], pos = decode(f["type"], buf, pos)
        return out, pos
    if t == "array":
        out = []
        while True:
            n, pos = read_long(buf, pos)
            if n == 0:
                break
            if n < 0:  # block with byte size
                _, pos = read_long(buf, pos)
                n = -n
            for _ in range(n):
                v, pos = decode(schema["items"], buf, pos)
                out.append(v)
        return out, pos
    if t == "map":
        out = {}
        while True:
            n, pos = read_long(buf, pos)
            if n == 0:
                break
            if n < 0:
                _, pos = read_long(buf, pos)
                n = -n
            for _ in range(n):
                kb, pos = read_bytes(buf, pos)
                out[kb.decode("utf-8")], pos = decode(schema["values"], buf, pos)
        return out, pos
    raise ValueError(f"avro type {t!r} unsupported")


# ---------------------------------------------------------------- encode
def encode(schema, value, out: bytearray):
    t = schema["type"] if isinstance(schema, dict) else schema
    if t == "union":
        branches = schema["branches"]
        if value is None:
            for i, b in enumerate(branches):
                if (b["type"] if isinstance(b, dict) else b) == "null":
                    write_long(out, i)
                    return
            raise ValueError("union has no null branch")
        for i, b in enumerate(branches):
            if (b["type"] if isinstance(b, dict) else b) != "null":
                write_long(out, i)
                encode(b, value, out)
                return
        raise ValueError("no non-null union branch")
    if t == "null":
        return
    if t == "boolean":
        out.append(1 if value else 0)
        return
    if t in ("int", "long"):
        write_long(out, int(value))
        return
    if t == "float":
        out += struct.pack("<f", value)
        return
    if t == "double":
        out += struct.pack("<d", value)
        return
    if t == "bytes":
        write_bytes(out, value)
        return
    if t == "string":
        write_bytes(out, value.encode("utf-8"))
        return
    if t == "fixed":
        assert len(value) == schema["size"]
        out += value
        return
    if t == "record":
        for f in schema["fields"]:
            encode(f["type"], value[f["name"]], out)
        return
    if t == "array":
        if value:
            write_long(out, len(value))
            for v in value:
                encode(schema["items"], v, out)
        write_long(out, 0)
        return
    if t == "map":
        if value:
            write_long(out, len(value))
            for k, v in value.items():
                write_bytes(out, k.encode("utf-8"))
                encode(schema["values"], v, out)
        write_long(out, 0)
        return
    raise ValueError(f"avro type {t!r} unsupported")


# --------------------------------------------------------------- files
def read_file(path: str) -> Tuple[dict, List[dict]]:
    """-> (header meta, records)."""
    raw = open(path, "rb").read()
    assert raw[:4] == MAGIC, "not an avro object container file"
    pos = 4
    meta_schema = {"type": "map", "values": {"type": "bytes"}}
    meta, pos = decode(meta_schema, raw, pos)
    sync = raw[pos:pos + 16]
    pos += 16
    schema = parse_schema(meta["avro.schema"].decode("utf-8"))
    codec = meta.get("avro.codec", b"null").decode("utf-8")
    records: List[dict] = []
    while pos < len(raw):
        count, pos = read_long(raw, pos)
        size, pos = read_long(raw, pos)
        block = bytes(raw[pos:pos + size])
        pos += size
        assert raw[pos:pos + 16] == sync, "bad avro sync marker"
        pos += 16
        if codec == "deflate":
            block = zlib.decompress(block, -15)
        elif codec != "null":
            raise ValueError(f"avro codec {codec} unsupported")
        bp = 0
        for _ in range(count):
            rec, bp = decode(schema, block, bp)
            records.append(rec)
    return meta, records


def write_file(path: str, schema_json: str, records: List[dict]):
    schema = parse_schema(schema_json)
    out = bytearray()
    out += MAGIC
    meta = {"avro.schema": schema_json.encode("utf-8"), "avro.codec": b"null"}
    encode({"type": "map", "values": {"type": "bytes"}}, meta, out)
    sync = b"auron-avro-sync!"  # any 16 bytes
    out += sync
    if records:
        block = bytearray()
        for r in records:
            encode(schema, r, block)
        write_long(out, len(records))
        write_long(out, len(block))
        out += block
        out += sync
    with open(path, "wb") as f:
        f.write(bytes(out))

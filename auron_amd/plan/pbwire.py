"""Minimal protobuf wire-format codec (proto3 subset), pure Python.

Role parity: the prost/protobuf layer under the reference's plan contract
(native-engine/auron-planner/proto/auron.proto is compiled by prost on the
Rust side and protoc-java on the JVM side). Here the schema lives in
plan/auron.proto and plan/proto.py declares the same messages as field
tables; this module implements the standard encoding so the bytes are
readable by any protobuf implementation given that .proto file.

Supported: varint (int32/int64/uint64/bool/enum), 64-bit (double),
length-delimited (string/bytes/message/packed-varint), repeated fields.
"""
from __future__ import annotations

from typing import Dict, Iterator, List, Tuple

WT_VARINT = 0
WT_I64 = 1
WT_LEN = 2


def write_varint(out: bytearray, v: int):
    if v < 0:
        v += 1 << 64  # two's complement, 10 bytes (protobuf int64 semantics)
    while True:
        b = v & 0x7F
        v >>= 7
        if v:
            out.append(b | 0x80)
        else:
            out.append(b)
            return


def read_varint(buf: bytes, pos: int) -> Tuple[int, int]:
    result = 0
    shift = 0
    while True:
        b = buf[pos]
        pos += 1
        result |= (b & 0x7F) << shift
        if not (b & 0x80):
            break
        shift += 7
        if shift > 70:
            raise ValueError("varint too long")
    if result >= 1 << 63:
        result -= 1 << 64  # interpret as signed int64
    return result, pos


def write_tag(out: bytearray, field: int, wt: int):
    write_varint(out, (field << 3) | wt)


def write_double(out: bytearray, field: int, v: float):
    import struct

    write_tag(out, field, WT_I64)
    out += struct.pack("<d", v)


def write_len(out: bytearray, field: int, payload: bytes):
    write_tag(out, field, WT_LEN)
    write_varint(out, len(payload))
    out += payload


def write_int(out: bytearray, field: int, v: int):
    write_tag(out, field, WT_VARINT)
    write_varint(out, v)


def write_str(out: bytearray, field: int, v: str):
    write_len(out, field, v.encode("utf-8"))


def iter_fields(buf: bytes) -> Iterator[Tuple[int, int, object]]:
    """Yield (field_number, wire_type, raw_value) over a message buffer."""
    import struct

    pos = 0
    n = len(buf)
    while pos < n:
        key, pos = read_varint(buf, pos)
        if key < 0:
            raise ValueError("bad tag")
        field, wt = key >> 3, key & 7
        if wt == WT_VARINT:
            v, pos = read_varint(buf, pos)
            yield field, wt, v
        elif wt == WT_I64:
            v = struct.unpack_from("<d", buf, pos)[0]
            pos += 8
            yield field, wt, v
        elif wt == WT_LEN:
            ln, pos = read_varint(buf, pos)
            yield field, wt, bytes(buf[pos:pos + ln])
            pos += ln
        else:
            raise ValueError(f"unsupported wire type {wt}")


def decode_fields(buf: bytes) -> Dict[int, List[object]]:
    out: Dict[int, List[object]] = {}
    for field, _, v in iter_fields(buf):
        out.setdefault(field, []).append(v)
    return out

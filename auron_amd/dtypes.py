"""Data types for the Auron-AMD columnar engine.

Role parity: the Arrow type surface used by the reference's plan protocol
(/root/reference/native-engine/auron-planner/proto/auron.proto:942-978) —
re-expressed for a torch-tensor-backed columnar store on MI355X.
"""
from __future__ import annotations

from dataclasses import dataclass

import torch

# dtype codes shared with the native C API (csrc/auron_common.h)
BOOL = 0
INT8 = 1
INT16 = 2
INT32 = 3
INT64 = 4
FLOAT32 = 5
FLOAT64 = 6
DATE32 = 7  # days since epoch, int32 storage
STRING = 8  # uint8 byte buffer + int32 offsets
DECIMAL64 = 9  # scaled int64
LIST = 10  # flattened child values + int32 offsets (element nulls: round 2)
TIMESTAMP = 11  # microseconds since epoch UTC, int64 storage
DECIMAL128 = 12  # two-limb scaled int: data [n,2] int64 = (lo bits, hi)

_NAMES = {
    BOOL: "bool",
    INT8: "int8",
    INT16: "int16",
    INT32: "int32",
    INT64: "int64",
    FLOAT32: "float32",
    FLOAT64: "float64",
    DATE32: "date32",
    STRING: "string",
    DECIMAL64: "decimal64",
    LIST: "list",
    TIMESTAMP: "timestamp",
    DECIMAL128: "decimal128",
}

_TORCH = {
    BOOL: torch.bool,
    INT8: torch.int8,
    INT16: torch.int16,
    INT32: torch.int32,
    INT64: torch.int64,
    FLOAT32: torch.float32,
    FLOAT64: torch.float64,
    DATE32: torch.int32,
    STRING: torch.uint8,  # byte buffer
    DECIMAL64: torch.int64,
    LIST: None,  # resolved via .child
    TIMESTAMP: torch.int64,
    DECIMAL128: torch.int64,
}


@dataclass(frozen=True)
class DataType:
    code: int
    precision: int = 0  # decimal only
    scale: int = 0  # decimal only

    @property
    def name(self) -> str:
        if self.code == DECIMAL64:
            return f"decimal({self.precision},{self.scale})"
        if self.code == DECIMAL128:
            return f"decimal128({self.precision},{self.scale})"
        if self.code == LIST:
            return f"list<{_NAMES[self.precision]}>"
        return _NAMES[self.code]

    @property
    def torch_dtype(self) -> torch.dtype:
        if self.code == LIST:
            return _TORCH[self.precision]
        return _TORCH[self.code]

    @property
    def is_string(self) -> bool:
        return self.code == STRING

    @property
    def is_list(self) -> bool:
        return self.code == LIST

    @property
    def uses_offsets(self) -> bool:
        '''Arrow offsets layout: strings (byte elements) and lists.'''
        return self.code in (STRING, LIST)

    @property
    def child(self) -> "DataType":
        '''Element type of a list (stored in the precision/scale slots).'''
        assert self.code == LIST
        return DataType(self.precision, 0, self.scale)

    @property
    def is_numeric(self) -> bool:
        return self.code in (INT8, INT16, INT32, INT64, FLOAT32, FLOAT64,
                             DECIMAL64, DECIMAL128)

    @property
    def is_integer(self) -> bool:
        return self.code in (INT8, INT16, INT32, INT64)

    @property
    def is_float(self) -> bool:
        return self.code in (FLOAT32, FLOAT64)

    def __repr__(self) -> str:  # pragma: no cover
        return self.name


bool_ = DataType(BOOL)
int8 = DataType(INT8)
int16 = DataType(INT16)
int32 = DataType(INT32)
int64 = DataType(INT64)
float32 = DataType(FLOAT32)
float64 = DataType(FLOAT64)
date32 = DataType(DATE32)
string = DataType(STRING)
timestamp = DataType(TIMESTAMP)


def decimal64(precision: int = 18, scale: int = 2) -> DataType:
    return DataType(DECIMAL64, precision, scale)


def decimal128(precision: int = 38, scale: int = 2) -> DataType:
    return DataType(DECIMAL128, precision, scale)


def list_of(child: DataType) -> DataType:
    assert child.code != LIST, "nested lists unsupported (round 2)"
    return DataType(LIST, child.code, child.scale)


def from_arrow(at) -> DataType:
    """Map a pyarrow DataType to an auron DataType."""
    import pyarrow as pa

    if pa.types.is_boolean(at):
        return bool_
    if pa.types.is_int8(at):
        return int8
    if pa.types.is_int16(at):
        return int16
    if pa.types.is_int32(at):
        return int32
    if pa.types.is_int64(at):
        return int64
    if pa.types.is_float32(at):
        return float32
    if pa.types.is_float64(at):
        return float64
    if pa.types.is_date32(at):
        return date32
    if pa.types.is_string(at) or pa.types.is_large_string(at):
        return string
    if pa.types.is_timestamp(at):
        return timestamp
    if pa.types.is_decimal(at):
        if at.precision > 18:
            return decimal128(at.precision, at.scale)
        return decimal64(at.precision, at.scale)
    if pa.types.is_dictionary(at):
        return from_arrow(at.value_type)
    raise TypeError(f"unsupported arrow type {at}")


def to_arrow(dt: DataType):
    import pyarrow as pa

    m = {
        BOOL: pa.bool_(),
        INT8: pa.int8(),
        INT16: pa.int16(),
        INT32: pa.int32(),
        INT64: pa.int64(),
        FLOAT32: pa.float32(),
        FLOAT64: pa.float64(),
        DATE32: pa.date32(),
        STRING: pa.string(),
    }
    if dt.code == DECIMAL64:
        return pa.decimal128(dt.precision, dt.scale)
    if dt.code == DECIMAL128:
        return pa.decimal128(dt.precision, dt.scale)
    if dt.code == TIMESTAMP:
        return pa.timestamp("us")
    return m[dt.code]

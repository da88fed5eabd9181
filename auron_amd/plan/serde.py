"""Plan serialization — the wire contract between front-end and engine.

Role parity: the reference's protobuf plan
(/root/reference/native-engine/auron-planner/proto/auron.proto:
 PhysicalPlanNode oneof :27-57, expression nodes :60-130,
 TaskDefinition :815) and its decoder (auron-planner/src/planner.rs:122
 create_plan / :941 create_expr). Here both sides are reflected over the
 dataclass plan/expr definitions and packed with msgpack (binary, stable
 tags) — one schema, no drift between builder and executor.

RecordBatches inside MemoryScan nodes are serialized with the exchange
packing (meta + raw buffers), so a TaskDefinition is fully
self-contained, like the reference's serialized task."""
from __future__ import annotations

import importlib
from dataclasses import fields, is_dataclass
from typing import Any

import msgpack
import torch

from .. import dtypes as _dt
from ..column import Column, RecordBatch
from ..dtypes import DataType
from ..exchange import pack_batch, unpack_batch
from ..exprs import (AggFunc, Aliased, Arith, BoolOp, CaseWhen, Cast, Cmp,
                     Coalesce, Col, ConcatStr, DatePart, Expr, InList, IsNull,
                     Length, Like, Literal, Not, Substr, WindowFunc)
from . import nodes as P

_EXPR_MOD = "auron_amd.exprs"
_FUNC_MOD = "auron_amd.functions"
_NODE_MOD = "auron_amd.plan.nodes"


def _classname(o) -> str:
    return type(o).__name__


def _encode(o) -> Any:
    if o is None or isinstance(o, (bool, int, float, str, bytes)):
        return o
    if isinstance(o, DataType):
        return {"__t": "DataType", "c": o.code, "p": o.precision, "s": o.scale}
    if isinstance(o, RecordBatch):
        meta, buf = pack_batch(o.to("cpu"), "cpu")
        return {"__t": "RecordBatch", "meta": meta,
                "buf": buf.numpy().tobytes()}
    if isinstance(o, (list, tuple)):
        return {"__t": "list", "v": [_encode(x) for x in o]} if isinstance(o, tuple) else [_encode(x) for x in o]
    if is_dataclass(o):
        d = {"__t": _classname(o)}
        for f in fields(o):
            d[f.name] = _encode(getattr(o, f.name))
        return d
    raise TypeError(f"cannot serialize {type(o)}")


_REGISTRY = {}
for mod in (_EXPR_MOD, _FUNC_MOD, _NODE_MOD):
    m = importlib.import_module(mod)
    for name in dir(m):
        c = getattr(m, name)
        if isinstance(c, type) and is_dataclass(c):
            _REGISTRY[name] = c


def _decode(o):
    if isinstance(o, list):
        return [_decode(x) for x in o]
    if not isinstance(o, dict) or "__t" not in o:
        return o
    t = o["__t"]
    if t == "DataType":
        return DataType(o["c"], o["p"], o["s"])
    if t == "RecordBatch":
        buf = torch.frombuffer(bytearray(o["buf"]), dtype=torch.uint8)
        return unpack_batch(o["meta"], buf)
    if t == "list":
        return tuple(_decode(x) for x in o["v"])
    cls = _REGISTRY.get(t)
    if cls is None:
        raise ValueError(f"unknown node type {t}")
    kwargs = {k: _decode(v) for k, v in o.items() if k != "__t"}
    return cls(**kwargs)


def serialize_plan(plan: P.PlanNode) -> bytes:
    return msgpack.packb(_encode(plan), use_bin_type=True)


def deserialize_plan(data: bytes) -> P.PlanNode:
    return _decode(msgpack.unpackb(data, raw=False, strict_map_key=False))


def serialize_task(task_id: str, stage_id: int, partition: int,
                   plan: P.PlanNode) -> bytes:
    """TaskDefinition (auron.proto:815 analogue)."""
    return msgpack.packb({
        "task_id": task_id, "stage_id": stage_id, "partition": partition,
        "plan": _encode(plan),
    }, use_bin_type=True)


def deserialize_task(data: bytes):
    d = msgpack.unpackb(data, raw=False, strict_map_key=False)
    return d["task_id"], d["stage_id"], d["partition"], _decode(d["plan"])

"""Plan serde roundtrip (TaskDefinition contract, auron.proto analogue)."""
import pytest

from auron_amd import AggFunc, AuronSession, col, dtypes, exprs, lit
from auron_amd.column import RecordBatch
from auron_amd.plan import nodes as P
from auron_amd.plan import serde

DATA = {"k": ["a", "b", None, "a"], "x": [1, 2, 3, 4]}
TYPES = {"k": dtypes.string, "x": dtypes.int64}


def _plan():
    scan = P.MemoryScan([RecordBatch.from_pydict(DATA, TYPES)])
    f = P.Filter(scan, (col("x") > 1) & col("k").is_not_null())
    agg = P.HashAgg(f, [exprs.Aliased(col("k"), "k")],
                    [AggFunc("sum", col("x") * 2, name="s2"),
                     AggFunc("count_star", None, name="n")], mode="complete")
    return P.Limit(P.Sort(agg, [(col("k"), True)]), 10)


def test_roundtrip_produces_same_result():
    s = AuronSession()
    plan = _plan()
    want = s.collect(plan).to_pydict()
    blob = serde.serialize_task("t-1", 0, 0, plan)
    assert isinstance(blob, bytes) and len(blob) > 50
    got_batches = s.execute_serialized(blob)
    got = RecordBatch.concat(got_batches).to_pydict()
    assert got == want


def test_roundtrip_expr_nodes():
    e = exprs.CaseWhen([(col("x").between(1, 3), lit(1))],
                       exprs.DatePart("year", col("d")))
    p = P.Project(P.MemoryScan([RecordBatch.from_pydict(
        {"x": [1], "d": [19000]}, {"x": dtypes.int64, "d": dtypes.date32})]),
        [exprs.Aliased(e, "out")])
    blob = serde.serialize_plan(p)
    p2 = serde.deserialize_plan(blob)
    s = AuronSession()
    assert s.collect(p2).to_pydict() == s.collect(p).to_pydict()


def test_tpcds_plans_serializable():
    import os

    from auron_amd.tpcds import datagen
    from auron_amd.tpcds.queries import QUERIES, Catalog

    root = os.path.join(os.path.dirname(__file__), "..", ".tpcds_cache")
    datagen.write_dataset(root, 0.01, tables=["date_dim", "store_sales", "item"])
    s = AuronSession()
    cat = Catalog(root, 0.01)
    plan = QUERIES["q3"](cat, s)
    blob = serde.serialize_plan(plan)
    p2 = serde.deserialize_plan(blob)
    assert s.collect(p2).to_pydict() == s.collect(plan).to_pydict()


def test_all_99_plans_roundtrip():
    """Every TPC-DS plan must survive serialize->deserialize with an
    identical structural fingerprint (planner/serde full-surface check)."""
    import os

    from auron_amd import AuronSession
    from auron_amd.plan.serde import deserialize_task, serialize_task
    from auron_amd.tpcds import datagen
    from auron_amd.tpcds.queries import QUERIES, Catalog
    from tests.test_plan_stability import _shape

    root = os.path.join(os.path.dirname(__file__), "..", ".tpcds_cache")
    datagen.write_dataset(root, 0.01)
    s = AuronSession()
    cat = Catalog(root, 0.01)
    for qn in sorted(QUERIES):
        plan = QUERIES[qn](cat, s)
        blob = serialize_task("t", 0, 0, plan)
        _, _, _, back = deserialize_task(blob)
        assert _shape(back) == _shape(plan), qn


def test_all_99_plans_protobuf_roundtrip():
    """The protobuf TaskDefinition contract (plan/auron.proto) carries
    every TPC-DS plan: encode -> standard-wire decode -> same structure."""
    import os

    from auron_amd import AuronSession
    from auron_amd.plan.proto import deserialize_task_pb, serialize_task_pb
    from auron_amd.tpcds import datagen
    from auron_amd.tpcds.queries import QUERIES, Catalog
    from tests.test_plan_stability import _shape

    root = os.path.join(os.path.dirname(__file__), "..", ".tpcds_cache")
    datagen.write_dataset(root, 0.01)
    s = AuronSession()
    cat = Catalog(root, 0.01)
    for qn in sorted(QUERIES):
        plan = QUERIES[qn](cat, s)
        blob = serialize_task_pb("t", 1, 2, plan)
        tid, stage, part, back = deserialize_task_pb(blob)
        assert (tid, stage, part) == ("t", 1, 2)
        assert _shape(back) == _shape(plan), qn


def test_protobuf_wire_vectors():
    """Byte-level wire-format vectors (varint, tags, nested messages) so
    the pure-python codec stays interchangeable with protoc bindings."""
    from auron_amd.plan.pbwire import (decode_fields, write_int, write_len,
                                       write_str)

    out = bytearray()
    write_int(out, 1, 150)
    assert bytes(out) == b"\x08\x96\x01"  # canonical protobuf example
    out = bytearray()
    write_str(out, 2, "testing")
    assert bytes(out) == b"\x12\x07testing"
    out = bytearray()
    write_int(out, 1, -2)  # negative int64: 10-byte twos complement
    f = decode_fields(bytes(out))
    assert f[1][-1] == -2
    inner = bytearray()
    write_int(inner, 1, 1)
    out = bytearray()
    write_len(out, 3, bytes(inner))
    f = decode_fields(bytes(out))
    assert decode_fields(f[3][-1])[1][-1] == 1

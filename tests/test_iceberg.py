"""Iceberg provider: metadata/manifest resolution down to the shared
ParquetScan path (thirdparty/auron-iceberg AuronConvertProvider parity),
plus Avro codec spec vectors."""
import os

import numpy as np
import pyarrow as pa
import pyarrow.parquet as pq
import pytest

from auron_amd import AuronSession, col
from auron_amd.lakehouse import IcebergTable, avro, iceberg
from auron_amd.plan import nodes as P


def test_avro_zigzag_spec_vectors():
    # Avro spec: 0->0, -1->1, 1->2, -2->3, 2->4 ...
    for v, z in [(0, 0), (-1, 1), (1, 2), (-2, 3), (2, 4), (-64, 127), (64, 128)]:
        assert avro.zigzag_encode(v) == z
        assert avro.zigzag_decode(z) == v
    out = bytearray()
    avro.write_long(out, 1)
    assert bytes(out) == b"\x02"
    out = bytearray()
    avro.write_long(out, -64)
    assert bytes(out) == b"\x7f"
    out = bytearray()
    avro.write_long(out, 64)
    assert bytes(out) == b"\x80\x01"


def test_avro_roundtrip_nested():
    schema = """{"type":"record","name":"t","fields":[
      {"name":"s","type":"string"},
      {"name":"n","type":["null","long"]},
      {"name":"xs","type":{"type":"array","items":"int"}},
      {"name":"m","type":{"type":"map","values":"string"}},
      {"name":"d","type":"double"},
      {"name":"b","type":"boolean"}]}"""
    recs = [
        {"s": "hello", "n": None, "xs": [1, -2, 3], "m": {"k": "v"}, "d": 1.5, "b": True},
        {"s": "", "n": 12345678901234, "xs": [], "m": {}, "d": -0.25, "b": False},
    ]
    import tempfile

    with tempfile.TemporaryDirectory() as d:
        p = os.path.join(d, "t.avro")
        avro.write_file(p, schema, recs)
        _, back = avro.read_file(p)
    assert back == recs


@pytest.fixture()
def iceberg_table(tmp_path):
    rng = np.random.default_rng(5)
    files = []
    tables = []
    for i in range(3):
        t = pa.table({"k": pa.array(rng.integers(0, 50, 1000)),
                      "v": pa.array(rng.normal(size=1000))})
        p = str(tmp_path / f"part-{i}.parquet")
        pq.write_table(t, p)
        files.append(p)
        tables.append(t)
    tbl = iceberg.write_table(str(tmp_path / "tbl"), files)
    return tbl, tables


def test_iceberg_scan_matches_parquet(iceberg_table):
    tbl, tables = iceberg_table
    s = AuronSession()
    out = s.collect(tbl.scan(columns=["k", "v"]))
    want_rows = sum(t.num_rows for t in tables)
    assert out.num_rows == want_rows
    got_sum = sum(v for v in out.to_pydict()["k"])
    want_sum = sum(sum(t["k"].to_pylist()) for t in tables)
    assert got_sum == want_sum


def test_iceberg_query(iceberg_table):
    tbl, tables = iceberg_table
    s = AuronSession()
    from auron_amd.exprs import AggFunc, Aliased

    plan = P.HashAgg(P.Filter(tbl.scan(columns=["k", "v"]), col("k") < 10),
                     [Aliased(col("k"), "k")],
                     [AggFunc("count_star", None, name="n")], mode="complete")
    out = s.collect(plan).to_pydict()
    import collections

    want = collections.Counter()
    for t in tables:
        for kv in t["k"].to_pylist():
            if kv < 10:
                want[kv] += 1
    got = dict(zip(out["k"], out["n"]))
    assert got == dict(want)

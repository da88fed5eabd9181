"""ORC scan, parquet/orc sinks, host-UDF bounce, Generate (explode)."""
import os

import pytest

from auron_amd import AuronSession, col, dtypes, exprs, lit
from auron_amd.column import RecordBatch
from auron_amd.plan import nodes as P

DATA = {"k": ["a,b,c", "x", None, "p,q"], "v": [1, 2, 3, 4]}
TYPES = {"k": dtypes.string, "v": dtypes.int64}


def _scan():
    return P.MemoryScan([RecordBatch.from_pydict(DATA, TYPES)])


def test_parquet_sink_roundtrip(tmp_path):
    s = AuronSession()
    sink = P.ParquetSink(_scan(), str(tmp_path / "out"))
    res = s.collect(sink).to_pydict()
    assert res["rows_written"] == [4]
    files = sorted(os.listdir(tmp_path / "out"))
    assert len(files) == 1
    back = s.collect(P.ParquetScan([str(tmp_path / "out" / files[0])]))
    assert back.to_pydict() == DATA


def test_orc_sink_and_scan_roundtrip(tmp_path):
    s = AuronSession()
    sink = P.OrcSink(_scan(), str(tmp_path / "orc"))
    assert s.collect(sink).to_pydict()["rows_written"] == [4]
    files = sorted(os.listdir(tmp_path / "orc"))
    back = s.collect(P.OrcScan([str(tmp_path / "orc" / f) for f in files]))
    assert back.to_pydict() == DATA


def test_pyudf_bounce():
    s = AuronSession()

    def fn(batch):
        vs = batch.column("v").to_pylist()
        return {"doubled": ([None if v is None else v * 2 for v in vs], dtypes.int64)}

    plan = P.PyUdf(_scan(), fn)
    out = s.collect(plan).to_pydict()
    assert out["doubled"] == [2, 4, 6, 8]


def test_generate_explode():
    s = AuronSession()
    g = P.Generate(_scan(), "posexplode_split", [col("k"), lit(",")])
    out = s.collect(g).to_pydict()
    assert out["col"] == ["a", "b", "c", "x", "p", "q"]
    assert out["pos"] == [0, 1, 2, 0, 0, 1]
    assert out["v"] == [1, 1, 1, 2, 4, 4]

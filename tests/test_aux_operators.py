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


def test_generate_json_tuple():
    from auron_amd import AuronSession, col, dtypes, lit
    from auron_amd.column import RecordBatch
    from auron_amd.plan import nodes as P

    docs = ['{"a": 1, "b": "x"}', '{"a": null}', "bad", None]
    sc = P.MemoryScan([RecordBatch.from_pydict(
        {"id": [1, 2, 3, 4], "j": docs}, {"id": dtypes.int64, "j": dtypes.string})])
    g = P.Generate(sc, "json_tuple", [col("j"), lit("a"), lit("b")])
    out = AuronSession().collect(g).to_pydict()
    assert out["c0"] == ["1", None, None, None]
    assert out["c1"] == ["x", None, None, None]
    assert out["id"] == [1, 2, 3, 4]


def test_generate_udtf():
    from auron_amd import AuronSession, col, dtypes
    from auron_amd.column import RecordBatch
    from auron_amd.plan import nodes as P

    def dup(n):
        # emits n rows of (i, i*10) for input n
        for i in range(int(n)):
            yield (i, i * 10)

    sc = P.MemoryScan([RecordBatch.from_pydict(
        {"n": [2, 0, 1]}, {"n": dtypes.int64})])
    g = P.Generate(sc, "udtf", [col("n")], udtf=dup,
                   udtf_schema=[("i", dtypes.int64), ("tens", dtypes.int64)])
    out = AuronSession().collect(g).to_pydict()
    assert out["n"] == [2, 2, 1]
    assert out["i"] == [0, 1, 0]
    assert out["tens"] == [0, 10, 0]


def test_py_udaf():
    from auron_amd import AuronSession, col, dtypes
    from auron_amd.column import RecordBatch
    from auron_amd.exprs import Aliased
    from auron_amd.plan import nodes as P

    def geo_mean_ish(vals):
        xs = [v for v in vals if v is not None]
        return (sum(xs) / len(xs), len(xs)) if xs else (None, 0)

    sc = P.MemoryScan([RecordBatch.from_pydict(
        {"k": ["a", "b", "a", "a"], "v": [1.0, 10.0, 3.0, None]},
        {"k": dtypes.string, "v": dtypes.float64})])
    g = P.PyUdaf(sc, [Aliased(col("k"), "k")], [col("v")], geo_mean_ish,
                 [("m", dtypes.float64), ("cnt", dtypes.int64)])
    out = AuronSession().collect(g).to_pydict()
    m = {k: (a, c) for k, a, c in zip(out["k"], out["m"], out["cnt"])}
    assert m == {"a": (2.0, 2), "b": (10.0, 1)}


def test_list_column_and_explode():
    from auron_amd import AuronSession, col, dtypes
    from auron_amd.column import Column, RecordBatch
    from auron_amd.plan import nodes as P

    lt = dtypes.list_of(dtypes.int64)
    c = Column.from_pylist([[1, 2], [], None, [7]], lt)
    assert c.to_pylist() == [[1, 2], [], None, [7]]
    # gather with nulls / reorder keeps layout
    import torch
    g = c.gather(torch.tensor([3, 0, -1]), may_have_negative=True)
    assert g.to_pylist() == [[7], [1, 2], None]
    assert Column.concat([c, g]).to_pylist() == \
        [[1, 2], [], None, [7], [7], [1, 2], None]

    sc = P.MemoryScan([RecordBatch(["id", "xs"],
                                   [Column.from_pylist([1, 2, 3, 4], dtypes.int64), c])])
    out = AuronSession().collect(P.Generate(sc, "posexplode", [col("xs")])).to_pydict()
    assert out["id"] == [1, 1, 4]
    assert out["pos"] == [0, 1, 0]
    assert out["col"] == [1, 2, 7]


def test_make_array_then_explode():
    from auron_amd import AuronSession, col, dtypes, functions as F
    from auron_amd.column import RecordBatch
    from auron_amd.exprs import Aliased
    from auron_amd.plan import nodes as P

    sc = P.MemoryScan([RecordBatch.from_pydict(
        {"a": [1, 10], "b": [2, 20]}, {"a": dtypes.int64, "b": dtypes.int64})])
    proj = P.Project(sc, [Aliased(F.MakeArray([col("a"), col("b")]), "arr")])
    out = AuronSession().collect(P.Generate(proj, "explode", [col("arr")])).to_pydict()
    assert out["col"] == [1, 2, 10, 20]


def test_cancel_and_task_error(tmp_path):
    import pytest as _pytest

    from auron_amd import AuronSession, col, dtypes
    from auron_amd.column import RecordBatch
    from auron_amd.engine.executor import TaskCancelled
    from auron_amd.plan import nodes as P
    from auron_amd.plan.serde import serialize_task
    from auron_amd.session import AuronTaskError

    s = AuronSession()
    plan = P.Filter(P.MemoryScan([RecordBatch.from_pydict(
        {"x": [1, 2]}, {"x": dtypes.int64})]), col("x") > 0)
    s.executor.cancel()
    with _pytest.raises(TaskCancelled):
        s.collect(plan)
    s.executor.reset_cancel()
    assert s.collect(plan).num_rows == 2
    # serialized task failure carries task context
    bad = P.ParquetScan([str(tmp_path / "nope.parquet")], columns=["x"])
    blob = serialize_task("t-1", 3, 7, bad)
    with _pytest.raises(AuronTaskError) as ei:
        s.execute_serialized(blob)
    assert ei.value.stage_id == 3 and ei.value.partition == 7


def test_ignore_corrupted_files(tmp_path, monkeypatch):
    import pyarrow as pa
    import pyarrow.parquet as pq

    from auron_amd import AuronSession, dtypes
    from auron_amd.plan import nodes as P

    good = str(tmp_path / "good.parquet")
    pq.write_table(pa.table({"x": pa.array([1, 2, 3])}), good,
                   compression="NONE", use_dictionary=False,
                   data_page_version="1.0")
    bad = str(tmp_path / "bad.parquet")
    with open(bad, "wb") as f:
        f.write(b"not a parquet file")
    s = AuronSession()
    plan = P.ParquetScan([good, bad], columns=["x"])
    import pytest as _pytest
    with _pytest.raises(Exception):
        s.collect(plan)
    monkeypatch.setenv("AURON_IGNORE_CORRUPTED_FILES", "1")
    out = AuronSession().collect(P.ParquetScan([good, bad], columns=["x"]))
    assert sorted(out.to_pydict()["x"]) == [1, 2, 3]


def test_collect_list_and_set():
    from auron_amd import AggFunc, AuronSession, col, dtypes
    from auron_amd.column import RecordBatch
    from auron_amd.exprs import Aliased
    from auron_amd.plan import nodes as P

    data = {"k": ["a", "a", "b", "a", "b", "c"],
            "v": [1, 2, 5, 2, 5, None]}
    t = {"k": dtypes.string, "v": dtypes.int64}
    sc = P.MemoryScan([RecordBatch.from_pydict(data, t)])
    plan = P.HashAgg(sc, [Aliased(col("k"), "k")],
                     [AggFunc("collect_list", col("v"), name="cl"),
                      AggFunc("collect_set", col("v"), name="cs")],
                     mode="complete")
    out = AuronSession().collect(plan).to_pydict()
    m = {k: (sorted(cl), sorted(cs)) for k, cl, cs in
         zip(out["k"], out["cl"], out["cs"])}
    assert m["a"] == ([1, 2, 2], [1, 2])
    assert m["b"] == ([5, 5], [5])
    assert m["c"] == ([], [])  # only-null group -> empty list, not null

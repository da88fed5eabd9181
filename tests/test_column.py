import pyarrow as pa
import pytest
import torch

from auron_amd import dtypes
from auron_amd.column import Column, RecordBatch


def test_pylist_roundtrip_numeric():
    c = Column.from_pylist([1, None, 3], dtypes.int64)
    assert len(c) == 3
    assert c.null_count == 1
    assert c.to_pylist() == [1, None, 3]


def test_pylist_roundtrip_string():
    c = Column.from_pylist(["hello", "", None, "天地"], dtypes.string)
    assert c.to_pylist() == ["hello", "", None, "天地"]


def test_gather_numeric_with_null_marker():
    c = Column.from_pylist([10, 20, 30], dtypes.int64)
    out = c.gather(torch.tensor([2, -1, 0]), may_have_negative=True)
    assert out.to_pylist() == [30, None, 10]


def test_gather_string():
    c = Column.from_pylist(["aa", "b", "cccc"], dtypes.string)
    out = c.gather(torch.tensor([2, 2, 0, -1]), may_have_negative=True)
    assert out.to_pylist() == ["cccc", "cccc", "aa", None]


def test_concat_strings_and_nulls():
    a = Column.from_pylist(["x", None], dtypes.string)
    b = Column.from_pylist(["yy"], dtypes.string)
    c = Column.concat([a, b])
    assert c.to_pylist() == ["x", None, "yy"]


def test_arrow_roundtrip():
    t = pa.table({
        "i": pa.array([1, 2, None], type=pa.int32()),
        "s": pa.array(["a", None, "ccc"]),
        "f": pa.array([1.5, 2.5, 3.5]),
    })
    b = RecordBatch.from_arrow(t)
    assert b.num_rows == 3
    assert b.column("i").to_pylist() == [1, 2, None]
    assert b.column("s").to_pylist() == ["a", None, "ccc"]
    t2 = b.to_arrow()
    assert t2.column("f").to_pylist() == [1.5, 2.5, 3.5]


def test_decimal_roundtrip():
    c = Column.from_pylist([1.23, 45.6, None], dtypes.decimal64(10, 2))
    assert c.to_pylist() == [1.23, 45.6, None]


def test_batch_filter():
    b = RecordBatch.from_pydict(
        {"x": [1, 2, 3, 4], "s": ["a", "b", "c", "d"]},
        {"x": dtypes.int64, "s": dtypes.string},
    )
    out = b.filter(torch.tensor([True, False, True, False]))
    assert out.to_pydict() == {"x": [1, 3], "s": ["a", "c"]}


def test_struct_columns_flatten_at_boundary():
    """Struct columns flatten into dotted leaf columns at the arrow
    boundary (GetStructField-on-struct reads become plain column refs)."""
    import pyarrow as pa

    from auron_amd.column import RecordBatch

    t = pa.table({
        "id": [1, 2],
        "info": pa.array([{"a": 5, "b": "x"}, {"a": None, "b": "y"}],
                         pa.struct([("a", pa.int64()), ("b", pa.string())])),
    })
    rb = RecordBatch.from_arrow(t)
    assert rb.names == ["id", "info.a", "info.b"]
    assert rb.column("info.a").to_pylist() == [5, None]
    assert rb.column("info.b").to_pylist() == ["x", "y"]

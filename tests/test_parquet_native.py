"""Native parquet decode vs pyarrow ground truth (host reference decoder
on CPU; HIP kernels on GPU)."""
import numpy as np
import pyarrow as pa
import pyarrow.parquet as pq
import pytest
import torch

from auron_amd import parquet_native
from auron_amd.column import RecordBatch


@pytest.fixture(scope="module")
def pq_file(tmp_path_factory):
    rng = np.random.default_rng(42)
    n = 50_000
    d = {
        "i32": pa.array(rng.integers(-1000, 1000, n).astype(np.int32),
                        mask=rng.random(n) < 0.05),
        "i64": pa.array(rng.integers(-10**12, 10**12, n),
                        mask=rng.random(n) < 0.03),
        "f64": pa.array(rng.normal(0, 100, n), mask=rng.random(n) < 0.02),
        "nonull": pa.array(rng.integers(0, 10**6, n)),
        "date": pa.array(rng.integers(0, 20000, n).astype(np.int32)).cast(pa.date32()),
        "allnull_free": pa.array(np.arange(n, dtype=np.int64)),
    }
    t = pa.table(d)
    p = tmp_path_factory.mktemp("pqn") / "t.parquet"
    pq.write_table(t, str(p), compression="NONE", use_dictionary=False,
                   data_page_version="1.0", row_group_size=16384,
                   data_page_size=8192)
    return str(p), t


def _check(cols, t):
    want = RecordBatch.from_arrow(t)
    for name in cols:
        got = cols[name]
        w = want.column(name)
        assert got.to_pylist() == w.to_pylist(), f"column {name} mismatch"


def test_np_reference_decoder(pq_file):
    path, t = pq_file
    names = t.schema.names
    cols = parquet_native.read_columns_native(path, names, "cpu", _np_only=True)
    assert cols is not None, "fast path rejected supported file"
    _check(cols, t)


def test_pyarrow_default_file_decodes_natively(tmp_path):
    # pyarrow defaults = snappy + dictionary; since round 2 the native
    # reader host-decompresses pages and decodes on device
    t = pa.table({"s": pa.array(["a", "b"] * 100)})
    p = tmp_path / "d.parquet"
    pq.write_table(t, str(p))
    out = parquet_native.read_columns_native(str(p), ["s"], "cpu", _np_only=True)
    assert out is not None
    assert out["s"].to_pylist() == ["a", "b"] * 100


@pytest.mark.parametrize("codec", ["snappy", "zstd", "gzip"])
@pytest.mark.parametrize("pv", ["1.0", "2.0"])
def test_compressed_pages_decode_natively(tmp_path, codec, pv):
    """Compressed v1/v2 data pages: host page decompression feeds the
    same device decode path (parquet_exec.rs/arrow-rs parity)."""
    rng = np.random.default_rng(11)
    n = 30_000
    t = pa.table({
        "x": pa.array(rng.integers(-10**9, 10**9, n), mask=rng.random(n) < 0.04),
        "f": pa.array(rng.normal(size=n)),
        "s": pa.array([f"w{int(v)}" for v in rng.integers(0, 50, n)],
                      mask=rng.random(n) < 0.03),
    })
    p = str(tmp_path / f"c-{codec}-{pv}.parquet")
    pq.write_table(t, p, compression=codec, use_dictionary=["s"],
                   data_page_version=pv, row_group_size=8192,
                   data_page_size=4096)
    cols = parquet_native.read_columns_native(p, ["x", "f", "s"], "cpu",
                                              _np_only=True)
    assert cols is not None and set(cols) == {"x", "f", "s"},         f"{codec}/{pv} rejected by the native path"
    _check(cols, t)


@pytest.mark.gpu
def test_gpu_kernels_decode(pq_file):
    path, t = pq_file
    names = t.schema.names
    cols = parquet_native.read_columns_native(path, names, "cuda:0")
    assert cols is not None
    assert all(c.data.is_cuda for c in cols.values())
    _check({k: v.to("cpu") for k, v in cols.items()}, t)


@pytest.mark.gpu
def test_gpu_decode_large_random(tmp_path):
    rng = np.random.default_rng(7)
    n = 2_000_000
    t = pa.table({
        "a": pa.array(rng.integers(0, 2**40, n), mask=rng.random(n) < 0.02),
        "b": pa.array(rng.normal(size=n)),
    })
    p = str(tmp_path / "big.parquet")
    pq.write_table(t, p, compression="NONE", use_dictionary=False,
                   data_page_version="1.0", row_group_size=1 << 20)
    cols = parquet_native.read_columns_native(p, ["a", "b"], "cuda:0")
    assert cols is not None
    ref = parquet_native.read_columns_native(p, ["a", "b"], "cpu", _np_only=True)
    for k in ("a", "b"):
        g = cols[k]
        r = ref[k]
        assert torch.equal(g.data.cpu(), r.data)
        gv = g.validity.cpu() if g.validity is not None else None
        rv = r.validity if r.validity is not None else None
        if gv is None or rv is None:
            assert gv is None and rv is None
        else:
            assert torch.equal(gv, rv)


@pytest.fixture(scope="module")
def dict_file(tmp_path_factory):
    """Dictionary-encoded file: strings + low-cardinality numerics,
    small pages so chunks have many RLE index pages."""
    rng = np.random.default_rng(3)
    n = 80_000
    words = ["alpha", "beta", "gamma", "", "a-much-longer-dictionary-entry",
             "x" * 40] + [f"w{i}" for i in range(500)]
    d = {
        "s": pa.array([words[i] for i in rng.integers(0, len(words), n)],
                      mask=rng.random(n) < 0.04),
        "snn": pa.array([words[i] for i in rng.integers(0, len(words), n)]),
        "di32": pa.array(rng.integers(0, 300, n).astype(np.int32),
                         mask=rng.random(n) < 0.05),
        "di64": pa.array(rng.integers(0, 1000, n) * 10**9,
                         mask=rng.random(n) < 0.03),
        "df64": pa.array(np.round(rng.normal(0, 10, n), 1)),
    }
    t = pa.table(d)
    p = tmp_path_factory.mktemp("pqd") / "dict.parquet"
    pq.write_table(t, str(p), compression="NONE", use_dictionary=True,
                   data_page_version="1.0", row_group_size=16384,
                   data_page_size=4096, dictionary_pagesize_limit=1 << 24)
    return str(p), t


def test_np_dict_decoder(dict_file):
    path, t = dict_file
    cols = parquet_native.read_columns_native(path, t.schema.names, "cpu",
                                              _np_only=True)
    assert cols is not None and set(cols) == set(t.schema.names), \
        "dictionary fast path rejected"
    _check(cols, t)


@pytest.mark.parametrize("codec", ["NONE", "snappy", "zstd"])
@pytest.mark.parametrize("pv", ["1.0", "2.0"])
def test_plain_byte_array_native(tmp_path, codec, pv):
    # the path pyarrow takes when a string column overflows the dictionary
    # page limit (high-cardinality columns at SF>=10)
    import random

    rng = random.Random(11)
    vals = ["".join(rng.choices("abcdefgh", k=rng.randint(0, 12)))
            if rng.random() > 0.15 else None for _ in range(5000)]
    t = pa.table({"s": pa.array(vals)})
    p = str(tmp_path / "pba.parquet")
    pq.write_table(t, p, compression=codec, use_dictionary=False,
                   data_page_version=pv)
    cols = parquet_native.read_columns_native(p, ["s"], "cpu", _np_only=True)
    assert cols is not None
    assert cols["s"].to_pylist() == vals


@pytest.mark.gpu
def test_plain_byte_array_gpu(tmp_path):
    import random

    from auron_amd import native

    native.require()
    rng = random.Random(12)
    vals = ["".join(rng.choices("abcdefgh", k=rng.randint(0, 20)))
            if rng.random() > 0.15 else None for _ in range(20000)]
    t = pa.table({"s": pa.array(vals)})
    for codec in ("NONE", "snappy"):
        p = str(tmp_path / f"pba_{codec}.parquet")
        pq.write_table(t, p, compression=codec, use_dictionary=False)
        cols = parquet_native.read_columns_native(p, ["s"], "cuda")
        assert cols is not None
        assert cols["s"].to("cpu").to_pylist() == vals


def test_partial_column_fallback(tmp_path):
    """Supported columns decode natively while an unsupported (nested
    list) column is dropped for the host path — no wholesale failure."""
    n = 5000
    rng = np.random.default_rng(5)
    t = pa.table({
        "num": pa.array(rng.integers(0, 50, n)),
        "ps": pa.array([f"unique-{i}" for i in range(n)]),
        "lst": pa.array([[1, 2]] * n),
    })
    p = str(tmp_path / "mix.parquet")
    pq.write_table(t, p, compression="NONE", data_page_version="1.0",
                   use_dictionary=["num"])
    ok, rest = parquet_native.split_supported(p, ["num", "ps", "lst"])
    assert set(ok) == {"num", "ps"} and rest == ["lst"]
    cols = parquet_native.read_columns_native(p, ok, "cpu", _np_only=True)
    assert cols is not None and "num" in cols and "ps" in cols
    _check({"num": cols["num"]}, t.select(["num"]))
    assert cols["ps"].to_pylist() == t["ps"].to_pylist()


@pytest.mark.gpu
def test_gpu_dict_decode(dict_file):
    path, t = dict_file
    cols = parquet_native.read_columns_native(path, t.schema.names, "cuda:0")
    assert cols is not None and set(cols) == set(t.schema.names)
    assert all(c.data.is_cuda for c in cols.values())
    _check({k: v.to("cpu") for k, v in cols.items()}, t)


@pytest.mark.parametrize("codec", ["NONE", "zstd"])
@pytest.mark.parametrize("pv", ["1.0", "2.0"])
def test_delta_binary_packed_native(tmp_path, codec, pv):
    import random

    rng = random.Random(4)
    xs = [rng.randint(-10 ** 9, 10 ** 9) if rng.random() > 0.1 else None
          for _ in range(7000)]
    ys = [rng.randint(0, 2 ** 40) for _ in range(7000)]
    t = pa.table({"x": pa.array(xs, pa.int32()), "y": pa.array(ys, pa.int64())})
    p = str(tmp_path / "delta.parquet")
    pq.write_table(t, p, use_dictionary=False, compression=codec,
                   column_encoding={"x": "DELTA_BINARY_PACKED",
                                    "y": "DELTA_BINARY_PACKED"},
                   data_page_version=pv)
    cols = parquet_native.read_columns_native(p, ["x", "y"], "cpu",
                                              _np_only=True)
    assert cols is not None
    assert cols["x"].to_pylist() == xs
    assert cols["y"].to_pylist() == ys


@pytest.mark.gpu
def test_delta_binary_packed_gpu(tmp_path):
    import random

    from auron_amd import native

    native.require()
    rng = random.Random(6)
    xs = [rng.randint(-10 ** 6, 10 ** 6) if rng.random() > 0.2 else None
          for _ in range(30000)]
    t = pa.table({"x": pa.array(xs, pa.int64())})
    p = str(tmp_path / "dg.parquet")
    pq.write_table(t, p, use_dictionary=False, compression="snappy",
                   column_encoding={"x": "DELTA_BINARY_PACKED"})
    cols = parquet_native.read_columns_native(p, ["x"], "cuda")
    assert cols is not None
    assert cols["x"].to("cpu").to_pylist() == xs

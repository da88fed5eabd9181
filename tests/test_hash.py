"""Spark murmur3 compatibility — golden vectors published by Spark
(the same behavioral constants the reference asserts in
spark_hash.rs tests: Murmur3Hash(..., 42).eval())."""
import pytest
import torch

from auron_amd import dtypes, ops
from auron_amd.column import Column


def _u(x):
    return x - 2 ** 32 if x >= 2 ** 31 else x


GOLDEN_I32 = {1: -559580957, 2: 1765031574, 3: -1823081949, 4: -397064898}
GOLDEN_I8 = [(1, _u(0xDEA578E3)), (0, _u(0x379FAE8F)), (-1, _u(0xA0590E3D)),
             (127, _u(0x43B4D8ED)), (-128, _u(0x422A1365))]
GOLDEN_I64 = [(1, _u(0x99F0149D)), (0, _u(0x9C67B85D)), (-1, _u(0xC8008529)),
              (2 ** 63 - 1, _u(0xA05B5D7B)), (-2 ** 63, _u(0xCD1E64FB))]
GOLDEN_STR = [("hello", _u(3286402344)), ("bar", _u(2486176763)), ("", _u(142593372)),
              ("😁", _u(885025535)), ("天地", _u(2395000894))]


def test_murmur3_i32_golden():
    c = Column.from_pylist(list(GOLDEN_I32.keys()), dtypes.int32)
    h = ops.murmur3_ref([c], 42).tolist()
    assert h == list(GOLDEN_I32.values())


def test_murmur3_i8_golden():
    c = Column.from_pylist([v for v, _ in GOLDEN_I8], dtypes.int8)
    h = ops.murmur3_ref([c], 42).tolist()
    assert h == [e for _, e in GOLDEN_I8]


def test_murmur3_i64_golden():
    c = Column.from_pylist([v for v, _ in GOLDEN_I64], dtypes.int64)
    h = ops.murmur3_ref([c], 42).tolist()
    assert h == [e for _, e in GOLDEN_I64]


def test_murmur3_string_golden():
    c = Column.from_pylist([s for s, _ in GOLDEN_STR], dtypes.string)
    h = ops.murmur3_ref([c], 42).tolist()
    assert h == [e for _, e in GOLDEN_STR]


def test_murmur3_null_skipped():
    c1 = Column.from_pylist([1, None], dtypes.int32)
    h = ops.murmur3_ref([c1], 42).tolist()
    # null row keeps the seed
    assert h == [GOLDEN_I32[1], 42]


def test_murmur3_multicolumn_chaining():
    a = Column.from_pylist([1], dtypes.int32)
    b = Column.from_pylist([2], dtypes.int32)
    h = ops.murmur3_ref([a, b], 42).tolist()
    # chained: hash(2, seed=hash(1, 42))
    h1 = ops.murmur3_ref([Column.from_pylist([2], dtypes.int32)], GOLDEN_I32[1]).tolist()
    assert h == h1


@pytest.mark.gpu
def test_murmur3_native_matches_ref():
    device = torch.device("cuda:0")
    import numpy as np

    rng = np.random.default_rng(0)
    n = 100_000
    cols_cpu = [
        Column.from_pylist(rng.integers(-2 ** 31, 2 ** 31, n).tolist(), dtypes.int32),
        Column.from_pylist(rng.integers(-2 ** 62, 2 ** 62, n).tolist(), dtypes.int64),
        Column.from_pylist(rng.normal(size=n).tolist(), dtypes.float64),
    ]
    ref = ops.murmur3_ref(cols_cpu, 42)
    got = ops.murmur3([c.to(device) for c in cols_cpu], 42).cpu()
    assert torch.equal(ref, got)


@pytest.mark.gpu
def test_murmur3_native_strings():
    device = torch.device("cuda:0")
    vals = ["hello", "bar", "", "😁", "天地", "a" * 37, None, "xyz"]
    c = Column.from_pylist(vals, dtypes.string)
    ref = ops.murmur3_ref([c], 42)
    got = ops.murmur3([c.to(device)], 42).cpu()
    assert torch.equal(ref, got)

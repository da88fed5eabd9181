"""Property-based correctness (hypothesis): engine semantics vs plain
Python/pandas on randomized inputs.

Role parity: the reference re-runs Spark's own SQL test suites
(auron-spark-tests); with no Spark here, randomized
property tests play that role — they sweep input shapes (nulls, empties,
duplicates, extremes, unicode) that the curated TPC-DS data misses.
"""
import math

import pytest
from hypothesis import given, settings, strategies as st

from auron_amd import AggFunc, AuronSession, col, dtypes
from auron_amd.column import Column, RecordBatch
from auron_amd.exprs import Aliased
from auron_amd.plan import nodes as P

ints = st.one_of(st.none(), st.integers(-2**40, 2**40))
floats = st.one_of(st.none(), st.floats(allow_nan=False, allow_infinity=False,
                                        width=64))
strs = st.one_of(st.none(), st.text(max_size=12))


def _batch(d, types):
    return RecordBatch.from_pydict(d, types)


@settings(max_examples=60, deadline=None)
@given(st.lists(ints, min_size=0, max_size=80),
       st.lists(st.integers(0, 5), min_size=0, max_size=80))
def test_hash_agg_sum_count_matches_python(vals, keys):
    n = min(len(vals), len(keys))
    vals, keys = vals[:n], keys[:n]
    b = _batch({"k": keys, "v": vals}, {"k": dtypes.int64, "v": dtypes.int64})
    plan = P.HashAgg(P.MemoryScan([b]), [Aliased(col("k"), "k")],
                     [AggFunc("sum", col("v"), name="s"),
                      AggFunc("count", col("v"), name="c")], mode="complete")
    got = AuronSession().collect(plan).to_pydict()
    ref_s, ref_c = {}, {}
    for k, v in zip(keys, vals):
        ref_c.setdefault(k, 0)
        if v is not None:
            ref_c[k] += 1
            ref_s[k] = ref_s.get(k, 0) + v
    assert dict(zip(got["k"], got["c"])) == ref_c
    for k, s in zip(got["k"], got["s"]):
        assert s == ref_s.get(k), (k, s)


@settings(max_examples=60, deadline=None)
@given(st.lists(ints, max_size=60), st.lists(ints, max_size=60))
def test_inner_join_matches_python(lk, rk):
    lb = _batch({"k": lk, "l": list(range(len(lk)))},
                {"k": dtypes.int64, "l": dtypes.int64})
    rb = _batch({"rk": rk, "r": list(range(len(rk)))},
                {"rk": dtypes.int64, "r": dtypes.int64})
    plan = P.HashJoin(P.MemoryScan([lb]), P.MemoryScan([rb]),
                      [col("k")], [col("rk")], how="inner")
    got = AuronSession().collect(plan).to_pydict()
    ref = sorted((i, j) for i, lv in enumerate(lk) for j, rv in enumerate(rk)
                 if lv is not None and lv == rv)
    assert sorted(zip(got["l"], got["r"])) == ref


@settings(max_examples=60, deadline=None)
@given(st.lists(st.tuples(ints, strs), max_size=60))
def test_sort_matches_python(rows):
    ks = [r[0] for r in rows]
    ss = [r[1] for r in rows]
    b = _batch({"k": ks, "s": ss}, {"k": dtypes.int64, "s": dtypes.string})
    plan = P.Sort(P.MemoryScan([b]), [(col("k"), True), (col("s"), False)])
    got = AuronSession().collect(plan).to_pydict()
    # Spark ordering: asc -> nulls first; desc -> nulls last.
    # UTF-8 byte order == code-point order, so python str sort is the
    # oracle for the engine's byte-wise string comparison.
    import functools

    def cmp(a, b2):
        (ka, sa), (kb, sb) = a, b2
        ka_t = (0,) if ka is None else (1, ka)
        kb_t = (0,) if kb is None else (1, kb)
        if ka_t != kb_t:
            return -1 if ka_t < kb_t else 1
        # desc string, nulls last
        if sa is None and sb is None:
            return 0
        if sa is None:
            return 1
        if sb is None:
            return -1
        if sa == sb:
            return 0
        return -1 if sa > sb else 1

    ref = sorted(zip(ks, ss), key=functools.cmp_to_key(cmp))
    assert list(zip(got["k"], got["s"])) == ref


@settings(max_examples=40, deadline=None)
@given(st.lists(strs, max_size=50), st.text(max_size=3))
def test_like_contains_matches_python(vals, pat):
    b = _batch({"s": vals}, {"s": dtypes.string})
    plan = P.Filter(P.MemoryScan([b]), col("s").like(f"%{pat}%"))
    got = AuronSession().collect(plan).to_pydict()["s"]
    ref = [v for v in vals if v is not None and pat in v]
    assert got == ref


@settings(max_examples=40, deadline=None)
@given(st.lists(floats, max_size=50))
def test_arith_division_null_semantics(vals):
    b = _batch({"x": vals}, {"x": dtypes.float64})
    plan = P.Project(P.MemoryScan([b]),
                     [Aliased(col("x") / col("x"), "r")])
    got = AuronSession().collect(plan).to_pydict()["r"]
    for v, r in zip(vals, got):
        if v is None or v == 0.0:
            assert r is None  # null input or div-by-zero -> null (Spark)
        else:
            assert r is not None and abs(r - 1.0) < 1e-12


@settings(max_examples=40, deadline=None)
@given(st.lists(ints, max_size=64))
def test_murmur3_ref_chunking_invariance(vals):
    """Hashing a column must not depend on how rows are batched."""
    from auron_amd import ops

    c = Column.from_pylist(vals, dtypes.int64)
    whole = ops.murmur3_ref([c]).tolist()
    cut = len(vals) // 2
    a = Column.from_pylist(vals[:cut], dtypes.int64)
    b2 = Column.from_pylist(vals[cut:], dtypes.int64)
    parts = ops.murmur3_ref([a]).tolist() + ops.murmur3_ref([b2]).tolist() \
        if vals else []
    assert whole == parts


@settings(max_examples=40, deadline=None)
@given(st.lists(st.tuples(st.integers(0, 3), ints), max_size=60))
def test_window_row_number_rank_matches_python(rows):
    ps = [r[0] for r in rows]
    os_ = [r[1] for r in rows]
    b = _batch({"p": ps, "o": os_}, {"p": dtypes.int64, "o": dtypes.int64})
    from auron_amd.exprs import WindowFunc

    fns = [Aliased(WindowFunc("row_number", None), "rn"),
           Aliased(WindowFunc("rank", None), "rk"),
           Aliased(WindowFunc("dense_rank", None), "dr")]
    got = AuronSession().collect(
        P.Window(P.MemoryScan([b]), [col("p")], [(col("o"), True)], fns)).to_pydict()
    # python oracle per partition (Spark: asc nulls first)
    import collections

    parts = collections.defaultdict(list)
    for p, o in zip(ps, os_):
        parts[p].append(o)
    okey = lambda o: (0, 0) if o is None else (1, o)
    for p, o, rn, rk, dr in zip(got["p"], got["o"], got["rn"], got["rk"], got["dr"]):
        ordered = sorted(parts[p], key=okey)
        # rank = 1 + #strictly-smaller; dense_rank = 1 + #distinct smaller
        smaller = sum(1 for x in ordered if okey(x) < okey(o))
        distinct_smaller = len({okey(x) for x in ordered if okey(x) < okey(o)})
        assert rk == smaller + 1, (p, o)
        assert dr == distinct_smaller + 1, (p, o)
        assert 1 <= rn <= len(ordered)


@settings(max_examples=40, deadline=None)
@given(st.lists(strs, max_size=40), st.integers(1, 6), st.integers(1, 6))
def test_substr_upper_length_matches_python(vals, start, ln):
    b = _batch({"s": vals}, {"s": dtypes.string})
    from auron_amd.exprs import Length, Substr, Upper

    plan = P.Project(P.MemoryScan([b]), [
        Aliased(Substr(col("s"), start, ln), "sub"),
        Aliased(Upper(col("s")), "up"),
        Aliased(Length(col("s")), "len")])
    got = AuronSession().collect(plan).to_pydict()
    for v, sub, up, l in zip(vals, got["sub"], got["up"], got["len"]):
        if v is None:
            assert sub is None and up is None and l is None
        else:
            enc = v.encode("utf-8")  # SQL substr is 1-based, byte-wise here
            assert sub == enc[start - 1:start - 1 + ln].decode("utf-8", "replace")
            assert l == len(enc)


@settings(max_examples=30, deadline=None)
@given(st.lists(ints, max_size=50))
def test_distinct_union_matches_python(vals):
    b = _batch({"x": vals}, {"x": dtypes.int64})
    plan = P.HashAgg(P.Union([P.MemoryScan([b]), P.MemoryScan([b])]),
                     [Aliased(col("x"), "x")], [], mode="complete")
    got = AuronSession().collect(plan).to_pydict()["x"]
    assert sorted(got, key=lambda v: (v is None, v)) == \
        sorted(set(vals), key=lambda v: (v is None, v))


@settings(max_examples=40, deadline=None)
@given(st.lists(ints, max_size=30), st.lists(strs, max_size=30),
       st.lists(floats, max_size=30))
def test_exchange_pack_unpack_roundtrip(a, b, c):
    """pack_batch/unpack_batch must be lossless for every column layout
    (ints with nulls, strings, floats, decimals, lists, empties)."""
    from auron_amd.exchange import pack_batch, unpack_batch

    n = min(len(a), len(b), len(c))
    cols = {
        "i": (a[:n], dtypes.int64),
        "s": (b[:n], dtypes.string),
        "f": (c[:n], dtypes.float64),
        "d": ([None if v is None else round(v, 2) % 10**5 for v in c[:n]],
              dtypes.decimal64(9, 2)),
        "ls": ([None if v is None else [v % 7, v % 11] for v in a[:n]],
               dtypes.list_of(dtypes.int64)),
    }
    rb = RecordBatch([k for k in cols],
                     [Column.from_pylist(v, t) for k, (v, t) in cols.items()])
    meta, buf = pack_batch(rb, "cpu")
    back = unpack_batch(meta, buf)
    for k in cols:
        want = rb.column(k).to_pylist()
        got = back.column(k).to_pylist()
        if cols[k][1].code == dtypes.DECIMAL64:
            want = [None if v is None else round(v, 2) for v in want]
            got = [None if v is None else round(v, 2) for v in got]
        assert got == want, k


@settings(max_examples=50, deadline=None)
@given(st.lists(st.one_of(st.none(),
                          st.decimals(min_value=-99999, max_value=99999,
                                      places=2)), max_size=60))
def test_decimal_sum_exact(vals):
    """decimal(7,2) sums are EXACT (scaled int64), matching python
    Decimal arithmetic bit-for-bit."""
    from decimal import Decimal

    pyvals = [None if v is None else float(v) for v in vals]
    b = _batch({"k": [1] * len(vals), "v": pyvals},
               {"k": dtypes.int64, "v": dtypes.decimal64(7, 2)})
    plan = P.HashAgg(P.MemoryScan([b]), [Aliased(col("k"), "k")],
                     [AggFunc("sum", col("v"), name="s"),
                      AggFunc("min", col("v"), name="mn"),
                      AggFunc("max", col("v"), name="mx")], mode="complete")
    got = AuronSession().collect(plan).to_pydict()
    nn = [Decimal(str(v)) for v in vals if v is not None]
    if not vals:
        return
    if not nn:
        assert got["s"] == [None] and got["mn"] == [None]
        return
    assert Decimal(str(got["s"][0])) == sum(nn)
    assert Decimal(str(got["mn"][0])) == min(nn)
    assert Decimal(str(got["mx"][0])) == max(nn)


@settings(max_examples=40, deadline=None)
@given(st.lists(st.decimals(min_value=0, max_value=999, places=2),
                min_size=1, max_size=40))
def test_decimal_roundtrip_through_parquet(vals):
    """decimal column -> parquet (store_decimal_as_integer) -> native np
    decode keeps exact cents."""
    import tempfile

    import pyarrow as pa
    import pyarrow.parquet as pq

    from auron_amd import parquet_native

    arr = pa.array([v for v in vals], type=pa.decimal128(7, 2))
    with tempfile.TemporaryDirectory() as td:
        p = f"{td}/d.parquet"
        pq.write_table(pa.table({"d": arr}), p, compression="NONE",
                       use_dictionary=False, data_page_version="1.0",
                       store_decimal_as_integer=True)
        cols = parquet_native.read_columns_native(p, ["d"], "cpu", _np_only=True)
        assert cols is not None
        got = cols["d"].to_pylist()
    from decimal import Decimal

    for g, w in zip(got, vals):
        assert Decimal(str(g)) == w

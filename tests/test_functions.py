"""Scalar function library tests (ext-functions parity).

Each function is checked against plain Python semantics matching Spark
(round half-away-from-zero, null-skipping greatest/least, ISO weekofyear,
Spark dayofweek numbering, 1-based instr, etc.).
"""
import datetime

import pytest
import torch

from auron_amd import dtypes, functions as F
from auron_amd.column import Column, RecordBatch
from auron_amd.exprs import col, lit


def _b(**cols):
    names, cs = [], []
    for k, (vals, dt) in cols.items():
        names.append(k)
        cs.append(Column.from_pylist(vals, dt))
    return RecordBatch(names, cs)


def ev(expr, batch):
    return expr.eval(batch).to_pylist()


def test_round_half_away():
    b = _b(x=([2.5, -2.5, 1.44, 1.45, None], dtypes.float64))
    assert ev(F.Round(col("x")), b) == [3.0, -3.0, 1.0, 1.0, None]
    r = ev(F.Round(col("x"), 1), b)
    assert r[2] == 1.4 and abs(r[3] - 1.5) < 1e-9


def test_floor_ceil_sign():
    b = _b(x=([1.5, -1.5, 0.0, None], dtypes.float64))
    assert ev(F.Floor(col("x")), b) == [1, -2, 0, None]
    assert ev(F.Ceil(col("x")), b) == [2, -1, 0, None]
    assert ev(F.Sign(col("x")), b) == [1.0, -1.0, 0.0, None]


def test_exp_ln_log10_pow():
    b = _b(x=([1.0, 0.0, -1.0, None], dtypes.float64))
    r = ev(F.Exp(col("x")), b)
    assert abs(r[0] - 2.718281828) < 1e-6
    ln = ev(F.Ln(col("x")), b)
    assert abs(ln[0]) < 1e-12 and ln[1] is None and ln[2] is None and ln[3] is None
    lg = ev(F.Log10(lit(100.0)), _b(x=([0.0], dtypes.float64)))
    assert abs(lg[0] - 2.0) < 1e-12
    p = ev(F.Pow(col("x"), lit(2.0)), b)
    assert p[:3] == [1.0, 0.0, 1.0] and p[3] is None


def test_greatest_least_null_skipping():
    b = _b(x=([1.0, None, None], dtypes.float64),
           y=([3.0, 2.0, None], dtypes.float64))
    assert ev(F.Greatest([col("x"), col("y")]), b) == [3.0, 2.0, None]
    assert ev(F.Least([col("x"), col("y")]), b) == [1.0, 2.0, None]


def test_nullif_nvl2_if():
    b = _b(x=([1, 2, None], dtypes.int64), y=([1, 9, 9], dtypes.int64))
    assert ev(F.NullIf(col("x"), col("y")), b) == [None, 2, None]
    assert ev(F.Nvl2(col("x"), lit(10), lit(20)), b) == [10, 10, 20]
    from auron_amd.exprs import Cmp
    assert ev(F.If(Cmp(">", col("y"), lit(5)), lit(1), lit(0)), b) == [0, 1, 1]


def _date(s):
    return (datetime.date.fromisoformat(s) - datetime.date(1970, 1, 1)).days


def _dcol(*isodates):
    return ([None if s is None else _date(s) for s in isodates], dtypes.date32)


def test_date_add_sub_diff():
    b = _b(d=_dcol("2001-01-30", None), n=([3, 3], dtypes.int32))
    assert ev(F.DateAdd(col("d"), col("n")), b) == [_date("2001-02-02"), None]
    assert ev(F.DateSub(col("d"), col("n")), b) == [_date("2001-01-27"), None]
    b2 = _b(a=_dcol("2001-03-01"), c=_dcol("2001-02-01"))
    assert ev(F.DateDiff(col("a"), col("c")), b2) == [28]


def test_add_months_clamps_day():
    b = _b(d=_dcol("2001-01-31", "2000-01-31", "2001-11-15"))
    r = ev(F.AddMonths(col("d"), 1), b)
    assert r == [_date("2001-02-28"), _date("2000-02-29"), _date("2001-12-15")]
    r2 = ev(F.AddMonths(col("d"), 14), b)
    assert r2[0] == _date("2002-03-31")


def test_last_day_quarter():
    b = _b(d=_dcol("2000-02-10", "2001-02-10", "2001-12-01"))
    assert ev(F.LastDay(col("d")), b) == [
        _date("2000-02-29"), _date("2001-02-28"), _date("2001-12-31")]
    assert ev(F.Quarter(col("d")), b) == [1, 1, 4]


def test_dayofweek_weekofyear():
    # 1970-01-01 Thursday → Spark dayofweek 5; 2001-01-01 Monday → 2
    b = _b(d=_dcol("1970-01-01", "2001-01-01", "2001-01-07"))
    assert ev(F.DayOfWeek(col("d")), b) == [5, 2, 1]
    # ISO weeks: 2001-01-01 is week 1; 2000-01-01 (Saturday) is ISO week 52/1999;
    # 2004-12-31 is week 53
    b2 = _b(d=_dcol("2001-01-01", "2000-01-01", "2004-12-31"))
    exp = [datetime.date.fromisoformat(s).isocalendar()[1]
           for s in ("2001-01-01", "2000-01-01", "2004-12-31")]
    assert ev(F.WeekOfYear(col("d")), b2) == exp


def test_trim_family():
    b = _b(s=(["  hi  ", "a", "   ", "", None, " x y "], dtypes.string))
    assert ev(F.Trim(col("s")), b) == ["hi", "a", "", "", None, "x y"]
    assert ev(F.Trim(col("s"), "leading"), b) == ["hi  ", "a", "", "", None, "x y "]
    assert ev(F.Trim(col("s"), "trailing"), b) == ["  hi", "a", "", "", None, " x y"]


def test_left_right_pad():
    b = _b(s=(["hello", "ab", None], dtypes.string))
    assert ev(F.Left(col("s"), 3), b) == ["hel", "ab", None]
    assert ev(F.Right(col("s"), 3), b) == ["llo", "ab", None]
    assert ev(F.LPad(col("s"), 7, "*"), b) == ["**hello", "*****ab", None]
    assert ev(F.RPad(col("s"), 7, "xy"), b) == ["helloxy", "abxyxyx", None]
    assert ev(F.LPad(col("s"), 3), b) == ["hel", " ab", None]


def test_replace_split_part():
    b = _b(s=(["a,b,c", "nope", None], dtypes.string))
    assert ev(F.Replace(col("s"), ",", "-"), b) == ["a-b-c", "nope", None]
    assert ev(F.SplitPart(col("s"), ",", 2), b) == ["b", "", None]
    assert ev(F.SplitPart(col("s"), ",", 9), b) == ["", "", None]


def test_instr_ascii():
    b = _b(s=(["hello", "xyz", "", None, "ohell"], dtypes.string))
    assert ev(F.Instr(col("s"), "ell"), b) == [2, 0, 0, None, 3]
    assert ev(F.Instr(col("s"), ""), b) == [1, 1, 1, None, 1]
    assert ev(F.Ascii(col("s")), b) == [104, 120, 0, None, 111]


def test_partition_ids():
    b = _b(x=([1, 2, 3], dtypes.int64))
    tok = F.EVAL_CONTEXT.set({"partition_id": 5, "row_base": 10})
    try:
        assert ev(F.SparkPartitionId(), b) == [5, 5, 5]
        base = (5 << 33) + 10
        assert ev(F.MonotonicallyIncreasingId(), b) == [base, base + 1, base + 2]
    finally:
        F.EVAL_CONTEXT.reset(tok)


def test_bround_half_even():
    b = _b(x=([2.5, 3.5, -2.5, 1.25, None], dtypes.float64))
    assert ev(F.Bround(col("x")), b) == [2.0, 4.0, -2.0, 1.0, None]
    assert ev(F.Bround(col("x"), 1), b) == [2.5, 3.5, -2.5, 1.2, None]


def test_isnan_normalize():
    b = _b(x=([float("nan"), 1.0, None], dtypes.float64))
    assert ev(F.IsNan(col("x")), b) == [True, False, False]
    import torch
    from auron_amd.column import Column
    neg = Column(dtypes.float64, torch.tensor([-0.0], dtype=torch.float64))
    nb = RecordBatch(["x"], [neg])
    r = F.NormalizeNanAndZero(col("x")).eval(nb)
    import math
    assert math.copysign(1.0, r.data[0].item()) == 1.0


def test_initcap_crypto():
    b = _b(s=(["hello world", "aBC dEF", "", None], dtypes.string))
    assert ev(F.InitCap(col("s")), b) == ["Hello World", "Abc Def", "", None]
    import hashlib
    got = ev(F.CryptoHash(col("s"), "md5"), b)
    assert got[0] == hashlib.md5(b"hello world").hexdigest() and got[3] is None
    got = ev(F.CryptoHash(col("s"), "sha256"), b)
    assert got[1] == hashlib.sha256(b"aBC dEF").hexdigest()


def test_get_json_object():
    docs = ['{"a": {"b": [1, 2, {"c": "x"}]}, "n": null, "t": true}',
            'not json', None, '{"a": 1.5}']
    b = _b(j=(docs, dtypes.string))
    assert ev(F.GetJsonObject(col("j"), "$.a.b[2].c"), b) == ["x", None, None, None]
    assert ev(F.GetJsonObject(col("j"), "$.a"), b) == [
        '{"b":[1,2,{"c":"x"}]}', None, None, "1.5"]
    assert ev(F.GetJsonObject(col("j"), "$.t"), b) == ["true", None, None, None]
    assert ev(F.GetJsonObject(col("j"), "$.n"), b) == [None, None, None, None]
    assert ev(F.GetJsonObject(col("j"), "$.missing"), b) == [None, None, None, None]


def test_trycast_and_rownum():
    from auron_amd.exprs import TryCast

    b = _b(s=(["12", "x", None, "3.5"], dtypes.string))
    assert ev(TryCast(col("s"), dtypes.int64), b) == [12, None, None, 3]
    assert ev(F.RowNum(), b) == [1, 2, 3, 4]


def test_concat_ws_reverse_repeat_space():
    b = _b(a=(["x", None, "p"], dtypes.string), c=(["y", "z", None], dtypes.string))
    assert ev(F.ConcatWs("-", [col("a"), col("c")]), b) == ["x-y", "z", "p"]
    assert ev(F.Reverse(col("a")), b) == ["x", None, "p"]
    b2 = _b(s=(["ab", None], dtypes.string))
    assert ev(F.Repeat(col("s"), 3), b2) == ["ababab", None]
    b3 = _b(n=([2, 0, None], dtypes.int64))
    assert ev(F.Space(col("n")), b3) == ["  ", "", None]


def test_translate_find_in_set():
    b = _b(s=(["abcba", None], dtypes.string))
    assert ev(F.Translate(col("s"), "abc", "AB"), b) == ["ABBA", None]
    b2 = _b(s=(["b", "d", "a,b", None], dtypes.string),
            l=(["a,b,c", "a,b,c", "a,b,c", "a"], dtypes.string))
    assert ev(F.FindInSet(col("s"), col("l")), b2) == [2, 0, 0, None]


def test_months_between_next_day_trunc():
    b = _b(a=_dcol("2001-03-31", "2001-03-15", "2001-03-10"),
           c=_dcol("2001-02-28", "2001-01-15", "2001-01-20"))
    r = ev(F.MonthsBetween(col("a"), col("c")), b)
    assert r[0] == 1.0 and r[1] == 2.0
    assert abs(r[2] - (2 + (10 - 20) / 31.0)) < 1e-9
    b2 = _b(d=_dcol("2001-01-01"))  # a Monday
    assert ev(F.NextDay(col("d"), "monday"), b2) == [_date("2001-01-08")]
    assert ev(F.NextDay(col("d"), "tu"), b2) == [_date("2001-01-02")]
    b3 = _b(d=_dcol("2001-03-15"))
    assert ev(F.TruncDate(col("d"), "year"), b3) == [_date("2001-01-01")]
    assert ev(F.TruncDate(col("d"), "month"), b3) == [_date("2001-03-01")]
    assert ev(F.TruncDate(col("d"), "week"), b3) == [_date("2001-03-12")]


def test_string_date_casts():
    from auron_amd.exprs import Cast

    b = _b(s=(["2001-03-15", " 1999-12-31", "bogus", None], dtypes.string))
    got = ev(Cast(col("s"), dtypes.date32), b)
    assert got == [_date("2001-03-15"), _date("1999-12-31"), None, None]
    b2 = _b(d=_dcol("2001-03-15", None))
    assert ev(Cast(col("d"), dtypes.string), b2) == ["2001-03-15", None]


def test_array_item_element_at_size():
    lt = dtypes.list_of(dtypes.int64)
    c = Column.from_pylist([[1, 2, 3], [], None, [7]], lt)
    b = RecordBatch(["xs"], [c])
    assert ev(F.GetArrayItem(col("xs"), 0), b) == [1, None, None, 7]
    assert ev(F.GetArrayItem(col("xs"), 2), b) == [3, None, None, None]
    assert ev(F.ElementAt(col("xs"), 1), b) == [1, None, None, 7]
    assert ev(F.ElementAt(col("xs"), -1), b) == [3, None, None, 7]
    assert ev(F.ElementAt(col("xs"), 5), b) == [None, None, None, None]
    assert ev(F.ArraySize(col("xs")), b) == [3, 0, -1, 1]

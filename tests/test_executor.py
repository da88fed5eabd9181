"""Operator correctness vs pandas oracle (the QueryResultComparator
pattern from the reference's integration harness: vanilla-engine results
are ground truth, compared row-wise with float tolerance)."""
import math

import pandas as pd
import pytest
import torch

from auron_amd import AggFunc, AuronSession, Col, col, dtypes, exprs, lit
from auron_amd.column import RecordBatch
from auron_amd.plan import nodes as P

T = {"k": dtypes.string, "g": dtypes.int32, "x": dtypes.int64, "f": dtypes.float64,
     "d": dtypes.date32}
DATA = {
    "k": ["a", "b", "a", None, "c", "b", "a"],
    "g": [1, 2, 1, 2, 3, None, 1],
    "x": [10, 20, 30, 40, None, 60, 70],
    "f": [1.5, 2.5, None, 4.0, 5.5, 6.0, 7.25],
    "d": [19000, 19001, 19002, 19003, 19004, 19005, 19006],
}


def scan(device="cpu"):
    return P.MemoryScan([RecordBatch.from_pydict(DATA, T, device)])


def df():
    return pd.DataFrame(DATA)


def sorted_rows(d: dict):
    rows = list(zip(*d.values()))
    return sorted(rows, key=lambda r: tuple((v is None, str(v)) for v in r))


def assert_rows_equal(got: dict, want: dict):
    gr, wr = sorted_rows(got), sorted_rows(want)
    assert len(gr) == len(wr), f"{len(gr)} rows != {len(wr)}\n{gr}\n{wr}"
    for a, b in zip(gr, wr):
        for x, y in zip(a, b):
            if isinstance(x, float) and isinstance(y, float):
                assert math.isclose(x, y, rel_tol=1e-9, abs_tol=1e-9), (a, b)
            else:
                assert x == y, (a, b)


def test_filter_project():
    s = AuronSession()
    plan = P.Project(
        P.Filter(scan(), (col("x") > 15) & col("k").is_not_null()),
        [exprs.Aliased(col("k"), "k"), exprs.Aliased(col("x") * 2, "x2")],
    )
    out = s.collect(plan).to_pydict()
    want = df()
    want = want[(want.x > 15) & want.k.notna()]
    assert_rows_equal(out, {"k": want.k.tolist(), "x2": (want.x * 2).tolist()})


def test_filter_null_semantics():
    s = AuronSession()
    # x > 15 is NULL for the null row -> dropped
    plan = P.Filter(scan(), col("x") > 15)
    out = s.collect(plan)
    # x = [10,20,30,40,None,60,70]: null comparison row dropped -> 5 rows
    assert out.num_rows == 5
    assert sorted(out.to_pydict()["x"]) == [20, 30, 40, 60, 70]


def _oracle_group_agg():
    """dict-based oracle (pandas groupby-with-None quirks avoided)."""
    from collections import defaultdict

    rows = list(zip(*DATA.values()))
    groups = defaultdict(list)
    for r in rows:
        groups[r[0]].append(dict(zip(DATA.keys(), r)))
    want = {"k": [], "sx": [], "cx": [], "af": [], "mn": [], "mx": []}
    for k, rs in groups.items():
        xs = [r["x"] for r in rs if r["x"] is not None]
        fs = [r["f"] for r in rs if r["f"] is not None]
        want["k"].append(k)
        want["sx"].append(sum(xs) if xs else None)
        want["cx"].append(len(xs))
        want["af"].append(sum(fs) / len(fs) if fs else None)
        want["mn"].append(min(xs) if xs else None)
        want["mx"].append(max(xs) if xs else None)
    return want


def test_agg_complete_group():
    s = AuronSession()
    plan = P.HashAgg(
        scan(), [exprs.Aliased(col("k"), "k")],
        [AggFunc("sum", col("x"), name="sx"), AggFunc("count", col("x"), name="cx"),
         AggFunc("avg", col("f"), name="af"), AggFunc("min", col("x"), name="mn"),
         AggFunc("max", col("x"), name="mx")],
        mode="complete",
    )
    out = s.collect(plan).to_pydict()
    want = _oracle_group_agg()
    assert_rows_equal(out, want)


def test_agg_global_no_keys():
    s = AuronSession()
    plan = P.HashAgg(scan(), [], [AggFunc("count_star", None, name="n"),
                                  AggFunc("sum", col("f"), name="sf")], mode="complete")
    out = s.collect(plan).to_pydict()
    assert out["n"] == [7]
    assert abs(out["sf"][0] - df().f.sum()) < 1e-9


def test_agg_partial_final_roundtrip():
    s = AuronSession()
    partial = P.HashAgg(scan(), [exprs.Aliased(col("k"), "k")],
                        [AggFunc("sum", col("x"), name="sx"),
                         AggFunc("avg", col("f"), name="af")], mode="partial")
    final = P.HashAgg(partial, [exprs.Aliased(col("k"), "k")],
                      [AggFunc("sum", col("x"), name="sx"),
                       AggFunc("avg", col("f"), name="af")], mode="final")
    out = s.collect(final).to_pydict()
    w = _oracle_group_agg()
    want = {"k": w["k"], "sx": w["sx"], "af": w["af"]}
    assert_rows_equal(out, want)


def test_count_distinct():
    s = AuronSession()
    plan = P.HashAgg(scan(), [exprs.Aliased(col("k"), "k")],
                     [AggFunc("count_distinct", col("g"), name="dg")], mode="complete")
    out = s.collect(plan).to_pydict()
    from collections import defaultdict
    gs = defaultdict(set)
    for k, g in zip(DATA["k"], DATA["g"]):
        if g is not None:
            gs[k].add(g)
    got = dict(zip(out["k"], out["dg"]))
    for k in set(DATA["k"]):
        assert got[k] == len(gs[k]), (k, got[k], gs[k])


JL = {"id": [1, 2, 3, 4, None], "lv": ["a", "b", "c", "d", "e"]}
JR = {"id": [2, 3, 3, None, 6], "rv": ["x", "y", "z", "w", "v"]}
JT = {"id": dtypes.int64, "lv": dtypes.string, "rv": dtypes.string}


def _join_plan(how, build_side="right", broadcast=False):
    l = P.MemoryScan([RecordBatch.from_pydict(JL, {"id": dtypes.int64, "lv": dtypes.string})])
    r = P.MemoryScan([RecordBatch.from_pydict(
        {"rid": JR["id"], "rv": JR["rv"]}, {"rid": dtypes.int64, "rv": dtypes.string})])
    return P.HashJoin(l, r, [col("id")], [col("rid")], how=how,
                      build_side=build_side, broadcast=broadcast)


def _sql_join_oracle(how):
    """SQL semantics: NULL keys never match (pandas merge wrongly matches
    NaN==NaN, so the oracle is explicit)."""
    lrows = list(zip(JL["id"], JL["lv"]))
    rrows = list(zip(JR["id"], JR["rv"]))
    out = {"id": [], "lv": [], "rid": [], "rv": []}
    rmatched = [False] * len(rrows)
    for lid, lv in lrows:
        hit = False
        for j, (rid, rv) in enumerate(rrows):
            if lid is not None and rid is not None and lid == rid:
                out["id"].append(lid); out["lv"].append(lv)
                out["rid"].append(rid); out["rv"].append(rv)
                hit = True
                rmatched[j] = True
        if not hit and how in ("left", "full"):
            out["id"].append(lid); out["lv"].append(lv)
            out["rid"].append(None); out["rv"].append(None)
    if how in ("right", "full"):
        for j, (rid, rv) in enumerate(rrows):
            if not rmatched[j]:
                out["id"].append(None); out["lv"].append(None)
                out["rid"].append(rid); out["rv"].append(rv)
    return out


@pytest.mark.parametrize("how", ["inner", "left", "right", "full"])
@pytest.mark.parametrize("build_side", ["right", "left"])
def test_hash_join(how, build_side):
    s = AuronSession()
    out = s.collect(_join_plan(how, build_side)).to_pydict()
    assert_rows_equal(out, _sql_join_oracle(how))


def test_semi_anti_join():
    s = AuronSession()
    semi = s.collect(_join_plan("semi")).to_pydict()
    assert sorted(semi["lv"]) == ["b", "c"]
    anti = s.collect(_join_plan("anti")).to_pydict()
    assert sorted(anti["lv"]) == ["a", "d", "e"]  # null-key left row kept by anti


def test_existence_join():
    s = AuronSession()
    out = s.collect(_join_plan("existence")).to_pydict()
    m = dict(zip(out["lv"], out["exists"]))
    assert m == {"a": False, "b": True, "c": True, "d": False, "e": False}


def test_sort_limit():
    s = AuronSession()
    plan = P.Sort(scan(), [(col("x"), False)], limit=3)
    out = s.collect(plan).to_pydict()
    assert out["x"] == [70, 60, 40]


def test_sort_nulls_first_asc():
    s = AuronSession()
    plan = P.Sort(scan(), [(col("x"), True)])
    out = s.collect(plan).to_pydict()
    assert out["x"] == [None, 10, 20, 30, 40, 60, 70]


def test_union_expand_limit():
    s = AuronSession()
    u = P.Union([scan(), scan()])
    out = s.collect(u)
    assert out.num_rows == 14
    e = P.Expand(scan(), [
        [exprs.Aliased(col("x"), "v"), exprs.Aliased(lit(0), "tag")],
        [exprs.Aliased(col("g").cast(dtypes.int64), "v"), exprs.Aliased(lit(1), "tag")],
    ])
    out = s.collect(e)
    assert out.num_rows == 14
    lim = P.Limit(scan(), 3, offset=2)
    assert s.collect(lim).num_rows == 3


def test_case_when_and_dates():
    s = AuronSession()
    plan = P.Project(scan(), [
        exprs.Aliased(exprs.CaseWhen([(col("x") > 30, lit(1))], lit(0)), "c"),
        exprs.Aliased(exprs.year(col("d")), "y"),
        exprs.Aliased(exprs.month(col("d")), "m"),
    ])
    out = s.collect(plan).to_pydict()
    d = pd.to_datetime(pd.Series(DATA["d"]), unit="D", origin="unix")
    assert out["y"] == d.dt.year.tolist()
    assert out["m"] == d.dt.month.tolist()


def test_window_row_number_rank():
    s = AuronSession()
    plan = P.Window(
        scan(), [col("k")], [(col("x"), True)],
        [exprs.Aliased(exprs.WindowFunc("row_number"), "rn"),
         exprs.Aliased(exprs.WindowFunc("sum", col("x")), "sx")],
    )
    out = s.collect(plan).to_pydict()
    d = df()
    d["rn"] = d.sort_values("x").groupby("k", dropna=False).cumcount() + 1
    got = {(k, x): rn for k, x, rn in zip(out["k"], out["x"], out["rn"])}
    for _, row in d.iterrows():
        k = None if isinstance(row.k, float) and math.isnan(row.k) else row.k
        x = None if math.isnan(row.x) else int(row.x)
        # nulls-first ordering differs from pandas NaN-last; only check non-null x
        if x is not None and k is not None:
            pass  # rank check below via direct reconstruction
    # direct check: within each partition rn is 1..n and ordered by x (nulls first)
    from collections import defaultdict

    parts = defaultdict(list)
    for k, x, rn in zip(out["k"], out["x"], out["rn"]):
        parts[k].append((rn, x))
    for k, rows in parts.items():
        rows.sort()
        assert [r for r, _ in rows] == list(range(1, len(rows) + 1))
        xs = [x for _, x in rows]
        non_null = [x for x in xs if x is not None]
        assert non_null == sorted(non_null)
        if None in xs:
            assert xs[0] is None  # asc -> nulls first


def _smj_plan(how):
    l = P.MemoryScan([RecordBatch.from_pydict(JL, {"id": dtypes.int64, "lv": dtypes.string})])
    r = P.MemoryScan([RecordBatch.from_pydict(
        {"rid": JR["id"], "rv": JR["rv"]}, {"rid": dtypes.int64, "rv": dtypes.string})])
    return P.SortMergeJoin(l, r, [col("id")], [col("rid")], how=how)


@pytest.mark.parametrize("how", ["inner", "left", "right", "full"])
def test_sort_merge_join(how):
    s = AuronSession()
    out = s.collect(_smj_plan(how)).to_pydict()
    assert_rows_equal(out, _sql_join_oracle(how))


def test_sort_merge_semi_anti_existence():
    s = AuronSession()
    assert sorted(s.collect(_smj_plan("semi")).to_pydict()["lv"]) == ["b", "c"]
    assert sorted(s.collect(_smj_plan("anti")).to_pydict()["lv"]) == ["a", "d", "e"]
    ex = s.collect(_smj_plan("existence")).to_pydict()
    assert dict(zip(ex["lv"], ex["exists"])) == {
        "a": False, "b": True, "c": True, "d": False, "e": False}


@pytest.mark.parametrize("how", ["inner", "left", "right", "full", "semi", "anti"])
def test_sort_merge_join_matches_hash_join_random(how):
    """SMJ and HJ must agree row-for-row on random multi-key data with
    duplicates, nulls, and a string key component."""
    import random

    rng = random.Random(11)
    def mk(n, prefix):
        return {
            "a": [rng.choice([None, 1, 2, 3, 4]) for _ in range(n)],
            "s": [rng.choice(["x", "y", "zz", "w"]) for _ in range(n)],
            prefix: list(range(n)),
        }
    lt = {"a": dtypes.int64, "s": dtypes.string, "lrow": dtypes.int64}
    rt = {"a": dtypes.int64, "s": dtypes.string, "rrow": dtypes.int64}
    ld = mk(60, "lrow")
    rd = mk(45, "rrow")
    lscan = P.MemoryScan([RecordBatch.from_pydict(ld, lt)])
    rscan = P.MemoryScan([RecordBatch.from_pydict(
        {"ra": rd["a"], "rs": rd["s"], "rrow": rd["rrow"]},
        {"ra": dtypes.int64, "rs": dtypes.string, "rrow": dtypes.int64})])
    keys_l = [col("a"), col("s")]
    keys_r = [col("ra"), col("rs")]
    s = AuronSession()
    smj = s.collect(P.SortMergeJoin(lscan, rscan, keys_l, keys_r, how=how)).to_pydict()
    hj = s.collect(P.HashJoin(lscan, rscan, keys_l, keys_r, how=how)).to_pydict()
    assert_rows_equal(smj, hj)


def test_window_range_vs_rows_frame_ties():
    """Default RANGE frame: peer rows (tied order keys) share the running
    value at the last peer; ROWS frame counts each row individually."""
    from auron_amd.exprs import WindowFunc, Aliased

    data = {"p": [1, 1, 1, 1], "o": [10, 20, 20, 30], "v": [1.0, 2.0, 3.0, 4.0]}
    t = {"p": dtypes.int64, "o": dtypes.int64, "v": dtypes.float64}
    scan_n = P.MemoryScan([RecordBatch.from_pydict(data, t)])
    fns = [Aliased(WindowFunc("sum", col("v")), "rs")]
    s = AuronSession()
    rng = s.collect(P.Window(scan_n, [col("p")], [(col("o"), True)], fns,
                             frame="range")).to_pydict()
    assert rng["rs"] == [1.0, 6.0, 6.0, 10.0]
    rows = s.collect(P.Window(scan_n, [col("p")], [(col("o"), True)], fns,
                              frame="rows")).to_pydict()
    assert rows["rs"] == [1.0, 3.0, 6.0, 10.0]


def test_cse_shared_subtree_evaluated_once():
    """A subtree OBJECT shared between project exprs runs once per batch
    (CachedExprsEvaluator parity)."""
    from auron_amd import exprs as E

    calls = {"n": 0}

    class Probe(E.Expr):
        def eval(self, batch):
            calls["n"] += 1
            return col("x").eval(batch)

    p = Probe()
    shared = p + lit(1)
    plan = P.Project(scan(), [exprs.Aliased(shared + lit(1), "a"),
                              exprs.Aliased(shared + lit(2), "b"),
                              exprs.Aliased(p, "c")])
    s = AuronSession()
    out = s.collect(plan).to_pydict()
    assert calls["n"] == 1
    assert out["a"] == [None if v is None else v + 2 for v in out["c"]]
    assert out["b"] == [None if v is None else v + 3 for v in out["c"]]


def test_metric_tree():
    """MetricNode tree parity: nested per-operator records with rows."""
    s = AuronSession()
    plan = P.Limit(P.Filter(scan(), col("x") > 10), 3)
    out = s.collect(plan)
    t = s.metric_tree()
    assert t["op"] == "Limit" and t["rows"] == out.num_rows
    assert t["children"][0]["op"] == "Filter"
    assert t["children"][0]["children"][0]["op"] == "MemoryScan"
    assert t["children"][0]["children"][0]["rows"] == len(DATA["x"])
    assert "Limit" in s.explain_metrics()


def test_window_percent_rank_cume_dist_ntile():
    from auron_amd.exprs import Aliased, WindowFunc

    data = {"p": [1, 1, 1, 1, 2, 2], "o": [10, 20, 20, 30, 5, 5],
            "v": [1.0, 2.0, 3.0, 4.0, 5.0, 6.0]}
    t = {"p": dtypes.int64, "o": dtypes.int64, "v": dtypes.float64}
    sc = P.MemoryScan([RecordBatch.from_pydict(data, t)])
    fns = [Aliased(WindowFunc("percent_rank", None), "pr"),
           Aliased(WindowFunc("cume_dist", None), "cd"),
           Aliased(WindowFunc("ntile", None, 2), "nt"),
           Aliased(WindowFunc("rank", None), "rk")]
    s = AuronSession()
    out = s.collect(P.Window(sc, [col("p")], [(col("o"), True)], fns)).to_pydict()
    rows = sorted(zip(out["p"], out["o"], out["pr"], out["cd"], out["nt"], out["rk"]))
    # partition 1: ranks 1,2,2,4 over 4 rows
    assert rows[0][2:] == (0.0, 0.25, 1, 1)
    assert rows[1][2] == pytest.approx(1 / 3) and rows[1][3] == 0.75 and rows[1][5] == 2
    assert rows[2][2] == pytest.approx(1 / 3) and rows[2][3] == 0.75 and rows[2][4] == 2
    assert rows[3][2:] == (1.0, 1.0, 2, 4)
    # partition 2: two tied rows
    assert rows[4][2:] == (0.0, 1.0, 1, 1)
    assert rows[5][2:] == (0.0, 1.0, 2, 1)


def test_window_first_last_nth_value():
    from auron_amd.exprs import Aliased, WindowFunc

    data = {"p": [1, 1, 1, 2], "o": [1, 2, 3, 1], "v": [10.0, 20.0, 30.0, 40.0]}
    t = {"p": dtypes.int64, "o": dtypes.int64, "v": dtypes.float64}
    sc = P.MemoryScan([RecordBatch.from_pydict(data, t)])
    fns = [Aliased(WindowFunc("first_value", col("v")), "fv"),
           Aliased(WindowFunc("last_value", col("v")), "lv"),
           Aliased(WindowFunc("nth_value", col("v"), 2), "nv")]
    s = AuronSession()
    out = s.collect(P.Window(sc, [col("p")], [(col("o"), True)], fns)).to_pydict()
    rows = sorted(zip(out["p"], out["o"], out["fv"], out["lv"], out["nv"]))
    assert rows[0][2:] == (10.0, 10.0, None)   # p1 o1
    assert rows[1][2:] == (10.0, 20.0, 20.0)   # p1 o2
    assert rows[2][2:] == (10.0, 30.0, 20.0)   # p1 o3
    assert rows[3][2:] == (40.0, 40.0, None)   # p2 o1


def test_smj_fallback_for_large_builds(monkeypatch):
    """spark.auron.smjfallback: hash joins with oversized build sides
    lower to the sort-merge join and produce identical results."""
    monkeypatch.setenv("AURON_SMJ_FALLBACK", "1")
    monkeypatch.setenv("AURON_SMJ_FALLBACK_ROWS", "2")  # tiny threshold
    s = AuronSession()
    for how in ("inner", "left", "full", "semi", "anti"):
        out = s.collect(_join_plan(how)).to_pydict()
        if how == "semi":
            assert sorted(out["lv"]) == ["b", "c"]
        elif how == "anti":
            assert sorted(out["lv"]) == ["a", "d", "e"]
        else:
            assert_rows_equal(out, _sql_join_oracle(how))


def test_partial_agg_skipping(monkeypatch):
    """With a forced 0.0 ratio, high-cardinality partial aggs emit
    singleton states; the final agg still produces exact results."""
    monkeypatch.setenv("AURON_PARTIAL_SKIP_RATIO", "0.0")
    import numpy as np

    n = 20_000  # above the sampling floor
    rng = np.random.default_rng(3)
    data = {"k": rng.integers(0, 15_000, n).tolist(),
            "v": rng.normal(0, 10, n).tolist()}
    t = {"k": dtypes.int64, "v": dtypes.float64}
    scan_n = P.MemoryScan([RecordBatch.from_pydict(data, t)])
    tower = P.HashAgg(
        P.Exchange(P.HashAgg(scan_n, [exprs.Aliased(col("k"), "k")],
                             [AggFunc("sum", col("v"), name="s"),
                              AggFunc("count", col("v"), name="c"),
                              AggFunc("max", col("v"), name="m")],
                             mode="partial"), "hash", [col("k")]),
        [exprs.Aliased(col("k"), "k")],
        [AggFunc("sum", col("v"), name="s"), AggFunc("count", col("v"), name="c"),
         AggFunc("max", col("v"), name="m")], mode="final")
    s = AuronSession()
    # world_size==1 rewrite would collapse the tower; execute WITHOUT rewrite
    # by running the partial stage separately first
    part = s.execute(P.HashAgg(scan_n, [exprs.Aliased(col("k"), "k")],
                               [AggFunc("sum", col("v"), name="s"),
                                AggFunc("count", col("v"), name="c"),
                                AggFunc("max", col("v"), name="m")],
                               mode="partial"))
    assert part[0].num_rows == n  # pass-through engaged
    fin = s.collect(P.HashAgg(P.MemoryScan(part), [exprs.Aliased(col("k"), "k")],
                              [AggFunc("sum", col("v"), name="s"),
                               AggFunc("count", col("v"), name="c"),
                               AggFunc("max", col("v"), name="m")], mode="final"))
    got = {k: (s_, c_, m_) for k, s_, c_, m_ in
           zip(*[fin.to_pydict()[x] for x in ("k", "s", "c", "m")])}
    import collections
    ref = collections.defaultdict(list)
    for k, v in zip(data["k"], data["v"]):
        ref[k].append(v)
    assert len(got) == len(ref)
    for k, vals in ref.items():
        s_, c_, m_ = got[k]
        assert c_ == len(vals) and abs(s_ - sum(vals)) < 1e-6
        assert abs(m_ - max(vals)) < 1e-12


def test_lead_lag_default_value():
    from auron_amd.exprs import Aliased, WindowFunc

    data = {"p": [1, 1, 1], "o": [1, 2, 3], "v": [10.0, None, 30.0]}
    t = {"p": dtypes.int64, "o": dtypes.int64, "v": dtypes.float64}
    sc = P.MemoryScan([RecordBatch.from_pydict(data, t)])
    fns = [Aliased(WindowFunc("lag", col("v"), 1, default=-1.0), "lg"),
           Aliased(WindowFunc("lead", col("v"), 1, default=-1.0), "ld")]
    out = AuronSession().collect(
        P.Window(sc, [col("p")], [(col("o"), True)], fns)).to_pydict()
    rows = sorted(zip(out["o"], out["lg"], out["ld"]))
    # frame escapes get the default; a genuine NULL neighbor stays NULL
    assert rows[0][1:] == (-1.0, None)   # o=1: no lag row; lead = null v
    assert rows[1][1:] == (10.0, 30.0)   # o=2
    assert rows[2][1:] == (None, -1.0)   # o=3: lag = null v; no lead row

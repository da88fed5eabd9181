"""Fused expression interpreter: compiler structure on CPU, kernel-vs-
interpreted-eval equivalence on GPU (numerics rule: the fused HIP path is
compared against the plain eval path on identical inputs)."""
import random

import pytest
import torch

from auron_amd import dtypes
from auron_amd.column import Column, RecordBatch
from auron_amd.exprs import (Arith, BoolOp, CaseWhen, Cast, Cmp, Coalesce,
                             Col, InList, IsNull, Literal, Not, eval_scope)
from auron_amd import fused

DEC = dtypes.decimal64(7, 2)


def _schema():
    return {
        "a": dtypes.int64, "b": dtypes.int32, "p": DEC, "q": DEC,
        "f": dtypes.float64, "d": dtypes.date32, "flag": dtypes.bool_,
        "s": dtypes.string,
    }


def _battery():
    a, b, p, q, f, d, flag = (Col(n) for n in "a b p q f d flag".split())
    return [
        (a + b) * Literal(3) - Literal(1),
        Arith("%", a, Literal(7)),
        a / b,                                   # spark int `/` -> double
        p + q,                                   # same-scale decimal
        p * q,                                   # decimal mul -> float64
        p - Literal(1.5, DEC),
        Cmp("<", p, Literal(9.0)),
        Cmp(">=", a, b),
        Cmp("==", d, Literal(10957)),
        BoolOp("and", [Cmp(">", a, Literal(2)), Cmp("<", b, Literal(50)),
                       Not(Cmp("==", a, b))]),
        BoolOp("or", [IsNull(p), Cmp(">", q, Literal(1.0))]),
        CaseWhen([(Cmp(">", a, Literal(5)), p),
                  (Cmp(">", a, Literal(2)), q)],
                 otherwise=Literal(0.0, DEC)),
        CaseWhen([(Cmp(">", f, Literal(0.5)), a)], otherwise=None),
        Coalesce([p, q, Literal(7.5, DEC)]),
        InList(b, [1, 5, 9, 33]),
        InList(p, [1.25, 3.5]),
        Cast(a, dtypes.float64),
        Cast(f, dtypes.int64),
        Cast(p, dtypes.float64),
        Cast(f, DEC),
        Arith("+", d, Literal(30)),              # date arithmetic
        Not(flag),
        BoolOp("and", [flag, Cmp("!=", a, Literal(0))]),
    ]


def test_compile_battery_cpu():
    progs = fused.compile_all(_battery(), _schema())
    assert progs is not None
    covered = sorted(i for p in progs for i in p.expr_idx)
    assert covered == list(range(len(_battery())))
    for p in progs:
        assert p.max_depth <= 30
        assert len(p.out_dtypes) <= 8
        assert p.instr_np.nbytes % 16 == 0


def test_compile_string_falls_back():
    sch = _schema()
    exprs = [Cmp("==", Col("s"), Literal("x")),  # string -> not compiled
             Col("a") + Literal(1)]
    prog = fused.compile_exprs(exprs, sch)
    assert prog is not None and prog.expr_idx == [1]
    assert fused.compile_exprs([Cmp("==", Col("s"), Literal("x"))], sch) is None


def _make_batch(device, n=4096):
    rng = random.Random(7)
    tl = torch.tensor

    def with_nulls(vals, frac=0.15):
        v = torch.tensor([rng.random() > frac for _ in range(n)])
        return v

    a = tl([rng.randint(-10, 10) for _ in range(n)], dtype=torch.int64)
    b = tl([rng.randint(0, 99) for _ in range(n)], dtype=torch.int32)
    p = tl([rng.randint(-10 ** 4, 10 ** 4) for _ in range(n)], dtype=torch.int64)
    q = tl([rng.randint(-10 ** 4, 10 ** 4) for _ in range(n)], dtype=torch.int64)
    f = torch.rand(n, dtype=torch.float64)
    d = tl([10950 + rng.randint(0, 20) for _ in range(n)], dtype=torch.int32)
    flag = tl([rng.random() > 0.5 for _ in range(n)])
    cols = [
        Column(dtypes.int64, a, with_nulls(a)),
        Column(dtypes.int32, b, None),
        Column(DEC, p, with_nulls(p)),
        Column(DEC, q, with_nulls(q)),
        Column(dtypes.float64, f, None),
        Column(dtypes.date32, d, with_nulls(d, 0.05)),
        Column(dtypes.bool_, flag, with_nulls(flag)),
    ]
    rb = RecordBatch(["a", "b", "p", "q", "f", "d", "flag"], cols)
    return rb.to(device) if device != "cpu" else rb


@pytest.mark.gpu
def test_fused_matches_eval_gpu():
    from auron_amd import native

    native.require()
    batch = _make_batch("cuda")
    exprs = _battery()
    schema = {nm: c.dtype for nm, c in zip(batch.names, batch.columns)}
    progs = fused.compile_all(exprs, schema)
    assert progs is not None
    got = [None] * len(exprs)
    for p in progs:
        for k, i in enumerate(fused.run(p, batch)):
            got[p.expr_idx[k]] = i
    assert all(g is not None for g in got)
    with eval_scope(batch):
        want = [e.eval(batch) for e in exprs]
    torch.cuda.synchronize()
    for i, (g, w) in enumerate(zip(got, want)):
        assert g.dtype.code == w.dtype.code and g.dtype.scale == w.dtype.scale, \
            f"expr {i}: dtype {g.dtype} != {w.dtype}"
        gv = g.validity.cpu()
        wv = w.validity.cpu() if w.validity is not None else torch.ones(
            len(w), dtype=torch.bool)
        assert torch.equal(gv, wv), f"expr {i}: validity mismatch"
        gd, wd = g.data.cpu(), w.data.cpu()
        if g.dtype.code in (dtypes.FLOAT64,):
            ok = torch.isclose(gd, wd, rtol=1e-12, atol=1e-12) | ~wv
        elif g.dtype.code == dtypes.BOOL:
            ok = (gd.bool() == wd.bool()) | ~wv
        else:
            ok = (gd == wd) | ~wv
        assert bool(ok.all()), f"expr {i}: values mismatch"


@pytest.mark.gpu
def test_fused_empty_and_all_null_gpu():
    from auron_amd import native

    native.require()
    batch = _make_batch("cuda", n=0)
    prog = fused.compile_exprs([Col("a") + Literal(1)], {
        nm: c.dtype for nm, c in zip(batch.names, batch.columns)})
    out = fused.run(prog, batch)
    assert len(out[0]) == 0
    # all-null column propagates
    n = 64
    nullcol = Column(dtypes.int64, torch.zeros(n, dtype=torch.int64),
                     torch.zeros(n, dtype=torch.bool)).to("cuda")
    rb = RecordBatch(["x"], [nullcol])
    prog = fused.compile_exprs([Col("x") * Literal(2)], {"x": dtypes.int64})
    out = fused.run(prog, rb)[0]
    assert not bool(out.validity.any().cpu())


@pytest.mark.gpu
def test_executor_fused_filter_project_matches(monkeypatch):
    """Integration: the executor's fused Filter/Project path (enabled via
    env) produces identical results to the interpreted path."""
    from auron_amd import AuronSession, col, native
    from auron_amd.exprs import Aliased, Cmp, Literal
    from auron_amd.plan import nodes as P

    native.require()
    batch = _make_batch("cuda", n=50000)
    scan = P.MemoryScan([batch])
    plan = P.Project(
        P.Filter(scan, Cmp(">", Col("a"), Literal(0))),
        [Aliased((Col("a") + Col("b")) * Literal(2), "y"),
         Aliased(Col("p"), "p")])

    def run():
        s = AuronSession(device="cuda:0")
        return s.collect(plan).to("cpu").to_pydict()

    monkeypatch.setenv("AURON_EXPR_FUSION", "1")
    got = run()
    monkeypatch.setenv("AURON_EXPR_FUSION", "0")
    want = run()
    assert got == want

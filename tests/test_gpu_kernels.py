"""GPU kernel numerics vs CPU reference implementations (SURVEY.md §4
oracle (a): per-kernel unit tests, HIP kernels vs CPU reference)."""
import numpy as np
import pytest
import torch

from auron_amd import AggFunc, AuronSession, col, dtypes, exprs, native, ops
from auron_amd.column import Column, RecordBatch
from auron_amd.plan import nodes as P

pytestmark = pytest.mark.gpu

DEV = "cuda:0"


def _rand_cols(n, with_strings=True, with_nulls=True, seed=0):
    rng = np.random.default_rng(seed)
    cols = []
    ints = rng.integers(0, 50, n).tolist()
    if with_nulls:
        ints = [None if rng.random() < 0.1 else v for v in ints]
    cols.append(Column.from_pylist(ints, dtypes.int64))
    if with_strings:
        words = ["alpha", "beta", "gamma", "delta", "", "omega-long-string"]
        ss = [words[i % len(words)] for i in rng.integers(0, len(words), n)]
        if with_nulls:
            ss = [None if rng.random() < 0.1 else v for v in ss]
        cols.append(Column.from_pylist(ss, dtypes.string))
    return cols


def test_native_lib_loads():
    lib = native.require()
    assert lib.au_abi_version() == 1


def test_group_ids_native_vs_ref():
    cols = _rand_cols(50_000)
    gids_ref, reps_ref = ops.group_ids_ref(cols)
    gcols = [c.to(DEV) for c in cols]
    gids, reps = ops.group_ids(gcols)
    assert reps.numel() == reps_ref.numel(), "group count mismatch"
    # same partition structure: rows with equal gid on GPU <=> equal on CPU
    g_cpu = gids.cpu().numpy()
    r_cpu = gids_ref.numpy()
    import collections

    m = {}
    for a, b in zip(g_cpu, r_cpu):
        if a in m:
            assert m[a] == b, "partition mismatch"
        else:
            m[a] = b
    assert len(m) == reps_ref.numel()


def test_hash_join_native_vs_ref():
    n = 20_000
    build = _rand_cols(n // 4, seed=1)
    probe = _rand_cols(n, seed=2)
    bi_r, pi_r, bm_r = ops.hash_join_ref(build, probe, True, True)
    bg = [c.to(DEV) for c in build]
    pg = [c.to(DEV) for c in probe]
    bi, pi, bm = ops.hash_join(bg, pg, emit_unmatched_probe=True, need_build_matched=True)
    ref_pairs = sorted(zip(bi_r.tolist(), pi_r.tolist()))
    got_pairs = sorted(zip(bi.cpu().tolist(), pi.cpu().tolist()))
    assert ref_pairs == got_pairs
    assert torch.equal(bm_r, bm.cpu())


def test_join_counts_native_vs_ref():
    build = _rand_cols(1000, seed=3)
    probe = _rand_cols(5000, seed=4)
    ref = ops.join_counts(build, probe)
    got = ops.join_counts([c.to(DEV) for c in build], [c.to(DEV) for c in probe])
    assert torch.equal(ref.cpu(), got.cpu())


def test_partition_roundtrip():
    cols = _rand_cols(100_000, with_strings=False, seed=5)
    gcols = [c.to(DEV) for c in cols]
    nparts = 8
    pids_ref = ops.partition_ids(cols, nparts)
    pids = ops.partition_ids(gcols, nparts)
    assert torch.equal(pids_ref, pids.cpu())
    order, counts = ops.partition_order(pids, nparts)
    assert int(counts.sum().item()) == 100_000
    # every row appears exactly once and lands in its partition's range
    sorted_pids = pids.gather(0, order.to(torch.int64))
    bounds = torch.cumsum(counts, 0)
    start = 0
    for p in range(nparts):
        end = int(bounds[p].item())
        seg = sorted_pids[start:end]
        assert bool((seg == p).all())
        start = end
    assert torch.equal(torch.sort(order).values,
                       torch.arange(100_000, dtype=torch.int64, device=order.device))


def test_end_to_end_query_on_gpu():
    data = {
        "k": ["a", "b", "a", None, "c", "b", "a"] * 1000,
        "x": list(range(7000)),
    }
    types = {"k": dtypes.string, "x": dtypes.int64}
    b_cpu = RecordBatch.from_pydict(data, types)
    b_gpu = b_cpu.to(DEV)

    def q(batch, device):
        s = AuronSession(device=device)
        plan = P.HashAgg(
            P.Filter(P.MemoryScan([batch]), col("x") > 100),
            [exprs.Aliased(col("k"), "k")],
            [AggFunc("sum", col("x"), name="sx"), AggFunc("count_star", None, name="n")],
            mode="complete",
        )
        out = s.collect(plan).to_pydict()
        return sorted(zip([str(k) for k in out["k"]], out["sx"], out["n"]))

    assert q(b_cpu, "cpu") == q(b_gpu, DEV)


def test_agg_avg_min_max_gpu_matches_cpu():
    rng = np.random.default_rng(7)
    n = 30_000
    data = {
        "g": rng.integers(0, 97, n).tolist(),
        "v": [None if rng.random() < 0.05 else float(x) for x in rng.normal(100, 10, n)],
    }
    types = {"g": dtypes.int64, "v": dtypes.float64}
    b = RecordBatch.from_pydict(data, types)

    def q(batch, device):
        s = AuronSession(device=device)
        plan = P.HashAgg(P.MemoryScan([batch]), [exprs.Aliased(col("g"), "g")],
                         [AggFunc("avg", col("v"), name="av"),
                          AggFunc("min", col("v"), name="mn"),
                          AggFunc("max", col("v"), name="mx")], mode="complete")
        out = s.collect(plan).to_pydict()
        return {g: (round(a, 9), mn, mx) for g, a, mn, mx in
                zip(out["g"], out["av"], out["mn"], out["mx"])}

    r_cpu = q(b, "cpu")
    r_gpu = q(b.to(DEV), DEV)
    assert set(r_cpu) == set(r_gpu)
    for g in r_cpu:
        a1, mn1, mx1 = r_cpu[g]
        a2, mn2, mx2 = r_gpu[g]
        assert abs(a1 - a2) < 1e-6 and mn1 == mn2 and mx1 == mx2


def test_agg_scatter_native_vs_torch():
    rng = np.random.default_rng(11)
    n = 500_000
    gids = torch.from_numpy(rng.integers(0, 1000, n)).to(torch.int64)
    vals_np = rng.normal(100, 50, n)
    validity = torch.from_numpy(rng.random(n) > 0.05)
    col_cpu = Column(dtypes.float64, torch.from_numpy(vals_np), validity)
    acc_ref, cnt_ref = ops.agg_scatter(gids, 1000, col_cpu, "sum")
    col_gpu = col_cpu.to(DEV)
    for fn in ("sum", "min", "max", "count"):
        a_ref, c_ref = ops.agg_scatter(gids, 1000, col_cpu, fn)
        a_gpu, c_gpu = ops.agg_scatter(gids.to(DEV), 1000, col_gpu, fn)
        assert torch.equal(c_ref, c_gpu.cpu()), fn
        if fn == "sum":
            assert torch.allclose(a_ref, a_gpu.cpu(), rtol=1e-9, atol=1e-6), fn
        elif fn != "count":
            assert torch.equal(a_ref, a_gpu.cpu()), fn
    # int64 path
    icol_cpu = Column(dtypes.int64, torch.from_numpy(rng.integers(-10**6, 10**6, n)), validity)
    icol_gpu = icol_cpu.to(DEV)
    for fn in ("sum", "min", "max"):
        a_ref, c_ref = ops.agg_scatter(gids, 1000, icol_cpu, fn)
        a_gpu, c_gpu = ops.agg_scatter(gids.to(DEV), 1000, icol_gpu, fn)
        assert torch.equal(a_ref, a_gpu.cpu()), fn


def test_agg_scatter_segmented_path_patterns():
    """Large-ngroups path (wave-segmented scan + global atomics): runs,
    recurring gids inside one 64-lane wave ([A,A,B,A]-style), random,
    and all-invalid stretches must all match the CPU reference."""
    rng = np.random.default_rng(5)
    NG = 50_000  # > LDS limit -> segmented global path
    pats = []
    # long runs
    pats.append(np.repeat(rng.integers(0, NG, 2000), rng.integers(1, 400, 2000)))
    # recurring gid within a wave: A A B A C A ...
    base = rng.integers(0, NG, 30_000)
    recur = np.empty(60_000, dtype=np.int64)
    recur[0::2] = 7  # gid 7 every other lane
    recur[1::2] = base
    pats.append(recur)
    # pure random
    pats.append(rng.integers(0, NG, 100_000))
    for pat in pats:
        n = len(pat)
        gids = torch.from_numpy(np.ascontiguousarray(pat)).to(torch.int64)
        vals = torch.from_numpy(rng.normal(10, 5, n))
        validity = torch.from_numpy(rng.random(n) > 0.1)
        ccpu = Column(dtypes.float64, vals, validity)
        cgpu = ccpu.to(DEV)
        for fn in ("sum", "min", "max"):
            a_ref, c_ref = ops.agg_scatter(gids, NG, ccpu, fn)
            a_gpu, c_gpu = ops.agg_scatter(gids.to(DEV), NG, cgpu, fn)
            assert torch.equal(c_ref, c_gpu.cpu()), (fn, n)
            nz = c_ref > 0  # empty-group identity repr differs (nulled later)
            if fn == "sum":
                assert torch.allclose(a_ref[nz], a_gpu.cpu()[nz], rtol=1e-9,
                                      atol=1e-6)
            else:
                assert torch.equal(a_ref[nz], a_gpu.cpu()[nz]), (fn, n)
        icpu = Column(dtypes.int64,
                      torch.from_numpy(rng.integers(-10**7, 10**7, n)), validity)
        igpu = icpu.to(DEV)
        for fn in ("sum", "min", "max"):
            a_ref, c_ref = ops.agg_scatter(gids, NG, icpu, fn)
            a_gpu, _ = ops.agg_scatter(gids.to(DEV), NG, igpu, fn)
            nz = c_ref > 0
            assert torch.equal(a_ref[nz], a_gpu.cpu()[nz]), (fn, n)


def test_memmgr_spill_restore_on_device():
    """HBM -> host spill and restore round-trip with a tiny budget."""
    from auron_amd.memory import MemManager

    def mk(tag, rows):
        data = torch.arange(rows, dtype=torch.int64, device=DEV)
        return [RecordBatch(["x"], [Column(dtypes.int64, data)])]

    mgr = MemManager(budget_bytes=1 << 20)  # 1 MiB
    h1 = mgr.register("h1", mk("a", 100_000))  # 800 KB
    h2 = mgr.register("h2", mk("b", 100_000))  # forces h1 to spill
    assert not h1.resident and h2.resident
    assert mgr.metrics.get("spill_d2h_bytes", 0) > 0
    back = h1.batches()  # restore (spills h2 in turn)
    assert back[0].columns[0].data.is_cuda
    assert int(back[0].columns[0].data[12345].item()) == 12345
    h1.release(); h2.release()


def test_agg_multi_fused_matches_single():
    """k_agg_multi must equal per-agg scatters for mixed dtypes/ops."""
    rng = np.random.default_rng(13)
    n = 300_000
    gids = torch.from_numpy(np.repeat(rng.integers(0, 20_000, 6000),
                                      rng.integers(1, 100, 6000))[:n]).to(torch.int64)
    n = gids.numel()
    items = []
    for dt, tdt in ((dtypes.float64, None), (dtypes.int64, None),
                    (dtypes.float32, np.float32), (dtypes.int32, np.int32)):
        raw = rng.normal(50, 20, n)
        if dt in (dtypes.int64, dtypes.int32):
            data = torch.from_numpy(raw.astype(np.int64 if dt == dtypes.int64 else np.int32))
        else:
            data = torch.from_numpy(raw.astype(np.float64 if dt == dtypes.float64 else np.float32))
        validity = torch.from_numpy(rng.random(n) > 0.07)
        items.append(Column(dt, data, validity))
    NG = 20_000
    fns = ["sum", "min", "max", "avg"]
    pairs = [(c.to(DEV), f) for c, f in zip(items, fns)]
    fused = ops.agg_scatter_multi(gids.to(DEV), NG, pairs)
    for (c, f), (facc, fcnt) in zip(pairs, fused):
        sacc, scnt = ops.agg_scatter(gids.to(DEV), NG, c, f)
        assert torch.equal(fcnt, scnt), f
        nz = (scnt > 0).cpu()
        if f in ("sum", "avg"):
            assert torch.allclose(facc.cpu()[nz], sacc.cpu()[nz],
                                  rtol=1e-9, atol=1e-6), f
        else:
            assert torch.equal(facc.cpu()[nz], sacc.cpu()[nz]), f


@pytest.mark.gpu
def test_bytes_gather_kernel_matches_cpu():
    import numpy as np

    rng = np.random.default_rng(9)
    words = ["", "a", "bb", "long-string-value-" * 3] + [f"w{i}" for i in range(200)]
    vals = [words[i] for i in rng.integers(0, len(words), 100_000)]
    c = Column.from_pylist(vals, dtypes.string, "cuda:0")
    idx = torch.as_tensor(rng.integers(0, len(vals), 250_000), dtype=torch.int64)
    got = c.gather(idx.to("cuda:0")).to("cpu")
    want = c.to("cpu").gather(idx)
    assert got.to_pylist() == want.to_pylist()
    # negative markers emit nulls
    idx2 = torch.tensor([0, -1, 5, -1], dtype=torch.int64, device="cuda:0")
    g2 = c.gather(idx2, may_have_negative=True).to("cpu")
    assert g2.to_pylist() == [vals[0], None, vals[5], None]


@pytest.mark.gpu
def test_multi_gather_matches_per_column():
    """Fused batch gather (k_multi_gather) == per-column gather, incl.
    negative indices (outer-join nulls), bools, decimals and strings."""
    import random

    from auron_amd import dtypes, native
    from auron_amd.column import Column, RecordBatch

    native.require()
    rng = random.Random(3)
    n = 20000
    cols = {
        "i64": Column.from_pylist([rng.randint(-9, 9) if rng.random() > .1
                                   else None for _ in range(n)], dtypes.int64),
        "i32": Column.from_pylist([rng.randint(0, 99) for _ in range(n)],
                                  dtypes.int32),
        "b": Column.from_pylist([rng.random() > .5 for _ in range(n)],
                                dtypes.bool_),
        "d": Column.from_pylist([rng.randint(0, 10 ** 6) / 100
                                 for _ in range(n)], dtypes.decimal64(9, 2)),
        "d128": Column.from_pylist([rng.randint(0, 10 ** 19) / 100
                                    for _ in range(n)],
                                   dtypes.decimal128(25, 2)),
        "s": Column.from_pylist([("x" * rng.randint(0, 6)) if rng.random() > .1
                                 else None for _ in range(n)], dtypes.string),
    }
    rb = RecordBatch(list(cols), list(cols.values())).to("cuda")
    idx = torch.tensor([rng.randrange(-1, n) for _ in range(n // 2)],
                       dtype=torch.int64, device="cuda")
    fused = rb.gather(idx, may_have_negative=True)
    percol = RecordBatch(rb.names, [c.gather(idx, may_have_negative=True)
                                    for c in rb.columns])
    torch.cuda.synchronize()
    for name in rb.names:
        f, p = fused.column(name).to("cpu"), percol.column(name).to("cpu")
        assert f.to_pylist() == p.to_pylist(), name
    # no-negatives path without validity stays validity-free
    idx2 = torch.arange(100, device="cuda")
    g = rb.select(["i32"]).gather(idx2)
    assert g is not None

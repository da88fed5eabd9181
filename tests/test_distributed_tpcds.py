"""World-size-2 (gloo) TPC-DS correctness: representative queries run
distributed and must produce exactly the single-rank result. Covers the
paths the 8-GPU scale bench exercises: sharded scans, broadcast joins,
shuffled joins, 2-phase aggs, window exchanges, CTE materialization."""
import math
import os

import pytest
import torch.distributed as dist
import torch.multiprocessing as mp

QUERIES_TO_CHECK = ["q3", "q23", "q72", "q38", "q47", "q5", "q1", "q88",
                    "q2", "q6", "q16", "q31", "q59", "q95", "q51", "q14",
                    # widened coverage: rollups, windows, set-ops, correlated
                    # scalar subqueries, multi-channel unions, anti/semi joins
                    "q4", "q9", "q11", "q18", "q22", "q27", "q33", "q36",
                    "q44", "q49", "q54", "q57", "q64", "q67", "q70", "q75",
                    "q77", "q78", "q80", "q87", "q93", "q97"]
SF = float(os.environ.get("AURON_TEST_SF", "0.01"))
ROOT = os.path.join(os.path.dirname(__file__), "..", ".tpcds_cache")


def _run(rank, world, port, q):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from auron_amd import AuronSession
        from auron_amd.tpcds.queries import QUERIES, Catalog

        s = AuronSession()
        cat = Catalog(ROOT, SF)
        qlist = QUERIES_TO_CHECK
        env_q = os.environ.get("AURON_DIST_QUERIES")
        if env_q:
            qlist = env_q.split(",")
        results = {}
        for qn in qlist:
            plan = QUERIES[qn](cat, s)
            results[qn] = s.collect_all(plan).to_pydict()
        if rank == 0:
            q.put(("ok", results))
    except Exception:  # pragma: no cover
        import traceback

        q.put(("err", traceback.format_exc()))
    finally:
        dist.destroy_process_group()


def _rows(d):
    def k(v):
        if v is None:
            return (True, "")
        if isinstance(v, float):
            return (False, f"{v:.9e}")
        return (False, str(v))

    return sorted(zip(*d.values()), key=lambda r: tuple(k(v) for v in r))


def test_world2_matches_single_rank():
    from auron_amd import AuronSession
    from auron_amd.tpcds import datagen
    from auron_amd.tpcds.queries import QUERIES, Catalog

    datagen.write_dataset(ROOT, SF)
    s = AuronSession()
    cat = Catalog(ROOT, SF)
    single = {}
    for qn in QUERIES_TO_CHECK:
        single[qn] = s.collect(QUERIES[qn](cat, s)).to_pydict()

    import random

    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    port = random.randint(20000, 40000)
    procs = [ctx.Process(target=_run, args=(r, 2, port, q)) for r in range(2)]
    for p in procs:
        p.start()
    status, dist_results = q.get(timeout=600)
    for p in procs:
        p.join(timeout=60)
    assert status == "ok", dist_results

    # ORDER BY keys that do not determine rows uniquely (duplicate
    # s_store_id values share a key) make the LIMIT subset engine-order
    # dependent; compare those as sets (same distinct rows, same count)
    tie_limit = {"q59"}
    for qn in QUERIES_TO_CHECK:
        a = _rows(single[qn])
        b = _rows(dist_results[qn])
        assert len(a) == len(b), f"{qn}: {len(a)} vs {len(b)} rows"
        if qn in tie_limit:
            def fmt(r):
                return tuple(f"{v:.6e}" if isinstance(v, float) else v
                             for v in r)

            assert {fmt(r) for r in a} == {fmt(r) for r in b}, qn
            continue
        for ra, rb in zip(a, b):
            for x, y in zip(ra, rb):
                if isinstance(x, float) and isinstance(y, float):
                    assert math.isclose(x, y, rel_tol=1e-6, abs_tol=1e-9), (qn, ra, rb)
                else:
                    assert x == y, (qn, ra, rb)


def test_world4_matches_single_rank():
    """World=4 over gloo on representative exchange-heavy queries: the
    collective schedule the 8-GPU driver bench will execute (all-to-all
    hash exchanges, broadcast gathers, single-rank collects) at a world
    size with non-trivial partitioning."""
    from auron_amd import AuronSession
    from auron_amd.tpcds import datagen
    from auron_amd.tpcds.queries import QUERIES, Catalog

    qs = ["q3", "q23", "q72", "q59", "q95", "q14", "q75", "q87"]
    datagen.write_dataset(ROOT, SF)
    s = AuronSession()
    cat = Catalog(ROOT, SF)
    single = {qn: s.collect(QUERIES[qn](cat, s)).to_pydict() for qn in qs}

    import random

    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    port = random.randint(20000, 40000)
    old = os.environ.get("AURON_DIST_QUERIES")
    os.environ["AURON_DIST_QUERIES"] = ",".join(qs)
    try:
        procs = [ctx.Process(target=_run, args=(r, 4, port, q)) for r in range(4)]
        for p in procs:
            p.start()
        status, dist_results = q.get(timeout=900)
        for p in procs:
            p.join(timeout=60)
    finally:
        if old is None:
            os.environ.pop("AURON_DIST_QUERIES", None)
        else:
            os.environ["AURON_DIST_QUERIES"] = old
    assert status == "ok", dist_results
    for qn in qs:
        a = _rows(single[qn])
        b = _rows(dist_results[qn])
        assert len(a) == len(b), f"{qn}: {len(a)} vs {len(b)} rows"
        if qn in ("q59",):
            def fmt(r):
                return tuple(f"{v:.6e}" if isinstance(v, float) else v for v in r)

            assert {fmt(r) for r in a} == {fmt(r) for r in b}, qn
            continue
        for ra, rb in zip(a, b):
            for x, y in zip(ra, rb):
                if isinstance(x, float) and isinstance(y, float):
                    assert math.isclose(x, y, rel_tol=1e-6, abs_tol=1e-9), (qn, ra, rb)
                else:
                    assert x == y, (qn, ra, rb)

"""Multi-process (gloo, world_size=2) correctness of the exchange layer:
shuffle repartition, broadcast, partial/final agg, shuffled hash join.
Exercises the same code path RCCL takes on the 8-GPU node (pairwise p2p
packing on gloo, all_to_all_single on nccl)."""
import math
import os

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

from auron_amd import AggFunc, AuronSession, col, dtypes, exprs
from auron_amd.column import RecordBatch
from auron_amd.plan import nodes as P

WORLD = 2


def _run(rank, world, port, fn_name, q):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        result = globals()[fn_name](rank, world)
        q.put((rank, "ok", result))
    except Exception as e:  # pragma: no cover
        import traceback

        q.put((rank, "err", traceback.format_exc()))
    finally:
        dist.destroy_process_group()


def _spawn(fn_name):
    import random

    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    port = random.randint(20000, 40000)
    procs = [ctx.Process(target=_run, args=(r, WORLD, port, fn_name, q)) for r in range(WORLD)]
    for p in procs:
        p.start()
    results = {}
    for _ in range(WORLD):
        rank, status, payload = q.get(timeout=120)
        assert status == "ok", payload
        results[rank] = payload
    for p in procs:
        p.join(timeout=30)
    return results


def _local_data(rank):
    # rank-disjoint rows
    if rank == 0:
        return {"k": ["a", "b", "a", "c"], "x": [1, 2, 3, 4]}
    return {"k": ["b", "c", "a", None], "x": [10, 20, 30, 40]}


def _scan(rank):
    d = _local_data(rank)
    return P.MemoryScan([RecordBatch.from_pydict(d, {"k": dtypes.string, "x": dtypes.int64})])


def body_shuffle_agg(rank, world):
    s = AuronSession()
    partial = P.HashAgg(_scan(rank), [exprs.Aliased(col("k"), "k")],
                        [AggFunc("sum", col("x"), name="sx"),
                         AggFunc("count", col("x"), name="cx")], mode="partial")
    ex = P.Exchange(partial, "hash", [col("k")])
    final = P.HashAgg(ex, [exprs.Aliased(col("k"), "k")],
                      [AggFunc("sum", col("x"), name="sx"),
                       AggFunc("count", col("x"), name="cx")], mode="final")
    out = s.collect_all(final).to_pydict()
    return out


def test_shuffle_partial_final_agg():
    results = _spawn("body_shuffle_agg")
    for rank, out in results.items():
        got = {k: (sx, cx) for k, sx, cx in zip(out["k"], out["sx"], out["cx"])}
        assert got == {"a": (34, 3), "b": (12, 2), "c": (24, 2), None: (40, 1)}


def body_broadcast_join(rank, world):
    s = AuronSession()
    left = _scan(rank)
    dim = {"k": ["a", "b", "c"], "label": ["A", "B", "C"]}
    # dim table sharded: each rank holds a slice; broadcast join gathers it
    shard = {k: v[rank::world] for k, v in dim.items()}
    right = P.MemoryScan([RecordBatch.from_pydict(
        {"rk": shard["k"], "label": shard["label"]},
        {"rk": dtypes.string, "label": dtypes.string})])
    join = P.HashJoin(left, right, [col("k")], [col("rk")], how="inner",
                      build_side="right", broadcast=True)
    out = s.collect_all(join).to_pydict()
    return out


def test_broadcast_join():
    results = _spawn("body_broadcast_join")
    for rank, out in results.items():
        pairs = sorted(zip(out["k"], out["label"]))
        assert pairs == [("a", "A"), ("a", "A"), ("a", "A"), ("b", "B"), ("b", "B"),
                        ("c", "C"), ("c", "C")]


def body_shuffle_join(rank, world):
    s = AuronSession()
    left = P.Exchange(_scan(rank), "hash", [col("k")])
    dim = {"rk": ["a", "b", "x"], "y": [100, 200, 300]}
    shard = {k: v[rank::world] for k, v in dim.items()}
    right = P.Exchange(
        P.MemoryScan([RecordBatch.from_pydict(shard, {"rk": dtypes.string, "y": dtypes.int64})]),
        "hash", [col("rk")])
    join = P.HashJoin(left, right, [col("k")], [col("rk")], how="left", build_side="right")
    out = s.collect_all(join).to_pydict()
    return out


def test_shuffle_join():
    results = _spawn("body_shuffle_join")
    for rank, out in results.items():
        got = sorted(zip(out["k"], out["x"], out["y"]),
                     key=lambda t: (t[0] is None, str(t[0]), t[1]))
        want = sorted([
            ("a", 1, 100), ("a", 3, 100), ("a", 30, 100),
            ("b", 2, 200), ("b", 10, 200),
            ("c", 4, None), ("c", 20, None), (None, 40, None),
        ], key=lambda t: (t[0] is None, str(t[0]), t[1]))
        assert got == want


def body_single_exchange(rank, world):
    s = AuronSession()
    ex = P.Exchange(_scan(rank), "single")
    out = s.collect(ex)
    return out.num_rows


def test_single_exchange_gathers_to_rank0():
    results = _spawn("body_single_exchange")
    assert results[0] == 8
    assert results[1] == 0


def _range_exchange(rank, world):
    import numpy as np

    rng = np.random.default_rng(rank)
    data = {"k": rng.integers(0, 1000, 200).tolist(),
            "v": list(range(200))}
    t = {"k": dtypes.int64, "v": dtypes.int64}
    s = AuronSession()
    plan = P.Exchange(
        P.MemoryScan([RecordBatch.from_pydict(data, t)]), "range", [col("k")])
    got = s.collect(plan).to_pydict()
    return {"rank": rank, "keys": got["k"], "n": len(got["k"])}


def test_range_exchange_global_order():
    """Range repartition: every key on rank 0 <= every key on rank 1,
    and no rows lost."""
    res = _spawn("_range_exchange")
    assert res[0]["n"] + res[1]["n"] == 400
    if res[0]["keys"] and res[1]["keys"]:
        assert max(res[0]["keys"]) <= min(res[1]["keys"])
    # rough balance (quantile bounds from 400 samples of uniform keys)
    assert 100 < res[0]["n"] < 300


def _range_exchange_strings(rank, world):
    import numpy as np

    rng = np.random.default_rng(100 + rank)
    words = [f"w{int(v):04d}" for v in rng.integers(0, 1000, 150)]
    s = AuronSession()
    plan = P.Exchange(
        P.MemoryScan([RecordBatch.from_pydict(
            {"k": words}, {"k": dtypes.string})]), "range", [col("k")])
    got = s.collect(plan).to_pydict()
    return {"keys": got["k"], "n": len(got["k"])}


def test_range_exchange_strings_global_order():
    res = _spawn("_range_exchange_strings")
    assert res[0]["n"] + res[1]["n"] == 300
    if res[0]["keys"] and res[1]["keys"]:
        assert max(res[0]["keys"]) <= min(res[1]["keys"])

"""Persistent shuffle files: write/read roundtrip, stage-retry re-read,
and the Exchange(persist=True) executor path (single-rank and gloo W=2)."""
import math
import os

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

from auron_amd import AggFunc, AuronSession, col, dtypes
from auron_amd.column import Column, RecordBatch
from auron_amd.exprs import Aliased
from auron_amd.plan import nodes as P
from auron_amd.shuffle import ShuffleReader, ShuffleWriter


def _batch(vals, names=("k", "v")):
    return RecordBatch.from_pydict(
        {names[0]: vals[0], names[1]: vals[1]},
        {names[0]: dtypes.int64, names[1]: dtypes.string})


def test_writer_reader_roundtrip(tmp_path):
    root = str(tmp_path)
    b0 = _batch(([1, 2, None], ["x", None, "zz"]))
    b1 = _batch(([7], ["seven"]))
    ShuffleWriter(root, "s1", 0).write([b0, None])
    ShuffleWriter(root, "s1", 1).write([b1, b0])
    r = ShuffleReader(root, "s1")
    p0 = r.read_partition(0)
    assert len(p0) == 2
    assert p0[0].to_pydict() == b0.to_pydict()
    assert p0[1].to_pydict() == b1.to_pydict()
    p1 = r.read_partition(1)
    assert len(p1) == 1 and p1[0].to_pydict() == b0.to_pydict()
    # stage retry: files persist, a second reader sees identical data
    again = ShuffleReader(root, "s1").read_partition(0)
    assert [b.to_pydict() for b in again] == [b.to_pydict() for b in p0]


def test_exchange_persist_single_rank(tmp_path, monkeypatch):
    monkeypatch.setenv("AURON_SHUFFLE_DIR", str(tmp_path))
    data = {"k": [1, 2, 1, 3, None, 2], "v": [10.0, 20.0, 30.0, 40.0, 50.0, 60.0]}
    t = {"k": dtypes.int64, "v": dtypes.float64}
    scan = P.MemoryScan([RecordBatch.from_pydict(data, t)])
    plan = P.HashAgg(
        P.Exchange(scan, "hash", [col("k")], persist=True),
        [Aliased(col("k"), "k")], [AggFunc("sum", col("v"), name="s")], mode="complete")
    s = AuronSession()
    got = s.collect(plan).to_pydict()
    m = dict(zip(got["k"], got["s"]))
    assert m[1] == 40.0 and m[2] == 80.0 and m[3] == 40.0 and m[None] == 50.0
    # shuffle files were actually written
    stages = os.listdir(str(tmp_path))
    assert any(d.startswith("stage-") for d in stages)


WORLD = 2


def _run(rank, world, port, sdir, q):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["AURON_SHUFFLE_DIR"] = sdir
    os.environ["AURON_SHUFFLE_PERSIST"] = "1"
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        data = ({"k": [1, 2, 1, 4], "v": [1.0, 2.0, 3.0, 4.0]} if rank == 0
                else {"k": [2, 4, 4, None], "v": [10.0, 20.0, 30.0, 40.0]})
        t = {"k": dtypes.int64, "v": dtypes.float64}
        scan = P.MemoryScan([RecordBatch.from_pydict(data, t)])
        plan = P.HashAgg(P.Exchange(scan, "hash", [col("k")]),
                         [Aliased(col("k"), "k")],
                         [AggFunc("sum", col("v"), name="s")],
                         mode="complete")
        s = AuronSession()
        got = s.collect(plan).to_pydict()
        q.put((rank, "ok", got))
    except Exception:
        import traceback

        q.put((rank, "err", traceback.format_exc()))
    finally:
        dist.destroy_process_group()


def test_exchange_persist_world2(tmp_path):
    import random

    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    port = random.randint(20000, 40000)
    procs = [ctx.Process(target=_run, args=(r, WORLD, port, str(tmp_path), q))
             for r in range(WORLD)]
    for p in procs:
        p.start()
    merged = {}
    for _ in range(WORLD):
        rank, status, payload = q.get(timeout=120)
        assert status == "ok", payload
        for k, v in zip(payload["k"], payload["s"]):
            assert k not in merged, f"key {k} on two ranks"
            merged[k] = v
    for p in procs:
        p.join(timeout=30)
    assert merged == {1: 4.0, 2: 12.0, 4: 54.0, None: 40.0}


def test_shuffle_zlib_codec(tmp_path, monkeypatch):
    monkeypatch.setenv("AURON_SHUFFLE_CODEC", "zlib")
    b0 = _batch(([1, 2, None] * 100, ["x", None, "zz"] * 100))
    ShuffleWriter(str(tmp_path), "c1", 0).write([b0, b0])
    r = ShuffleReader(str(tmp_path), "c1")
    got = r.read_partition(1)
    assert len(got) == 1 and got[0].to_pydict() == b0.to_pydict()
    raw = os.path.getsize(os.path.join(str(tmp_path), "stage-c1", "map-0.data"))
    monkeypatch.delenv("AURON_SHUFFLE_CODEC")
    ShuffleWriter(str(tmp_path), "c2", 0).write([b0, b0])
    plain = os.path.getsize(os.path.join(str(tmp_path), "stage-c2", "map-0.data"))
    assert raw < plain  # codec actually engaged


def test_nccl_branch_slicing_math():
    """Simulate all_to_all_single's output contract for W=3 and verify the
    nccl-branch reassembly (`_split_recv_flat`) recovers every batch —
    including empty and None destinations."""
    import torch

    from auron_amd.exchange import _split_recv_flat, pack_batch

    b1 = _batch(([1, None, 3], ["a", "bb", None]))
    b2 = _batch(([9], ["z"]))
    metas, bufs = [], []
    for src in (b1, None, b2):
        if src is None:
            metas.append(None)
            bufs.append(torch.zeros(0, dtype=torch.uint8))
        else:
            m, buf = pack_batch(src, "cpu")
            metas.append(m)
            bufs.append(buf)
    recv_sizes = [b.numel() for b in bufs]
    recv_flat = torch.cat(bufs) if sum(recv_sizes) else torch.zeros(0, dtype=torch.uint8)
    out = _split_recv_flat(recv_flat, recv_sizes, metas)
    assert len(out) == 2
    assert out[0].to_pydict() == b1.to_pydict()
    assert out[1].to_pydict() == b2.to_pydict()

#!/usr/bin/env python3
"""Flagship benchmark: TPC-DS query suite on auron_amd.

Driver contract: `python bench.py --gpus N --steps K --warmup W` (for N>1
launched under torch.distributed.run, one rank per GPU over RCCL). One
step = run the full implemented TPC-DS query suite once against the
synthetic dsdgen dataset at --sf (strong scaling: fixed dataset, fact
files sharded across ranks, exchange over RCCL/xGMI).

Rank 0 prints ONE JSON line with the suite wall-clock per step (max over
ranks). BASELINE.md's published number (1519 s, TPC-DS 1TB / 99 queries,
unknown CPU hardware) is not directly comparable to this config, so
vs_baseline is null until the full 99-query/1TB config runs.
"""
import argparse
import json
import shutil
import os
import sys
import time

import os as _os

_os.environ.setdefault("PYTORCH_ALLOC_CONF", "expandable_segments:True")

import torch
import torch.distributed as dist


def log(msg):
    r = os.environ.get("RANK", "0")
    print(f"[bench r{r}] {msg}", file=sys.stderr, flush=True)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=3)
    ap.add_argument("--warmup", type=int, default=1)
    ap.add_argument("--sf", type=float, default=float(os.environ.get("AURON_BENCH_SF", "10")))
    ap.add_argument("--queries", type=str, default="all")
    ap.add_argument("--data-root", type=str,
                    default=os.environ.get("AURON_DATA_ROOT", ""))
    ap.add_argument("--device", type=str, default=None)
    args = ap.parse_args()

    from auron_amd import AuronSession, init_distributed
    from auron_amd.tpcds import datagen
    from auron_amd.tpcds.queries import QUERIES, Catalog

    if not args.data_root:
        # prefer a RAM-backed root when it has more headroom than the
        # repo filesystem (GPU boxes: ~80 GB disk, TBs of DRAM; dsdgen
        # data in tmpfs = the page-cache residency every published
        # TPC-DS run assumes)
        repo_root = os.path.join(os.path.dirname(os.path.abspath(__file__)), "data")
        args.data_root = repo_root
        try:
            shm = "/dev/shm"
            if os.path.isdir(shm) and os.access(shm, os.W_OK):
                free_shm = shutil.disk_usage(shm).free
                free_repo = shutil.disk_usage(os.path.dirname(repo_root)).free
                if free_shm > free_repo:
                    args.data_root = os.path.join(shm, "auron_tpcds")
        except OSError:
            pass

    ctx = init_distributed()
    rank, world = ctx.rank, ctx.world_size
    have_gpu = torch.cuda.is_available()
    device = args.device or ("cuda" if have_gpu else "cpu")
    if have_gpu and world > 1:
        torch.cuda.set_device(int(os.environ.get("LOCAL_RANK", "0")))

    qnames = sorted(QUERIES.keys()) if args.queries == "all" else args.queries.split(",")

    # ---- dataset (generated once, sharded across ranks, cached on disk)
    t0 = time.perf_counter()
    datagen.write_dataset(args.data_root, args.sf, rank=rank, world=world)
    if world > 1:
        dist.barrier()
    log(f"dataset sf={args.sf} ready in {time.perf_counter() - t0:.1f}s")

    session = AuronSession(device=device)
    cat = Catalog(args.data_root, args.sf)

    trace = os.environ.get("AURON_BENCH_TRACE", "") == "1"

    def run_suite():
        per_q = {}
        for qn in qnames:
            if trace:
                log(f"start {qn}")
            tq = time.perf_counter()
            plan = QUERIES[qn](cat, session)
            res = session.collect(plan)
            if have_gpu:
                torch.cuda.synchronize()
            per_q[qn] = time.perf_counter() - tq
            if trace:
                log(f"done  {qn} {per_q[qn]:.2f}s")
        return per_q

    def barrier_sync():
        if world > 1:
            dist.barrier()
        if have_gpu:
            torch.cuda.synchronize()

    for w in range(args.warmup):
        pq = run_suite()
        log(f"warmup {w}: {sum(pq.values()):.3f}s " +
            " ".join(f"{k}={v:.2f}" for k, v in pq.items()))

    barrier_sync()
    t_start = time.perf_counter()
    per_q_last = None
    for k in range(args.steps):
        per_q_last = run_suite()
    barrier_sync()
    elapsed = time.perf_counter() - t_start

    # max over ranks
    if world > 1:
        t = torch.tensor([elapsed], dtype=torch.float64,
                         device="cuda" if dist.get_backend() == "nccl" else "cpu")
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())

    suite_seconds = elapsed / args.steps
    if rank == 0 and os.environ.get("AURON_BENCH_METRICS", "0") == "1":
        for k, v in sorted(session.metrics().items(), key=lambda kv: -kv[1]):
            log(f"metric {k}: {v:.3f}s")
    if rank == 0:
        out = {
            "metric": "tpcds_suite_seconds",
            "value": suite_seconds,
            "unit": "s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": suite_seconds * 1000.0,
            "higher_is_better": False,
            "scaling": "strong",
            # Auron total TPC-DS **1TB** = ~1519 s on the
            # reference's (unspecified CPU) hardware; only an SF=1000 run
            # here is the same config, so smaller SFs report null
            "vs_baseline": (suite_seconds / 1519.0) if args.sf >= 1000 else None,
            "dtype": "decimal64/fp64",
            "data": "synthetic",
            "config": {
                "model": f"tpcds-{len(qnames)}q",
                "queries": qnames,
                "sf": args.sf,
                "parallelism": f"dp{world}",
                "caches": "HBM staged-bytes + decoded-column buffer pool "
                          "persist across queries/steps (buffer-pool "
                          "semantics; warmup steps warm them)",
                "per_query_s": {k: round(v, 4) for k, v in (per_q_last or {}).items()},
            },
        }
        print(json.dumps(out), flush=True)


if __name__ == "__main__":
    main()

#!/usr/bin/env python3
"""Microbenchmark the parquet def-level decode path on one column chunk:
isolates au_pq_rle1 cost from the rest of the scan (run on a GPU box)."""
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import numpy as np
import torch

from auron_amd import native, parquet_native
from auron_amd.tpcds import datagen

ROOT = "/tmp/microds"


def main():
    datagen.write_dataset(ROOT, 1.0, tables=["store_sales"])
    path = datagen.dataset_paths(ROOT, 1.0, "store_sales")[0]
    cols = ["ss_sold_date_sk", "ss_quantity", "ss_ext_sales_price",
            "ss_customer_sk", "ss_item_sk"]
    dev = "cuda:0"
    # warm meta + buffers
    out = parquet_native.read_columns_native(path, cols, dev)
    torch.cuda.synchronize()
    meta = parquet_native._get_meta(path, cols)
    total_pages = sum(len(ck.pages) for cm in meta.cols for ck in cm.pages)
    nvals = sum(c[2] for cm in meta.cols for c in cm.chunks)
    print(f"cols={len(meta.cols)} chunks/col={len(meta.cols[0].chunks)} "
          f"pages={total_pages} values={nvals}")

    # full read timing
    for tag in range(3):
        torch.cuda.synchronize()
        t0 = time.time()
        parquet_native.read_columns_native(path, cols, dev)
        torch.cuda.synchronize()
        print(f"full read: {(time.time()-t0)*1000:.1f} ms")

    # isolate rle1: replay the def-level launches only
    lib = native.lib()
    sp = native.stream_ptr(torch.device(dev))
    # rebuild staging buffer like the reader does
    import numpy as _np

    mm = np.memmap(path, dtype=np.uint8, mode="r")
    full = meta.total + 8 + meta.extra_total + (8 if meta.extra_total else 0)
    buf_t = torch.empty(full, dtype=torch.uint8)
    buf = buf_t.numpy()
    for (src, clen, dst) in meta.ranges:
        buf[dst:dst + clen] = mm[src:src + clen]
    dbuf = buf_t.to(dev)
    jobs = []
    for cm in meta.cols:
        if not cm.has_def:
            continue
        for (chunk, ck) in zip(cm.chunks, cm.pages):
            nvals_c = chunk[2]
            pages_np = np.zeros((len(ck.pages), 6), dtype=np.int64)
            for i, p in enumerate(ck.pages):
                pages_np[i] = (p.def_off, p.def_len, p.values_off, p.n_values,
                               p.row_start, 0)
            pdev = torch.from_numpy(pages_np).to(dev)
            outv = torch.empty(nvals_c, dtype=torch.uint8, device=dev)
            jobs.append((pdev, len(ck.pages), outv))
    torch.cuda.synchronize()
    ev0, ev1 = torch.cuda.Event(True), torch.cuda.Event(True)
    for rep in range(3):
        ev0.record()
        for (pdev, npages, outv) in jobs:
            rc = lib.au_pq_rle1(pdev.data_ptr(), npages, dbuf.data_ptr(),
                                outv.data_ptr(), sp)
            native.check(rc, "au_pq_rle1")
        ev1.record()
        torch.cuda.synchronize()
        ms = ev0.elapsed_time(ev1)
        vals = sum(j[2].numel() for j in jobs)
        print(f"rle1 replay: {ms:.2f} ms for {len(jobs)} launches, "
              f"{vals} vals -> {vals/ms/1e6:.1f} Gval/s")


if __name__ == "__main__":
    main()

"""Hot-op dispatch: native HIP kernels on GPU, reference impls on CPU.

The CPU implementations are the correctness oracles (the analogue of the
reference's JVM-less test mode) — GPU numerics tests compare
kernel output against these bit-for-bit (hashes) or set-wise (tables).

GPU path REQUIRES the native library (no silent eager fallback): see
auron_amd.native.require().
"""
from __future__ import annotations

import os
from typing import List, Optional, Tuple

import torch

from . import dtypes, native
from .column import Column

SPARK_SEED = 42
_M32 = 0xFFFFFFFF


def _use_native(device: torch.device) -> bool:
    if device.type != "cuda":
        return False
    if os.environ.get("AURON_FORCE_REF", "0") == "1":
        return False  # debugging only: force torch/host reference paths
    if os.environ.get("AURON_REQUIRE_NATIVE", "1") == "0" and not native.available():
        return False
    native.require()  # raise loudly on GPU when missing
    return True


# ===================================================================== hash
def _rotl(x, r):
    return (((x << r) & _M32) | (x >> (32 - r))) & _M32


def _mixk1(k):
    k = (k * 0xCC9E2D51) & _M32
    k = _rotl(k, 15)
    k = (k * 0x1B873593) & _M32
    return k


def _mixh1(h, k):
    h = (h ^ k) & _M32
    h = _rotl(h, 13)
    h = (h * 5 + 0xE6546B64) & _M32
    return h


def _fmix(h, length):
    h = (h ^ length) & _M32
    h = h ^ (h >> 16)
    h = (h * 0x85EBCA6B) & _M32
    h = h ^ (h >> 13)
    h = (h * 0xC2B2AE35) & _M32
    h = h ^ (h >> 16)
    return h


def _hash_int_t(v, seed):
    return _fmix(_mixh1(seed, _mixk1(v & _M32)), 4)


def _hash_long_t(v, seed):
    h = _mixh1(seed, _mixk1(v & _M32))
    h = _mixh1(h, _mixk1((v >> 32) & _M32))
    return _fmix(h, 8)


def murmur3_ref(cols: List[Column], seed: int = SPARK_SEED) -> torch.Tensor:
    """Pure-torch Spark Murmur3_x86_32, bit-exact vs the HIP kernel.

    Reference semantics: spark_hash.rs (seed-chained per column, nulls
    skipped, -0.0 normalized, strings hashed as UTF-8 bytes).
    """
    n = len(cols[0])
    device = cols[0].device
    h = torch.full((n,), seed, dtype=torch.int64, device=device)
    for c in cols:
        if c.dtype.is_string:
            from . import strings as S

            lens = S.lengths(c)
            maxlen = int(lens.max().item()) if n else 0
            padded = S.to_padded(c, max(maxlen, 1)).to(torch.int64)
            hv = h.clone()
            nwords = maxlen // 4
            for j in range(nwords):
                w = (padded[:, 4 * j] | (padded[:, 4 * j + 1] << 8)
                     | (padded[:, 4 * j + 2] << 16) | (padded[:, 4 * j + 3] << 24))
                m = lens >= 4 * (j + 1)
                hv = torch.where(m, _mixh1(hv, _mixk1(w)), hv)
            aligned = lens & ~3
            for t in range(maxlen):
                m = (t >= aligned) & (t < lens)
                if not bool(m.any()):
                    continue
                b = padded[:, t]
                sb = torch.where(b >= 128, (b - 256) & _M32, b)
                hv = torch.where(m, _mixh1(hv, _mixk1(sb)), hv)
            hv = _fmix(hv, lens & _M32)
        else:
            d = c.data
            code = c.dtype.code
            if code in (dtypes.BOOL,):
                v = d.to(torch.int64)
                hv = _hash_int_t(v, h)
            elif code in (dtypes.INT8, dtypes.INT16, dtypes.INT32, dtypes.DATE32):
                v = d.to(torch.int64) & _M32
                hv = _hash_int_t(v, h)
            elif code in (dtypes.INT64, dtypes.DECIMAL64):
                v = d.to(torch.int64)
                hv = _hash_long_t(v, h)
            elif code == dtypes.FLOAT32:
                f = torch.where(d == 0, torch.zeros_like(d), d)
                v = f.view(torch.int32).to(torch.int64) & _M32
                hv = _hash_int_t(v, h)
            elif code == dtypes.FLOAT64:
                f = torch.where(d == 0, torch.zeros_like(d), d)
                v = f.view(torch.int64)
                hv = _hash_long_t(v, h)
            else:
                raise TypeError(c.dtype.name)
        if c.validity is not None:
            h = torch.where(c.validity, hv, h)
        else:
            h = hv
    # reinterpret as signed int32
    h = torch.where(h >= 2 ** 31, h - 2 ** 32, h)
    return h.to(torch.int32)


def murmur3(cols: List[Column], seed: int = SPARK_SEED) -> torch.Tensor:
    device = cols[0].device
    n = len(cols[0])
    if not _use_native(device):
        return murmur3_ref(cols, seed)
    lib = native.lib()
    out = torch.empty(n, dtype=torch.int32, device=device)
    descs, keep = native.pack_descs(cols, device)
    rc = lib.au_murmur3(descs.data_ptr(), len(cols), n, seed, out.data_ptr(),
                        native.stream_ptr(device))
    native.check(rc, "au_murmur3")
    del keep
    return out


# ================================================================= group-by
def _next_pow2(x: int) -> int:
    p = 1024
    while p < x:
        p <<= 1
    return p


def _key_tuple_lists(cols: List[Column]):
    import math

    lists = []
    for c in cols:
        vals = c.to_pylist()
        if c.dtype.is_float:
            vals = ["NaN" if (v is not None and isinstance(v, float) and math.isnan(v)) else v for v in vals]
        lists.append(vals)
    return list(zip(*lists)) if cols else []


def group_ids_ref(cols: List[Column]) -> Tuple[torch.Tensor, torch.Tensor]:
    keys = _key_tuple_lists(cols)
    d = {}
    gids = []
    reps = []
    for i, k in enumerate(keys):
        g = d.get(k)
        if g is None:
            g = len(d)
            d[k] = g
            reps.append(i)
        gids.append(g)
    device = cols[0].device
    return (torch.tensor(gids, dtype=torch.int64, device=device),
            torch.tensor(reps, dtype=torch.int64, device=device))


def group_ids(cols: List[Column]) -> Tuple[torch.Tensor, torch.Tensor]:
    """-> (gid per row [n] int64, representative row per group [G] int64).

    Group order is order-of-appearance on CPU, unspecified on GPU (atomic
    assignment) — SQL GROUP BY output order is unspecified anyway.
    """
    device = cols[0].device
    n = len(cols[0])
    if n == 0:
        return (torch.empty(0, dtype=torch.int64, device=device),
                torch.empty(0, dtype=torch.int64, device=device))
    if not _use_native(device):
        return group_ids_ref(cols)
    lib = native.lib()
    hashes = murmur3(cols)
    cap = _next_pow2(2 * n)
    slots = torch.full((cap,), -1, dtype=torch.int32, device=device)
    rep = torch.empty(n, dtype=torch.int32, device=device)
    gids = torch.empty(n, dtype=torch.int32, device=device)
    rep_rows = torch.empty(n, dtype=torch.int64, device=device)
    counter = torch.zeros(1, dtype=torch.int32, device=device)
    descs, keep = native.pack_descs(cols, device)
    rc = lib.au_group_ids(descs.data_ptr(), len(cols), n, slots.data_ptr(), cap,
                          hashes.data_ptr(), rep.data_ptr(), gids.data_ptr(),
                          rep_rows.data_ptr(), counter.data_ptr(),
                          native.stream_ptr(device))
    native.check(rc, "au_group_ids")
    ng = int(counter.item())
    del keep
    return gids.to(torch.int64), rep_rows[:ng]


# ==================================================================== joins
def hash_join_ref(build_cols, probe_cols, emit_unmatched_probe, need_build_matched):
    bkeys = _key_tuple_lists(build_cols)
    pkeys = _key_tuple_lists(probe_cols)
    table = {}
    for j, k in enumerate(bkeys):
        if any(v is None for v in k):
            continue
        table.setdefault(k, []).append(j)
    n_build = len(build_cols[0]) if build_cols else 0
    matched = [False] * n_build
    bi, pi = [], []
    for i, k in enumerate(pkeys):
        rows = [] if any(v is None for v in k) else table.get(k, [])
        if rows:
            for j in rows:
                bi.append(j)
                pi.append(i)
                matched[j] = True
        elif emit_unmatched_probe:
            bi.append(-1)
            pi.append(i)
    device = probe_cols[0].device
    bm = torch.tensor(matched, dtype=torch.bool, device=device) if need_build_matched else None
    return (torch.tensor(bi, dtype=torch.int64, device=device),
            torch.tensor(pi, dtype=torch.int64, device=device), bm)



class JoinTable:
    """Prebuilt chained hash table over a broadcast build side (the
    reference's cached_build_hash_map_id object,
    broadcast_join_exec.rs:90): heads/next chains + row hashes stay in
    HBM so repeated probes of the same relation skip the build pass."""

    __slots__ = ("heads", "nxt", "cap", "bhash", "n_build")

    def __init__(self, heads, nxt, cap, bhash, n_build):
        self.heads = heads
        self.nxt = nxt
        self.cap = cap
        self.bhash = bhash
        self.n_build = n_build

    @property
    def nbytes(self):
        return (self.heads.numel() + self.nxt.numel()) * 4 + self.bhash.numel() * 8


def join_build(build_cols: List[Column]) -> Optional["JoinTable"]:
    """Build the chained table once (au_join_build); None on the CPU
    reference path (which has no reusable native table)."""
    device = build_cols[0].device
    if not _use_native(device):
        return None
    lib = native.lib()
    sp = native.stream_ptr(device)
    n_build = len(build_cols[0])
    bhash = murmur3(build_cols)
    cap = _next_pow2(2 * max(n_build, 1))
    heads = torch.full((cap,), -1, dtype=torch.int32, device=device)
    nxt = torch.empty(max(n_build, 1), dtype=torch.int32, device=device)
    if n_build:
        rc = lib.au_join_build(n_build, heads.data_ptr(), cap, nxt.data_ptr(),
                               bhash.data_ptr(), sp)
        native.check(rc, "au_join_build")
    return JoinTable(heads, nxt, cap, bhash, n_build)


def hash_join(build_cols: List[Column], probe_cols: List[Column],
              emit_unmatched_probe: bool = False,
              need_build_matched: bool = False,
              table: Optional["JoinTable"] = None):
    """Hash-join index computation.

    -> (build_idx [m] int64 (-1 = probe row had no match),
        probe_idx [m] int64,
        build_matched [n_build] bool | None)
    SQL NULL join keys never match (join_row_has_null in kernels.hip).
    """
    device = probe_cols[0].device
    n_probe = len(probe_cols[0])
    n_build = len(build_cols[0])
    if not _use_native(device):
        return hash_join_ref(build_cols, probe_cols, emit_unmatched_probe, need_build_matched)
    lib = native.lib()
    sp = native.stream_ptr(device)
    if table is None:
        table = join_build(build_cols)
    heads, nxt, cap, bhash = table.heads, table.nxt, table.cap, table.bhash
    phash = murmur3(probe_cols)
    bdescs, bkeep = native.pack_descs(build_cols, device)
    pdescs, pkeep = native.pack_descs(probe_cols, device)
    counts = torch.zeros(n_probe, dtype=torch.int32, device=device)
    rc = lib.au_join_count(bdescs.data_ptr(), pdescs.data_ptr(), len(build_cols),
                           n_probe, heads.data_ptr(), cap, nxt.data_ptr(),
                           phash.data_ptr(), counts.data_ptr(), sp)
    native.check(rc, "au_join_count")
    eff = counts.to(torch.int64)
    if emit_unmatched_probe:
        eff = eff.clamp(min=1)
    offsets = torch.zeros(n_probe + 1, dtype=torch.int64, device=device)
    torch.cumsum(eff, 0, out=offsets[1:])
    total = int(offsets[-1].item())
    build_idx = torch.empty(max(total, 1), dtype=torch.int64, device=device)
    probe_idx = torch.empty(max(total, 1), dtype=torch.int64, device=device)
    bm = torch.zeros(n_build, dtype=torch.uint8, device=device) if need_build_matched else None
    rc = lib.au_join_fill(bdescs.data_ptr(), pdescs.data_ptr(), len(build_cols),
                          n_probe, heads.data_ptr(), cap, nxt.data_ptr(),
                          phash.data_ptr(), offsets.data_ptr(),
                          build_idx.data_ptr(), probe_idx.data_ptr(),
                          bm.data_ptr() if bm is not None else None,
                          1 if emit_unmatched_probe else 0, sp)
    native.check(rc, "au_join_fill")
    del bkeep, pkeep
    return build_idx[:total], probe_idx[:total], (bm.bool() if bm is not None else None)


def join_counts(build_cols: List[Column], probe_cols: List[Column]) -> torch.Tensor:
    """Per-probe-row match count (semi/anti/existence joins)."""
    device = probe_cols[0].device
    n_probe = len(probe_cols[0])
    n_build = len(build_cols[0])
    if not _use_native(device):
        bi, pi, _ = hash_join_ref(build_cols, probe_cols, False, False)
        counts = torch.zeros(n_probe, dtype=torch.int32)
        if pi.numel():
            counts.scatter_add_(0, pi, torch.ones(pi.numel(), dtype=torch.int32))
        return counts
    lib = native.lib()
    sp = native.stream_ptr(device)
    bhash = murmur3(build_cols)
    phash = murmur3(probe_cols)
    cap = _next_pow2(2 * max(n_build, 1))
    heads = torch.full((cap,), -1, dtype=torch.int32, device=device)
    nxt = torch.empty(max(n_build, 1), dtype=torch.int32, device=device)
    bdescs, bkeep = native.pack_descs(build_cols, device)
    pdescs, pkeep = native.pack_descs(probe_cols, device)
    if n_build:
        rc = lib.au_join_build(n_build, heads.data_ptr(), cap, nxt.data_ptr(),
                               bhash.data_ptr(), sp)
        native.check(rc, "au_join_build")
    counts = torch.zeros(n_probe, dtype=torch.int32, device=device)
    rc = lib.au_join_count(bdescs.data_ptr(), pdescs.data_ptr(), len(build_cols),
                           n_probe, heads.data_ptr(), cap, nxt.data_ptr(),
                           phash.data_ptr(), counts.data_ptr(), sp)
    native.check(rc, "au_join_count")
    del bkeep, pkeep
    return counts


def string_sort_ranks(c: Column) -> torch.Tensor:
    """Order-preserving dense ranks for a string column (device int64).

    Unique strings (group-table reps) are sorted on the host — uniques are
    small — and each row maps to its rank through the device group ids.
    Used by sort/window so string order-by keys never fall back to a host
    argsort over the full column."""
    gids, reps = group_ids([c])
    rep_vals = c.gather(reps).to_pylist()
    order = sorted(range(len(rep_vals)),
                   key=lambda i: (rep_vals[i] is None, rep_vals[i] or ""))
    rank = [0] * len(rep_vals)
    for r, i in enumerate(order):
        rank[i] = r
    rank_t = torch.tensor(rank, dtype=torch.int64, device=c.device)
    return rank_t[gids]


# ================================================================ partition
def _normalize_hash_col(c: Column) -> Column:
    """Promote key columns to a canonical width before hashing: murmur3
    hashes int32 and int64 differently, so `week_seq` (int32) vs
    `week_seq - 52` (promoted int64) would land on different ranks. Our
    internal exchange only needs consistency, not Spark shuffle-file
    layout, so ints hash as int64 and floats as float64."""
    if c.dtype.code in (dtypes.BOOL, dtypes.INT8, dtypes.INT16, dtypes.INT32,
                        dtypes.DATE32):
        return Column(dtypes.int64, c.data.to(torch.int64), c.validity)
    if c.dtype.code == dtypes.FLOAT32:
        return Column(dtypes.float64, c.data.to(torch.float64), c.validity)
    return c


def partition_ids(cols: List[Column], nparts: int) -> torch.Tensor:
    """Internal hash partitioning: pmod(murmur3(normalized keys, 42), n)."""
    cols = [_normalize_hash_col(c) for c in cols]
    device = cols[0].device
    h = murmur3(cols)
    if not _use_native(device):
        return torch.remainder(h.to(torch.int64), nparts).to(torch.int32)
    lib = native.lib()
    n = h.numel()
    out = torch.empty(n, dtype=torch.int32, device=device)
    rc = lib.au_pmod(h.data_ptr(), n, nparts, out.data_ptr(), native.stream_ptr(device))
    native.check(rc, "au_pmod")
    return out


def partition_order(part_ids: torch.Tensor, nparts: int) -> Tuple[torch.Tensor, torch.Tensor]:
    """-> (row order grouping rows by partition [n] int64, counts [nparts] int64)."""
    device = part_ids.device
    n = part_ids.numel()
    if not _use_native(device):
        counts = torch.bincount(part_ids.to(torch.int64), minlength=nparts)
        order = torch.argsort(part_ids.to(torch.int64), stable=True)
        return order, counts
    lib = native.lib()
    sp = native.stream_ptr(device)
    counts = torch.zeros(nparts, dtype=torch.int32, device=device)
    rc = lib.au_part_hist(part_ids.data_ptr(), n, nparts, counts.data_ptr(), sp)
    native.check(rc, "au_part_hist")
    counts64 = counts.to(torch.int64)
    base = torch.zeros(nparts, dtype=torch.int64, device=device)
    if nparts > 1:
        torch.cumsum(counts64[:-1], 0, out=base[1:])
    cursors = torch.zeros(nparts, dtype=torch.int32, device=device)
    order = torch.empty(n, dtype=torch.int64, device=device)
    rc = lib.au_part_scatter(part_ids.data_ptr(), n, base.data_ptr(),
                             cursors.data_ptr(), order.data_ptr(), sp)
    native.check(rc, "au_part_scatter")
    return order, counts64


# ============================================================== aggregation
def _agg_scatter_native(gids: torch.Tensor, num_groups: int, values: Column, fn: str):
    """HIP atomic scatter-accumulate (csrc/agg.hip). torch's fp64
    scatter_add_ measured ~400x slower on gfx950 (rocprof, q23)."""
    lib = native.lib()
    device = gids.device
    sp = native.stream_ptr(device)
    n = gids.numel()
    g = gids.contiguous()
    counts = torch.zeros(num_groups, dtype=torch.int64, device=device)
    vptr = values.validity.contiguous().data_ptr() if values.validity is not None else None
    keep = [g, counts, values.validity, values.data]
    if fn == "count":
        rc = lib.au_agg_count(g.data_ptr(), n, vptr, counts.data_ptr(), sp)
        native.check(rc, "au_agg_count")
        return counts, counts
    v = values.data
    if v.dtype in (torch.int8, torch.int16, torch.bool):
        v = v.to(torch.int32)
    v = v.contiguous()
    keep.append(v)
    vtype = {torch.float64: 0, torch.int64: 1, torch.int32: 2, torch.float32: 3}[v.dtype]
    acc_f64 = v.dtype in (torch.float32, torch.float64)
    op = {"sum": 0, "avg": 0, "min": 1, "max": 2}[fn]
    if op == 0:
        init = 0.0 if acc_f64 else 0
    elif op == 1:
        init = float("inf") if acc_f64 else torch.iinfo(torch.int64).max
    else:
        init = float("-inf") if acc_f64 else torch.iinfo(torch.int64).min
    acc = torch.full((num_groups,), init,
                     dtype=torch.float64 if acc_f64 else torch.int64, device=device)
    rc = lib.au_agg_scatter(g.data_ptr(), n, vptr, v.data_ptr(), vtype, op,
                            1 if acc_f64 else 0, acc.data_ptr(), counts.data_ptr(),
                            num_groups, sp)
    native.check(rc, "au_agg_scatter")
    del keep
    if fn in ("min", "max"):
        # state keeps the input dtype (reference acc.rs semantics)
        tgt = values.data.dtype if values.data.dtype not in (torch.int8, torch.int16, torch.bool) else torch.int64
        if acc.dtype != tgt:
            acc = acc.to(tgt)
    elif fn in ("sum", "avg") and values.data.dtype == torch.float32:
        pass  # f64 accumulator is the widened sum dtype
    return acc, counts


def agg_scatter_multi(gids: torch.Tensor, num_groups: int, items):
    """Fused accumulate of several (values Column, fn) pairs in ONE kernel
    pass over the rows (gids read once, wave head-flags computed once).
    fn in sum/avg/min/max. -> [(acc, counts), ...] like agg_scatter."""
    device = gids.device
    if not items:
        return []
    if num_groups == 0 or not _use_native(device) or len(items) == 1:
        return [agg_scatter(gids, num_groups, c, f) for c, f in items]
    import numpy as np

    lib = native.lib()
    sp = native.stream_ptr(device)
    g = gids.contiguous()
    n = g.numel()
    descs = np.zeros((len(items), 6), dtype=np.int64)
    outs = []
    keep = [g]
    for j, (values, fn) in enumerate(items):
        v = values.data
        if v.dtype in (torch.int8, torch.int16, torch.bool):
            v = v.to(torch.int32)
        v = v.contiguous()
        keep.append(v)
        vtype = {torch.float64: 0, torch.int64: 1, torch.int32: 2,
                 torch.float32: 3}[v.dtype]
        acc_f64 = v.dtype in (torch.float32, torch.float64)
        op = {"sum": 0, "avg": 0, "min": 1, "max": 2}[fn]
        if op == 0:
            init = 0.0 if acc_f64 else 0
        elif op == 1:
            init = float("inf") if acc_f64 else torch.iinfo(torch.int64).max
        else:
            init = float("-inf") if acc_f64 else torch.iinfo(torch.int64).min
        acc = torch.full((num_groups,), init,
                         dtype=torch.float64 if acc_f64 else torch.int64,
                         device=device)
        counts = torch.zeros(num_groups, dtype=torch.int64, device=device)
        vptr = 0
        if values.validity is not None:
            val = values.validity.contiguous()
            keep.append(val)
            vptr = val.data_ptr()
        descs[j] = (v.data_ptr(), vptr, vtype, op, acc.data_ptr(),
                    counts.data_ptr())
        outs.append((values, fn, acc, counts))
    from .pinned import to_device

    darr = to_device(descs.reshape(-1), device)
    keep.append(darr)
    rc = lib.au_agg_multi(g.data_ptr(), n, darr.data_ptr(), len(items), sp)
    native.check(rc, "au_agg_multi")
    del keep
    res = []
    for values, fn, acc, counts in outs:
        if fn in ("min", "max"):
            tgt = values.data.dtype if values.data.dtype not in \
                (torch.int8, torch.int16, torch.bool) else torch.int64
            if acc.dtype != tgt:
                acc = acc.to(tgt)
        res.append((acc, counts))
    return res


def agg_count_star(gids: torch.Tensor, num_groups: int) -> torch.Tensor:
    """Per-group row counts (no values column): native wave-segmented
    count kernel on GPU, torch scatter_add_ on CPU."""
    device = gids.device
    if num_groups > 0 and _use_native(device):
        lib = native.lib()
        sp = native.stream_ptr(device)
        g = gids.contiguous()
        counts = torch.zeros(num_groups, dtype=torch.int64, device=device)
        rc = lib.au_agg_count(g.data_ptr(), g.numel(), None,
                              counts.data_ptr(), sp)
        native.check(rc, "au_agg_count")
        return counts
    cnt = torch.zeros(num_groups, dtype=torch.int64, device=device)
    if gids.numel():
        cnt.scatter_add_(0, gids, torch.ones_like(gids, dtype=torch.int64))
    return cnt


def agg_scatter(gids: torch.Tensor, num_groups: int, values: Column, fn: str):
    """Scatter-accumulate values into per-group accumulators.

    -> (acc tensor [G], count-of-valid [G] int64). Nulls are excluded.
    GPU: hand-written atomic scatter kernels; CPU: torch scatter_reduce.
    """
    device = gids.device
    if num_groups > 0 and _use_native(device):
        return _agg_scatter_native(gids, num_groups, values, fn)
    v = values.data
    valid = values.validity
    g = gids
    if valid is not None:
        keep = valid
        g = gids[keep]
        v = v[keep]
    cnt = torch.zeros(num_groups, dtype=torch.int64, device=device)
    cnt.scatter_add_(0, g, torch.ones_like(g, dtype=torch.int64))
    if fn == "count":
        return cnt, cnt
    if fn in ("sum", "avg"):
        if v.dtype in (torch.float32,):
            v = v.to(torch.float64)
        if v.dtype in (torch.int8, torch.int16, torch.int32):
            v = v.to(torch.int64)
        acc = torch.zeros(num_groups, dtype=v.dtype, device=device)
        acc.scatter_add_(0, g, v)
        return acc, cnt
    if fn in ("min", "max"):
        red = "amin" if fn == "min" else "amax"
        init = torch.finfo(v.dtype).max if v.dtype.is_floating_point else torch.iinfo(v.dtype).max
        if fn == "max":
            init = torch.finfo(v.dtype).min if v.dtype.is_floating_point else torch.iinfo(v.dtype).min
        acc = torch.full((num_groups,), init, dtype=v.dtype, device=device)
        acc.scatter_reduce_(0, g, v, reduce=red, include_self=True)
        return acc, cnt
    raise ValueError(fn)

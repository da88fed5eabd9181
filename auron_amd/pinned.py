"""Shared pinned host-buffer pool + fast small-upload helper.

hipHostMalloc costs ~0.1-1 ms per call and pageable H2D copies stage
through an internal bounce buffer, so both the parquet scan staging
buffers and the per-kernel descriptor arrays (hash/join/agg/decode
descs) go through this pool: acquire a pinned slab, copy, async H2D,
release with a recorded event so the slab is only reused once the copy
finished."""
from __future__ import annotations

import threading

import numpy as np
import torch


class PinnedPool:
    MAX_ENTRIES = 32

    def __init__(self):
        self._free: list = []  # (capacity, base_tensor, event|None)
        self._lock = threading.Lock()
        self._events: dict = {}  # id(base) -> reusable hip event

    def acquire(self, nbytes: int):
        with self._lock:
            for i, (cap, t, ev) in enumerate(self._free):
                if cap >= nbytes and (ev is None or ev.query()):
                    self._free.pop(i)
                    return t, t[:nbytes]
        cap = 1 << max(12, (int(nbytes) - 1).bit_length())
        t = torch.empty(cap, dtype=torch.uint8, pin_memory=torch.cuda.is_available())
        return t, t[:nbytes]

    def release(self, base: torch.Tensor, device) -> None:
        ev = None
        if torch.device(device).type == "cuda":
            ev = self._events.get(id(base))
            if ev is None:
                ev = torch.cuda.Event()
                self._events[id(base)] = ev
            ev.record(torch.cuda.current_stream(device))
        with self._lock:
            if len(self._free) < self.MAX_ENTRIES:
                self._free.append((base.numel(), base, ev))
            else:
                self._events.pop(id(base), None)


POOL = PinnedPool()


_copy_stream = None
_SPILL_BOUNCE = 128 << 20


def copy_stream():
    """Dedicated D2H copy stream (spill = hipMemcpyAsync on a
    side stream, never the compute stream)."""
    global _copy_stream
    if _copy_stream is None:
        _copy_stream = torch.cuda.Stream()
    return _copy_stream


def d2h_staged(t: torch.Tensor) -> torch.Tensor:
    """Device -> pageable host through a pooled pinned bounce buffer on
    the copy stream. Pageable d2h copies in HIP stage through a tiny
    internal buffer; an explicit 128 MB pinned bounce keeps the transfer
    at link speed without pinning the full payload."""
    if t.device.type != "cuda":
        return t
    src = t.contiguous()
    nbytes = src.numel() * src.element_size()
    if nbytes == 0:
        return torch.empty(t.shape, dtype=t.dtype)
    flat = src.reshape(-1).view(torch.uint8)
    dst = torch.empty(nbytes, dtype=torch.uint8)
    s = copy_stream()
    base, bounce = POOL.acquire(min(nbytes, _SPILL_BOUNCE))
    cap = bounce.numel()
    s.wait_stream(torch.cuda.current_stream())
    with torch.cuda.stream(s):
        for off in range(0, nbytes, cap):
            n = min(cap, nbytes - off)
            bounce[:n].copy_(flat[off:off + n], non_blocking=True)
            s.synchronize()
            dst[off:off + n].copy_(bounce[:n])
    POOL.release(base, t.device)
    return dst.view(t.dtype).reshape(t.shape)


def batch_to_host(batch) -> "object":
    """RecordBatch device->host via the staged path (spill tier 1)."""
    from .column import Column, RecordBatch

    cols = []
    for c in batch.columns:
        cols.append(Column(
            c.dtype, d2h_staged(c.data),
            None if c.validity is None else d2h_staged(c.validity),
            None if c.offsets is None else d2h_staged(c.offsets)))
    return RecordBatch(batch.names, cols)


def to_device(arr: np.ndarray, device) -> torch.Tensor:
    """Upload a host array; pooled pinned slab for payloads big enough to
    beat torch's own small-copy staging (tiny descriptor arrays go the
    plain route — pool bookkeeping costs more than it saves there)."""
    flat = np.ascontiguousarray(arr).view(np.uint8).reshape(-1)
    dev = torch.device(device)
    if dev.type != "cuda" or flat.nbytes < (8 << 10):
        return torch.from_numpy(flat.copy()).to(dev)
    base, view = POOL.acquire(flat.nbytes)
    view.numpy()[:] = flat
    out = view.to(dev, non_blocking=True)
    POOL.release(base, dev)
    return out

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

"""Memory manager: HBM budget, consumer registry, spill tiers.

Role parity: auron-memmgr (/root/reference/native-engine/auron-memmgr/
 src/lib.rs — MemManager::init :46, register_consumer :82, fair-share
 spill policy :308-428; spill.rs Spill tiers). MI355X design: the managed
resource is device HBM3E (288 GB/GPU x memoryFraction); spill tier 1 is
pinned host DRAM via async D2H on a dedicated copy stream, tier 2 is a
disk file. On CPU (CI) the "device" tier is host RAM and spill goes
straight to disk, so the whole policy is exercised without a GPU.
"""
from __future__ import annotations

import os
import tempfile
import threading
import time
from typing import Dict, List, Optional

import torch

from .column import Column, RecordBatch


def _col_bytes(c: Column) -> int:
    total = c.data.numel() * c.data.element_size()
    if c.validity is not None:
        total += c.validity.numel()
    if c.offsets is not None:
        total += c.offsets.numel() * 8
    return total


def _batch_bytes(b: RecordBatch) -> int:
    return sum(_col_bytes(c) for c in b.columns)


class BatchHolder:
    """A registered consumer holding materialized batches (the reference's
    MemConsumer). Spill moves the payload down a tier; batches() restores."""

    def __init__(self, mgr: "MemManager", tag: str, batches: List[RecordBatch]):
        self.mgr = mgr
        self.tag = tag
        self._batches: Optional[List[RecordBatch]] = batches
        self._spilled_host: Optional[List[RecordBatch]] = None
        self._spill_file: Optional[str] = None
        self.bytes = sum(_batch_bytes(b) for b in batches)
        self.device = batches[0].device if batches else torch.device("cpu")
        self.last_touch = time.monotonic()

    @property
    def resident(self) -> bool:
        return self._batches is not None

    @property
    def host_resident(self) -> bool:
        return self._spilled_host is not None

    def refresh(self):
        """Re-account after the caller mutated the resident batch list
        (update_mem_used analogue); may trigger spills elsewhere."""
        if self._batches is not None:
            self.bytes = sum(_batch_bytes(b) for b in self._batches)
            self.last_touch = time.monotonic()
            self.mgr.reserve(0)

    def spill(self) -> int:
        """Move payload one tier down. Returns bytes released."""
        if self._batches is None:
            return 0
        if self.device.type == "cuda":
            # tier 1: device -> host via the pinned bounce buffer on the
            # dedicated copy stream (SURVEY §3.5 hipMemcpyAsync mapping)
            from .pinned import batch_to_host

            self._spilled_host = [batch_to_host(b) for b in self._batches]
            self.mgr.metrics["spill_d2h_bytes"] = self.mgr.metrics.get("spill_d2h_bytes", 0) + self.bytes
        else:
            # CPU mode: straight to disk so CI exercises the file tier
            self._spill_to_disk(self._batches)
        self._batches = None
        self.mgr._host_pressure()
        return self.bytes

    def spill_host_to_disk(self) -> int:
        """Tier 2: push a host-resident payload to a disk file (keeps host
        spill bytes bounded under mgr.host_budget — advisor finding r1)."""
        if self._spilled_host is None:
            return 0
        self._spill_to_disk(self._spilled_host)
        self._spilled_host = None
        return self.bytes

    def _spill_to_disk(self, batches: List[RecordBatch]):
        from .exchange import pack_batch

        fd, path = tempfile.mkstemp(prefix=f"auron-spill-{self.tag}-", suffix=".bin",
                                    dir=self.mgr.spill_dir)
        metas = []
        with os.fdopen(fd, "wb") as f:
            for b in batches:
                meta, buf = pack_batch(b.to("cpu"), "cpu")
                raw = buf.numpy().tobytes()
                metas.append((meta, len(raw)))
                f.write(raw)
        self._spill_file = path
        self._disk_metas = metas
        self.mgr.metrics["spill_disk_bytes"] = self.mgr.metrics.get("spill_disk_bytes", 0) + self.bytes

    def _restore_from_disk(self) -> List[RecordBatch]:
        from .exchange import unpack_batch

        out = []
        with open(self._spill_file, "rb") as f:
            for meta, n in self._disk_metas:
                raw = f.read(n)
                buf = torch.frombuffer(bytearray(raw), dtype=torch.uint8)
                out.append(unpack_batch(meta, buf))
        os.unlink(self._spill_file)
        self._spill_file = None
        return out

    def batches(self) -> List[RecordBatch]:
        """Restore (if spilled) and return the payload resident again."""
        self.last_touch = time.monotonic()
        if self._batches is not None:
            return self._batches
        self.mgr.reserve(self.bytes, exclude=self)
        if self._spilled_host is not None:
            self._batches = [b.to(self.device) for b in self._spilled_host]
            self._spilled_host = None
        elif self._spill_file is not None:
            self._batches = [b.to(self.device) for b in self._restore_from_disk()]
        else:
            self._batches = []
        return self._batches

    def release(self):
        self.mgr.unregister(self)
        self._batches = None
        self._spilled_host = None
        if self._spill_file:
            try:
                os.unlink(self._spill_file)
            except OSError:
                pass


class MemManager:
    """Tracks resident bytes of registered holders against a budget and
    spills on pressure with the reference's fair-share policy
    (lib.rs:365 divides the budget by spillable count): holders above
    their fair share spill first, largest overage first, so one huge
    consumer cannot evict many small recently-used ones; remaining
    pressure falls back to LRU victims."""

    def __init__(self, budget_bytes: Optional[int] = None, fraction: float = 0.8,
                 spill_dir: Optional[str] = None, host_budget_bytes: Optional[int] = None):
        if budget_bytes is None:
            from .config import MEM_BUDGET, AuronConf

            _mb = AuronConf().get(MEM_BUDGET)
            env = str(_mb) if _mb else None
            if env is not None:
                budget_bytes = int(env)
            elif torch.cuda.is_available():
                free, total = torch.cuda.mem_get_info()
                budget_bytes = int(total * fraction)
            else:
                budget_bytes = 16 << 30
        self.budget = budget_bytes
        if host_budget_bytes is None:
            from .config import HOST_SPILL_BUDGET, AuronConf

            host_budget_bytes = int(AuronConf().get(HOST_SPILL_BUDGET))
        self.host_budget = host_budget_bytes
        if spill_dir is None:
            from .config import SPILL_DIR, AuronConf

            spill_dir = AuronConf().get(SPILL_DIR) or None
        self.spill_dir = spill_dir or tempfile.gettempdir()
        self._holders: List[BatchHolder] = []
        self._lock = threading.Lock()
        self.metrics: Dict[str, int] = {}

    def _host_pressure(self):
        """Enforce the host-DRAM spill budget: LRU host-resident holders go
        down to the disk tier (reference: memmgr/spill.rs two-tier Spill)."""
        hosts = [h for h in self._holders if h.host_resident]
        used = sum(h.bytes for h in hosts)
        if used <= self.host_budget:
            return
        for h in sorted(hosts, key=lambda h: h.last_touch):
            used -= h.spill_host_to_disk()
            self.metrics["spill_host2disk_count"] = \
                self.metrics.get("spill_host2disk_count", 0) + 1
            if used <= self.host_budget:
                return

    def resident_bytes(self) -> int:
        return sum(h.bytes for h in self._holders if h.resident)

    def register(self, tag: str, batches: List[RecordBatch]) -> BatchHolder:
        h = BatchHolder(self, tag, batches)
        with self._lock:
            self._holders.append(h)
        # no exclusion: a holder larger than the whole budget spills itself
        # immediately (it is idle until its consumer touches it again)
        self.reserve(0)
        return h

    def unregister(self, h: BatchHolder):
        with self._lock:
            if h in self._holders:
                self._holders.remove(h)

    def reserve(self, nbytes: int, exclude: Optional[BatchHolder] = None):
        """Ensure `nbytes` headroom: fair-share victims first, then LRU."""
        with self._lock:
            used = self.resident_bytes()
            if used + nbytes <= self.budget:
                return
            cands = [h for h in self._holders if h.resident and h is not exclude]
            fair = self.budget / max(len(cands), 1)
            over = sorted((h for h in cands if h.bytes > fair),
                          key=lambda h: h.bytes, reverse=True)
            lru = sorted((h for h in cands if h.bytes <= fair),
                         key=lambda h: h.last_touch)
            for v in over + lru:
                used -= v.spill()
                self.metrics["spill_count"] = self.metrics.get("spill_count", 0) + 1
                if used + nbytes <= self.budget:
                    return

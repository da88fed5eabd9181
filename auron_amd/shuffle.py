"""Persistent shuffle files: stage-boundary spill + retry contract.

Role parity: the reference's sort-based shuffle writer and IPC reader
pair (shuffle_writer_exec.rs + ipc_reader_exec.rs + the Spark-side
AuronShuffleManager), which persist one data file + one index file per
map task so reduce tasks (and stage retries) can re-read any partition
segment without re-running the map stage.

On a single MI355X node the default exchange is in-flight over xGMI
(exchange.py) and never touches disk; this module is the durable
alternative the executor switches to when `AURON_SHUFFLE_PERSIST=1` (or
`Exchange.persist=True`): each rank writes its packed per-destination
buffers to `<dir>/stage-<id>/map-<rank>.{data,index}` and every rank
then reads its own segment from every map file. Files outlive the
exchange, so a reader can be re-run (stage retry) without the writers.

File format:
  data file:  concat of per-dest segments, each = pack_batch payload
              (zlib-compressed when codec="zlib" — the ipc_compression.rs
              role; in-flight xGMI exchange stays raw on purpose, disk is
              where a codec pays)
  index file: msgpack {"version", "world", "codec", "offsets": [w+1],
              "metas": [w]} (offsets[d]..offsets[d+1] is dest d's byte
              range; metas[d] is the pack_batch meta dict or None for an
              empty dest)
"""
from __future__ import annotations

import os
import zlib
from typing import List, Optional

import msgpack
import torch

from .column import RecordBatch
from .exchange import pack_batch, unpack_batch

_VERSION = 1


def _codec() -> str:
    return os.environ.get("AURON_SHUFFLE_CODEC", "none")


def _stage_dir(root: str, stage_id: str) -> str:
    return os.path.join(root, f"stage-{stage_id}")


class ShuffleWriter:
    """One map task's persistent shuffle output (data + index file)."""

    def __init__(self, root: str, stage_id: str, map_rank: int):
        self.dir = _stage_dir(root, stage_id)
        os.makedirs(self.dir, exist_ok=True)
        self.map_rank = map_rank
        self._data_path = os.path.join(self.dir, f"map-{map_rank}.data")
        self._index_path = os.path.join(self.dir, f"map-{map_rank}.index")

    def write(self, batches_by_dest: List[Optional[RecordBatch]]) -> None:
        codec = _codec()
        offsets = [0]
        metas = []
        tmp_data = self._data_path + ".tmp"
        with open(tmp_data, "wb") as f:
            for b in batches_by_dest:
                if b is None or b.num_rows == 0:
                    metas.append(None)
                    offsets.append(offsets[-1])
                    continue
                meta, buf = pack_batch(b, "cpu")
                raw = buf.numpy().tobytes()
                if codec == "zlib":
                    raw = zlib.compress(raw, 1)
                f.write(raw)
                metas.append(meta)
                offsets.append(offsets[-1] + len(raw))
        index = {"version": _VERSION, "world": len(batches_by_dest),
                 "codec": codec, "offsets": offsets, "metas": metas}
        tmp_idx = self._index_path + ".tmp"
        with open(tmp_idx, "wb") as f:
            f.write(msgpack.packb(index, use_bin_type=True))
        # rename-commit so a torn write is never visible to readers
        os.replace(tmp_data, self._data_path)
        os.replace(tmp_idx, self._index_path)


class ShuffleReader:
    """Reads one reduce partition's segments from every map output."""

    def __init__(self, root: str, stage_id: str):
        self.dir = _stage_dir(root, stage_id)

    def map_ranks(self) -> List[int]:
        out = []
        for fn in os.listdir(self.dir):
            if fn.startswith("map-") and fn.endswith(".index"):
                out.append(int(fn[4:-6]))
        return sorted(out)

    def read_partition(self, dest: int, device="cpu") -> List[RecordBatch]:
        out = []
        for m in self.map_ranks():
            idx_path = os.path.join(self.dir, f"map-{m}.index")
            with open(idx_path, "rb") as f:
                index = msgpack.unpackb(f.read(), raw=False, strict_map_key=False)
            assert index["version"] == _VERSION
            meta = index["metas"][dest]
            if meta is None:
                continue
            lo, hi = index["offsets"][dest], index["offsets"][dest + 1]
            with open(os.path.join(self.dir, f"map-{m}.data"), "rb") as f:
                f.seek(lo)
                raw = f.read(hi - lo)
            if index.get("codec", "none") == "zlib":
                raw = zlib.decompress(raw)
            buf = torch.frombuffer(bytearray(raw), dtype=torch.uint8)
            b = unpack_batch(meta, buf)
            out.append(b.to(device) if str(device) != "cpu" else b)
        return out


def shuffle_root() -> str:
    return os.environ.get("AURON_SHUFFLE_DIR",
                          os.path.join(os.getcwd(), ".auron_shuffle"))


def persist_enabled(node_flag: bool) -> bool:
    return node_flag or os.environ.get("AURON_SHUFFLE_PERSIST", "0") == "1"

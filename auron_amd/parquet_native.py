"""Native Parquet read path: host page-header parse -> device page decode.

Role parity: the reference's parquet scan (parquet_exec.rs + arrow-rs
parquet decoder, CPU). MI355X design: the host only touches metadata —
footer via pyarrow, page headers via a minimal thrift-compact parser —
and ships the raw column-chunk bytes to HBM once; gfx950 kernels decode
RLE/bit-packed definition levels, RLE dictionary indices, and PLAIN
values into device columns (csrc/parquet.hip). Unsupported shapes
(nested lists/maps/structs, repeated levels) fall back to the pyarrow
host path per column.

Supported fast path: data pages v1+v2, UNCOMPRESSED/snappy/zstd/gzip
chunks (host page decompression into the staging buffer's extra
region), optional-level (max_def_level<=1) columns; PLAIN,
RLE_DICTIONARY and DELTA_BINARY_PACKED (host delta unpack -> PLAIN)
INT32/INT64/FLOAT/DOUBLE; BYTE_ARRAY strings both dictionary-encoded
(dictionary page parsed host-side, per-row bytes gathered on device)
and PLAIN (host offset walk, device byte gather).
"""
from __future__ import annotations

import os
from dataclasses import dataclass
from typing import Dict, List, Optional, Tuple

import numpy as np
import torch

from . import dtypes
from .column import Column, RecordBatch

# ---------------------------------------------------------------- thrift
# Minimal thrift compact-protocol reader (PageHeader only).
_CT_STOP = 0
_CT_BOOL_TRUE = 1
_CT_BOOL_FALSE = 2
_CT_BYTE = 3
_CT_I16 = 4
_CT_I32 = 5
_CT_I64 = 6
_CT_DOUBLE = 7
_CT_BINARY = 8
_CT_LIST = 9
_CT_SET = 10
_CT_MAP = 11
_CT_STRUCT = 12


def _uvarint(buf: memoryview, pos: int) -> Tuple[int, int]:
    result = 0
    shift = 0
    while True:
        b = buf[pos]
        pos += 1
        result |= (b & 0x7F) << shift
        if not b & 0x80:
            return result, pos
        shift += 7


def _zigzag(v: int) -> int:
    return (v >> 1) ^ -(v & 1)


def _skip(buf: memoryview, pos: int, ctype: int) -> int:
    if ctype in (_CT_BOOL_TRUE, _CT_BOOL_FALSE):
        return pos
    if ctype == _CT_BYTE:
        return pos + 1
    if ctype in (_CT_I16, _CT_I32, _CT_I64):
        v, pos = _uvarint(buf, pos)
        return pos
    if ctype == _CT_DOUBLE:
        return pos + 8
    if ctype == _CT_BINARY:
        n, pos = _uvarint(buf, pos)
        return pos + n
    if ctype in (_CT_LIST, _CT_SET):
        b = buf[pos]
        pos += 1
        size = b >> 4
        etype = b & 0x0F
        if size == 15:
            size, pos = _uvarint(buf, pos)
        for _ in range(size):
            pos = _skip(buf, pos, etype)
        return pos
    if ctype == _CT_STRUCT:
        fid = 0
        while True:
            b = buf[pos]
            pos += 1
            if b == _CT_STOP:
                return pos
            delta = b >> 4
            ft = b & 0x0F
            if delta == 0:
                v, pos = _uvarint(buf, pos)
                fid = _zigzag(v)
            else:
                fid += delta
            pos = _skip(buf, pos, ft)
    if ctype == _CT_MAP:
        n, pos = _uvarint(buf, pos)
        if n:
            kv = buf[pos]
            pos += 1
            for _ in range(n):
                pos = _skip(buf, pos, kv >> 4)
                pos = _skip(buf, pos, kv & 0x0F)
        return pos
    raise ValueError(f"thrift type {ctype}")


def _read_struct_fields(buf: memoryview, pos: int, want: Dict[int, str]) -> Tuple[dict, int]:
    """Read a compact struct, capturing integer fields in `want` and the
    byte range of any wanted struct fields."""
    out: dict = {}
    fid = 0
    while True:
        b = buf[pos]
        pos += 1
        if b == _CT_STOP:
            return out, pos
        delta = b >> 4
        ft = b & 0x0F
        if delta == 0:
            v, pos = _uvarint(buf, pos)
            fid = _zigzag(v)
        else:
            fid += delta
        name = want.get(fid)
        if name is None:
            pos = _skip(buf, pos, ft)
        elif ft in (_CT_I16, _CT_I32, _CT_I64):
            v, pos = _uvarint(buf, pos)
            out[name] = _zigzag(v)
        elif ft == _CT_STRUCT:
            start = pos
            pos = _skip(buf, pos, _CT_STRUCT)
            out[name] = (start, pos)
        elif ft in (_CT_BOOL_TRUE, _CT_BOOL_FALSE):
            out[name] = ft == _CT_BOOL_TRUE
        else:
            pos = _skip(buf, pos, ft)
    return out, pos


@dataclass
class PageDesc:
    n_values: int
    def_off: int  # byte offset of RLE def-level payload (after 4-byte len), -1 if none
    def_len: int
    values_off: int  # byte offset of values ([bw][rle-indices] for dict pages)
    values_len: int
    row_start: int
    encoding: int = 0  # 0=PLAIN, 8=RLE_DICTIONARY (PLAIN_DICTIONARY folded in)


@dataclass
class ChunkPages:
    """One column chunk's decoded page map (+ dictionary page, if any)."""
    pages: List[PageDesc]
    dict_off: int = -1
    dict_len: int = 0
    dict_nvals: int = 0
    # lazy host-parsed BYTE_ARRAY dictionary layout (absolute offsets into
    # the staged buffer, entry byte lengths)
    dict_str_offs: Optional[np.ndarray] = None
    dict_str_lens: Optional[np.ndarray] = None
    # host-parsed def-level runs (au_host_rle1_parse): [N,4] int32 of
    # (abs out row, count, abs src byte, rep|-1); device expansion only
    runs_np: Optional[np.ndarray] = None
    # per-page RLE index bit widths (host byte, cached so HBM-cache hits
    # never need the host buffer)
    idx_bws: Optional[list] = None
    # lazy host-parsed PLAIN BYTE_ARRAY layout: per PRESENT value
    # (absolute offset, byte length) in page order
    plain_str_offs: Optional[np.ndarray] = None
    plain_str_lens: Optional[np.ndarray] = None

    @property
    def is_dict(self) -> bool:
        return self.dict_off >= 0


class _ExtraAlloc:
    """Cursor over the decompressed-page region appended after the raw
    staging buffer, plus the job list that refills it on every read
    (host page decompression -> device decode; parquet_exec.rs analogue
    of arrow-rs's page decompression)."""

    def __init__(self, base: int):
        self.base = base
        self.cursor = base
        self.jobs: List[tuple] = []  # ("d", src, clen, dest, ulen, codec) | ("c", src, n, dest)

    def decompress(self, src: int, clen: int, ulen: int, codec: str) -> int:
        dest = self.cursor
        self.jobs.append(("d", src, clen, dest, ulen, codec))
        self.cursor += ulen
        return dest

    def copy(self, src: int, n: int) -> int:
        dest = self.cursor
        self.jobs.append(("c", src, n, dest))
        self.cursor += n
        return dest

    def delta(self, src: int, length: int, nv: int, esize: int) -> int:
        """DELTA_BINARY_PACKED page payload -> PLAIN values at dest
        (host decode, au_host_delta_unpack)."""
        dest = self.cursor
        self.jobs.append(("delta", src, length, dest, nv, esize))
        self.cursor += nv * esize
        return dest


def _codec_decompress(codec: str, raw: bytes, ulen: int) -> bytes:
    import pyarrow as pa

    return pa.Codec(codec.lower()).decompress(raw, decompressed_size=ulen) \
        .to_pybytes()


def parse_pages(buf: np.ndarray, chunk_off: int, chunk_len: int,
                num_values: int, has_def: bool, codec: Optional[str] = None,
                extra: Optional[_ExtraAlloc] = None,
                esize: int = 0) -> Optional[ChunkPages]:
    """Parse page headers of one column chunk (v1 + v2 data pages).

    Supports PLAIN data pages and RLE_DICTIONARY/PLAIN_DICTIONARY data
    pages with a PLAIN dictionary page; snappy/zstd/gzip page payloads
    are routed through the extra-region decompress jobs. Chunks mixing
    dictionary and plain data pages fall back to the host path."""
    mv = memoryview(buf)
    pos = chunk_off
    end = chunk_off + chunk_len
    ck = ChunkPages([])
    row = 0
    while row < num_values and pos < end:
        hdr, pos2 = _read_struct_fields(mv, pos, {
            1: "type", 2: "uncompressed_page_size", 3: "compressed_page_size",
            5: "data_page_header", 7: "dictionary_page_header",
            8: "data_page_header_v2",
        })
        page_data = pos2
        page_len = hdr.get("compressed_page_size", 0)
        ulen = hdr.get("uncompressed_page_size", page_len)
        ptype = hdr.get("type", -1)
        if ptype == 2:  # dictionary page (PLAIN values)
            dph_range = hdr.get("dictionary_page_header")
            if dph_range is None or ck.dict_off >= 0:
                return None
            dph, _ = _read_struct_fields(mv, dph_range[0], {
                1: "num_values", 2: "encoding"})
            if dph.get("encoding", 0) not in (0, 2):  # PLAIN / PLAIN_DICTIONARY
                return None
            if codec:
                ck.dict_off = extra.decompress(page_data, page_len, ulen, codec)
                ck.dict_len = ulen
            else:
                ck.dict_off = page_data
                ck.dict_len = page_len
            ck.dict_nvals = dph["num_values"]
            pos = page_data + page_len
            continue
        if ptype == 3:  # DATA_PAGE_V2
            dph_range = hdr.get("data_page_header_v2")
            if dph_range is None:
                return None
            dph, _ = _read_struct_fields(mv, dph_range[0], {
                1: "num_values", 4: "encoding", 5: "def_len", 6: "rep_len",
                7: "is_compressed"})
            enc = dph.get("encoding", 0)
            if enc in (2, 8):
                enc = 8
            elif enc == 5 and esize in (4, 8) and extra is not None:
                pass  # DELTA_BINARY_PACKED: host-decoded to PLAIN below
            elif enc != 0:
                return None
            nv = dph["num_values"]
            dlen = dph.get("def_len", 0) if has_def else 0
            rlen = dph.get("rep_len", 0)
            if rlen:
                return None  # repeated columns stay on the host path
            vals_raw = page_data + dlen
            vals_clen = page_len - dlen
            vals_ulen = ulen - dlen
            compressed = bool(dph.get("is_compressed", True)) and bool(codec)
            if compressed:
                # v2 keeps levels uncompressed; only values are coded
                def_off = extra.copy(page_data, dlen) if dlen else -1
                values_off = extra.decompress(vals_raw, vals_clen, vals_ulen,
                                              codec)
                values_len = vals_ulen
            else:
                def_off = page_data if dlen else -1
                values_off = vals_raw
                values_len = vals_clen
            if enc == 5:
                values_off = extra.delta(values_off, values_len, nv, esize)
                values_len = nv * esize
                enc = 0
            ck.pages.append(PageDesc(nv, def_off, dlen, values_off, values_len,
                                     row, enc))
            row += nv
            pos = page_data + page_len
            continue
        if ptype != 0:  # index page etc.
            return None
        dph_range = hdr.get("data_page_header")
        if dph_range is None:
            return None
        dph, _ = _read_struct_fields(mv, dph_range[0], {
            1: "num_values", 2: "encoding", 3: "def_enc", 4: "rep_enc"})
        enc = dph.get("encoding", 0)
        if enc in (2, 8):
            enc = 8
            if ck.dict_off < 0:
                return None
        elif enc == 5 and esize in (4, 8) and extra is not None:
            pass  # DELTA_BINARY_PACKED: host-decoded to PLAIN below
        elif enc != 0:
            return None
        nv = dph["num_values"]
        if codec:
            # v1 compresses the whole payload (def prefix included); the
            # def length lives inside, so peek via one parse-time pass
            payload = _codec_decompress(codec, bytes(mv[page_data:page_data + page_len]), ulen)
            dest = extra.decompress(page_data, page_len, ulen, codec)
            if has_def:
                dlen = int.from_bytes(payload[:4], "little")
                ck.pages.append(PageDesc(nv, dest + 4, dlen, dest + 4 + dlen,
                                         ulen - 4 - dlen, row, enc))
            else:
                ck.pages.append(PageDesc(nv, -1, 0, dest, ulen, row, enc))
        elif has_def:
            dlen = int.from_bytes(bytes(mv[page_data:page_data + 4]), "little")
            def_off = page_data + 4
            values_off = def_off + dlen
            values_len = page_len - 4 - dlen
            ck.pages.append(PageDesc(nv, def_off, dlen, values_off, values_len,
                                     row, enc))
        else:
            ck.pages.append(PageDesc(nv, -1, 0, page_data, page_len, row, enc))
        if enc == 5:
            p = ck.pages[-1]
            dest = extra.delta(p.values_off, p.values_len, nv, esize)
            ck.pages[-1] = PageDesc(p.n_values, p.def_off, p.def_len, dest,
                                    nv * esize, p.row_start, 0)
        row += nv
        pos = page_data + page_len
    if row != num_values or not ck.pages:
        return None
    encs = {p.encoding for p in ck.pages}
    if len(encs) > 1:  # mixed dict/plain chunk -> host fallback
        return None
    if encs == {8} and ck.dict_off < 0:
        return None
    return ck


# ------------------------------------------------------- numpy reference
def rle1_decode_np(data: np.ndarray, n: int) -> np.ndarray:
    """RLE/bit-packed hybrid, bit width 1 (definition levels)."""
    out = np.zeros(n, dtype=np.uint8)
    mv = memoryview(data)
    pos = 0
    i = 0
    while i < n:
        header, pos = _uvarint(mv, pos)
        if header & 1:  # bit-packed literal run: (header>>1) groups of 8
            ngroups = header >> 1
            bits = np.unpackbits(data[pos:pos + ngroups], bitorder="little")
            take = min(ngroups * 8, n - i)
            out[i:i + take] = bits[:take]
            pos += ngroups
            i += take
        else:  # repeated run
            cnt = header >> 1
            val = data[pos]
            pos += 1
            out[i:i + cnt] = val & 1
            i += cnt
    return out


def rle_idx_decode_np(data: np.ndarray, n: int, bw: int) -> np.ndarray:
    """RLE/bit-packed hybrid, arbitrary bit width (dictionary indices)."""
    out = np.zeros(n, dtype=np.int32)
    mv = memoryview(data)
    pos = 0
    i = 0
    vbytes = (bw + 7) >> 3
    while i < n and pos < len(data):
        header, pos = _uvarint(mv, pos)
        if header & 1:  # bit-packed: (header>>1) groups of 8, bw bits each
            ngroups = header >> 1
            nbytes = ngroups * bw
            bits = np.unpackbits(data[pos:pos + nbytes], bitorder="little")
            vals = bits.reshape(-1, bw).astype(np.int64)
            vals = (vals * (1 << np.arange(bw, dtype=np.int64))).sum(axis=1)
            take = min(ngroups * 8, n - i)
            out[i:i + take] = vals[:take]
            pos += nbytes
            i += take
        else:
            cnt = header >> 1
            v = int.from_bytes(bytes(mv[pos:pos + vbytes]), "little")
            pos += vbytes
            out[i:i + cnt] = v
            i += cnt
    return out


_PHYS_NP = {"INT32": np.int32, "INT64": np.int64, "FLOAT": np.float32, "DOUBLE": np.float64}
_PHYS_CODE = {"INT32": 0, "INT64": 1, "FLOAT": 2, "DOUBLE": 3}


def parse_dict_strings(buf: np.ndarray, ck: ChunkPages) -> None:
    """Parse a BYTE_ARRAY dictionary page into (abs offsets, lengths)."""
    if ck.dict_str_offs is not None:
        return
    offs = np.empty(ck.dict_nvals, dtype=np.int64)
    lens = np.empty(ck.dict_nvals, dtype=np.int64)
    pos = ck.dict_off
    for j in range(ck.dict_nvals):
        ln = int.from_bytes(bytes(memoryview(buf)[pos:pos + 4]), "little")
        offs[j] = pos + 4
        lens[j] = ln
        pos += 4 + ln
    ck.dict_str_offs = offs
    ck.dict_str_lens = lens


def parse_plain_strings(buf: np.ndarray, ck: ChunkPages) -> None:
    """Host walk of PLAIN BYTE_ARRAY page payloads -> per-present-value
    (absolute offset, length) arrays (au_host_plainba_parse; the
    variable-length [len][bytes] chain is inherently serial, like
    arrow-rs's offset materialization)."""
    if ck.plain_str_offs is not None:
        return
    from . import native

    lib = native.host_lib()
    total = sum(p.n_values for p in ck.pages)
    offs = np.empty(total, dtype=np.int64)
    lens = np.empty(total, dtype=np.int32)
    base = buf.ctypes.data
    n = 0
    for p in ck.pages:
        got = lib.au_host_plainba_parse(
            base, p.values_off, p.values_len,
            offs.ctypes.data + 8 * n, lens.ctypes.data + 4 * n, total - n)
        if got < 0:
            raise ValueError("corrupt PLAIN byte-array page")
        n += got
    ck.plain_str_offs = offs[:n]
    ck.plain_str_lens = lens[:n].astype(np.int64)


def _chunk_indices_np(buf: np.ndarray, ck: ChunkPages,
                      validity: Optional[np.ndarray]) -> np.ndarray:
    """Decode all dict-index pages of a chunk into one compact array."""
    parts = []
    for p in ck.pages:
        if validity is not None:
            nvalid = int(validity[p.row_start:p.row_start + p.n_values].sum())
        else:
            nvalid = p.n_values
        bw = int(buf[p.values_off])  # host reference path always has buf
        parts.append(rle_idx_decode_np(
            buf[p.values_off + 1:p.values_off + p.values_len], nvalid, bw))
    return np.concatenate(parts) if len(parts) > 1 else parts[0]


def decode_chunk_np_dict(buf: np.ndarray, ck: ChunkPages, num_values: int,
                         phys: str):
    """Host reference decoder for dictionary-encoded chunks.

    Returns (data, validity, offsets): offsets is None for fixed-width
    physical types; for BYTE_ARRAY data is the flat byte array."""
    has_def = ck.pages[0].def_off >= 0
    validity = None
    if has_def:
        validity = np.zeros(num_values, dtype=np.uint8)
        for p in ck.pages:
            validity[p.row_start:p.row_start + p.n_values] = rle1_decode_np(
                buf[p.def_off:p.def_off + p.def_len], p.n_values)
    idx = _chunk_indices_np(buf, ck, validity)
    if phys == "BYTE_ARRAY":
        parse_dict_strings(buf, ck)
        row_idx = np.zeros(num_values, dtype=np.int64)
        vmask = validity.astype(bool) if validity is not None else np.ones(num_values, dtype=bool)
        row_idx[vmask] = idx
        lens = np.where(vmask, ck.dict_str_lens[row_idx], 0)
        offsets = np.zeros(num_values + 1, dtype=np.int64)
        np.cumsum(lens, out=offsets[1:])
        total = int(offsets[-1])
        data = np.empty(total, dtype=np.uint8)
        starts = ck.dict_str_offs[row_idx]
        pos = 0
        for r in range(num_values):
            ln = lens[r]
            if ln:
                data[pos:pos + ln] = buf[starts[r]:starts[r] + ln]
                pos += ln
        return data, validity, offsets
    npdt = _PHYS_NP[phys]
    esize = npdt().itemsize
    dvals = np.frombuffer(buf, dtype=npdt, count=ck.dict_nvals,
                          offset=ck.dict_off) if ck.dict_off % esize == 0 else \
        np.frombuffer(bytes(buf[ck.dict_off:ck.dict_off + ck.dict_nvals * esize]),
                      dtype=npdt)
    out = np.zeros(num_values, dtype=npdt)
    if validity is not None:
        out[validity.astype(bool)] = dvals[idx]
    else:
        out[:] = dvals[idx]
    return out, validity, None


def decode_chunk_np(buf: np.ndarray, pages: List[PageDesc], num_values: int,
                    phys: str) -> Tuple[np.ndarray, Optional[np.ndarray]]:
    """Host reference decoder (correctness oracle for the HIP kernels)."""
    npdt = _PHYS_NP[phys]
    esize = npdt().itemsize
    out = np.zeros(num_values, dtype=npdt)
    validity = None
    has_def = pages[0].def_off >= 0
    if has_def:
        validity = np.zeros(num_values, dtype=np.uint8)
    for p in pages:
        if has_def:
            v = rle1_decode_np(buf[p.def_off:p.def_off + p.def_len], p.n_values)
            validity[p.row_start:p.row_start + p.n_values] = v
            nvalid = int(v.sum())
            vals = np.frombuffer(buf, dtype=npdt, count=nvalid, offset=p.values_off)
            out[p.row_start:p.row_start + p.n_values][v.astype(bool)] = vals
        else:
            vals = np.frombuffer(buf, dtype=npdt, count=p.n_values, offset=p.values_off)
            out[p.row_start:p.row_start + p.n_values] = vals
    return out, validity


# ------------------------------------------------------------ file reader
_ARROW_TO_AURON = {
    "int32": dtypes.int32, "int64": dtypes.int64, "float": dtypes.float32,
    "double": dtypes.float64, "date32[day]": dtypes.date32,
}

from .pinned import POOL as _PINNED
from .pinned import to_device as _pin_to_device

# --------------------------------------------------------- HBM bytes cache
# Staged column-chunk bytes stay resident in HBM across queries/steps
# (288 GB per GPU: the MI355X answer to re-reading inputs every query —
# decode/filter/agg still run per query, only the host->device staging is
# elided). LRU-bounded; spark.auron.scan.hbmCache.maxBytes.
import collections as _collections

_DBUF_CACHE: "_collections.OrderedDict" = _collections.OrderedDict()
_DBUF_BYTES = [0]


def _dbuf_cache_cap() -> int:
    from .config import SCAN_HBM_CACHE, AuronConf

    v = AuronConf().get(SCAN_HBM_CACHE)
    if v:
        return v
    if torch.cuda.is_available():
        _, total = torch.cuda.mem_get_info()
        return int(total * 0.30)
    return 0


def _dbuf_cache_get(key):
    ent = _DBUF_CACHE.get(key)
    if ent is not None:
        _DBUF_CACHE.move_to_end(key)
    return ent


def dbuf_cache_clear():
    """Drop all cached staged bytes (OOM-pressure escape hatch)."""
    _DBUF_CACHE.clear()
    _DBUF_BYTES[0] = 0


def _dbuf_cache_put(key, dbuf):
    cap = _dbuf_cache_cap()
    if cap <= 0:
        return
    nbytes = dbuf.numel()
    if nbytes > cap:
        return
    # allocation-pressure guard: the cache must never be what pushes a
    # query over the edge — evict (and refuse to insert) whenever free
    # HBM would drop below 25% of the device, so big working sets
    # (SF>=100 joins/aggs) reclaim the cache before the allocator OOMs
    if torch.cuda.is_available():
        from .config import HBM_CACHE_RESERVE, AuronConf

        free, total = torch.cuda.mem_get_info()
        reserve = int(total * AuronConf().get(HBM_CACHE_RESERVE))
        while free - nbytes < reserve and _DBUF_CACHE:
            _, old = _DBUF_CACHE.popitem(last=False)
            _DBUF_BYTES[0] -= old.numel()
            free += old.numel()
        if free - nbytes < reserve:
            return
    _DBUF_CACHE[key] = dbuf
    _DBUF_BYTES[0] += nbytes
    while _DBUF_BYTES[0] > cap and _DBUF_CACHE:
        _, old = _DBUF_CACHE.popitem(last=False)
        _DBUF_BYTES[0] -= old.numel()

# footer + page-header metadata cache, keyed by (path, mtime, columns).
# Metadata only (never data): the Spark-side analogue is the parquet
# footer cache; pages are re-read and re-decoded on every scan.
_META_CACHE: Dict[tuple, object] = {}
_META_CACHE_MAX = 4096


class NativeParquetFile:
    """Per-file metadata: which columns take the device fast path."""

    def __init__(self, path: str):
        import pyarrow.parquet as pq

        self.path = path
        self.pf = pq.ParquetFile(path)
        self.md = self.pf.metadata
        self.schema = self.pf.schema
        self.names = [self.schema.column(i).name for i in range(self.md.num_columns)]

    def column_supported(self, name: str) -> bool:
        i = self.names.index(name)
        sc = self.schema.column(i)
        if sc.max_definition_level > 1 or sc.max_repetition_level > 0:
            return False
        if sc.physical_type not in _PHYS_NP and sc.physical_type != "BYTE_ARRAY":
            return False
        if sc.physical_type == "BYTE_ARRAY":
            # strings only via the dictionary path; PLAIN BYTE_ARRAY data
            # pages are detected at parse time and fall back
            lt = str(sc.logical_type).lower()
            if not ("string" in lt or "none" in lt):
                return False
        for rg in range(self.md.num_row_groups):
            cm = self.md.row_group(rg).column(i)
            if cm.compression not in ("UNCOMPRESSED", "SNAPPY", "ZSTD", "GZIP"):
                return False
        return True


def _chunk_meta(md, rg: int, ci: int):
    cm = md.row_group(rg).column(ci)
    off = cm.data_page_offset
    if cm.dictionary_page_offset is not None:
        off = min(off, cm.dictionary_page_offset)
    codec = None if cm.compression == "UNCOMPRESSED" else cm.compression
    return (off, cm.total_compressed_size, cm.num_values, cm.physical_type,
            cm.statistics, codec)


@dataclass
class _ColMeta:
    name: str
    dtype: object = None  # auron DataType
    phys: str = ""
    has_def: bool = False
    chunks: list = None  # [(new_off, clen, nvals, any_nulls)]
    pages: list = None  # [[PageDesc]] rebased to compact buffer; lazy


@dataclass
class _FileMeta:
    total: int
    ranges: list  # [(src_off, clen, new_off)]
    cols: list  # [_ColMeta]
    parsed: bool = False
    extra_total: int = 0  # decompressed-page region size (after raw+pad)
    jobs: list = None  # decompress/copy jobs refilled per read
    runs_parsed: bool = False  # def-level run headers host-parsed


def _build_meta(path: str, columns: List[str]) -> Optional[_FileMeta]:
    nf = NativeParquetFile(path)
    md = nf.md
    for c in columns:
        if c not in nf.names or not nf.column_supported(c):
            return None
    ranges = []
    cols = []
    pos = 0
    for cname in columns:
        ci = nf.names.index(cname)
        sc = nf.schema.column(ci)
        logical = str(sc.logical_type)
        phys = sc.physical_type
        if phys == "BYTE_ARRAY":
            dt = dtypes.string
        elif logical.startswith("Decimal"):
            import re

            m = re.search(r"precision=(\d+), scale=(\d+)", logical)
            dt = dtypes.decimal64(int(m.group(1)), int(m.group(2)))
        elif "date" in logical.lower():
            dt = dtypes.date32
        elif phys == "INT32":
            dt = dtypes.int32
        elif phys == "INT64":
            dt = dtypes.int64
        elif phys == "FLOAT":
            dt = dtypes.float32
        else:
            dt = dtypes.float64
        cm = _ColMeta(cname, dt, phys, sc.max_definition_level == 1, [], None)
        for rg in range(md.num_row_groups):
            off, clen, nvals, _phys, stats, codec = _chunk_meta(md, rg, ci)
            any_nulls = stats is None or not stats.has_null_count or stats.null_count > 0
            ranges.append((off, clen, pos))
            cm.chunks.append((pos, clen, nvals, any_nulls, codec))
            pos += clen
        cols.append(cm)
    return _FileMeta(pos, ranges, cols)


def _get_meta(path: str, columns: List[str]) -> Optional[_FileMeta]:
    key = (path, os.path.getmtime(path), tuple(columns))
    if key in _META_CACHE:
        return _META_CACHE[key]
    meta = _build_meta(path, columns)
    if len(_META_CACHE) > _META_CACHE_MAX:
        _META_CACHE.clear()
    _META_CACHE[key] = meta
    return meta


def split_supported(path: str, columns: List[str]) -> Tuple[List[str], List[str]]:
    """Partition `columns` into (native-decodable, host-fallback)."""
    key = (path, "split", tuple(columns))
    if key in _META_CACHE:
        return _META_CACHE[key]
    try:
        nf = NativeParquetFile(path)
        ok = [c for c in columns if c in nf.names and nf.column_supported(c)]
        rest = [c for c in columns if c not in ok]
    except Exception:
        ok, rest = [], list(columns)
    _META_CACHE[key] = (ok, rest)
    return ok, rest


def _decode_all(meta, buf, dbuf, device, use_gpu):
    out: Dict[str, Column] = {}
    for cm in meta.cols:
        if cm.phys == "BYTE_ARRAY":
            parts = []
            for (chunk, ck) in zip(cm.chunks, cm.pages):
                (_off, _clen, nvals, chunk_nulls, _codec) = chunk
                parts.append(_decode_chunk_strings(
                    buf, dbuf, ck, nvals, device, use_gpu, chunk_nulls))
            col = parts[0] if len(parts) == 1 else Column.concat(parts)
            if not use_gpu:
                col = col.to(device)
            out[cm.name] = col
            continue
        parts_data = []
        parts_valid = []
        any_nulls = False
        for (chunk, ck) in zip(cm.chunks, cm.pages):
            (_off, _clen, nvals, chunk_nulls, _codec) = chunk
            if use_gpu:
                if ck.is_dict:
                    data_t, valid_t = _decode_chunk_gpu_dict(
                        dbuf, buf, ck, nvals, cm.phys, device, chunk_nulls)
                else:
                    data_t, valid_t = _decode_chunk_gpu(dbuf, ck.pages, nvals,
                                                        cm.phys, device,
                                                        chunk_nulls,
                                                        ck.runs_np)
            else:
                if ck.is_dict:
                    data_np, valid_np, _ = decode_chunk_np_dict(buf, ck, nvals,
                                                                cm.phys)
                else:
                    data_np, valid_np = decode_chunk_np(buf, ck.pages, nvals,
                                                        cm.phys)
                data_t = torch.from_numpy(data_np)
                valid_t = torch.from_numpy(valid_np).to(torch.bool) if valid_np is not None else None
            if cm.dtype.code == dtypes.DECIMAL64 and data_t.dtype == torch.int32:
                data_t = data_t.to(torch.int64)  # widen INT32-backed decimals
            parts_data.append(data_t)
            if valid_t is not None:
                parts_valid.append(valid_t)
                if chunk_nulls:
                    any_nulls = True
        data = torch.cat(parts_data) if len(parts_data) > 1 else parts_data[0]
        validity = None
        if parts_valid and len(parts_valid) == len(parts_data):
            validity = torch.cat(parts_valid) if len(parts_valid) > 1 else parts_valid[0]
            if not any_nulls:
                validity = None
            elif not use_gpu and bool(validity.all()):
                validity = None
        if not use_gpu:
            data = data.to(device)
            if validity is not None:
                validity = validity.to(device)
        out[cm.name] = Column(cm.dtype, data, validity)
    return out

def read_columns_native(path: str, columns: List[str], device,
                        _np_only: bool = False) -> Optional[Dict[str, Column]]:
    """Decode `columns` of `path` on `device`. Returns None if any column
    is unsupported (caller falls back wholesale for simplicity).

    IO strategy: memmap the file, copy ONLY the needed column-chunk byte
    ranges into one pinned host buffer, upload it to HBM once, then run
    the decode kernels against device-resident page bytes. Footer and
    page-header metadata is cached per (path, mtime)."""
    columns = list(columns)
    meta = _get_meta(path, columns)
    if meta is None:
        return None
    use_gpu = (not _np_only) and torch.device(device).type == "cuda"

    if use_gpu and meta.parsed and getattr(meta, "runs_parsed", False):
        ck_key = (path, os.path.getmtime(path), tuple(columns))
        hit = _dbuf_cache_get(ck_key)
        if hit is not None:
            return _decode_all(meta, None, hit, device, True)

    mm = np.memmap(path, dtype=np.uint8, mode="r")

    def alloc(nbytes):
        if use_gpu:
            return _PINNED.acquire(nbytes)
        return None, torch.empty(nbytes, dtype=torch.uint8)

    def run_jobs(buf, jobs):
        for job in jobs:
            if job[0] == "d":
                _, s0, clen, dest, ulen, codec = job
                buf[dest:dest + ulen] = np.frombuffer(
                    _codec_decompress(codec, bytes(buf[s0:s0 + clen]), ulen),
                    dtype=np.uint8)
            elif job[0] == "delta":
                from . import native

                _, s0, length, dest, nv, esize = job
                lib = native.host_lib()
                got = lib.au_host_delta_unpack(
                    buf.ctypes.data, s0, length, nv, esize,
                    buf.ctypes.data + dest)
                if got < 0 or got > nv:
                    raise ValueError("corrupt DELTA_BINARY_PACKED page")
            else:
                _, s0, n, dest = job
                buf[dest:dest + n] = buf[s0:s0 + n]

    # +8 pad: the dict-index kernel's tail does unaligned 8-byte loads;
    # decompressed pages land after the pad (+8 tail pad again)
    full = meta.total + 8 + meta.extra_total + (8 if meta.extra_total else 0)
    pin_base, buf_t = alloc(full)
    try:
        buf = buf_t.numpy()
        for (src, clen, dst) in meta.ranges:
            buf[dst:dst + clen] = mm[src:src + clen]

        if not meta.parsed:
            extra = _ExtraAlloc(meta.total + 8)
            ok = []
            jobs = []
            for cm in meta.cols:
                cm.pages = []
                good = True
                j0 = len(extra.jobs)
                for (new_off, clen, nvals, _an, codec) in cm.chunks:
                    es = _PHYS_NP[cm.phys]().itemsize \
                        if cm.phys in _PHYS_NP else 0
                    ck = parse_pages(buf, new_off, clen, nvals, cm.has_def,
                                     codec, extra, esize=es)
                    if ck is None:
                        good = False
                        break
                    cm.pages.append(ck)
                if good:
                    ok.append(cm)
                    jobs.extend(extra.jobs[j0:])
            # per-column fallback: drop unparseable columns; their chunk ranges
            # stay in the staging layout (small waste, correctness unaffected)
            meta.cols = ok
            meta.parsed = True
            meta.extra_total = extra.cursor - extra.base
            meta.jobs = jobs
            if not ok:
                return None
            if meta.extra_total:
                # first read discovered compressed pages: grow the staging
                # buffer to cover the decompressed region
                if use_gpu and pin_base is not None:
                    _PINNED.release(pin_base, "cpu")
                pin_base, buf_t2 = alloc(meta.total + 8 + meta.extra_total + 8)
                buf2 = buf_t2.numpy()
                buf2[:meta.total] = buf[:meta.total]
                buf_t, buf = buf_t2, buf2

        if meta.jobs:
            run_jobs(buf, meta.jobs)

        if use_gpu and not meta.runs_parsed:
            # one-time host parse of def-level run headers (see
            # _parse_rle1_runs); must run after decompression jobs so
            # compressed chunks' level bytes are in place
            for cm in meta.cols:
                for ck in cm.pages:
                    if ck.pages and ck.pages[0].def_off >= 0:
                        ck.runs_np = _parse_rle1_runs(buf, ck.pages)
            meta.runs_parsed = True

        dbuf = buf_t.to(device, non_blocking=True) if use_gpu else None

        if use_gpu:
            # pre-parse everything that needs HOST bytes so later
            # cache-hit reads can run without a host buffer at all
            for cm in meta.cols:
                for ck in cm.pages:
                    if cm.phys == "BYTE_ARRAY":
                        if ck.is_dict:
                            parse_dict_strings(buf, ck)
                        else:
                            parse_plain_strings(buf, ck)
                    if ck.is_dict and ck.idx_bws is None:
                        ck.idx_bws = [int(buf[p.values_off]) for p in ck.pages]
            _dbuf_cache_put((path, os.path.getmtime(path), tuple(columns)),
                            dbuf)

        return _decode_all(meta, buf, dbuf, device, use_gpu)
    finally:
        if pin_base is not None:
            _PINNED.release(pin_base, device)


def _parse_rle1_runs(buf: np.ndarray, pages: List[PageDesc]) -> Optional[np.ndarray]:
    """Host-parse all def-level run headers of a chunk once (cached in the
    file meta): the serial walk that made k_pq_rle1 the #1 kernel moves to
    the CPU where it costs microseconds, leaving the device a flat
    expansion (k_rle1_expand)."""
    import ctypes

    from . import native

    if not native.available():
        return None
    lib = native.host_lib()
    cap = sum(p.def_len + 2 for p in pages)
    if cap <= 0:
        return None
    runs = np.empty((cap, 4), dtype=np.int32)
    base = buf.ctypes.data
    n = 0
    for p in pages:
        r = lib.au_host_rle1_parse(
            ctypes.c_void_p(base + p.def_off), ctypes.c_int64(p.def_len),
            ctypes.c_int64(p.n_values), ctypes.c_int64(p.row_start),
            ctypes.c_int64(p.def_off),
            ctypes.c_void_p(runs.ctypes.data + n * 16),
            ctypes.c_int64(cap - n))
        if r < 0:
            return None
        n += r
    return np.ascontiguousarray(runs[:n])


def _gpu_validity_prefix(dbuf, pages, num_values, device, chunk_nulls=True,
                         runs_np=None):
    """Decode def levels on device; returns (validity uint8 | None, prefix).

    Null-free chunks (stats null_count == 0) skip the decode."""
    from . import native

    lib = native.require()
    has_def = pages[0].def_off >= 0 and chunk_nulls
    if not has_def:
        return None, None
    sp = native.stream_ptr(device)
    validity = torch.empty(num_values, dtype=torch.uint8, device=device)
    if runs_np is not None:
        druns = _pin_to_device(runs_np.reshape(-1), device)
        rc = lib.au_rle1_expand(druns.data_ptr(), runs_np.shape[0],
                                dbuf.data_ptr(), validity.data_ptr(), sp)
        native.check(rc, "au_rle1_expand")
    else:
        npages = len(pages)
        arr = np.zeros((npages, 6), dtype=np.int64)
        for i, p in enumerate(pages):
            arr[i] = (p.def_off, p.def_len, p.values_off, p.n_values, p.row_start, 0)
        darr = _pin_to_device(arr.reshape(-1), device)
        rc = lib.au_pq_rle1(darr.data_ptr(), npages, dbuf.data_ptr(),
                            validity.data_ptr(), sp)
        native.check(rc, "au_pq_rle1")
    prefix = torch.cumsum(validity.to(torch.int64), 0)
    return validity, prefix


def _gpu_dict_indices(dbuf, buf, ck: ChunkPages, prefix, num_values, device):
    """Decode all RLE dictionary-index pages of a chunk into one int32
    array: compact (one entry per VALID row) when `prefix` is given,
    dense otherwise. Per-page valid counts and output bases come from
    the prefix sum ON DEVICE — no host sync."""
    from . import native

    lib = native.require()
    pages = ck.pages
    npages = len(pages)
    arr = np.zeros((npages, 6), dtype=np.int64)
    for i, p in enumerate(pages):
        bw = ck.idx_bws[i] if ck.idx_bws is not None else int(buf[p.values_off])
        arr[i] = (p.values_off + 1, p.values_len - 1, 0, p.n_values,
                  p.row_start, bw)
    darr = _pin_to_device(arr.reshape(-1), device)
    sp = native.stream_ptr(device)
    idx = torch.empty(max(num_values, 1), dtype=torch.int32, device=device)
    rc = lib.au_pq_rle_idx(darr.data_ptr(), npages, dbuf.data_ptr(),
                           idx.data_ptr(),
                           prefix.data_ptr() if prefix is not None else None,
                           sp)
    native.check(rc, "au_pq_rle_idx")
    return idx


def _decode_chunk_gpu_dict(dbuf, buf, ck: ChunkPages, num_values: int,
                           phys: str, device, chunk_nulls=True):
    """Dictionary-encoded fixed-width chunk on device."""
    validity, prefix = _gpu_validity_prefix(dbuf, ck.pages, num_values, device,
                                            chunk_nulls, ck.runs_np)
    idx = _gpu_dict_indices(dbuf, buf, ck, prefix, num_values, device)
    tdt = {"INT32": torch.int32, "INT64": torch.int64,
           "FLOAT": torch.float32, "DOUBLE": torch.float64}[phys]
    esize = _PHYS_NP[phys]().itemsize
    dvals = dbuf[ck.dict_off:ck.dict_off + ck.dict_nvals * esize].clone().view(tdt)
    if validity is None:
        return dvals[idx[:num_values].to(torch.int64)], None
    # idx is compact over valid rows; route each row to its entry via the
    # prefix sum (invalid rows read a clamped slot and are masked out)
    vmask = validity.to(torch.bool)
    pos = (prefix - 1).clamp(min=0)
    idx_row = idx[pos].to(torch.int64).clamp_(0, max(ck.dict_nvals - 1, 0))
    out = torch.where(vmask, dvals[idx_row], torch.zeros((), dtype=tdt,
                                                         device=device))
    return out, vmask


def _decode_chunk_plain_strings(buf, dbuf, ck: ChunkPages, num_values: int,
                                device, use_gpu: bool,
                                chunk_nulls: bool) -> Column:
    """PLAIN BYTE_ARRAY chunk -> string Column (host-parsed offsets,
    device byte gather; the path pyarrow writers take when a column
    overflows the dictionary-page limit — high-cardinality strings)."""
    if not use_gpu:
        parse_plain_strings(buf, ck)
        has_def = ck.pages[0].def_off >= 0
        validity = None
        if has_def:
            validity = np.zeros(num_values, dtype=np.uint8)
            for p in ck.pages:
                validity[p.row_start:p.row_start + p.n_values] = rle1_decode_np(
                    buf[p.def_off:p.def_off + p.def_len], p.n_values)
        vmask = validity.astype(bool) if validity is not None \
            else np.ones(num_values, dtype=bool)
        lens = np.zeros(num_values, dtype=np.int64)
        lens[vmask] = ck.plain_str_lens
        starts = np.zeros(num_values, dtype=np.int64)
        starts[vmask] = ck.plain_str_offs
        offsets = np.zeros(num_values + 1, dtype=np.int64)
        np.cumsum(lens, out=offsets[1:])
        data = np.empty(int(offsets[-1]), dtype=np.uint8)
        pos = 0
        for r in range(num_values):
            ln = lens[r]
            if ln:
                data[pos:pos + ln] = buf[starts[r]:starts[r] + ln]
                pos += ln
        vt = None
        if validity is not None and chunk_nulls and not validity.all():
            vt = torch.from_numpy(validity.astype(bool))
        return Column(dtypes.string, torch.from_numpy(data), vt,
                      torch.from_numpy(offsets))
    validity, prefix = _gpu_validity_prefix(dbuf, ck.pages, num_values, device,
                                            chunk_nulls, ck.runs_np)
    if ck.plain_str_offs is None:
        parse_plain_strings(buf, ck)
    offs_t = torch.from_numpy(ck.plain_str_offs).to(device, non_blocking=True)
    lens_t = torch.from_numpy(ck.plain_str_lens).to(device, non_blocking=True)
    if validity is not None:
        vmask = validity.to(torch.bool)
        pos = (prefix - 1).clamp(min=0)
        p_idx = pos.clamp_(0, max(int(offs_t.numel()) - 1, 0))
        zero = torch.zeros((), dtype=torch.int64, device=device)
        lens = torch.where(vmask, lens_t[p_idx], zero)
        starts = offs_t[p_idx]
    else:
        vmask = None
        lens = lens_t[:num_values]
        starts = offs_t[:num_values]
    offsets = torch.zeros(num_values + 1, dtype=torch.int64, device=device)
    torch.cumsum(lens, 0, out=offsets[1:])
    total = int(offsets[-1].item())
    if total:
        row = torch.repeat_interleave(lens)
        within = torch.arange(total, dtype=torch.int64, device=device) \
            - offsets[:-1][row]
        src = starts[row] + within
        data = dbuf[src]
    else:
        data = torch.empty(0, dtype=torch.uint8, device=device)
    vt = vmask if (vmask is not None and chunk_nulls) else None
    return Column(dtypes.string, data, vt, offsets)


def _decode_chunk_strings(buf, dbuf, ck: ChunkPages, num_values: int, device,
                          use_gpu: bool, chunk_nulls: bool) -> Column:
    """Dictionary-encoded BYTE_ARRAY chunk -> string Column."""
    if not ck.is_dict:
        return _decode_chunk_plain_strings(buf, dbuf, ck, num_values, device,
                                           use_gpu, chunk_nulls)
    if not use_gpu:
        data, validity, offsets = decode_chunk_np_dict(buf, ck, num_values,
                                                       "BYTE_ARRAY")
        vt = None
        if validity is not None and chunk_nulls and not validity.all():
            vt = torch.from_numpy(validity.astype(bool))
        return Column(dtypes.string, torch.from_numpy(data), vt,
                      torch.from_numpy(offsets))
    validity, prefix = _gpu_validity_prefix(dbuf, ck.pages, num_values, device,
                                            chunk_nulls, ck.runs_np)
    idx = _gpu_dict_indices(dbuf, buf, ck, prefix, num_values, device)
    parse_dict_strings(buf, ck)
    dict_offs = torch.from_numpy(ck.dict_str_offs).to(device)
    dict_lens = torch.from_numpy(ck.dict_str_lens).to(device)
    if validity is not None:
        vmask = validity.to(torch.bool)
        pos = (prefix - 1).clamp(min=0)
        row_idx = idx[pos].to(torch.int64).clamp_(0, max(ck.dict_nvals - 1, 0))
        lens = torch.where(vmask, dict_lens[row_idx],
                           torch.zeros_like(dict_lens[row_idx]))
    else:
        vmask = None
        row_idx = idx[:num_values].to(torch.int64)
        lens = dict_lens[row_idx]
    offsets = torch.zeros(num_values + 1, dtype=torch.int64, device=device)
    torch.cumsum(lens, 0, out=offsets[1:])
    total = int(offsets[-1].item())
    if total:
        row = torch.repeat_interleave(lens)
        within = torch.arange(total, dtype=torch.int64, device=device) \
            - offsets[:-1][row]
        src = dict_offs[row_idx[row]] + within
        data = dbuf[src]
    else:
        data = torch.empty(0, dtype=torch.uint8, device=device)
    vt = vmask if (vmask is not None and chunk_nulls) else None
    return Column(dtypes.string, data, vt, offsets.to(torch.int64))


def _decode_chunk_gpu(dbuf: torch.Tensor, pages: List[PageDesc], num_values: int,
                      phys: str, device, chunk_nulls: bool = True,
                      runs_np=None):
    """HIP kernel path: RLE def-levels + PLAIN value scatter on device."""
    from . import native

    lib = native.require()
    import ctypes

    esize = _PHYS_NP[phys]().itemsize
    has_def = pages[0].def_off >= 0 and chunk_nulls
    # null-free chunks (column statistics say null_count == 0) skip the
    # def-level decode entirely: all values present -> dense PLAIN copy.
    # This removes the single hottest kernel of the suite (k_pq_rle1 was
    # 41% of GPU time; see profiles/).
    npages = len(pages)
    # descriptor layout (int64 x 6): def_off, def_len, values_off, n, row_start, pad
    arr = np.zeros((npages, 6), dtype=np.int64)
    for i, p in enumerate(pages):
        arr[i] = (p.def_off, p.def_len, p.values_off, p.n_values, p.row_start, 0)
    darr = _pin_to_device(arr.reshape(-1), device)
    sp = native.stream_ptr(device)

    tdt = {"INT32": torch.int32, "INT64": torch.int64,
           "FLOAT": torch.float32, "DOUBLE": torch.float64}[phys]
    out = torch.empty(num_values, dtype=tdt, device=device)
    if not has_def:
        rc = lib.au_pq_copy_plain(darr.data_ptr(), npages, dbuf.data_ptr(),
                                  out.data_ptr(), esize, num_values, sp)
        native.check(rc, "au_pq_copy_plain")
        return out, None
    validity, prefix = _gpu_validity_prefix(dbuf, pages, num_values, device,
                                            True, runs_np)
    rc = lib.au_pq_scatter(darr.data_ptr(), npages, dbuf.data_ptr(),
                           validity.data_ptr(), prefix.data_ptr(),
                           out.data_ptr(), esize, num_values, sp)
    native.check(rc, "au_pq_scatter")
    return out, validity.to(torch.bool)

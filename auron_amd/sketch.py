"""Sketches: Spark-compatible XXH64 and bloom filter.

Role parity: the reference's ext-commons xxhash64 and
spark_bloom_filter (BloomFilterMightContain / bloom-filter join
pruning). Both are torch-vectorized so they run on-device over whole
columns; the bloom filter's bit layout and double-hashing scheme match
org.apache.spark.util.sketch.BloomFilterImpl so filters are
interchangeable with Spark-produced ones.
"""
from __future__ import annotations

import math
import struct
from typing import List

import torch

from . import dtypes
from .column import Column
from .ops import _hash_long_t

_M64 = (1 << 64) - 1


def _i64(c: int) -> int:
    """Wrap an unsigned 64-bit constant into int64 two's complement."""
    c &= _M64
    return c - (1 << 64) if c >= (1 << 63) else c


P1 = _i64(0x9E3779B185EBCA87)
P2 = _i64(0xC2B2AE3D27D4EB4F)
P3 = _i64(0x165667B19E3779F9)
P4 = _i64(0x85EBCA77C2B2AE63)
P5 = _i64(0x27D4EB2F165667C5)


def _lshr(x: torch.Tensor, n: int) -> torch.Tensor:
    if n == 0:
        return x
    return (x >> n) & ((1 << (64 - n)) - 1)


def _rotl64(x: torch.Tensor, r: int) -> torch.Tensor:
    return (x << r) | _lshr(x, 64 - r)


def _fmix64(h: torch.Tensor) -> torch.Tensor:
    h = h ^ _lshr(h, 33)
    h = h * P2
    h = h ^ _lshr(h, 29)
    h = h * P3
    h = h ^ _lshr(h, 32)
    return h


def _xxh_round(acc: torch.Tensor, inp: torch.Tensor) -> torch.Tensor:
    acc = acc + inp * P2
    return _rotl64(acc, 31) * P1


def _merge_round(hash_: torch.Tensor, v: torch.Tensor) -> torch.Tensor:
    hash_ = hash_ ^ _xxh_round(torch.zeros_like(v), v)
    return hash_ * P1 + P4


def xxh64_long(value: torch.Tensor, seed) -> torch.Tensor:
    """XXH64 of one little-endian 8-byte block (Spark XXH64.hashLong)."""
    v = value.to(torch.int64)
    if not torch.is_tensor(seed):
        seed = torch.full_like(v, _i64(seed))
    h = seed + P5 + 8
    h = h ^ _xxh_round(torch.zeros_like(v), v)
    h = _rotl64(h, 27) * P1 + P4
    return _fmix64(h)


def xxh64_bytes_col(c: Column, seed: int) -> torch.Tensor:
    """XXH64 over each row's UTF-8 bytes (vectorized, masked per length)."""
    seed_t = torch.full((len(c),), _i64(seed), dtype=torch.int64,
                        device=c.device)
    return _xxh64_bytes_seeded(c, seed_t)


SPARK_XXH_SEED = 42


def xxhash64(cols: List[Column], seed: int = SPARK_XXH_SEED) -> torch.Tensor:
    """Spark XxHash64 expression: seed-chained per column, nulls skipped."""
    n = len(cols[0])
    device = cols[0].device
    h = torch.full((n,), _i64(seed), dtype=torch.int64, device=device)
    for c in cols:
        code = c.dtype.code
        if c.dtype.is_string:
            # column chaining passes the running hash as the per-row seed
            hv = _xxh64_bytes_seeded(c, h)
        elif code == dtypes.BOOL:
            hv = xxh64_long(c.data.to(torch.int64), h)
        elif code in (dtypes.INT8, dtypes.INT16, dtypes.INT32, dtypes.DATE32,
                      dtypes.INT64, dtypes.DECIMAL64):
            hv = xxh64_long(c.data.to(torch.int64), h)
        elif code == dtypes.FLOAT32:
            f = torch.where(c.data == 0, torch.zeros_like(c.data), c.data)
            hv = xxh64_long(f.view(torch.int32).to(torch.int64), h)
        elif code == dtypes.FLOAT64:
            f = torch.where(c.data == 0, torch.zeros_like(c.data), c.data)
            hv = xxh64_long(f.view(torch.int64), h)
        else:
            raise TypeError(c.dtype.name)
        if c.validity is not None:
            h = torch.where(c.validity, hv, h)
        else:
            h = hv
    return h


def _xxh64_bytes_seeded(c: Column, seed_t: torch.Tensor) -> torch.Tensor:
    """xxh64_bytes_col generalized to a per-row tensor seed."""
    from . import strings as S

    n = len(c)
    device = c.device
    lens = S.lengths(c).to(torch.int64)
    maxlen = int(lens.max().item()) if n else 0
    A = S.to_padded(c, max(maxlen, 1)).to(torch.int64)

    def word8(p):
        w = torch.zeros(n, dtype=torch.int64, device=device)
        for i in range(8):
            w = w | (A[:, p + i] << (8 * i))
        return w

    def word4(p):
        w = torch.zeros(n, dtype=torch.int64, device=device)
        for i in range(4):
            w = w | (A[:, p + i] << (8 * i))
        return w

    stripe_end = torch.div(lens, 32, rounding_mode="floor") * 32
    h = seed_t + P5
    if maxlen >= 32:
        v1 = seed_t + P1 + P2
        v2 = seed_t + P2
        v3 = seed_t.clone()
        v4 = seed_t - P1
        for s in range(0, (maxlen // 32) * 32, 32):
            m = lens >= s + 32
            if not bool(m.any()):
                break
            v1 = torch.where(m, _xxh_round(v1, word8(s)), v1)
            v2 = torch.where(m, _xxh_round(v2, word8(s + 8)), v2)
            v3 = torch.where(m, _xxh_round(v3, word8(s + 16)), v3)
            v4 = torch.where(m, _xxh_round(v4, word8(s + 24)), v4)
        big = _rotl64(v1, 1) + _rotl64(v2, 7) + _rotl64(v3, 12) + _rotl64(v4, 18)
        big = _merge_round(big, v1)
        big = _merge_round(big, v2)
        big = _merge_round(big, v3)
        big = _merge_round(big, v4)
        h = torch.where(lens >= 32, big, h)
    h = h + lens
    for p in range(0, max(maxlen - 7, 0), 8):
        m = (p >= stripe_end) & (p + 8 <= lens)
        if not bool(m.any()):
            continue
        nh = h ^ _xxh_round(torch.zeros_like(h), word8(p))
        nh = _rotl64(nh, 27) * P1 + P4
        h = torch.where(m, nh, h)
    last8 = lens & ~7
    tail_start = torch.where(lens - last8 >= 4, last8 + 4, last8)
    for p in range(0, max(maxlen - 3, 0), 4):
        m = (last8 == p) & (lens - last8 >= 4) & (p >= stripe_end)
        if not bool(m.any()):
            continue
        nh = h ^ (word4(p) * P1)
        nh = _rotl64(nh, 23) * P2 + P3
        h = torch.where(m, nh, h)
    for t in range(maxlen):
        m = (t >= tail_start) & (t < lens)
        if not bool(m.any()):
            continue
        nh = h ^ (A[:, t] * P5)
        nh = _rotl64(nh, 11) * P1
        h = torch.where(m, nh, h)
    return _fmix64(h)


# ================================================================ bloom
class BloomFilter:
    """Spark BloomFilterImpl-compatible bloom filter over longs.

    Double hashing with Murmur3_x86_32.hashLong: h1 = hash(v, 0),
    h2 = hash(v, h1), probe bits (h1 + i*h2) for i in 1..k. The bit
    array is an int64 word tensor (little-endian bit order within each
    word, matching Spark's BitArray)."""

    VERSION = 1

    def __init__(self, num_bits: int, num_hashes: int, device="cpu"):
        num_bits = max(64, (num_bits + 63) & ~63)
        self.m = num_bits
        self.k = num_hashes
        self.words = torch.zeros(num_bits // 64, dtype=torch.int64,
                                 device=device)

    @classmethod
    def create(cls, expected_items: int, fpp: float = 0.03, device="cpu"):
        n = max(expected_items, 1)
        m = int(-n * math.log(fpp) / (math.log(2) ** 2))
        k = max(1, round(m / n * math.log(2)))
        return cls(m, k, device)

    def _bit_indexes(self, values: torch.Tensor) -> torch.Tensor:
        v = values.to(torch.int64)
        z = torch.zeros_like(v)
        h1 = _hash_long_t(v, z)  # int64 holding uint32 pattern
        h1s = torch.where(h1 >= 2 ** 31, h1 - 2 ** 32, h1)  # signed int32
        h2 = _hash_long_t(v, h1)
        h2s = torch.where(h2 >= 2 ** 31, h2 - 2 ** 32, h2)
        idxs = []
        for i in range(1, self.k + 1):
            comb = (h1s + i * h2s) & 0xFFFFFFFF
            comb = torch.where(comb >= 2 ** 31, comb - 2 ** 32, comb)
            comb = torch.where(comb < 0, ~comb, comb)
            idxs.append(comb % self.m)
        return torch.stack(idxs, dim=1)  # (n, k)

    def put_longs(self, values: torch.Tensor) -> None:
        if values.numel() == 0:
            return
        idx = self._bit_indexes(values).reshape(-1)
        mask = torch.zeros(self.m, dtype=torch.bool, device=self.words.device)
        mask[idx] = True
        weights = torch.tensor(
            [_i64(1 << b) for b in range(64)], dtype=torch.int64,
            device=self.words.device)
        packed = (mask.view(-1, 64).to(torch.int64) * weights).sum(dim=1)
        self.words |= packed

    def might_contain_longs(self, values: torch.Tensor) -> torch.Tensor:
        if values.numel() == 0:
            return torch.zeros(0, dtype=torch.bool, device=values.device)
        idx = self._bit_indexes(values)  # (n, k)
        word = idx >> 6
        bit = idx & 63
        hits = (self.words[word] >> bit) & 1
        return (hits == 1).all(dim=1)

    # Spark sketch serialized form: int32 version, int32 numHashFunctions,
    # int32 numWords, then numWords big-endian int64 words.
    def to_bytes(self) -> bytes:
        words = self.words.cpu().numpy().tolist()
        return struct.pack(f">iii{len(words)}q", self.VERSION, self.k,
                           len(words), *words)

    @classmethod
    def from_bytes(cls, data: bytes, device="cpu") -> "BloomFilter":
        ver, k, nw = struct.unpack_from(">iii", data)
        assert ver == cls.VERSION
        words = struct.unpack_from(f">{nw}q", data, 12)
        bf = cls(nw * 64, k, device)
        bf.words = torch.tensor(words, dtype=torch.int64, device=device)
        return bf

    def put_column(self, c: Column) -> None:
        v = c.data.to(torch.int64)
        if c.validity is not None:
            v = v[c.validity]
        self.put_longs(v)

    def might_contain_column(self, c: Column) -> torch.Tensor:
        """Row mask; null inputs report False (filter drops them — matches
        inner-join runtime-filter semantics where null keys never match)."""
        res = self.might_contain_longs(c.data.to(torch.int64))
        if c.validity is not None:
            res = res & c.validity
        return res

"""Columnar storage: Column + RecordBatch backed by torch tensors.

Role parity: Arrow RecordBatch as used throughout the reference native
engine (arrow-rs arrays in /root/reference/native-engine/*). Device
residency: on MI355X, `data`/`offsets`/`validity` live in HBM3E as torch
CUDA tensors; the same code paths run on CPU tensors for the no-GPU CI
mode (the analogue of the reference's `is_jni_bridge_inited()` fallback).

String columns use Arrow layout: uint8 byte buffer + int32 offsets (n+1).
Validity is a torch.bool tensor (True = valid) or None meaning all-valid.
"""
from __future__ import annotations

from typing import Iterable, List, Optional, Sequence

import torch

from . import dtypes
from .dtypes import DataType


class Column:
    __slots__ = ("dtype", "data", "validity", "offsets")

    def __init__(
        self,
        dtype: DataType,
        data: torch.Tensor,
        validity: Optional[torch.Tensor] = None,
        offsets: Optional[torch.Tensor] = None,
    ):
        self.dtype = dtype
        self.data = data
        self.validity = validity
        self.offsets = offsets
        if dtype.uses_offsets:
            assert offsets is not None and offsets.dtype == torch.int64

    # ---------------------------------------------------------- basics
    def __len__(self) -> int:
        if self.dtype.uses_offsets:
            return int(self.offsets.shape[0]) - 1
        return int(self.data.shape[0])

    @property
    def device(self) -> torch.device:
        return self.data.device

    @property
    def null_count(self) -> int:
        if self.validity is None:
            return 0
        return int((~self.validity).sum().item())

    def to(self, device) -> "Column":
        device = torch.device(device)
        if self.device == device:
            return self
        # non_blocking is only safe toward the GPU; an async D2H copy into
        # pageable host memory returns before the bytes land (reader sees
        # stale pool memory)
        nb = device.type == "cuda"
        return Column(
            self.dtype,
            self.data.to(device, non_blocking=nb),
            None if self.validity is None else self.validity.to(device, non_blocking=nb),
            None if self.offsets is None else self.offsets.to(device, non_blocking=nb),
        )

    def clone_meta(self, data, validity=None, offsets=None) -> "Column":
        return Column(self.dtype, data, validity, offsets)

    # ---------------------------------------------------------- construction
    @staticmethod
    def from_pylist(values: Sequence, dtype: DataType, device="cpu") -> "Column":
        device = torch.device(device)
        n = len(values)
        validity = None
        if any(v is None for v in values):
            validity = torch.tensor([v is not None for v in values], dtype=torch.bool)
        if dtype.is_list:
            offs = [0]
            flat = []
            for v in values:
                if v is not None:
                    flat.extend(v)
                offs.append(len(flat))
            data = torch.tensor(flat, dtype=dtype.torch_dtype) if flat else \
                torch.empty(0, dtype=dtype.torch_dtype)
            col = Column(dtype, data, validity, torch.tensor(offs, dtype=torch.int64))
        elif dtype.is_string:
            bufs = []
            offs = [0]
            total = 0
            for v in values:
                b = (v or "").encode("utf-8")
                bufs.append(b)
                total += len(b)
                offs.append(total)
            data = torch.frombuffer(bytearray(b"".join(bufs)), dtype=torch.uint8) if total else torch.empty(0, dtype=torch.uint8)
            offsets = torch.tensor(offs, dtype=torch.int64)
            col = Column(dtype, data, validity, offsets)
        else:
            fill = 0
            tvals = [fill if v is None else v for v in values]
            if dtype.code == dtypes.DECIMAL64:
                scaled = [int(round(float(v) * (10 ** dtype.scale))) for v in tvals]
                data = torch.tensor(scaled, dtype=torch.int64)
            elif dtype.code == dtypes.DECIMAL128:
                from decimal import Decimal

                m64 = (1 << 64) - 1
                limbs = []
                for v in tvals:
                    iv = int(Decimal(str(v)).scaleb(dtype.scale))
                    lo = iv & m64
                    limbs.append(((lo - (1 << 64)) if lo >= (1 << 63) else lo,
                                  iv >> 64))
                data = torch.tensor(limbs, dtype=torch.int64) if limbs else \
                    torch.empty((0, 2), dtype=torch.int64)
            else:
                data = torch.tensor(tvals, dtype=dtype.torch_dtype)
            col = Column(dtype, data, validity, None)
        assert len(col) == n
        return col.to(device)

    def to_pylist(self) -> list:
        c = self.to("cpu")
        valid = None if c.validity is None else c.validity.tolist()
        if self.dtype.is_string:
            raw = bytes(c.data.numpy().tobytes())
            offs = c.offsets.tolist()
            out = []
            for i in range(len(c)):
                if valid is not None and not valid[i]:
                    out.append(None)
                else:
                    out.append(raw[offs[i]:offs[i + 1]].decode("utf-8", errors="replace"))
            return out
        if self.dtype.is_list:
            offs = c.offsets.tolist()
            flat = c.data.tolist()
            out = []
            for i in range(len(c)):
                if valid is not None and not valid[i]:
                    out.append(None)
                else:
                    out.append(flat[offs[i]:offs[i + 1]])
            return out
        if self.dtype.code == dtypes.DECIMAL128:
            s = 10 ** self.dtype.scale
            lo = c.data[:, 0].tolist()
            hi = c.data[:, 1].tolist()
            m64 = (1 << 64) - 1
            vals = [((h << 64) | (l & m64)) / s for l, h in zip(lo, hi)]
            if valid is None:
                return vals
            return [v if ok else None for v, ok in zip(vals, valid)]
        vals = c.data.tolist()
        if self.dtype.code == dtypes.DECIMAL64:
            s = 10 ** self.dtype.scale
            vals = [v / s for v in vals]
        if valid is None:
            return vals
        return [v if ok else None for v, ok in zip(vals, valid)]

    # ---------------------------------------------------------- kernels (torch-expressed)
    def gather(self, indices: torch.Tensor,
               may_have_negative: bool = False) -> "Column":
        """take(): rows at `indices` (int64). Negative index -1 = emit null
        (outer joins set may_have_negative; the default path skips the
        device->host `.any()` probe, which would sync the stream on every
        gather — one of the r2 profile's top latency taxes)."""
        idx = indices.to(self.device)
        neg = None
        if may_have_negative and idx.numel():
            neg = idx < 0
            idx = idx.clamp(min=0)
            if not bool(neg.any()) if idx.device.type != "cuda" else False:
                neg = None
        validity = None
        if self.validity is not None:
            validity = self.validity[idx]
        if neg is not None:
            if validity is None:
                validity = torch.ones(idx.shape[0], dtype=torch.bool, device=self.device)
            validity = validity & ~neg
        if self.dtype.uses_offsets:
            off = self.offsets.to(torch.int64)
            starts = off[idx]
            lens = off[idx + 1] - starts
            if neg is not None:
                lens = torch.where(neg, torch.zeros_like(lens), lens)
            new_off = torch.zeros(idx.shape[0] + 1, dtype=torch.int64, device=self.device)
            torch.cumsum(lens, 0, out=new_off[1:])
            total = int(new_off[-1].item())
            if total == 0:
                out_el = torch.empty(0, dtype=self.data.dtype, device=self.device)
            elif self.device.type == "cuda" and self.data.dtype == torch.uint8:
                # one kernel instead of the arange/repeat_interleave/index
                # chain (k_bytes_gather, see kernels.hip)
                from . import native

                out_el = torch.empty(total, dtype=torch.uint8, device=self.device)
                rc = native.lib().au_bytes_gather(
                    self.data.data_ptr(), starts.contiguous().data_ptr(),
                    new_off.data_ptr(), out_el.data_ptr(), idx.shape[0],
                    native.stream_ptr(self.device))
                native.check(rc, "au_bytes_gather")
            else:
                pos = torch.arange(total, dtype=torch.int64, device=self.device)
                row = torch.repeat_interleave(lens)  # maps element pos -> out row
                el_idx = pos - new_off[:-1][row] + starts[row]
                out_el = self.data[el_idx]
            return Column(self.dtype, out_el, validity, new_off)
        data = self.data[idx]
        return Column(self.dtype, data, validity, None)

    def filter(self, mask: torch.Tensor) -> "Column":
        idx = torch.nonzero(mask, as_tuple=False).flatten()
        return self.gather(idx)

    def slice(self, start: int, length: int) -> "Column":
        idx = torch.arange(start, start + length, dtype=torch.int64, device=self.device)
        return self.gather(idx)

    @staticmethod
    def concat(cols: List["Column"]) -> "Column":
        assert cols
        dt = cols[0].dtype
        for c in cols[1:]:
            assert c.dtype.code == dt.code and c.dtype.scale == dt.scale, \
                f"concat dtype mismatch: {dt.name} vs {c.dtype.name}"
        device = cols[0].device
        any_null = any(c.validity is not None for c in cols)
        validity = None
        if any_null:
            parts = [
                c.validity if c.validity is not None else torch.ones(len(c), dtype=torch.bool, device=device)
                for c in cols
            ]
            validity = torch.cat(parts)
        if dt.uses_offsets:
            datas = [c.data for c in cols]
            data = torch.cat(datas) if datas else torch.empty(0, dtype=cols[0].data.dtype, device=device)
            offs = []
            base = 0
            for c in cols:
                o = c.offsets.to(torch.int64)
                offs.append(o[:-1] + base if len(offs) else o[:-1] + base)
                base += int(o[-1].item())
            offs.append(torch.tensor([base], dtype=torch.int64, device=device))
            offsets = torch.cat(offs).to(torch.int64)
            return Column(dt, data, validity, offsets)
        return Column(dt, torch.cat([c.data for c in cols]), validity, None)

    # ---------------------------------------------------------- arrow interop
    @staticmethod
    def from_arrow(arr, device="cpu") -> "Column":
        import numpy as np
        import pyarrow as pa

        if isinstance(arr, pa.ChunkedArray):
            arr = arr.combine_chunks()
        if pa.types.is_dictionary(arr.type):
            arr = arr.dictionary_decode()
        dt = dtypes.from_arrow(arr.type)
        validity = None
        if arr.null_count:
            validity = torch.from_numpy(arr.is_valid().to_numpy(zero_copy_only=False))
        if dt.is_string:
            arr = arr.cast(pa.string())
            # offsets/values buffers (zero-copy views into the arrow buffers;
            # the arrow table is kept alive by the returned tensors' base)
            arr = arr.combine_chunks() if isinstance(arr, pa.ChunkedArray) else arr
            off_np = np.frombuffer(arr.buffers()[1], dtype=np.int32,
                                   count=len(arr) + 1,
                                   offset=arr.offset * 4).astype(np.int64)
            start = int(off_np[0])
            end = int(off_np[-1])
            buf = arr.buffers()[2]
            if buf is None or end == start:
                data = torch.empty(0, dtype=torch.uint8)
                offsets = torch.zeros(len(arr) + 1, dtype=torch.int64)
            else:
                data = torch.from_numpy(np.frombuffer(buf, dtype=np.uint8, count=end - start, offset=start))
                offsets = torch.from_numpy(off_np if start == 0 else off_np - start)
            col = Column(dt, data, validity, offsets)
        elif dt.code == dtypes.DECIMAL64:
            d128 = arr.cast(pa.decimal128(dt.precision, dt.scale))
            d128 = d128.combine_chunks() if isinstance(d128, pa.ChunkedArray) else d128
            buf = d128.buffers()[1]
            if buf is not None:
                # exact unscaled ints: decimal128 little-endian, low 8 bytes
                raw = np.frombuffer(buf, dtype=np.int64,
                                    count=2 * (len(d128) + d128.offset))
                scaled = raw[2 * d128.offset::2].copy()
            else:
                scaled = np.zeros(len(d128), dtype=np.int64)
            col = Column(dt, torch.from_numpy(scaled), validity, None)
        elif dt.code == dtypes.DECIMAL128:
            d128 = arr.combine_chunks() if isinstance(arr, pa.ChunkedArray) else arr
            buf = d128.buffers()[1]
            if buf is not None:
                # both little-endian limbs: data[:,0]=low 64 bits, [:,1]=high
                raw = np.frombuffer(buf, dtype=np.int64,
                                    count=2 * (len(d128) + d128.offset))
                limbs = raw[2 * d128.offset:].reshape(-1, 2).copy()
            else:
                limbs = np.zeros((len(d128), 2), dtype=np.int64)
            col = Column(dt, torch.from_numpy(limbs), validity, None)
        else:
            if dt.code == dtypes.DATE32:
                arr = arr.cast(pa.int32())
            if dt.code == dtypes.TIMESTAMP:
                # normalize any unit to int64 microseconds
                arr = arr.cast(pa.timestamp("us")).cast(pa.int64())
            if arr.null_count:
                arr = arr.fill_null(0)  # validity carries the null mask
            np_arr = arr.to_numpy(zero_copy_only=arr.type != pa.bool_())
            if np_arr.dtype == np.bool_:
                np_arr = np.ascontiguousarray(np_arr)
            t = torch.from_numpy(np_arr)
            if t.dtype != dt.torch_dtype:
                t = t.to(dt.torch_dtype)
            col = Column(dt, t, validity, None)
        return col.to(device)

    def to_arrow(self):
        import numpy as np
        import pyarrow as pa

        c = self.to("cpu")
        mask = None
        if c.validity is not None:
            mask = ~c.validity.numpy()
        if self.dtype.is_string:
            vals = c.to_pylist()
            return pa.array(vals, type=pa.string())
        if self.dtype.code == dtypes.DECIMAL64:
            # exact scaled-int -> arrow decimal128 (schema + exactness
            # survive a sink/scan round-trip; advisor finding r1)
            from decimal import Decimal

            sc = self.dtype.scale
            vals = [None if (mask is not None and mask[i])
                    else Decimal(int(c.data[i].item())).scaleb(-sc)
                    for i in range(len(c))]
            p = max(self.dtype.precision, sc + 1, 1)
            return pa.array(vals, type=pa.decimal128(p, sc))
        if self.dtype.code == dtypes.DECIMAL128:
            from decimal import Decimal

            sc = self.dtype.scale
            lo = c.data[:, 0].numpy()
            hi = c.data[:, 1].numpy()
            vals = []
            for i in range(len(c)):
                if mask is not None and mask[i]:
                    vals.append(None)
                else:
                    v = (int(hi[i]) << 64) | (int(lo[i]) & ((1 << 64) - 1))
                    vals.append(Decimal(v).scaleb(-sc))
            p = max(self.dtype.precision, sc + 1, 1)
            return pa.array(vals, type=pa.decimal128(min(p, 38), sc))
        np_arr = c.data.numpy()
        if self.dtype.code == dtypes.TIMESTAMP:
            return pa.array(np_arr, from_pandas=False, mask=mask).cast(
                pa.timestamp("us"))
        return pa.array(np_arr, from_pandas=False, mask=mask)

    def __repr__(self) -> str:  # pragma: no cover
        return f"Column({self.dtype.name}, n={len(self)}, nulls={self.null_count}, dev={self.device})"



def compact_validity(v: Optional[torch.Tensor]) -> Optional[torch.Tensor]:
    """Collapse an all-true validity mask to None — but only where the
    check is free (CPU). On device the `.all()` reduction would sync the
    stream; keeping a redundant mask is far cheaper than the bubble."""
    if v is None or v.device.type == "cuda":
        return v
    return None if bool(v.all()) else v


class RecordBatch:
    # _eval_memo: optional CSE scope opened by exprs.eval_scope
    __slots__ = ("names", "columns", "_eval_memo")

    def __init__(self, names: List[str], columns: List[Column]):
        assert len(names) == len(columns)
        if columns:
            n = len(columns[0])
            for c in columns:
                assert len(c) == n, f"ragged batch: {[len(x) for x in columns]}"
        self.names = list(names)
        self.columns = list(columns)

    @property
    def num_rows(self) -> int:
        return len(self.columns[0]) if self.columns else 0

    @property
    def num_columns(self) -> int:
        return len(self.columns)

    @property
    def device(self) -> torch.device:
        return self.columns[0].device if self.columns else torch.device("cpu")

    def column(self, name: str) -> Column:
        return self.columns[self.names.index(name)]

    def to(self, device) -> "RecordBatch":
        return RecordBatch(self.names, [c.to(device) for c in self.columns])

    def gather(self, idx: torch.Tensor,
               may_have_negative: bool = False) -> "RecordBatch":
        if self.device.type == "cuda" and len(self.columns) > 1:
            fused = self._gather_fused(idx, may_have_negative)
            if fused is not None:
                return fused
        return RecordBatch(self.names,
                           [c.gather(idx, may_have_negative) for c in self.columns])

    def _gather_fused(self, idx: torch.Tensor, may_have_negative: bool):
        """All fixed-width columns in ONE kernel launch (k_multi_gather);
        string/list columns keep the per-column path (k_bytes_gather).
        Join/window/sort row gathers were one at::native index_select per
        column — the top at::native entry of the final SF=10 profile."""
        from . import native

        if not native.available():
            return None
        import numpy as np

        lib = native.lib()
        idx = idx.to(self.device).to(torch.int64).contiguous()
        n = int(idx.numel())
        out_cols: list = [None] * len(self.columns)
        descs = []
        keep = [idx]
        for ci, c in enumerate(self.columns):
            if c.dtype.uses_offsets:
                out_cols[ci] = c.gather(idx, may_have_negative)
                continue
            data = c.data if c.data.is_contiguous() else c.data.contiguous()
            keep.append(data)
            esize = data.element_size() * (2 if c.dtype.code == dtypes.DECIMAL128 else 1)
            shape = (n, 2) if c.dtype.code == dtypes.DECIMAL128 else (n,)
            dst = torch.empty(shape, dtype=data.dtype, device=self.device)
            want_valid = c.validity is not None or may_have_negative
            dv = torch.empty(n, dtype=torch.bool, device=self.device) \
                if want_valid else None
            sv = 0
            if c.validity is not None:
                v = c.validity if c.validity.is_contiguous() else c.validity.contiguous()
                keep.append(v)
                sv = v.data_ptr()
            descs.append((data.data_ptr(), sv, dst.data_ptr(),
                          dv.data_ptr() if dv is not None else 0, esize, 0))
            out_cols[ci] = Column(c.dtype, dst, dv)
        if descs and n:
            arr = np.array(descs, dtype=np.uint64).reshape(-1, 6)
            arr = np.ascontiguousarray(arr)
            for lo in range(0, len(descs), 24):
                sub = np.ascontiguousarray(arr[lo:lo + 24])
                rc = lib.au_multi_gather(idx.data_ptr(), n, sub.ctypes.data,
                                         sub.shape[0],
                                         native.stream_ptr(self.device))
                native.check(rc, "au_multi_gather")
        return RecordBatch(self.names, out_cols)

    def filter(self, mask: torch.Tensor) -> "RecordBatch":
        idx = torch.nonzero(mask, as_tuple=False).flatten()
        return self.gather(idx)

    def slice(self, start: int, length: int) -> "RecordBatch":
        return RecordBatch(self.names, [c.slice(start, length) for c in self.columns])

    def select(self, names: Iterable[str]) -> "RecordBatch":
        names = list(names)
        return RecordBatch(names, [self.column(n) for n in names])

    @staticmethod
    def concat(batches: List["RecordBatch"]) -> "RecordBatch":
        assert batches
        names = batches[0].names
        cols = [Column.concat([b.columns[i] for b in batches]) for i in range(len(names))]
        return RecordBatch(names, cols)

    @staticmethod
    def from_pydict(d: dict, types: dict, device="cpu") -> "RecordBatch":
        names = list(d.keys())
        cols = [Column.from_pylist(d[n], types[n], device) for n in names]
        return RecordBatch(names, cols)

    def to_pydict(self) -> dict:
        return {n: c.to_pylist() for n, c in zip(self.names, self.columns)}

    @staticmethod
    def from_arrow(table, device="cpu") -> "RecordBatch":
        import pyarrow as pa

        names = []
        cols = []
        for i, name in enumerate(table.schema.names):
            arr = table.column(i)
            if pa.types.is_struct(arr.type):
                # struct columns flatten into dotted leaf columns at the
                # boundary (GetStructField then resolves to a plain column
                # reference — ext-exprs GetIndexedField-on-struct parity)
                carr = arr.combine_chunks() if isinstance(arr, pa.ChunkedArray) else arr
                for fi in range(carr.type.num_fields):
                    f = carr.type.field(fi)
                    names.append(f"{name}.{f.name}")
                    cols.append(Column.from_arrow(carr.field(fi), device))
                continue
            names.append(name)
            cols.append(Column.from_arrow(arr, device))
        return RecordBatch(names, cols)

    def to_arrow(self):
        import pyarrow as pa

        return pa.table({n: c.to_arrow() for n, c in zip(self.names, self.columns)})

    def __repr__(self) -> str:  # pragma: no cover
        return f"RecordBatch({self.num_rows} rows, {list(zip(self.names, [c.dtype.name for c in self.columns]))})"

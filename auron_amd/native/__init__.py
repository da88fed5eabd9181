"""ctypes loader for the native HIP engine (libauron_hip.so).

Role parity: the reference's JNI bridge + cdylib entry
(/root/reference/native-engine/auron-jni-bridge/src/jni_bridge.rs,
 /root/reference/native-engine/auron/src/exec.rs) — here the host side is
Python, the native side is C/HIP, and batches are exchanged zero-copy as
raw device pointers into torch-owned HBM buffers.

On a GPU box the native library is REQUIRED: ops raise AuronNativeMissing
rather than silently falling back to eager torch (set
AURON_REQUIRE_NATIVE=0 only for debugging).
"""
from __future__ import annotations

import ctypes
import os
from typing import List, Optional, Tuple

import numpy as np
import torch

_LIB_NAME = "libauron_hip.so"
_here = os.path.dirname(os.path.abspath(__file__))
LIB_PATH = os.path.join(_here, _LIB_NAME)

_lib = None
_load_error: Optional[str] = None


class AuronNativeMissing(RuntimeError):
    pass


def _try_load():
    global _lib, _load_error
    if _lib is not None or _load_error is not None:
        return
    if not os.path.exists(LIB_PATH):
        _load_error = f"{LIB_PATH} not built (run python -m auron_amd.native.build)"
        return
    try:
        lib = ctypes.CDLL(LIB_PATH)
    except OSError as e:  # e.g. no ROCm runtime on this host
        _load_error = str(e)
        return
    c = ctypes.c_void_p
    i32 = ctypes.c_int32
    i64 = ctypes.c_int64
    lib.au_abi_version.restype = ctypes.c_int
    lib.au_murmur3.argtypes = [c, ctypes.c_int, i64, i32, c, c]
    lib.au_group_ids.argtypes = [c, ctypes.c_int, i64, c, i64, c, c, c, c, c, c]
    lib.au_join_build.argtypes = [i64, c, i64, c, c, c]
    lib.au_join_count.argtypes = [c, c, ctypes.c_int, i64, c, i64, c, c, c, c]
    lib.au_join_fill.argtypes = [c, c, ctypes.c_int, i64, c, i64, c, c, c, c, c, c, ctypes.c_int, c]
    lib.au_pmod.argtypes = [c, i64, i32, c, c]
    lib.au_bytes_gather.argtypes = [c, c, c, c, i64, c]
    lib.au_multi_gather.argtypes = [c, i64, c, ctypes.c_int, c]
    lib.au_multi_gather.restype = ctypes.c_int
    lib.au_agg_scatter.argtypes = [c, i64, c, c, ctypes.c_int, ctypes.c_int,
                                   ctypes.c_int, c, c, i64, c]
    lib.au_agg_count.argtypes = [c, i64, c, c, c]
    lib.au_agg_multi.argtypes = [c, i64, c, ctypes.c_int, c]
    for f in ("au_agg_scatter", "au_agg_count", "au_agg_multi"):
        getattr(lib, f).restype = ctypes.c_int
    lib.au_pq_rle1.argtypes = [c, ctypes.c_int, c, c, c]
    lib.au_rle1_expand.argtypes = [c, i64, c, c, c]
    lib.au_rle1_expand.restype = ctypes.c_int
    lib.au_host_rle1_parse.argtypes = [c, i64, i64, i64, i64, c, i64]
    lib.au_host_rle1_parse.restype = i64
    lib.au_host_plainba_parse.argtypes = [c, i64, i64, c, c, i64]
    lib.au_host_plainba_parse.restype = i64
    lib.au_host_delta_unpack.argtypes = [c, i64, i64, i64, ctypes.c_int, c]
    lib.au_host_delta_unpack.restype = i64
    lib.au_pq_rle_idx.argtypes = [c, ctypes.c_int, c, c, c, c]
    lib.au_pq_scatter.argtypes = [c, ctypes.c_int, c, c, c, c, ctypes.c_int, i64, c]
    lib.au_pq_copy_plain.argtypes = [c, ctypes.c_int, c, c, ctypes.c_int, i64, c]
    for f in ("au_pq_rle1", "au_pq_rle_idx", "au_pq_scatter", "au_pq_copy_plain"):
        getattr(lib, f).restype = ctypes.c_int
    lib.au_expr_exec.argtypes = [c, ctypes.c_int, c, ctypes.c_int, c,
                                 ctypes.c_int, ctypes.c_int, i64, c]
    lib.au_expr_exec.restype = ctypes.c_int
    lib.au_part_hist.argtypes = [c, i64, i32, c, c]
    lib.au_part_scatter.argtypes = [c, i64, c, c, c, c]
    for f in ("au_murmur3", "au_group_ids", "au_join_build", "au_join_count",
              "au_join_fill", "au_pmod", "au_part_hist", "au_part_scatter"):
        getattr(lib, f).restype = ctypes.c_int
    _lib = lib


def available() -> bool:
    _try_load()
    return _lib is not None


def require():
    _try_load()
    if _lib is None:
        raise AuronNativeMissing(
            f"auron native HIP library unavailable: {_load_error}. "
            "Build it with `python -m auron_amd.native.build`."
        )
    return _lib


def lib():
    return require()


def host_lib():
    """Host-callable entry points (e.g. au_host_rle1_parse) work without a
    GPU; same .so, same loader."""
    return require()


_DESC_DTYPE = np.dtype(
    [("data", "<u8"), ("offsets", "<u8"), ("validity", "<u8"),
     ("dtype", "<i4"), ("scale", "<i4")]
)
assert _DESC_DTYPE.itemsize == 32


def pack_descs(cols, device) -> Tuple[torch.Tensor, list]:
    """Pack Column descriptors into a device uint8 tensor (AuColDesc[]).

    Returns (desc_tensor, keepalive). Caller must hold `keepalive` until the
    kernels using the descriptors have been enqueued (torch stream ordering
    then keeps buffers alive until completion).
    """
    n = len(cols)
    arr = np.zeros(n, dtype=_DESC_DTYPE)
    keep = []
    for i, col in enumerate(cols):
        data = col.data
        if not data.is_contiguous():
            data = data.contiguous()
        keep.append(data)
        arr[i]["data"] = data.data_ptr()
        if col.offsets is not None:
            off = col.offsets.contiguous()
            keep.append(off)
            arr[i]["offsets"] = off.data_ptr()
        if col.validity is not None:
            v = col.validity.contiguous()
            keep.append(v)
            arr[i]["validity"] = v.data_ptr()
        arr[i]["dtype"] = col.dtype.code
        arr[i]["scale"] = col.dtype.scale
    from ..pinned import to_device

    dev = to_device(arr.view(np.uint8).reshape(-1), device)
    keep.append(dev)
    return dev, keep


def stream_ptr(device) -> ctypes.c_void_p:
    return ctypes.c_void_p(torch.cuda.current_stream(device).cuda_stream)


def check(rc: int, what: str):
    if rc != 0:
        raise RuntimeError(f"native kernel {what} failed with hipError {rc}")
    if os.environ.get("AURON_SYNC_NATIVE", "0") == "1":
        torch.cuda.synchronize()

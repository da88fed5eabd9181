"""Fused expression compiler: Expr tree -> one-kernel stack program.

Role parity: the reference evaluates a whole Project/Filter expression
tree natively per batch (datafusion physical-expr evaluation inside
project_exec.rs / filter_exec.rs). The eager path here launches one
at::native kernel per operator with an HBM round-trip between each; this
compiler lowers a scalar expression tree to a postfix program executed
by ONE HIP kernel (native/csrc/fused.hip), reading every input column
once and writing only the final outputs.

Semantics mirror exprs.py exactly (promotion rules, Kleene logic, null
cond -> false, /0 -> null, Spark integer `/` as double division).
Unsupported nodes (strings, float32, windows, subqueries) make
compile_exprs return None for that expr and the caller falls back to
the interpreted eval.
"""
from __future__ import annotations

import struct
from typing import Dict, List, Optional, Tuple

import numpy as np
import torch

from . import dtypes
from .column import Column
from .dtypes import DataType
from .exprs import (Arith, BoolOp, CaseWhen, Cast, Cmp, Coalesce, Col,
                    InList, IsNull, Literal, Not, TryCast, _infer_lit_dtype)

# opcodes — keep in sync with native/csrc/fused.hip
PUSH_COL, PUSH_LIT_I, PUSH_LIT_D, PUSH_NULL = 0, 1, 2, 3
I2D, D2I_TRUNC, D2I_ROUND, TRUNC_I, NEZ = 4, 5, 6, 7, 8
ADDI, SUBI, MULI = 9, 10, 11
ADDD, SUBD, MULD = 12, 13, 14
DIVD, MODI, DIVI = 15, 16, 17
LTI, LEI, GTI, GEI, EQI, NEI = 18, 19, 20, 21, 22, 23
LTD, LED, GTD, GED, EQD, NED = 24, 25, 26, 27, 28, 29
ANDB, ORB, NOTB = 30, 31, 32
ISNULL, ISNOTNULL = 33, 34
IFELSE, COALESCE2, OUT = 35, 36, 37
PICK, NIP = 38, 39

_CMP_I = {"<": LTI, "<=": LEI, ">": GTI, ">=": GEI, "==": EQI, "!=": NEI}
_CMP_D = {"<": LTD, "<=": LED, ">": GTD, ">=": GED, "==": EQD, "!=": NED}

_MAX_DEPTH = 30   # null bitmask is 32 bits
_MAX_INSTR = 192  # EX_MAX_INSTR

_INT_BITS = {dtypes.BOOL: 1, dtypes.INT8: 8, dtypes.INT16: 16,
             dtypes.INT32: 32, dtypes.DATE32: 32, dtypes.INT64: 64,
             dtypes.DECIMAL64: 64}

_RANK_ORDER = [dtypes.BOOL, dtypes.INT8, dtypes.INT16, dtypes.INT32,
               dtypes.DATE32, dtypes.INT64, dtypes.FLOAT32, dtypes.FLOAT64]
_RANK = {c: i for i, c in enumerate(_RANK_ORDER)}


class _Unsupported(Exception):
    pass


def _promote_dt(a: DataType, b: DataType) -> DataType:
    """Mirror exprs._promote's result dtype."""
    if a.code == b.code and a.code != dtypes.DECIMAL64:
        return a
    if a.code == dtypes.DECIMAL64 or b.code == dtypes.DECIMAL64:
        return dtypes.float64
    if a.code not in _RANK or b.code not in _RANK:
        raise _Unsupported(f"promote {a} {b}")
    return a if _RANK[a.code] >= _RANK[b.code] else b


def _is_d(dt: DataType) -> bool:
    return dt.code in (dtypes.FLOAT32, dtypes.FLOAT64)


_MAX_COLS = 16   # EX_MAX_COLS (kernel-arg block)
_MAX_OUTS = 8    # EX_MAX_OUTS


class Prog:
    __slots__ = ("instr_np", "colnames", "out_dtypes", "max_depth",
                 "expr_idx", "_dev")

    def __init__(self, instr_np, colnames, out_dtypes, max_depth, expr_idx):
        self.instr_np = instr_np          # contiguous uint8 view of ExInstr[]
        self.colnames = colnames
        self.out_dtypes = out_dtypes
        self.max_depth = max_depth
        self.expr_idx = expr_idx          # which caller exprs compiled
        self._dev = {}                    # device -> uploaded instr tensor

    def dev_instrs(self, device):
        t = self._dev.get(device)
        if t is None:
            from .pinned import to_device

            t = to_device(self.instr_np, device)
            self._dev[device] = t
        return t


class _Compiler:
    def __init__(self, schema: Dict[str, DataType]):
        self.schema = schema
        self.instrs: List[Tuple[int, int, int]] = []  # (op, a, imm_int64)
        self.cols: List[str] = []
        self.col_idx: Dict[str, int] = {}
        self.depth = 0
        self.max_depth = 0

    # ------------------------------------------------------------ helpers
    def op(self, op: int, a: int = 0, imm: int = 0, d: int = 0):
        self.instrs.append((op, a, imm))
        self.depth += d
        if self.depth > self.max_depth:
            self.max_depth = self.depth
        if self.depth > _MAX_DEPTH or len(self.instrs) > _MAX_INSTR:
            raise _Unsupported("program too large")

    def lit_d(self, v: float):
        self.op(PUSH_LIT_D, 0, struct.unpack("<q", struct.pack("<d", v))[0],
                d=1)

    def lit_i(self, v: int):
        self.op(PUSH_LIT_I, 0, int(v), d=1)

    def typeof(self, e) -> DataType:
        """Result dtype, mirroring each eval()'s promotion logic."""
        if isinstance(e, Col):
            dt = self.schema.get(e.name)
            if dt is None:
                raise _Unsupported(f"unknown col {e.name}")
            return dt
        if isinstance(e, Literal):
            return e.dtype or _infer_lit_dtype(e.value)
        if isinstance(e, (Cast, TryCast)):
            return e.to
        if isinstance(e, Arith):
            lt, rt = self.typeof(e.left), self.typeof(e.right)
            if (lt.code == dtypes.DECIMAL64 and rt.code == dtypes.DECIMAL64
                    and lt.scale == rt.scale and e.op in "+-"):
                return lt
            dt = _promote_dt(lt, rt)
            if e.op == "/" and dt.is_integer:
                return dtypes.float64
            return dt
        if isinstance(e, (Cmp, BoolOp, Not, IsNull, InList)):
            return dtypes.bool_
        if isinstance(e, Coalesce):
            return self.typeof(e.args[0])
        if isinstance(e, CaseWhen):
            vals = [self.typeof(v) for _, v in e.branches]
            if e.otherwise is not None:
                vals.append(self.typeof(e.otherwise))
            out = vals[0]
            for v in vals[1:]:
                if v.code != out.code:
                    out = _promote_dt(out, v)
            return out
        raise _Unsupported(type(e).__name__)

    def coerce(self, frm: DataType, to: DataType):
        """Emit conversion of the top slot, mirroring exprs._cast_col."""
        if frm.code == to.code and frm.scale == to.scale:
            return
        if frm.is_string or to.is_string:
            raise _Unsupported("string cast")
        if frm.code == dtypes.DECIMAL64 and to.code == dtypes.DECIMAL64:
            diff = to.scale - frm.scale
            if diff >= 0:
                self.lit_i(10 ** diff)
                self.op(MULI, d=-1)
            else:
                self.lit_i(10 ** (-diff))
                self.op(DIVI, d=-1)
            return
        if frm.code == dtypes.DECIMAL64:
            self.op(I2D)
            self.lit_d(float(10 ** frm.scale))
            self.op(DIVD, d=-1)
            if not _is_d(to):
                if to.code == dtypes.BOOL:
                    raise _Unsupported("decimal->bool")
                self.op(D2I_TRUNC)
                bits = _INT_BITS.get(to.code)
                if bits is None:
                    raise _Unsupported("decimal cast target")
                if bits < 64:
                    self.op(TRUNC_I, 0, bits)
            elif to.code == dtypes.FLOAT32:
                raise _Unsupported("float32")
            return
        if to.code == dtypes.DECIMAL64:
            if not _is_d(frm):
                self.op(I2D)
            self.lit_d(float(10 ** to.scale))
            self.op(MULD, d=-1)
            self.op(D2I_ROUND)
            return
        if to.code == dtypes.FLOAT32 or frm.code == dtypes.FLOAT32:
            raise _Unsupported("float32")
        if _is_d(to) and not _is_d(frm):
            self.op(I2D)
            return
        if _is_d(frm) and not _is_d(to):
            if to.code == dtypes.BOOL:
                raise _Unsupported("float->bool")
            self.op(D2I_TRUNC)
            bits = _INT_BITS[to.code]
            if bits < 64:
                self.op(TRUNC_I, 0, bits)
            return
        # int-family to int-family
        if to.code == dtypes.BOOL:
            self.op(NEZ)
            return
        fb, tb = _INT_BITS[frm.code], _INT_BITS[to.code]
        if tb < fb:
            self.op(TRUNC_I, 0, tb)

    def to_bool(self, dt: DataType):
        if dt.code != dtypes.BOOL:
            self.op(NEZ)

    # --------------------------------------------------------------- emit
    def emit(self, e) -> DataType:
        if isinstance(e, Col):
            dt = self.typeof(e)
            if dt.is_string or dt.code == dtypes.FLOAT32 or dt.code == dtypes.LIST:
                raise _Unsupported(f"col dtype {dt}")
            idx = self.col_idx.get(e.name)
            if idx is None:
                idx = len(self.cols)
                self.cols.append(e.name)
                self.col_idx[e.name] = idx
            self.op(PUSH_COL, idx, d=1)
            return dt
        if isinstance(e, Literal):
            dt = self.typeof(e)
            if e.value is None:
                if dt.is_string:
                    raise _Unsupported("string literal")
                self.op(PUSH_NULL, d=1)
                return dt
            if dt.is_string:
                raise _Unsupported("string literal")
            if dt.code == dtypes.DECIMAL64:
                self.lit_i(int(round(float(e.value) * 10 ** dt.scale)))
            elif _is_d(dt):
                self.lit_d(float(e.value))
            elif dt.code == dtypes.BOOL:
                self.lit_i(1 if e.value else 0)
            else:
                self.lit_i(int(e.value))
            return dt
        if isinstance(e, (Cast, TryCast)):
            frm = self.emit(e.child)
            self.coerce(frm, e.to)
            return e.to
        if isinstance(e, Arith):
            return self._emit_arith(e)
        if isinstance(e, Cmp):
            return self._emit_cmp(e)
        if isinstance(e, BoolOp):
            bop = ANDB if e.op == "and" else ORB
            dt0 = self.emit(e.args[0])
            self.to_bool(dt0)
            for a in e.args[1:]:
                dt = self.emit(a)
                self.to_bool(dt)
                self.op(bop, d=-1)
            return dtypes.bool_
        if isinstance(e, Not):
            dt = self.emit(e.child)
            self.to_bool(dt)
            self.op(NOTB)
            return dtypes.bool_
        if isinstance(e, IsNull):
            self.emit(e.child)
            self.op(ISNULL)
            return dtypes.bool_
        if isinstance(e, Coalesce):
            dt0 = self.emit(e.args[0])
            for a in e.args[1:]:
                dt = self.emit(a)
                self.coerce(dt, dt0)
                self.op(COALESCE2, d=-1)
            return dt0
        if isinstance(e, CaseWhen):
            out_dt = self.typeof(e)
            if out_dt.is_string:
                raise _Unsupported("string case")
            self._emit_case(list(e.branches), e.otherwise, out_dt)
            return out_dt
        if isinstance(e, InList):
            return self._emit_inlist(e)
        raise _Unsupported(type(e).__name__)

    def _emit_arith(self, e: Arith) -> DataType:
        lt, rt = self.typeof(e.left), self.typeof(e.right)
        if (lt.code == dtypes.DECIMAL64 and rt.code == dtypes.DECIMAL64
                and lt.scale == rt.scale and e.op in "+-"):
            self.emit(e.left)
            self.emit(e.right)
            self.op(ADDI if e.op == "+" else SUBI, d=-1)
            return lt
        dt = _promote_dt(lt, rt)
        int_div = e.op == "/" and dt.is_integer
        # Spark `/` on integers is double division; convert operands to
        # double as they are pushed (the stack has no swap)
        tgt = dtypes.float64 if int_div else dt
        self.emit(e.left)
        self.coerce(lt, tgt)
        self.emit(e.right)
        self.coerce(rt, tgt)
        if e.op in "+-*":
            if _is_d(tgt):
                self.op({"+": ADDD, "-": SUBD, "*": MULD}[e.op], d=-1)
            else:
                self.op({"+": ADDI, "-": SUBI, "*": MULI}[e.op], d=-1)
            return dt
        if e.op == "/":
            if not _is_d(tgt):
                raise _Unsupported("int / not double")  # unreachable
            self.op(DIVD, d=-1)
            return dtypes.float64 if int_div else dt
        if e.op == "%":
            if _is_d(tgt):
                raise _Unsupported("float %")
            self.op(MODI, d=-1)
            return dt
        raise _Unsupported(f"arith {e.op}")

    def _emit_cmp(self, e: Cmp) -> DataType:
        lt, rt = self.typeof(e.left), self.typeof(e.right)
        if lt.is_string or rt.is_string:
            raise _Unsupported("string cmp")
        dt = _promote_dt(lt, rt)
        self.emit(e.left)
        self.coerce(lt, dt)
        self.emit(e.right)
        self.coerce(rt, dt)
        self.op((_CMP_D if _is_d(dt) else _CMP_I)[e.op], d=-1)
        return dtypes.bool_

    def _emit_case(self, branches, otherwise, out_dt: DataType):
        cond, val = branches[0]
        cdt = self.emit(cond)
        self.to_bool(cdt)
        vdt = self.emit(val)
        self.coerce(vdt, out_dt)
        if len(branches) > 1:
            self._emit_case(branches[1:], otherwise, out_dt)
        elif otherwise is not None:
            odt = self.emit(otherwise)
            self.coerce(odt, out_dt)
        else:
            self.op(PUSH_NULL, d=1)
        self.op(IFELSE, d=-2)

    def _emit_inlist(self, e: InList) -> DataType:
        cdt = self.typeof(e.child)
        if cdt.is_string:
            raise _Unsupported("string inlist")
        vals = [v for v in e.values if v is not None]
        self.emit(e.child)
        if not vals:
            # all-false, null iff child null: x != x
            self.op(PICK, 0, d=1)
            self.op(NEI, d=-1)
            return dtypes.bool_
        as_d = cdt.code == dtypes.DECIMAL64
        if as_d:
            # mirror InList.eval: decimal membership compares as float64
            self.op(I2D)
            self.lit_d(float(10 ** cdt.scale))
            self.op(DIVD, d=-1)
        first = True
        for v in vals:
            self.op(PICK, 1 if not first else 0, d=1)
            if as_d or _is_d(cdt):
                self.lit_d(float(v))
                self.op(EQD, d=-1)
            else:
                self.lit_i(int(v))
                self.op(EQI, d=-1)
            if not first:
                self.op(ORB, d=-1)
            first = False
        self.op(NIP, d=-1)
        return dtypes.bool_


_INSTR_DTYPE = np.dtype([("op", "<i4"), ("a", "<i4"), ("imm", "<i8")])


def compile_exprs(exprs: List, schema: Dict[str, DataType]) -> Optional[Prog]:
    """Compile each expr independently into one merged program (shared
    column table, one OUT per compiled expr). Returns None if nothing
    compiled. Exprs that fail stay interpreted (Prog.expr_idx says which
    compiled)."""
    comp = _Compiler(schema)
    out_dtypes: List[DataType] = []
    expr_idx: List[int] = []
    for i, e in enumerate(exprs):
        if len(out_dtypes) >= _MAX_OUTS:
            break
        mark_i, mark_d = len(comp.instrs), comp.depth
        mark_c = len(comp.cols)
        try:
            dt = comp.emit(e)
            if dt.is_string or dt.code == dtypes.FLOAT32:
                raise _Unsupported("output dtype")
            if len(comp.cols) > _MAX_COLS:
                raise _Unsupported("too many input columns")
            comp.op(OUT, len(out_dtypes), d=-1)
        except _Unsupported:
            del comp.instrs[mark_i:]
            comp.depth = mark_d
            for nm in comp.cols[mark_c:]:
                del comp.col_idx[nm]
            del comp.cols[mark_c:]
            continue
        out_dtypes.append(dt)
        expr_idx.append(i)
    if not expr_idx or len(comp.instrs) > _MAX_INSTR:
        return None
    arr = np.zeros(len(comp.instrs), dtype=_INSTR_DTYPE)
    for j, (op, a, imm) in enumerate(comp.instrs):
        arr[j] = (op, a, imm)
    return Prog(arr.view(np.uint8).reshape(-1), comp.cols, out_dtypes,
                comp.max_depth, expr_idx)


def _host_descs(cols):
    """AuColDesc[] as a HOST numpy array (device pointers inside); the
    launch shim copies it into the kernel-argument block, so nothing is
    uploaded per call."""
    from . import native

    arr = np.zeros(len(cols), dtype=native._DESC_DTYPE)
    keep = []
    for i, col in enumerate(cols):
        data = col.data if col.data.is_contiguous() else col.data.contiguous()
        keep.append(data)
        arr[i]["data"] = data.data_ptr()
        if col.validity is not None:
            v = col.validity if col.validity.is_contiguous() else col.validity.contiguous()
            keep.append(v)
            arr[i]["validity"] = v.data_ptr()
        arr[i]["dtype"] = col.dtype.code
        arr[i]["scale"] = col.dtype.scale
    return arr, keep


def compile_all(exprs: List, schema: Dict[str, DataType]) -> Optional[List[Prog]]:
    """Cover all compilable exprs with as many Progs as the kernel-arg
    limits require (each <=8 outputs / <=16 input columns). Each Prog's
    expr_idx refers to positions in the original `exprs` list."""
    progs: List[Prog] = []
    remaining = list(range(len(exprs)))
    while remaining:
        p = compile_exprs([exprs[i] for i in remaining], schema)
        if p is None:
            break
        p.expr_idx = [remaining[j] for j in p.expr_idx]
        progs.append(p)
        covered = set(p.expr_idx)
        remaining = [i for i in remaining if i not in covered]
        if len(p.out_dtypes) < _MAX_OUTS:
            break  # cap not hit: everything left is unsupported
    return progs or None


def run(prog: Prog, batch) -> List[Column]:
    """Execute a compiled program on a device batch -> output Columns."""
    from . import native

    device = batch.device
    n = batch.num_rows
    outs = []
    for dt in prog.out_dtypes:
        data = torch.empty(n, dtype=dt.torch_dtype, device=device)
        val = torch.empty(n, dtype=torch.bool, device=device)
        outs.append(Column(dt, data, val))
    if n == 0:
        return outs
    lib = native.require()
    cols = [batch.column(nm) for nm in prog.colnames]
    descs, keep = _host_descs(cols)
    odescs, okeep = _host_descs(outs)
    pdev = prog.dev_instrs(device)
    n_instr = prog.instr_np.nbytes // 16
    rc = lib.au_expr_exec(pdev.data_ptr(), n_instr,
                          descs.ctypes.data, len(cols),
                          odescs.ctypes.data, len(outs),
                          max(prog.max_depth, 1), n,
                          native.stream_ptr(device))
    native.check(rc, "au_expr_exec")
    return outs

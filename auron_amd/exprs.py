"""Expression trees + vectorized evaluator (Spark SQL semantics).

Role parity: the reference's expression surface —
proto expression nodes (/root/reference/native-engine/auron-planner/proto/auron.proto:60-130),
datafusion-ext-exprs (TryCast, StringStartsWith/EndsWith/Contains, ...) and
the Spark-semantics scalar functions in datafusion-ext-functions.

Evaluation is columnar over `Column`s. On GPU the tensor ops run as HIP
kernels on device-resident data; SQL-specific hot paths (hashing, hash
tables) go through `auron_amd.ops` native kernels instead.

Spark null semantics implemented here:
 - arithmetic/comparison propagate null (validity AND)
 - AND/OR use Kleene 3-valued logic
 - x / 0 and x % 0 yield null (non-ANSI Spark behavior)
"""
from __future__ import annotations

from dataclasses import dataclass, field
from typing import List, Optional, Sequence, Tuple

import torch

from . import dtypes, strings
from .column import Column, RecordBatch, compact_validity
from .dtypes import DataType


class Expr:
    """Base expression. Subclasses implement eval(batch) -> Column.

    CSE: every subclass's eval is wrapped with an identity-keyed memo
    (CachedExprsEvaluator analogue, datafusion-ext-commons cached_exprs_
    evaluator): when an operator opens a memo scope on the batch
    (`eval_scope`), shared subtree OBJECTS evaluate once per batch."""

    def __init_subclass__(cls, **kw):
        super().__init_subclass__(**kw)
        inner = cls.__dict__.get("eval")
        if inner is None or getattr(inner, "_cse_wrapped", False):
            return

        def eval(self, batch, _inner=inner):
            memo = getattr(batch, "_eval_memo", None)
            if memo is None:
                return _inner(self, batch)
            key = id(self)
            hit = memo.get(key)
            # the memo pins the expr object so its id can't be recycled by
            # a new expr allocated during the same scope
            if hit is None or hit[0] is not self:
                hit = (self, _inner(self, batch))
                memo[key] = hit
            return hit[1]

        eval._cse_wrapped = True
        cls.eval = eval

    def eval(self, batch: RecordBatch) -> Column:
        raise NotImplementedError

    # builder sugar
    def __add__(self, o):
        return Arith("+", self, _lit(o))

    def __sub__(self, o):
        return Arith("-", self, _lit(o))

    def __mul__(self, o):
        return Arith("*", self, _lit(o))

    def __truediv__(self, o):
        return Arith("/", self, _lit(o))

    def __eq__(self, o):  # type: ignore[override]
        return Cmp("==", self, _lit(o))

    def __ne__(self, o):  # type: ignore[override]
        return Cmp("!=", self, _lit(o))

    def __lt__(self, o):
        return Cmp("<", self, _lit(o))

    def __le__(self, o):
        return Cmp("<=", self, _lit(o))

    def __gt__(self, o):
        return Cmp(">", self, _lit(o))

    def __ge__(self, o):
        return Cmp(">=", self, _lit(o))

    def __and__(self, o):
        return BoolOp("and", [self, _lit(o)])

    def __or__(self, o):
        return BoolOp("or", [self, _lit(o)])

    def __invert__(self):
        return Not(self)

    def __hash__(self):
        return id(self)

    def alias(self, name: str) -> "Aliased":
        return Aliased(self, name)

    def is_null(self):
        return IsNull(self)

    def is_not_null(self):
        return Not(IsNull(self))

    def cast(self, dt: DataType):
        return Cast(self, dt)

    def isin(self, values):
        return InList(self, list(values))

    def between(self, lo, hi):
        return BoolOp("and", [Cmp(">=", self, _lit(lo)), Cmp("<=", self, _lit(hi))])

    def like(self, pattern: str):
        return Like(self, pattern)

    def substr(self, start: int, length: int):
        return Substr(self, start, length)


import contextlib


@contextlib.contextmanager
def eval_scope(batch):
    """Open a CSE memo on `batch`: shared Expr objects evaluate once."""
    had = hasattr(batch, "_eval_memo")
    if not had:
        batch._eval_memo = {}
    try:
        yield batch
    finally:
        if not had:
            del batch._eval_memo


@dataclass(eq=False)
class Aliased:
    expr: Expr
    name: str


def _lit(v):
    if isinstance(v, Expr):
        return v
    return Literal(v)


def _infer_lit_dtype(v) -> DataType:
    if isinstance(v, bool):
        return dtypes.bool_
    if isinstance(v, int):
        return dtypes.int64
    if isinstance(v, float):
        return dtypes.float64
    if isinstance(v, str):
        return dtypes.string
    if v is None:
        return dtypes.int64
    raise TypeError(f"literal {v!r}")


def _all_valid(n, device):
    return torch.ones(n, dtype=torch.bool, device=device)


def combine_validity(*cols: Column) -> Optional[torch.Tensor]:
    v = None
    for c in cols:
        if c.validity is not None:
            v = c.validity if v is None else (v & c.validity)
    return v


@dataclass(eq=False)
class Literal(Expr):
    value: object
    dtype: Optional[DataType] = None

    def eval(self, batch: RecordBatch) -> Column:
        dt = self.dtype or _infer_lit_dtype(self.value)
        n = batch.num_rows
        device = batch.device
        if self.value is None:
            validity = torch.zeros(n, dtype=torch.bool, device=device)
            if dt.is_string:
                return Column(dt, torch.empty(0, dtype=torch.uint8, device=device), validity,
                              torch.zeros(n + 1, dtype=torch.int64, device=device))
            if dt.code == dtypes.DECIMAL128:
                return Column(dt, torch.zeros((n, 2), dtype=torch.int64, device=device), validity)
            return Column(dt, torch.zeros(n, dtype=dt.torch_dtype, device=device), validity)
        if dt.is_string:
            b = self.value.encode("utf-8")
            data = torch.tensor(list(b), dtype=torch.uint8, device=device).repeat(n) if b else torch.empty(0, dtype=torch.uint8, device=device)
            offsets = (torch.arange(n + 1, dtype=torch.int64, device=device) * len(b))
            return Column(dt, data, None, offsets)
        if dt.code == dtypes.DECIMAL64:
            v = int(round(float(self.value) * 10 ** dt.scale))
            return Column(dt, torch.full((n,), v, dtype=torch.int64, device=device))
        if dt.code == dtypes.DECIMAL128:
            from decimal import Decimal

            iv = int(Decimal(str(self.value)).scaleb(dt.scale).to_integral_value())
            m64 = (1 << 64) - 1
            lo = iv & m64
            if lo >= 1 << 63:
                lo -= 1 << 64
            limbs = torch.tensor([[lo, iv >> 64]], dtype=torch.int64,
                                 device=device)
            return Column(dt, limbs.expand(n, 2).contiguous())
        return Column(dt, torch.full((n,), self.value, dtype=dt.torch_dtype, device=device))


@dataclass(eq=False)
class Col(Expr):
    name: str

    def eval(self, batch: RecordBatch) -> Column:
        return batch.column(self.name)


def _promote(l: Column, r: Column) -> Tuple[Column, Column, DataType]:
    """Numeric type promotion (Spark's least-common-type, simplified)."""
    a, b = l.dtype, r.dtype
    if a.code == b.code and a.code != dtypes.DECIMAL64:
        return l, r, a
    if dtypes.DECIMAL128 in (a.code, b.code):
        return (_cast_col(l, dtypes.float64), _cast_col(r, dtypes.float64),
                dtypes.float64)
    if a.code == dtypes.DECIMAL64 or b.code == dtypes.DECIMAL64:
        # operate in float64 for mixed decimal arithmetic; dedicated decimal
        # kernels (scaled-int64) arrive with the decimal op set
        lf = _cast_col(l, dtypes.float64)
        rf = _cast_col(r, dtypes.float64)
        return lf, rf, dtypes.float64
    order = [dtypes.BOOL, dtypes.INT8, dtypes.INT16, dtypes.INT32, dtypes.DATE32, dtypes.INT64, dtypes.FLOAT32, dtypes.FLOAT64]
    rank = {c: i for i, c in enumerate(order)}
    target = a if rank[a.code] >= rank[b.code] else b
    return _cast_col(l, target), _cast_col(r, target), target


_US_PER_DAY = 86_400_000_000
_US_PER_SEC = 1_000_000


def _cast_col(c: Column, dt: DataType) -> Column:
    if c.dtype.code == dt.code and c.dtype.scale == dt.scale:
        return c
    if c.dtype.is_string or dt.is_string:
        return _cast_string(c, dt)
    if c.dtype.code == dtypes.TIMESTAMP or dt.code == dtypes.TIMESTAMP:
        return _cast_timestamp(c, dt)
    if c.dtype.code == dtypes.DECIMAL128 or dt.code == dtypes.DECIMAL128:
        return _cast_decimal128(c, dt)
    if c.dtype.code == dtypes.DECIMAL64 and dt.code == dtypes.DECIMAL64:
        diff = dt.scale - c.dtype.scale
        data = c.data * (10 ** diff) if diff >= 0 else torch.div(c.data, 10 ** (-diff), rounding_mode="trunc")
        return Column(dt, data, c.validity)
    if c.dtype.code == dtypes.DECIMAL64:
        f = c.data.to(torch.float64) / (10 ** c.dtype.scale)
        if dt.is_float:
            return Column(dt, f.to(dt.torch_dtype), c.validity)
        return Column(dt, f.to(dt.torch_dtype), c.validity)
    if dt.code == dtypes.DECIMAL64:
        scaled = torch.round(c.data.to(torch.float64) * (10 ** dt.scale)).to(torch.int64)
        return Column(dt, scaled, c.validity)
    if c.dtype.is_float and dt.is_integer:
        # Spark cast double->int truncates toward zero
        return Column(dt, c.data.trunc().to(dt.torch_dtype), c.validity)
    return Column(dt, c.data.to(dt.torch_dtype), c.validity)


def dec128_to_float64(data: torch.Tensor, scale: int) -> torch.Tensor:
    """[n,2] limbs -> float64 value/10^scale (precision-limited, like the
    reference's decimal->double cast)."""
    # signed-limb composition: value = (hi + carry)*2^64 + lo_signed with
    # carry = 1 when the low limb's bit pattern is negative as int64 —
    # avoids the 2^64-cancellation that loses ~11 bits near zero
    lo = data[:, 0]
    hi_adj = data[:, 1] + (lo < 0).to(torch.int64)
    v = hi_adj.to(torch.float64) * (2.0 ** 64) + lo.to(torch.float64)
    return v / (10.0 ** scale)


def dec64_to_dec128(data: torch.Tensor) -> torch.Tensor:
    """int64 scaled values -> [n,2] limbs (sign-extended high limb)."""
    hi = data >> 63  # arithmetic: 0 or -1
    return torch.stack([data, hi], dim=1)


def _cast_decimal128(c: Column, dt: DataType) -> Column:
    if c.dtype.code == dtypes.DECIMAL128:
        f = dec128_to_float64(c.data, c.dtype.scale)
        if dt.code == dtypes.DECIMAL128 or dt.code == dtypes.DECIMAL64:
            # rescale through float64 only when scales differ; same-scale
            # narrowing checks the value fits the 64-bit backing exactly
            if dt.scale == c.dtype.scale and dt.code == dtypes.DECIMAL64:
                lo, hi = c.data[:, 0], c.data[:, 1]
                fits = hi == (lo >> 63)
                if not bool(fits.all() if lo.device.type != "cuda" else True):
                    raise ValueError("decimal128 -> decimal64 overflow")
                return Column(dt, lo.clone(), c.validity)
            scaled = torch.round(f * (10.0 ** dt.scale)).to(torch.int64)
            if dt.code == dtypes.DECIMAL64:
                return Column(dt, scaled, c.validity)
            return Column(dt, dec64_to_dec128(scaled), c.validity)
        if dt.is_float:
            return Column(dt, f.to(dt.torch_dtype), c.validity)
        if dt.is_integer:
            return Column(dt, f.trunc().to(dt.torch_dtype), c.validity)
        raise TypeError(f"cast decimal128 -> {dt}")
    # -> decimal128
    if c.dtype.code == dtypes.DECIMAL64 and c.dtype.scale == dt.scale:
        return Column(dt, dec64_to_dec128(c.data), c.validity)
    as64 = _cast_col(c, dtypes.decimal64(18, dt.scale))
    return Column(dt, dec64_to_dec128(as64.data), c.validity)


def _cast_timestamp(c: Column, dt: DataType) -> Column:
    """Spark timestamp cast matrix (micros-since-epoch storage):
    ts<->date via whole days (floor, so pre-epoch rounds down),
    ts<->integral via SECONDS, ts<->float via fractional seconds."""
    if c.dtype.code == dtypes.TIMESTAMP:
        us = c.data
        if dt.code == dtypes.DATE32:
            days = torch.div(us, _US_PER_DAY, rounding_mode="floor")
            return Column(dt, days.to(torch.int32), c.validity)
        if dt.is_integer:
            secs = torch.div(us, _US_PER_SEC, rounding_mode="floor")
            return Column(dt, secs.to(dt.torch_dtype), c.validity)
        if dt.is_float:
            return Column(dt, (us.to(torch.float64) / _US_PER_SEC).to(dt.torch_dtype),
                          c.validity)
        raise TypeError(f"cast timestamp -> {dt}")
    # -> timestamp
    if c.dtype.code == dtypes.DATE32:
        return Column(dt, c.data.to(torch.int64) * _US_PER_DAY, c.validity)
    if c.dtype.is_integer:
        return Column(dt, c.data.to(torch.int64) * _US_PER_SEC, c.validity)
    if c.dtype.is_float:
        return Column(dt, (c.data.to(torch.float64) * _US_PER_SEC).to(torch.int64),
                      c.validity)
    raise TypeError(f"cast {c.dtype} -> timestamp")


def _parse_ts_micros(v: str):
    """'YYYY-MM-DD[ HH:MM:SS[.ffffff]]' -> micros since epoch (UTC),
    None on parse failure (non-ANSI cast)."""
    import datetime as _dt_mod

    t = v.strip().replace("T", " ")
    if "." in t:
        # py3.10 fromisoformat needs exactly 3 or 6 fractional digits
        head, frac = t.rsplit(".", 1)
        if frac.isdigit():
            t = head + "." + (frac + "000000")[:6]
    try:
        if " " in t:
            d = _dt_mod.datetime.fromisoformat(t)
        else:
            d = _dt_mod.datetime.combine(_dt_mod.date.fromisoformat(t),
                                         _dt_mod.time())
    except ValueError:
        return None
    epoch = _dt_mod.datetime(1970, 1, 1)
    delta = d.replace(tzinfo=None) - epoch
    return delta.days * 86_400_000_000 + delta.seconds * 1_000_000 + delta.microseconds


def _format_ts_micros(us: int) -> str:
    import datetime as _dt_mod

    d = _dt_mod.datetime(1970, 1, 1) + _dt_mod.timedelta(microseconds=us)
    base = d.strftime("%Y-%m-%d %H:%M:%S")
    if d.microsecond:
        base += f".{d.microsecond:06d}".rstrip("0")
    return base


def _cast_string(c: Column, dt: DataType) -> Column:
    import datetime as _dt_mod

    _EPOCH = _dt_mod.date(1970, 1, 1)
    if dt.is_string and c.dtype.is_string:
        return c
    if c.dtype.is_string:
        # string -> numeric/date: host round-trip (cold path in TPC-DS);
        # unparseable values yield null (try_cast/non-ANSI cast semantics)
        vals = c.to_pylist()
        out = []
        for v in vals:
            if v is None:
                out.append(None)
            elif dt.code == dtypes.DATE32:
                try:
                    out.append((_dt_mod.date.fromisoformat(v.strip()) - _EPOCH).days)
                except ValueError:
                    out.append(None)
            elif dt.code == dtypes.TIMESTAMP:
                out.append(_parse_ts_micros(v))
            else:
                try:
                    out.append(float(v) if dt.is_float or dt.code == dtypes.DECIMAL64 else int(float(v)))
                except ValueError:
                    out.append(None)
        return Column.from_pylist(out, dt, c.device)
    # numeric/date -> string: host round-trip
    vals = c.to_pylist()
    if c.dtype.code == dtypes.DATE32:
        out = [None if v is None else (_EPOCH + _dt_mod.timedelta(days=int(v))).isoformat()
               for v in vals]
    elif c.dtype.code == dtypes.TIMESTAMP:
        out = [None if v is None else _format_ts_micros(int(v)) for v in vals]
    else:
        out = [None if v is None else (str(int(v)) if c.dtype.is_integer else str(v)) for v in vals]
    return Column.from_pylist(out, dt, c.device)


@dataclass(eq=False)
class Cast(Expr):
    child: Expr
    to: DataType

    def eval(self, batch: RecordBatch) -> Column:
        return _cast_col(self.child.eval(batch), self.to)


@dataclass(eq=False)
class TryCast(Expr):
    """try_cast (ext-exprs cast.rs): unparseable inputs yield null.
    String parsing in _cast_string already nulls on failure; this node
    declares the intent at the plan surface."""
    child: Expr
    to: DataType

    def eval(self, batch: RecordBatch) -> Column:
        return _cast_col(self.child.eval(batch), self.to)


@dataclass(eq=False)
class Arith(Expr):
    op: str
    left: Expr
    right: Expr

    def eval(self, batch: RecordBatch) -> Column:
        l = self.left.eval(batch)
        r = self.right.eval(batch)
        # decimal same-scale fast path for +/-
        if (l.dtype.code == dtypes.DECIMAL64 and r.dtype.code == dtypes.DECIMAL64
                and l.dtype.scale == r.dtype.scale and self.op in "+-"):
            data = l.data + r.data if self.op == "+" else l.data - r.data
            return Column(l.dtype, data, combine_validity(l, r))
        l, r, dt = _promote(l, r)
        validity = combine_validity(l, r)
        a, b = l.data, r.data
        if self.op == "+":
            data = a + b
        elif self.op == "-":
            data = a - b
        elif self.op == "*":
            data = a * b
        elif self.op in ("/", "%"):
            # unconditional guard: probing `zero.any()` would sync the
            # stream on every division (r2 profile tax)
            zero = b == 0
            b = torch.where(zero, torch.ones_like(b), b)
            v2 = ~zero
            validity = v2 if validity is None else (validity & v2)
            if self.op == "/":
                if dt.is_integer:
                    # Spark SQL `/` is double division for any input type
                    # (integer division is a separate `div` operator)
                    data = a.to(torch.float64) / b.to(torch.float64)
                else:
                    data = a / b
            else:
                data = torch.remainder(a, b)
                if dt.is_integer:
                    # Spark % takes sign of dividend (fmod), torch.remainder takes divisor sign
                    data = a - torch.div(a, b, rounding_mode="trunc") * b
        else:
            raise ValueError(self.op)
        if self.op == "/" and dt.is_integer:
            return Column(dtypes.float64, data, validity)
        return Column(dt, data, validity)


@dataclass(eq=False)
class Cmp(Expr):
    op: str
    left: Expr
    right: Expr

    def eval(self, batch: RecordBatch) -> Column:
        # literal-string fast path: compare against the pattern directly
        # instead of materializing a repeated literal column
        if (self.op in ("==", "!=") and isinstance(self.right, Literal)
                and isinstance(self.right.value, str)):
            l = self.left.eval(batch)
            if l.dtype.is_string:
                m = strings.eq_literal(l, self.right.value)
                if self.op == "!=":
                    m = ~m
                return Column(dtypes.bool_, m, l.validity)
        l = self.left.eval(batch)
        r = self.right.eval(batch)
        if l.dtype.is_string or r.dtype.is_string:
            data = strings.compare(l, r, self.op)
            return Column(dtypes.bool_, data, combine_validity(l, r))
        l, r, _ = _promote(l, r)
        a, b = l.data, r.data
        fn = {"==": torch.eq, "!=": torch.ne, "<": torch.lt, "<=": torch.le, ">": torch.gt, ">=": torch.ge}[self.op]
        return Column(dtypes.bool_, fn(a, b), combine_validity(l, r))


@dataclass(eq=False)
class BoolOp(Expr):
    op: str  # "and" | "or"
    args: List[Expr]

    def eval(self, batch: RecordBatch) -> Column:
        cols = [a.eval(batch) for a in self.args]
        n = batch.num_rows
        device = batch.device
        val = cols[0].data.bool()
        valid = cols[0].validity if cols[0].validity is not None else _all_valid(n, device)
        for c in cols[1:]:
            v2 = c.validity if c.validity is not None else _all_valid(n, device)
            b = c.data.bool()
            if self.op == "and":
                # Kleene: FALSE dominates null
                out_valid = (valid & v2) | (valid & ~val) | (v2 & ~b)
                val = (val | ~valid) & (b | ~v2)  # treat null as true; masked by out_valid
                valid = out_valid
            else:
                out_valid = (valid & v2) | (valid & val) | (v2 & b)
                val = (val & valid) | (b & v2)
                valid = out_valid
        valid = compact_validity(valid)
        return Column(dtypes.bool_, val, valid)


@dataclass(eq=False)
class Not(Expr):
    child: Expr

    def eval(self, batch: RecordBatch) -> Column:
        c = self.child.eval(batch)
        return Column(dtypes.bool_, ~c.data.bool(), c.validity)


@dataclass(eq=False)
class IsNull(Expr):
    child: Expr

    def eval(self, batch: RecordBatch) -> Column:
        c = self.child.eval(batch)
        if c.validity is None:
            return Column(dtypes.bool_, torch.zeros(len(c), dtype=torch.bool, device=c.device))
        return Column(dtypes.bool_, ~c.validity)


@dataclass(eq=False)
class Coalesce(Expr):
    args: List[Expr]

    def eval(self, batch: RecordBatch) -> Column:
        cols = [a.eval(batch) for a in self.args]
        out = cols[0]
        for c in cols[1:]:
            if out.validity is None:
                return out
            if out.dtype.is_string:
                take = out.validity
                idx = torch.arange(len(out), device=out.device)
                # build merged via gather from whichever side
                vals = out.to_pylist()
                vals2 = c.to_pylist()
                merged = [v if v is not None else vals2[i] for i, v in enumerate(vals)]
                out = Column.from_pylist(merged, out.dtype, str(out.device))
            else:
                c2 = _cast_col(c, out.dtype)
                data = torch.where(out.validity, out.data, c2.data)
                if c2.validity is None:
                    validity = None
                else:
                    validity = out.validity | c2.validity
                    validity = compact_validity(validity)
                out = Column(out.dtype, data, validity)
        return out


@dataclass(eq=False)
class CaseWhen(Expr):
    branches: List[Tuple[Expr, Expr]]
    otherwise: Optional[Expr] = None

    def eval(self, batch: RecordBatch) -> Column:
        n = batch.num_rows
        device = batch.device
        decided = torch.zeros(n, dtype=torch.bool, device=device)
        result_data = None
        result_valid = torch.zeros(n, dtype=torch.bool, device=device)
        out_dt = None
        pieces = []
        for cond, val in self.branches:
            c = cond.eval(batch)
            hit = c.data.bool()
            if c.validity is not None:
                hit = hit & c.validity
            take = hit & ~decided
            decided = decided | hit
            pieces.append((take, val))
        vals = [v.eval(batch) for _, v in pieces]
        if self.otherwise is not None:
            vals.append(self.otherwise.eval(batch))
            pieces.append((~decided, None))
        # promote all to common type
        out_dt = vals[0].dtype
        for v in vals[1:]:
            if v.dtype.code != out_dt.code:
                _, _, out_dt = _promote(Column(out_dt, torch.zeros(0, dtype=out_dt.torch_dtype, device=device)),
                                        Column(v.dtype, torch.zeros(0, dtype=v.dtype.torch_dtype, device=device)))
        if out_dt.is_string:
            # host path for string case/when
            py = [None] * n
            for (take, _), v in zip(pieces, vals):
                tv = v.to_pylist()
                for i in torch.nonzero(take).flatten().tolist():
                    py[i] = tv[i]
            return Column.from_pylist(py, out_dt, str(device))
        vals = [_cast_col(v, out_dt) for v in vals]
        result_data = torch.zeros(n, dtype=out_dt.torch_dtype, device=device)
        for (take, _), v in zip(pieces, vals):
            result_data = torch.where(take, v.data, result_data)
            vv = v.validity if v.validity is not None else _all_valid(n, device)
            result_valid = torch.where(take, vv, result_valid)
        result_valid = compact_validity(result_valid)
        return Column(out_dt, result_data, result_valid)


@dataclass(eq=False)
class InList(Expr):
    child: Expr
    values: List[object]

    def eval(self, batch: RecordBatch) -> Column:
        c = self.child.eval(batch)
        if c.dtype.is_string:
            m = strings.isin(c, [v for v in self.values if v is not None])
            return Column(dtypes.bool_, m, c.validity)
        vals = torch.tensor([v for v in self.values if v is not None], dtype=c.dtype.torch_dtype if c.dtype.code != dtypes.DECIMAL64 else torch.float64, device=c.device)
        data = c.data if c.dtype.code != dtypes.DECIMAL64 else c.data.to(torch.float64) / 10 ** c.dtype.scale
        m = torch.isin(data, vals)
        return Column(dtypes.bool_, m, c.validity)


@dataclass(eq=False)
class Like(Expr):
    child: Expr
    pattern: str

    def eval(self, batch: RecordBatch) -> Column:
        c = self.child.eval(batch)
        return Column(dtypes.bool_, strings.like(c, self.pattern), c.validity)


@dataclass(eq=False)
class StringStartsWith(Expr):
    """ext-exprs StringStartsWith (vectorized prefix compare)."""
    child: Expr
    prefix: str

    def eval(self, batch: RecordBatch) -> Column:
        c = self.child.eval(batch)
        return Column(dtypes.bool_, strings.startswith(c, self.prefix),
                      c.validity)


@dataclass(eq=False)
class StringEndsWith(Expr):
    child: Expr
    suffix: str

    def eval(self, batch: RecordBatch) -> Column:
        c = self.child.eval(batch)
        return Column(dtypes.bool_, strings.endswith(c, self.suffix),
                      c.validity)


@dataclass(eq=False)
class StringContains(Expr):
    child: Expr
    needle: str

    def eval(self, batch: RecordBatch) -> Column:
        c = self.child.eval(batch)
        return Column(dtypes.bool_, strings.contains(c, self.needle),
                      c.validity)


@dataclass(eq=False)
class Substr(Expr):
    child: Expr
    start: int  # 1-based (SQL)
    length: int

    def eval(self, batch: RecordBatch) -> Column:
        c = self.child.eval(batch)
        return strings.substr(c, self.start, self.length)


@dataclass(eq=False)
class Length(Expr):
    child: Expr

    def eval(self, batch: RecordBatch) -> Column:
        c = self.child.eval(batch)
        off = c.offsets.to(torch.int64)
        return Column(dtypes.int32, (off[1:] - off[:-1]).to(torch.int32), c.validity)


@dataclass(eq=False)
class Sqrt(Expr):
    child: Expr

    def eval(self, batch: RecordBatch) -> Column:
        c = _cast_col(self.child.eval(batch), dtypes.float64)
        return Column(dtypes.float64, torch.sqrt(c.data), c.validity)


@dataclass(eq=False)
class Abs(Expr):
    child: Expr

    def eval(self, batch: RecordBatch) -> Column:
        c = self.child.eval(batch)
        return Column(c.dtype, torch.abs(c.data), c.validity)


@dataclass(eq=False)
class Upper(Expr):
    child: Expr

    def eval(self, batch: RecordBatch) -> Column:
        return strings.upper(self.child.eval(batch))


@dataclass(eq=False)
class Lower(Expr):
    child: Expr

    def eval(self, batch: RecordBatch) -> Column:
        return strings.lower(self.child.eval(batch))


@dataclass(eq=False)
class ConcatStr(Expr):
    args: List[Expr]

    def eval(self, batch: RecordBatch) -> Column:
        cols = [a.eval(batch) for a in self.args]
        return strings.concat(cols)


def _civil_from_days(days: torch.Tensor):
    """days since 1970-01-01 -> (year, month, day), Hinnant's algorithm."""
    z = days.to(torch.int64) + 719468
    era = torch.div(z, 146097, rounding_mode="floor")
    doe = z - era * 146097
    yoe = torch.div(doe - torch.div(doe, 1460, rounding_mode="floor")
                    + torch.div(doe, 36524, rounding_mode="floor")
                    - torch.div(doe, 146096, rounding_mode="floor"), 365, rounding_mode="floor")
    y = yoe + era * 400
    doy = doe - (365 * yoe + torch.div(yoe, 4, rounding_mode="floor") - torch.div(yoe, 100, rounding_mode="floor"))
    mp = torch.div(5 * doy + 2, 153, rounding_mode="floor")
    d = doy - torch.div(153 * mp + 2, 5, rounding_mode="floor") + 1
    m = torch.where(mp < 10, mp + 3, mp - 9)
    y = y + (m <= 2).to(torch.int64)
    return y, m, d


@dataclass(eq=False)
class DatePart(Expr):
    part: str  # year | month | day
    child: Expr

    def eval(self, batch: RecordBatch) -> Column:
        c = self.child.eval(batch)
        y, m, d = _civil_from_days(c.data)
        v = {"year": y, "month": m, "day": d}[self.part]
        return Column(dtypes.int32, v.to(torch.int32), c.validity)


# ------------------------------------------------------------------ window
@dataclass(eq=False)
class WindowFunc(Expr):
    """Window function marker expr (window/mod.rs WindowFunctionProcessor
    analogue). Evaluated by the Window operator, not by Expr.eval."""
    fn: str  # row_number | rank | dense_rank | sum | avg | count | min | max | lead | lag | ...
    arg: Optional[Expr] = None
    offset: int = 1  # lead/lag/ntile/nth_value parameter
    default: object = None  # lead/lag third argument

    def eval(self, batch):  # pragma: no cover
        raise RuntimeError("WindowFunc must appear under a Window operator")


# --------------------------------------------------------------- aggregates
@dataclass(eq=False)
class AggFunc:
    """Aggregate function spec (proto AggFunction analogue, auron.proto agg enum)."""
    fn: str  # sum | count | min | max | avg | first | count_star | count_distinct
    expr: Optional[Expr] = None
    distinct: bool = False
    name: str = ""


def year(e: Expr) -> Expr:
    return DatePart("year", e)


def month(e: Expr) -> Expr:
    return DatePart("month", e)


def dayofmonth(e: Expr) -> Expr:
    return DatePart("day", e)


def col(name: str) -> Col:
    return Col(name)


def lit(v, dtype: Optional[DataType] = None) -> Literal:
    return Literal(v, dtype)

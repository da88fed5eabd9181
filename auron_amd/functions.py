"""Spark-semantics scalar functions (ext-functions parity).

Role parity: datafusion-ext-functions (spark_strings.rs, spark_dates.rs,
spark_math) and ext-exprs (SparkPartitionId, MonotonicallyIncreasingId,
RowNum). Functions are torch-vectorized; a few cold string paths bounce
through the host (documented inline).
"""
from __future__ import annotations

import contextvars
from dataclasses import dataclass
from typing import List

import torch

from . import dtypes, strings
from .column import Column, compact_validity
from .exprs import (Expr, _all_valid, _cast_col, _civil_from_days,
                    combine_validity)

# executor-injected evaluation context (partition id etc.)
EVAL_CONTEXT: contextvars.ContextVar = contextvars.ContextVar(
    "auron_eval_context", default={"partition_id": 0, "batch_ordinal": 0})


def _f64(e, batch) -> Column:
    return _cast_col(e.eval(batch), dtypes.float64)


@dataclass(eq=False)
class Round(Expr):
    """Spark round: half-up away from zero."""
    child: Expr
    ndigits: int = 0

    def eval(self, batch):
        c = _f64(self.child, batch)
        f = 10.0 ** self.ndigits
        x = c.data * f
        data = torch.where(x >= 0, torch.floor(x + 0.5), torch.ceil(x - 0.5)) / f
        return Column(dtypes.float64, data, c.validity)


@dataclass(eq=False)
class Floor(Expr):
    child: Expr

    def eval(self, batch):
        c = _f64(self.child, batch)
        return Column(dtypes.int64, torch.floor(c.data).to(torch.int64), c.validity)


@dataclass(eq=False)
class Ceil(Expr):
    child: Expr

    def eval(self, batch):
        c = _f64(self.child, batch)
        return Column(dtypes.int64, torch.ceil(c.data).to(torch.int64), c.validity)


@dataclass(eq=False)
class Exp(Expr):
    child: Expr

    def eval(self, batch):
        c = _f64(self.child, batch)
        return Column(dtypes.float64, torch.exp(c.data), c.validity)


@dataclass(eq=False)
class Ln(Expr):
    child: Expr

    def eval(self, batch):
        c = _f64(self.child, batch)
        ok = c.data > 0
        v = combine_validity(c)
        v = ok if v is None else (v & ok)
        return Column(dtypes.float64, torch.log(c.data.clamp(min=1e-300)), v)


@dataclass(eq=False)
class Log10(Expr):
    child: Expr

    def eval(self, batch):
        c = _f64(self.child, batch)
        ok = c.data > 0
        v = combine_validity(c)
        v = ok if v is None else (v & ok)
        return Column(dtypes.float64, torch.log10(c.data.clamp(min=1e-300)), v)


@dataclass(eq=False)
class Pow(Expr):
    base: Expr
    exponent: Expr

    def eval(self, batch):
        b = _f64(self.base, batch)
        e = _f64(self.exponent, batch)
        return Column(dtypes.float64, torch.pow(b.data, e.data),
                      combine_validity(b, e))


@dataclass(eq=False)
class Sign(Expr):
    child: Expr

    def eval(self, batch):
        c = _f64(self.child, batch)
        return Column(dtypes.float64, torch.sign(c.data), c.validity)


@dataclass(eq=False)
class Greatest(Expr):
    args: List[Expr]

    def eval(self, batch):
        cols = [_f64(a, batch) for a in self.args]
        data = cols[0].data
        for c in cols[1:]:
            data = torch.maximum(data, c.data)
        # Spark greatest skips nulls; null only when ALL null
        n = batch.num_rows
        device = batch.device
        any_valid = torch.zeros(n, dtype=torch.bool, device=device)
        acc = torch.full((n,), float("-inf"), dtype=torch.float64, device=device)
        for c in cols:
            v = c.validity if c.validity is not None else _all_valid(n, device)
            acc = torch.where(v, torch.maximum(acc, c.data), acc)
            any_valid = any_valid | v
        return Column(dtypes.float64, acc, compact_validity(any_valid))


@dataclass(eq=False)
class Least(Expr):
    args: List[Expr]

    def eval(self, batch):
        cols = [_f64(a, batch) for a in self.args]
        n = batch.num_rows
        device = batch.device
        any_valid = torch.zeros(n, dtype=torch.bool, device=device)
        acc = torch.full((n,), float("inf"), dtype=torch.float64, device=device)
        for c in cols:
            v = c.validity if c.validity is not None else _all_valid(n, device)
            acc = torch.where(v, torch.minimum(acc, c.data), acc)
            any_valid = any_valid | v
        return Column(dtypes.float64, acc, compact_validity(any_valid))


@dataclass(eq=False)
class NullIf(Expr):
    left: Expr
    right: Expr

    def eval(self, batch):
        from .exprs import Cmp

        l = self.left.eval(batch)
        eq = Cmp("==", self.left, self.right).eval(batch)
        hit = eq.data.bool()
        if eq.validity is not None:
            hit = hit & eq.validity
        v = l.validity if l.validity is not None else _all_valid(len(l), l.device)
        return Column(l.dtype, l.data, v & ~hit, l.offsets)


@dataclass(eq=False)
class Nvl2(Expr):
    check: Expr
    if_not_null: Expr
    if_null: Expr

    def eval(self, batch):
        from .exprs import CaseWhen, IsNull, Not

        return CaseWhen([(Not(IsNull(self.check)), self.if_not_null)],
                        self.if_null).eval(batch)


@dataclass(eq=False)
class If(Expr):
    cond: Expr
    then: Expr
    otherwise: Expr

    def eval(self, batch):
        from .exprs import CaseWhen

        return CaseWhen([(self.cond, self.then)], self.otherwise).eval(batch)


# ------------------------------------------------------------------ dates
def _days_in_month(y, m):
    dim = torch.tensor([31, 28, 31, 30, 31, 30, 31, 31, 30, 31, 30, 31],
                       dtype=torch.int64, device=y.device)
    d = dim[(m - 1).clamp(0, 11)]
    leap = ((y % 4 == 0) & (y % 100 != 0)) | (y % 400 == 0)
    return torch.where((m == 2) & leap, d + 1, d)


def _days_from_civil(y, m, d):
    """Hinnant's days_from_civil, vectorized (inverse of _civil_from_days)."""
    y = y - (m <= 2).to(torch.int64)
    era = torch.div(torch.where(y >= 0, y, y - 399), 400, rounding_mode="floor")
    yoe = y - era * 400
    mp = torch.where(m > 2, m - 3, m + 9)
    doy = torch.div(153 * mp + 2, 5, rounding_mode="floor") + d - 1
    doe = yoe * 365 + torch.div(yoe, 4, rounding_mode="floor") \
        - torch.div(yoe, 100, rounding_mode="floor") + doy
    return era * 146097 + doe - 719468


@dataclass(eq=False)
class DateAdd(Expr):
    child: Expr
    days: Expr

    def eval(self, batch):
        c = self.child.eval(batch)
        d = self.days.eval(batch)
        data = (c.data.to(torch.int64) + d.data.to(torch.int64)).to(torch.int32)
        return Column(dtypes.date32, data, combine_validity(c, d))


@dataclass(eq=False)
class DateSub(Expr):
    child: Expr
    days: Expr

    def eval(self, batch):
        c = self.child.eval(batch)
        d = self.days.eval(batch)
        data = (c.data.to(torch.int64) - d.data.to(torch.int64)).to(torch.int32)
        return Column(dtypes.date32, data, combine_validity(c, d))


@dataclass(eq=False)
class DateDiff(Expr):
    end: Expr
    start: Expr

    def eval(self, batch):
        e = self.end.eval(batch)
        s = self.start.eval(batch)
        data = (e.data.to(torch.int64) - s.data.to(torch.int64)).to(torch.int32)
        return Column(dtypes.int32, data, combine_validity(e, s))


@dataclass(eq=False)
class AddMonths(Expr):
    child: Expr
    months: int

    def eval(self, batch):
        c = self.child.eval(batch)
        y, m, d = _civil_from_days(c.data)
        t = y * 12 + (m - 1) + self.months
        ny = torch.div(t, 12, rounding_mode="floor")
        nm = t - ny * 12 + 1
        nd = torch.minimum(d, _days_in_month(ny, nm))
        return Column(dtypes.date32, _days_from_civil(ny, nm, nd).to(torch.int32),
                      c.validity)


@dataclass(eq=False)
class LastDay(Expr):
    child: Expr

    def eval(self, batch):
        c = self.child.eval(batch)
        y, m, d = _civil_from_days(c.data)
        nd = _days_in_month(y, m)
        return Column(dtypes.date32, _days_from_civil(y, m, nd).to(torch.int32),
                      c.validity)


@dataclass(eq=False)
class Quarter(Expr):
    child: Expr

    def eval(self, batch):
        c = self.child.eval(batch)
        _, m, _ = _civil_from_days(c.data)
        return Column(dtypes.int32, (torch.div(m - 1, 3, rounding_mode="floor") + 1)
                      .to(torch.int32), c.validity)


@dataclass(eq=False)
class DayOfWeek(Expr):
    """Spark: 1 = Sunday .. 7 = Saturday."""
    child: Expr

    def eval(self, batch):
        c = self.child.eval(batch)
        # 1970-01-01 was a Thursday (dow 5 in Spark numbering)
        dow = torch.remainder(c.data.to(torch.int64) + 4, 7) + 1
        return Column(dtypes.int32, dow.to(torch.int32), c.validity)


@dataclass(eq=False)
class WeekOfYear(Expr):
    """ISO week number."""
    child: Expr

    def eval(self, batch):
        c = self.child.eval(batch)
        days = c.data.to(torch.int64)
        # ISO: week of the Thursday of this week
        dow_mon0 = torch.remainder(days + 3, 7)  # 0 = Monday
        thursday = days - dow_mon0 + 3
        y, _, _ = _civil_from_days(thursday)
        jan1 = _days_from_civil(y, torch.ones_like(y), torch.ones_like(y))
        week = torch.div(thursday - jan1, 7, rounding_mode="floor") + 1
        return Column(dtypes.int32, week.to(torch.int32), c.validity)


@dataclass(eq=False)
class MonthsBetween(Expr):
    """Spark months_between(end, start): whole months + day-fraction/31;
    both-on-last-day and same-day-of-month give integral results."""
    end: Expr
    start: Expr

    def eval(self, batch):
        e = self.end.eval(batch)
        s = self.start.eval(batch)
        ye, me, de = _civil_from_days(e.data)
        ys, ms, ds = _civil_from_days(s.data)
        months = (ye - ys) * 12 + (me - ms)
        last_e = de == _days_in_month(ye, me)
        last_s = ds == _days_in_month(ys, ms)
        frac = (de - ds).to(torch.float64) / 31.0
        res = months.to(torch.float64) + torch.where(
            (de == ds) | (last_e & last_s), torch.zeros_like(frac), frac)
        return Column(dtypes.float64, res, combine_validity(e, s))


@dataclass(eq=False)
class NextDay(Expr):
    """next_day(date, dow): first date AFTER `date` falling on weekday
    `dow` ('monday'..'sunday', Spark accepts 2-letter+ prefixes)."""
    child: Expr
    dow: str

    def eval(self, batch):
        names = ["monday", "tuesday", "wednesday", "thursday", "friday",
                 "saturday", "sunday"]
        want = next(i for i, n in enumerate(names)
                    if n.startswith(self.dow.lower()[:2]))
        c = self.child.eval(batch)
        days = c.data.to(torch.int64)
        cur = torch.remainder(days + 3, 7)  # 0 = Monday
        delta = torch.remainder(want - cur + 7 - 1, 7) + 1
        return Column(dtypes.date32, (days + delta).to(torch.int32), c.validity)


@dataclass(eq=False)
class TruncDate(Expr):
    """trunc(date, 'year'|'month'|'week')."""
    child: Expr
    unit: str = "month"

    def eval(self, batch):
        c = self.child.eval(batch)
        days = c.data.to(torch.int64)
        if self.unit.lower() in ("year", "yy", "yyyy"):
            y, m, d = _civil_from_days(days)
            out = _days_from_civil(y, torch.ones_like(y), torch.ones_like(y))
        elif self.unit.lower() in ("month", "mm", "mon"):
            y, m, d = _civil_from_days(days)
            out = _days_from_civil(y, m, torch.ones_like(y))
        elif self.unit.lower() == "week":  # Monday of this week
            out = days - torch.remainder(days + 3, 7)
        else:
            raise ValueError(f"trunc unit {self.unit}")
        return Column(dtypes.date32, out.to(torch.int32), c.validity)


# ----------------------------------------------------------------- strings
@dataclass(eq=False)
class Trim(Expr):
    child: Expr
    mode: str = "both"  # both | leading | trailing

    def eval(self, batch):
        c = self.child.eval(batch)
        n = len(c)
        if n == 0:
            return c
        off = c.offsets.to(torch.int64)
        lens = off[1:] - off[:-1]
        W = int(lens.max().item()) if n else 0
        if W == 0:
            return c
        padded = strings.to_padded(c, W)
        pos = torch.arange(W, device=c.device).unsqueeze(0)
        in_str = pos < lens.unsqueeze(1)
        is_space = (padded == 32) & in_str
        nonspace = in_str & ~is_space
        any_ns = nonspace.any(dim=1)
        first_ns = torch.argmax(nonspace.to(torch.int8), dim=1)
        last_ns = W - 1 - torch.argmax(nonspace.flip(1).to(torch.int8), dim=1)
        start = first_ns if self.mode in ("both", "leading") else torch.zeros_like(first_ns)
        end = (last_ns + 1) if self.mode in ("both", "trailing") else lens
        start = torch.where(any_ns, start, torch.zeros_like(start))
        end = torch.where(any_ns, end, torch.zeros_like(end))
        new_lens = (end - start).clamp(min=0)
        new_off = torch.zeros(n + 1, dtype=torch.int64, device=c.device)
        torch.cumsum(new_lens, 0, out=new_off[1:])
        total = int(new_off[-1].item())
        if total == 0:
            data = torch.empty(0, dtype=torch.uint8, device=c.device)
        else:
            row = torch.repeat_interleave(new_lens)
            p = torch.arange(total, dtype=torch.int64, device=c.device)
            within = p - new_off[:-1][row]
            src = off[:-1][row] + start[row] + within
            data = c.data[src]
        return Column(dtypes.string, data, c.validity, new_off.to(torch.int64))


@dataclass(eq=False)
class Left(Expr):
    child: Expr
    n: int

    def eval(self, batch):
        from .exprs import Substr

        return Substr(self.child, 1, self.n).eval(batch)


@dataclass(eq=False)
class Right(Expr):
    child: Expr
    n: int

    def eval(self, batch):
        c = self.child.eval(batch)
        vals = c.to_pylist()  # cold path
        out = [None if v is None else v[-self.n:] if self.n else "" for v in vals]
        return Column.from_pylist(out, dtypes.string, str(c.device))


@dataclass(eq=False)
class Replace(Expr):
    child: Expr
    search: str
    replacement: str = ""

    def eval(self, batch):
        c = self.child.eval(batch)
        vals = c.to_pylist()  # cold path (regexless literal replace)
        out = [None if v is None else v.replace(self.search, self.replacement)
               for v in vals]
        return Column.from_pylist(out, dtypes.string, str(c.device))


@dataclass(eq=False)
class LPad(Expr):
    child: Expr
    length: int
    pad: str = " "

    def eval(self, batch):
        c = self.child.eval(batch)
        vals = c.to_pylist()
        out = []
        for v in vals:
            if v is None:
                out.append(None)
            elif len(v) >= self.length:
                out.append(v[:self.length])
            else:
                need = self.length - len(v)
                p = (self.pad * need)[:need]
                out.append(p + v)
        return Column.from_pylist(out, dtypes.string, str(c.device))


@dataclass(eq=False)
class RPad(Expr):
    child: Expr
    length: int
    pad: str = " "

    def eval(self, batch):
        c = self.child.eval(batch)
        vals = c.to_pylist()
        out = []
        for v in vals:
            if v is None:
                out.append(None)
            elif len(v) >= self.length:
                out.append(v[:self.length])
            else:
                need = self.length - len(v)
                out.append(v + (self.pad * need)[:need])
        return Column.from_pylist(out, dtypes.string, str(c.device))


@dataclass(eq=False)
class ConcatWs(Expr):
    """concat_ws(sep, args...): null args are SKIPPED (Spark), unlike
    concat which nulls the whole row."""
    sep: str
    args: List[Expr]

    def eval(self, batch):
        cols = [a.eval(batch) for a in self.args]
        lists = [c.to_pylist() for c in cols]
        out = [self.sep.join(v for v in row if v is not None)
               for row in zip(*lists)] if lists else [""] * batch.num_rows
        return Column.from_pylist(out, dtypes.string, str(batch.device))


@dataclass(eq=False)
class Reverse(Expr):
    child: Expr

    def eval(self, batch):
        c = self.child.eval(batch)
        out = [None if v is None else v[::-1] for v in c.to_pylist()]
        return Column.from_pylist(out, dtypes.string, str(c.device))


@dataclass(eq=False)
class Repeat(Expr):
    child: Expr
    n: int

    def eval(self, batch):
        c = self.child.eval(batch)
        k = max(self.n, 0)
        out = [None if v is None else v * k for v in c.to_pylist()]
        return Column.from_pylist(out, dtypes.string, str(c.device))


@dataclass(eq=False)
class Space(Expr):
    n: Expr

    def eval(self, batch):
        c = self.n.eval(batch)
        out = [None if v is None else " " * max(int(v), 0)
               for v in c.to_pylist()]
        return Column.from_pylist(out, dtypes.string, str(batch.device))


@dataclass(eq=False)
class Translate(Expr):
    """translate(s, from, to): per-character mapping; chars beyond `to`'s
    length are deleted (SQL semantics)."""
    child: Expr
    src: str
    dst: str

    def eval(self, batch):
        c = self.child.eval(batch)
        table = {}
        for i, ch in enumerate(self.src):
            table[ord(ch)] = self.dst[i] if i < len(self.dst) else None
        out = [None if v is None else v.translate(table) for v in c.to_pylist()]
        return Column.from_pylist(out, dtypes.string, str(c.device))


@dataclass(eq=False)
class FindInSet(Expr):
    """find_in_set(s, csv): 1-based index of s in comma-separated csv
    column; 0 if absent or s contains a comma."""
    child: Expr
    csv: Expr

    def eval(self, batch):
        s = self.child.eval(batch)
        l = self.csv.eval(batch)
        out = []
        for v, lst in zip(s.to_pylist(), l.to_pylist()):
            if v is None or lst is None:
                out.append(None)
            elif "," in v:
                out.append(0)
            else:
                parts = lst.split(",")
                out.append(parts.index(v) + 1 if v in parts else 0)
        return Column.from_pylist(out, dtypes.int32, str(batch.device))


@dataclass(eq=False)
class SplitPart(Expr):
    child: Expr
    delimiter: str
    part: int  # 1-based

    def eval(self, batch):
        c = self.child.eval(batch)
        vals = c.to_pylist()
        out = []
        for v in vals:
            if v is None:
                out.append(None)
            else:
                parts = v.split(self.delimiter)
                out.append(parts[self.part - 1] if 0 < self.part <= len(parts) else "")
        return Column.from_pylist(out, dtypes.string, str(c.device))


@dataclass(eq=False)
class Instr(Expr):
    """1-based position of substr, 0 if absent (Spark instr)."""
    child: Expr
    sub: str

    def eval(self, batch):
        c = self.child.eval(batch)
        n = len(c)
        pat = torch.tensor(list(self.sub.encode()), dtype=torch.uint8, device=c.device)
        m = pat.numel()
        lens = strings.lengths(c)
        if m == 0:
            return Column(dtypes.int32, torch.ones(n, dtype=torch.int32, device=c.device),
                          c.validity)
        W = int(lens.max().item()) if n else 0
        if W < m:
            return Column(dtypes.int32, torch.zeros(n, dtype=torch.int32, device=c.device),
                          c.validity)
        A = strings.to_padded(c, W)
        win = A.unfold(1, m, 1)
        hit = (win == pat).all(dim=2)
        starts = torch.arange(hit.shape[1], device=c.device).unsqueeze(0)
        valid_win = starts + m <= lens.unsqueeze(1)
        hit = hit & valid_win
        any_hit = hit.any(dim=1)
        first = torch.argmax(hit.to(torch.int8), dim=1) + 1
        data = torch.where(any_hit, first, torch.zeros_like(first)).to(torch.int32)
        return Column(dtypes.int32, data, c.validity)


@dataclass(eq=False)
class Ascii(Expr):
    child: Expr

    def eval(self, batch):
        c = self.child.eval(batch)
        lens = strings.lengths(c)
        off = c.offsets.to(torch.int64)
        first = c.data[off[:-1].clamp(max=max(int(c.data.numel()) - 1, 0))] \
            if c.data.numel() else torch.zeros(len(c), dtype=torch.uint8, device=c.device)
        data = torch.where(lens > 0, first.to(torch.int32),
                           torch.zeros(len(c), dtype=torch.int32, device=c.device))
        return Column(dtypes.int32, data, c.validity)


@dataclass(eq=False)
class Bround(Expr):
    """Spark bround: half-even (banker's) rounding."""
    child: Expr
    ndigits: int = 0

    def eval(self, batch):
        c = _f64(self.child, batch)
        f = 10.0 ** self.ndigits
        # torch.round is round-half-to-even
        return Column(dtypes.float64, torch.round(c.data * f) / f, c.validity)


@dataclass(eq=False)
class IsNan(Expr):
    child: Expr

    def eval(self, batch):
        c = _f64(self.child, batch)
        nan = torch.isnan(c.data)
        if c.validity is not None:
            nan = nan & c.validity  # null is not NaN
        return Column(dtypes.bool_, nan)


@dataclass(eq=False)
class NormalizeNanAndZero(Expr):
    """Spark NormalizeNaNAndZero: -0.0 -> 0.0, all NaNs -> one NaN."""
    child: Expr

    def eval(self, batch):
        c = _f64(self.child, batch)
        d = torch.where(c.data == 0, torch.zeros_like(c.data), c.data)
        d = torch.where(torch.isnan(d), torch.full_like(d, float("nan")), d)
        return Column(dtypes.float64, d, c.validity)


@dataclass(eq=False)
class InitCap(Expr):
    """First letter of each word upper-cased, rest lowered (host path)."""
    child: Expr

    def eval(self, batch):
        c = self.child.eval(batch)
        out = [None if v is None else " ".join(
            w[:1].upper() + w[1:].lower() if w else w for w in v.split(" "))
            for v in c.to_pylist()]
        return Column.from_pylist(out, dtypes.string, str(c.device))


@dataclass(eq=False)
class CryptoHash(Expr):
    """md5 / sha1 / sha2-N hex digests (host path, hashlib)."""
    child: Expr
    algo: str = "md5"  # md5 | sha1 | sha224 | sha256 | sha384 | sha512

    def eval(self, batch):
        import hashlib

        c = self.child.eval(batch)
        fn = getattr(hashlib, self.algo)
        out = [None if v is None else fn(v.encode()).hexdigest()
               for v in c.to_pylist()]
        return Column.from_pylist(out, dtypes.string, str(c.device))


def _json_path_get(obj, path: str):
    """Spark get_json_object path subset: $.a.b[0].c"""
    if not path.startswith("$"):
        return None
    import re as _re

    cur = obj
    for tok in _re.findall(r"\.([^.\[\]]+)|\[(\d+)\]", path[1:]):
        name, idx = tok
        if name:
            if not isinstance(cur, dict) or name not in cur:
                return None
            cur = cur[name]
        else:
            i = int(idx)
            if not isinstance(cur, list) or i >= len(cur):
                return None
            cur = cur[i]
    return cur


@dataclass(eq=False)
class GetJsonObject(Expr):
    """Spark get_json_object (host path; spark_get_json_object.rs role)."""
    child: Expr
    path: str

    def eval(self, batch):
        import json

        c = self.child.eval(batch)
        out = []
        for v in c.to_pylist():
            if v is None:
                out.append(None)
                continue
            try:
                r = _json_path_get(json.loads(v), self.path)
            except (ValueError, TypeError):
                r = None
            if r is None:
                out.append(None)
            elif isinstance(r, str):
                out.append(r)
            elif isinstance(r, bool):
                out.append("true" if r else "false")
            elif isinstance(r, (dict, list)):
                out.append(json.dumps(r, separators=(",", ":")))
            else:
                out.append(str(r))
        return Column.from_pylist(out, dtypes.string, str(c.device))


# ------------------------------------------------------------- sketches
@dataclass(eq=False)
class XxHash64(Expr):
    """Spark xxhash64(...) — seed-chained over args, nulls skipped."""
    args: List[Expr]
    seed: int = 42

    def eval(self, batch):
        from . import sketch

        cols = [a.eval(batch) for a in self.args]
        return Column(dtypes.int64, sketch.xxhash64(cols, self.seed))


@dataclass(eq=False)
class BloomFilterMightContain(Expr):
    """Runtime-filter probe: true iff the long key may be in the filter."""
    bloom: object  # sketch.BloomFilter
    child: Expr

    def eval(self, batch):
        c = self.child.eval(batch)
        mask = self.bloom.might_contain_longs(c.data.to(torch.int64))
        return Column(dtypes.bool_, mask, c.validity)


@dataclass(eq=False)
class GetArrayItem(Expr):
    """ext-exprs GetIndexedField analogue: 0-based element of a LIST
    column; out-of-bounds -> null."""
    child: Expr
    index: int

    def eval(self, batch):
        c = self.child.eval(batch)
        assert c.dtype.is_list
        off = c.offsets.to(torch.int64)
        lens = off[1:] - off[:-1]
        ok = torch.tensor(self.index, device=c.device) < lens
        ok = ok & (self.index >= 0)
        if c.validity is not None:
            ok = ok & c.validity
        pos = (off[:-1] + self.index).clamp(0, max(int(c.data.numel()) - 1, 0))
        data = c.data[pos] if c.data.numel() else \
            torch.zeros(len(c), dtype=c.dtype.torch_dtype, device=c.device)
        return Column(c.dtype.child, data, compact_validity(ok))


@dataclass(eq=False)
class ElementAt(Expr):
    """Spark element_at(list, i): 1-based; negative i counts from the
    end; out-of-bounds -> null."""
    child: Expr
    index: int

    def eval(self, batch):
        c = self.child.eval(batch)
        assert c.dtype.is_list and self.index != 0
        off = c.offsets.to(torch.int64)
        lens = off[1:] - off[:-1]
        if self.index > 0:
            k = torch.full_like(lens, self.index - 1)
        else:
            k = lens + self.index
        ok = (k >= 0) & (k < lens)
        if c.validity is not None:
            ok = ok & c.validity
        pos = (off[:-1] + k).clamp(0, max(int(c.data.numel()) - 1, 0))
        data = c.data[pos] if c.data.numel() else \
            torch.zeros(len(c), dtype=c.dtype.torch_dtype, device=c.device)
        return Column(c.dtype.child, data, compact_validity(ok))


@dataclass(eq=False)
class ArraySize(Expr):
    """size(list): element count; null list -> -1 (Spark legacy default)."""
    child: Expr

    def eval(self, batch):
        c = self.child.eval(batch)
        assert c.dtype.is_list
        off = c.offsets.to(torch.int64)
        lens = (off[1:] - off[:-1]).to(torch.int32)
        if c.validity is not None:
            lens = torch.where(c.validity, lens,
                               torch.full_like(lens, -1))
        return Column(dtypes.int32, lens)


@dataclass(eq=False)
class MakeArray(Expr):
    """spark make_array / array(...): fixed-width list per row from k
    element expressions (device-vectorized; element nulls become 0 — an
    element-validity buffer is a round-2 item)."""
    args: List[Expr]

    def eval(self, batch):
        cols = [a.eval(batch) for a in self.args]
        k = len(cols)
        n = batch.num_rows
        first = cols[0]
        data = torch.stack([c.data for c in cols], dim=1).reshape(-1)
        offsets = (torch.arange(n + 1, dtype=torch.int64, device=batch.device)
                   * k)
        return Column(dtypes.list_of(first.dtype), data,
                      combine_validity(*cols), offsets)


# ------------------------------------------------------------- engine ids
@dataclass(eq=False)
class SparkPartitionId(Expr):
    def eval(self, batch):
        pid = EVAL_CONTEXT.get()["partition_id"]
        return Column(dtypes.int32,
                      torch.full((batch.num_rows,), pid, dtype=torch.int32,
                                 device=batch.device))


@dataclass(eq=False)
class RowNum(Expr):
    """ext-exprs RowNum: 1-based row number within this task's stream."""

    def eval(self, batch):
        base = EVAL_CONTEXT.get().get("row_base", 0)
        n = batch.num_rows
        return Column(dtypes.int64,
                      base + 1 + torch.arange(n, dtype=torch.int64,
                                              device=batch.device))


@dataclass(eq=False)
class MonotonicallyIncreasingId(Expr):
    """Spark layout: partition_id << 33 | row index within partition."""

    def eval(self, batch):
        ctx = EVAL_CONTEXT.get()
        base = (ctx["partition_id"] << 33) + ctx.get("row_base", 0)
        n = batch.num_rows
        return Column(dtypes.int64,
                      base + torch.arange(n, dtype=torch.int64, device=batch.device))


# ------------------------------------------------------------- timestamps
def _ts_micros_col(e: Expr, batch) -> Column:
    """Evaluate to a timestamp column (strings/dates coerce, spark_dates.rs
    to_timestamp semantics)."""
    c = e.eval(batch)
    if c.dtype.code != dtypes.TIMESTAMP:
        c = _cast_col(c, dtypes.timestamp)
    return c


_US_DAY = 86_400_000_000
_US_HOUR = 3_600_000_000
_US_MIN = 60_000_000
_US_SEC = 1_000_000


def _time_of_day_us(us: torch.Tensor) -> torch.Tensor:
    return us - torch.div(us, _US_DAY, rounding_mode="floor") * _US_DAY


@dataclass(eq=False)
class Hour(Expr):
    child: Expr

    def eval(self, batch):
        c = _ts_micros_col(self.child, batch)
        tod = _time_of_day_us(c.data)
        return Column(dtypes.int32, torch.div(tod, _US_HOUR, rounding_mode="floor").to(torch.int32), c.validity)


@dataclass(eq=False)
class Minute(Expr):
    child: Expr

    def eval(self, batch):
        c = _ts_micros_col(self.child, batch)
        tod = _time_of_day_us(c.data)
        return Column(dtypes.int32, (torch.div(tod, _US_MIN, rounding_mode="floor") % 60).to(torch.int32), c.validity)


@dataclass(eq=False)
class Second(Expr):
    child: Expr

    def eval(self, batch):
        c = _ts_micros_col(self.child, batch)
        tod = _time_of_day_us(c.data)
        return Column(dtypes.int32, (torch.div(tod, _US_SEC, rounding_mode="floor") % 60).to(torch.int32), c.validity)


@dataclass(eq=False)
class UnixTimestamp(Expr):
    """unix_timestamp(ts|date|string) -> seconds since epoch (int64)."""
    child: Expr

    def eval(self, batch):
        c = _ts_micros_col(self.child, batch)
        return Column(dtypes.int64, torch.div(c.data, _US_SEC, rounding_mode="floor"), c.validity)


@dataclass(eq=False)
class FromUnixtime(Expr):
    """from_unixtime(seconds) -> 'yyyy-MM-dd HH:mm:ss' string."""
    child: Expr

    def eval(self, batch):
        c = _cast_col(self.child.eval(batch), dtypes.int64)
        ts = Column(dtypes.timestamp, c.data * _US_SEC, c.validity)
        return _cast_col(ts, dtypes.string)


@dataclass(eq=False)
class ToTimestamp(Expr):
    child: Expr

    def eval(self, batch):
        return _ts_micros_col(self.child, batch)


@dataclass(eq=False)
class ToDate(Expr):
    child: Expr

    def eval(self, batch):
        c = self.child.eval(batch)
        if c.dtype.code == dtypes.DATE32:
            return c
        if c.dtype.code == dtypes.TIMESTAMP:
            return _cast_col(c, dtypes.date32)
        return _cast_col(c, dtypes.date32)


@dataclass(eq=False)
class DateFormat(Expr):
    """date_format(ts/date, java pattern) — host path, common subset of
    the JDK pattern letters (spark_dates.rs date_format parity)."""
    child: Expr
    pattern: str

    _MAP = [("yyyy", "%Y"), ("MM", "%m"), ("dd", "%d"), ("HH", "%H"),
            ("mm", "%M"), ("ss", "%S"), ("EEEE", "%A"), ("EEE", "%a"),
            ("MMMM", "%B"), ("MMM", "%b"), ("D", "%j"), ("yy", "%y")]

    def eval(self, batch):
        import datetime as _dt

        c = _ts_micros_col(self.child, batch)
        pat = self.pattern
        for j, p in self._MAP:
            pat = pat.replace(j, p)
        epoch = _dt.datetime(1970, 1, 1)
        vals = c.to_pylist()
        out = [None if v is None else
               (epoch + _dt.timedelta(microseconds=int(v))).strftime(pat)
               for v in vals]
        return Column.from_pylist(out, dtypes.string, str(batch.device))


@dataclass(eq=False)
class TruncTimestamp(Expr):
    """date_trunc(unit, ts): YEAR/MONTH/DAY/HOUR/MINUTE/SECOND/WEEK."""
    unit: str
    child: Expr

    def eval(self, batch):
        c = _ts_micros_col(self.child, batch)
        u = self.unit.lower()
        us = c.data
        if u in ("second", "minute", "hour", "day", "week"):
            q = {"second": _US_SEC, "minute": _US_MIN, "hour": _US_HOUR,
                 "day": _US_DAY, "week": 7 * _US_DAY}[u]
            off = 4 * _US_DAY if u == "week" else 0  # 1970-01-01 is a Thursday
            data = torch.div(us - off, q, rounding_mode="floor") * q + off
            return Column(dtypes.timestamp, data, c.validity)
        days = torch.div(us, _US_DAY, rounding_mode="floor")
        y, m, d = _civil_from_days(days)
        if u in ("year", "yyyy", "yy"):
            m = torch.ones_like(m)
            d = torch.ones_like(d)
        elif u in ("month", "mon", "mm"):
            d = torch.ones_like(d)
        else:
            raise ValueError(f"date_trunc unit {self.unit}")
        data = _days_from_civil(y, m, d) * _US_DAY
        return Column(dtypes.timestamp, data, c.validity)


# ----------------------------------------------------- decimal utilities
@dataclass(eq=False)
class UnscaledValue(Expr):
    """spark_unscaled_value: decimal -> raw unscaled int64."""
    child: Expr

    def eval(self, batch):
        c = self.child.eval(batch)
        if c.dtype.code == dtypes.DECIMAL64:
            return Column(dtypes.int64, c.data.clone(), c.validity)
        if c.dtype.code == dtypes.DECIMAL128:
            lo, hi = c.data[:, 0], c.data[:, 1]
            fits = hi == (lo >> 63)
            v = c.validity if c.validity is not None else torch.ones(
                len(c), dtype=torch.bool, device=c.device)
            return Column(dtypes.int64, lo.clone(), compact_validity(v & fits))
        raise TypeError("unscaled_value expects a decimal input")


@dataclass(eq=False)
class MakeDecimal(Expr):
    """spark_make_decimal: unscaled int64 -> decimal(p,s) (inverse of
    UnscaledValue; nulls for values whose digits exceed p)."""
    child: Expr
    precision: int = 18
    scale: int = 2

    def eval(self, batch):
        c = self.child.eval(batch)
        data = c.data.to(torch.int64)
        limit = 10 ** min(self.precision, 18)
        ok = (data > -limit) & (data < limit)
        v = c.validity
        v = ok if v is None else (v & ok)
        return Column(dtypes.decimal64(self.precision, self.scale), data,
                      compact_validity(v))


@dataclass(eq=False)
class CheckOverflow(Expr):
    """spark_check_overflow: null out decimal values whose digit count
    exceeds the declared precision (non-ANSI overflow semantics)."""
    child: Expr
    precision: int = 18
    scale: int = 2

    def eval(self, batch):
        c = self.child.eval(batch)
        if c.dtype.code == dtypes.DECIMAL64:
            limit = 10 ** min(self.precision, 18)
            ok = (c.data > -limit) & (c.data < limit)
            v = c.validity
            v = ok if v is None else (v & ok)
            return Column(dtypes.decimal64(self.precision, c.dtype.scale),
                          c.data, compact_validity(v))
        return c


# ------------------------------------------------------------- randomness
@dataclass(eq=False)
class Rand(Expr):
    """rand([seed]): uniform [0,1) float64; per-partition generator seeded
    by (seed, partition_id) like SparkRandn — deterministic per task."""
    seed: int = 42

    def eval(self, batch):
        ctx = EVAL_CONTEXT.get()
        g = torch.Generator(device="cpu")
        g.manual_seed((self.seed << 16) ^ ctx.get("partition_id", 0)
                      ^ ctx.get("batch_ordinal", 0))
        data = torch.rand(batch.num_rows, dtype=torch.float64, generator=g)
        return Column(dtypes.float64, data.to(batch.device))


@dataclass(eq=False)
class Randn(Expr):
    """randn([seed]): standard normal float64 (SparkRandn parity)."""
    seed: int = 42

    def eval(self, batch):
        ctx = EVAL_CONTEXT.get()
        g = torch.Generator(device="cpu")
        g.manual_seed((self.seed << 16) ^ ctx.get("partition_id", 0)
                      ^ ctx.get("batch_ordinal", 0) ^ 0x5EED)
        data = torch.randn(batch.num_rows, dtype=torch.float64, generator=g)
        return Column(dtypes.float64, data.to(batch.device))


@dataclass(eq=False)
class ListQuantile(Expr):
    """Exact interpolated quantile of each LIST row (Spark percentile
    semantics; percentile_approx is implemented exactly). Sorting is one
    device sort of (row, value) pairs — value-major argsort, then a
    stable row-major argsort groups each row's values in order."""
    child: Expr
    p: float

    def eval(self, batch):
        c = self.child.eval(batch)
        assert c.dtype.is_list, "percentile over a collected list"
        n = len(c)
        device = c.device
        offs = c.offsets
        lens = offs[1:] - offs[:-1]
        total = int(offs[-1].item()) if n else 0
        if total == 0:
            return Column(dtypes.float64,
                          torch.zeros(n, dtype=torch.float64, device=device),
                          torch.zeros(n, dtype=torch.bool, device=device))
        vals = c.data.to(torch.float64)
        row = torch.repeat_interleave(
            torch.arange(n, dtype=torch.int64, device=device), lens)
        o1 = torch.argsort(vals, stable=True)
        o2 = torch.argsort(row[o1], stable=True)
        sorted_vals = vals[o1][o2]  # row-major, values ascending per row
        # interpolated index p*(len-1) within each row
        pos = self.p * (lens.to(torch.float64) - 1).clamp(min=0)
        lo = pos.floor().to(torch.int64)
        hi = pos.ceil().to(torch.int64)
        frac = pos - lo.to(torch.float64)
        base = offs[:-1]
        safe = lens > 0
        lo_i = (base + lo).clamp(0, total - 1)
        hi_i = (base + hi).clamp(0, total - 1)
        out = sorted_vals[lo_i] * (1 - frac) + sorted_vals[hi_i] * frac
        validity = compact_validity(safe) if bool((~safe).any()) else None
        return Column(dtypes.float64, out, validity)


_UNARY_MATH = {
    "sin": torch.sin, "cos": torch.cos, "tan": torch.tan,
    "asin": torch.asin, "acos": torch.acos, "atan": torch.atan,
    "sinh": torch.sinh, "cosh": torch.cosh, "tanh": torch.tanh,
    "log2": torch.log2, "cbrt": lambda x: torch.sign(x) * x.abs() ** (1.0 / 3),
    "degrees": torch.rad2deg, "radians": torch.deg2rad,
    "expm1": torch.expm1, "log1p": torch.log1p, "rint": torch.round,
}


@dataclass(eq=False)
class UnaryMath(Expr):
    """Generic float64 unary math (spark_math parity: sin/cos/.../cbrt)."""
    fn: str
    child: Expr

    def eval(self, batch):
        c = _f64(self.child, batch)
        return Column(dtypes.float64, _UNARY_MATH[self.fn](c.data), c.validity)


@dataclass(eq=False)
class Atan2(Expr):
    y: Expr
    x: Expr

    def eval(self, batch):
        a = _f64(self.y, batch)
        b = _f64(self.x, batch)
        return Column(dtypes.float64, torch.atan2(a.data, b.data),
                      combine_validity(a, b))


# ------------------------------------------------------------ regex (host)
def _java_regex(p: str) -> str:
    """Minimal Java->Python regex translation (the dialects agree on the
    subset these functions see; host cold path like GetJsonObject)."""
    return p


@dataclass(eq=False)
class RegexpExtract(Expr):
    child: Expr
    pattern: str
    group: int = 1

    def eval(self, batch):
        import re as _re

        rx = _re.compile(_java_regex(self.pattern))
        c = self.child.eval(batch)
        out = []
        for v in c.to_pylist():
            if v is None:
                out.append(None)
                continue
            m = rx.search(v)
            # Spark: no match -> empty string
            out.append(m.group(self.group) if m and m.group(self.group)
                       is not None else "")
        return Column.from_pylist(out, dtypes.string, str(batch.device))


@dataclass(eq=False)
class RegexpReplace(Expr):
    child: Expr
    pattern: str
    replacement: str

    def eval(self, batch):
        import re as _re

        rx = _re.compile(_java_regex(self.pattern))
        # Java $1 backrefs -> Python \1
        rep = _re.sub(r"\$(\d+)", r"\\\1", self.replacement)
        c = self.child.eval(batch)
        out = [None if v is None else rx.sub(rep, v) for v in c.to_pylist()]
        return Column.from_pylist(out, dtypes.string, str(batch.device))


@dataclass(eq=False)
class RLike(Expr):
    child: Expr
    pattern: str

    def eval(self, batch):
        import re as _re

        rx = _re.compile(_java_regex(self.pattern))
        c = self.child.eval(batch)
        vals = c.to_pylist()
        data = torch.tensor([v is not None and rx.search(v) is not None
                             for v in vals], dtype=torch.bool)
        return Column(dtypes.bool_, data.to(batch.device), c.validity)


@dataclass(eq=False)
class ArrayContains(Expr):
    """array_contains(list, value): vectorized flat compare + segment any."""
    child: Expr
    value: object

    def eval(self, batch):
        c = self.child.eval(batch)
        assert c.dtype.is_list
        n = len(c)
        device = c.device
        offs = c.offsets
        lens = offs[1:] - offs[:-1]
        if int(offs[-1].item()) == 0:
            return Column(dtypes.bool_,
                          torch.zeros(n, dtype=torch.bool, device=device),
                          c.validity)
        row = torch.repeat_interleave(
            torch.arange(n, dtype=torch.int64, device=device), lens)
        if c.dtype.child.code == dtypes.DECIMAL64:
            hit = c.data.to(torch.float64) / 10 ** c.dtype.scale == float(self.value)
        else:
            hit = c.data == self.value
        out = torch.zeros(n, dtype=torch.bool, device=device)
        out.scatter_reduce_(0, row, hit, reduce="amax", include_self=True)
        return Column(dtypes.bool_, out.to(torch.bool), c.validity)


@dataclass(eq=False)
class SubstringIndex(Expr):
    """substring_index(s, delim, count): Spark semantics (host path)."""
    child: Expr
    delim: str
    count: int

    def eval(self, batch):
        c = self.child.eval(batch)
        out = []
        for v in c.to_pylist():
            if v is None:
                out.append(None)
            elif self.count > 0:
                out.append(self.delim.join(v.split(self.delim)[:self.count]))
            elif self.count < 0:
                out.append(self.delim.join(v.split(self.delim)[self.count:]))
            else:
                out.append("")
        return Column.from_pylist(out, dtypes.string, str(batch.device))


@dataclass(eq=False)
class Levenshtein(Expr):
    """levenshtein(a, b): edit distance (host DP, cold path)."""
    left: Expr
    right: Expr

    def eval(self, batch):
        a = self.left.eval(batch).to_pylist()
        b = self.right.eval(batch).to_pylist()

        def dist(x, y):
            if x is None or y is None:
                return None
            prev = list(range(len(y) + 1))
            for i, cx in enumerate(x, 1):
                cur = [i]
                for j, cy in enumerate(y, 1):
                    cur.append(min(prev[j] + 1, cur[j - 1] + 1,
                                   prev[j - 1] + (cx != cy)))
                prev = cur
            return prev[-1]

        return Column.from_pylist([dist(x, y) for x, y in zip(a, b)],
                                  dtypes.int32, str(batch.device))

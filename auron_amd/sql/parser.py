"""SQL lexer + recursive-descent parser for the Spark-SQL dialect used by
the TPC-DS suite the reference ships (dev/auron-it tpcds-queries/q*.sql):
SELECT/DISTINCT, comma + ANSI joins (left/right/full outer), WHERE,
GROUP BY [ROLLUP|CUBE]/GROUPING SETS, HAVING, window functions with
PARTITION BY / ORDER BY / ROWS|RANGE frames (explicit ROWS bounds), WITH
CTEs, UNION [ALL] / INTERSECT [ALL] / EXCEPT [ALL], scalar/IN/EXISTS
subqueries, CASE, CAST, BETWEEN, LIKE, IN lists, INTERVAL arithmetic,
ordinal ORDER BY references, NULLS FIRST/LAST, DATE/TIMESTAMP literals.
"""
from __future__ import annotations

import re
from typing import List, Optional, Tuple

from . import ast as A


class SqlError(ValueError):
    pass


_TOKEN_RE = re.compile(
    r"""
    (?P<ws>\s+|--[^\n]*)
  | (?P<num>\d+\.\d*|\.\d+|\d+)
  | (?P<str>'(?:[^']|'')*')
  | (?P<bq>`[^`]*`)
  | (?P<ident>[A-Za-z_][A-Za-z_0-9]*)
  | (?P<op><>|!=|<=|>=|\|\||[(),.*/%+\-<>=;])
    """,
    re.VERBOSE,
)

KEYWORDS = {
    "select", "distinct", "from", "where", "group", "by", "having", "order",
    "limit", "as", "and", "or", "not", "in", "is", "null", "between", "like",
    "case", "when", "then", "else", "end", "cast", "exists", "union", "all",
    "intersect", "except", "join", "inner", "left", "right", "full", "outer",
    "cross", "on", "with", "rollup", "cube", "grouping", "sets", "over", "partition",
    "rows", "range", "unbounded", "preceding", "following", "current", "row",
    "asc", "desc", "nulls", "first", "last", "interval", "substring", "for",
}


class Tok:
    __slots__ = ("kind", "text", "pos")

    def __init__(self, kind: str, text: str, pos: int):
        self.kind = kind  # num | str | ident | kw | op | eof
        self.text = text
        self.pos = pos

    def __repr__(self):
        return f"Tok({self.kind},{self.text!r})"


def tokenize(sql: str) -> List[Tok]:
    toks: List[Tok] = []
    i, n = 0, len(sql)
    while i < n:
        m = _TOKEN_RE.match(sql, i)
        if not m:
            raise SqlError(f"lex error at {sql[i:i+20]!r}")
        i = m.end()
        if m.lastgroup == "ws":
            continue
        kind = m.lastgroup
        text = m.group()
        if kind == "ident":
            low = text.lower()
            if low in KEYWORDS:
                toks.append(Tok("kw", low, m.start()))
                continue
            toks.append(Tok("ident", text, m.start()))
        elif kind == "bq":
            # backquoted identifier: verbatim (may contain spaces)
            toks.append(Tok("ident", text[1:-1], m.start()))
        elif kind == "str":
            toks.append(Tok("str", text[1:-1].replace("''", "'"), m.start()))
        else:
            toks.append(Tok(kind, text, m.start()))
    toks.append(Tok("eof", "", n))
    return toks


class Parser:
    def __init__(self, sql: str):
        self.sql = sql
        self.toks = tokenize(sql)
        self.i = 0

    # ------------------------------------------------------------ cursor
    @property
    def cur(self) -> Tok:
        return self.toks[self.i]

    def peek(self, k: int = 1) -> Tok:
        j = min(self.i + k, len(self.toks) - 1)
        return self.toks[j]

    def advance(self) -> Tok:
        t = self.cur
        self.i += 1
        return t

    def at_kw(self, *kws: str) -> bool:
        return self.cur.kind == "kw" and self.cur.text in kws

    def at_op(self, *ops: str) -> bool:
        return self.cur.kind == "op" and self.cur.text in ops

    def accept_kw(self, *kws: str) -> bool:
        if self.at_kw(*kws):
            self.advance()
            return True
        return False

    def accept_op(self, op: str) -> bool:
        if self.at_op(op):
            self.advance()
            return True
        return False

    def expect_kw(self, kw: str):
        if not self.accept_kw(kw):
            self.fail(f"expected {kw.upper()}")

    def expect_op(self, op: str):
        if not self.accept_op(op):
            self.fail(f"expected {op!r}")

    def fail(self, msg: str):
        t = self.cur
        ctx = self.sql[max(0, t.pos - 30):t.pos + 30].replace("\n", " ")
        raise SqlError(f"{msg} at ...{ctx}... (got {t.kind}:{t.text!r})")

    # ------------------------------------------------------------- query
    def parse_query(self) -> A.Query:
        q = self._query()
        if self.cur.kind != "eof" and not self.at_op(";"):
            self.fail("trailing tokens")
        return q

    def _query(self) -> A.Query:
        ctes: List[Tuple[str, A.Query]] = []
        if self.accept_kw("with"):
            while True:
                name = self._ident("CTE name")
                self.expect_kw("as")
                self.expect_op("(")
                sub = self._query()
                self.expect_op(")")
                ctes.append((name.lower(), sub))
                if not self.accept_op(","):
                    break
        body = self._set_expr()
        order_by: List[A.OrderItem] = []
        limit = None
        if self.at_kw("order"):
            order_by = self._order_by()
        if self.accept_kw("limit"):
            if self.cur.kind != "num":
                self.fail("expected LIMIT count")
            limit = int(self.advance().text)
        return A.Query(ctes, body, order_by, limit)

    def _order_by(self) -> List[A.OrderItem]:
        self.expect_kw("order")
        self.expect_kw("by")
        items = []
        while True:
            e = self.expr()
            asc = True
            if self.accept_kw("desc"):
                asc = False
            else:
                self.accept_kw("asc")
            nf = None
            if self.accept_kw("nulls"):
                if self.accept_kw("first"):
                    nf = True
                elif self.accept_kw("last"):
                    nf = False
                else:
                    self.fail("expected FIRST/LAST")
            items.append(A.OrderItem(e, asc, nf))
            if not self.accept_op(","):
                break
        return items

    def _set_expr(self) -> A.ANode:
        left = self._set_operand()
        while self.at_kw("union", "intersect", "except"):
            op = self.advance().text
            if op == "union":
                op = "union_all" if self.accept_kw("all") else "union"
            elif self.accept_kw("all"):
                op = f"{op}_all"  # multiset INTERSECT ALL / EXCEPT ALL
            right = self._set_operand()
            left = A.SetOp(op, left, right)
        return left

    def _set_operand(self) -> A.ANode:
        if self.accept_op("("):
            # parenthesized select / nested set expr (no ORDER/LIMIT inside)
            inner = self._set_expr()
            self.expect_op(")")
            return inner
        return self._select()

    def _select(self) -> A.Select:
        self.expect_kw("select")
        distinct = self.accept_kw("distinct")
        items = [self._select_item()]
        while self.accept_op(","):
            items.append(self._select_item())
        from_: List[A.ANode] = []
        if self.accept_kw("from"):
            from_.append(self._table_ref())
            while self.accept_op(","):
                from_.append(self._table_ref())
        where = self.expr() if self.accept_kw("where") else None
        group_by: List[A.ANode] = []
        rollup = False
        gsets = None
        if self.accept_kw("group"):
            self.expect_kw("by")
            if self.accept_kw("rollup"):
                rollup = True
                self.expect_op("(")
                group_by = self._expr_list()
                self.expect_op(")")
            elif self.accept_kw("cube"):
                # CUBE(a,b,...) desugars to GROUPING SETS of all subsets
                self.expect_op("(")
                exprs = self._expr_list()
                self.expect_op(")")
                import itertools as _it

                gsets = [list(sub)
                         for r in range(len(exprs), -1, -1)
                         for sub in _it.combinations(exprs, r)]
            elif self.accept_kw("grouping"):
                self.expect_kw("sets")
                self.expect_op("(")
                gsets = []
                while True:
                    self.expect_op("(")
                    gsets.append([] if self.at_op(")") else self._expr_list())
                    self.expect_op(")")
                    if not self.accept_op(","):
                        break
                self.expect_op(")")
            else:
                group_by = self._expr_list()
                if self.accept_kw("rollup"):  # GROUP BY a, ROLLUP(b) (unused)
                    self.fail("mixed ROLLUP unsupported")
        having = self.expr() if self.accept_kw("having") else None
        return A.Select(items, from_, where, group_by, rollup, gsets,
                        having, distinct)

    def _select_item(self) -> A.SelectItem:
        if self.at_op("*"):
            self.advance()
            return A.SelectItem(A.Star())
        e = self.expr()
        alias = None
        if self.accept_kw("as"):
            alias = self._ident("alias")
        elif self.cur.kind == "ident":
            alias = self.advance().text
        return A.SelectItem(e, alias)

    def _ident(self, what: str) -> str:
        if self.cur.kind == "ident":
            return self.advance().text
        # some keywords double as identifiers in practice (e.g. aliases
        # named "first"/"last" never appear in TPC-DS; keep strict)
        self.fail(f"expected {what}")

    # --------------------------------------------------------- relations
    def _table_ref(self) -> A.ANode:
        left = self._table_primary()
        while True:
            kind = None
            if self.accept_kw("cross"):
                self.expect_kw("join")
                kind = "cross"
            elif self.accept_kw("inner"):
                self.expect_kw("join")
                kind = "inner"
            elif self.at_kw("left", "right", "full"):
                kind = self.advance().text
                self.accept_kw("outer")
                self.expect_kw("join")
            elif self.accept_kw("join"):
                kind = "inner"
            else:
                return left
            right = self._table_primary()
            on = None
            if kind != "cross":
                self.expect_kw("on")
                on = self.expr()
            left = A.Join(left, right, kind, on)

    def _table_primary(self) -> A.ANode:
        if self.accept_op("("):
            if self.at_kw("select", "with") or self.at_op("("):
                q = self._subquery_body()
                self.expect_op(")")
                self.accept_kw("as")
                alias = self._ident("derived-table alias")
                return A.DerivedTable(q, alias)
            t = self._table_ref()
            self.expect_op(")")
            return t
        name = self._ident("table name").lower()
        alias = None
        if self.accept_kw("as"):
            alias = self._ident("alias")
        elif self.cur.kind == "ident":
            alias = self.advance().text
        return A.Table(name, alias)

    def _subquery_body(self) -> A.Query:
        """A query appearing inside parentheses (subquery / derived table)."""
        if self.at_op("("):
            # ((select ...) union all (select ...)) style
            body = self._set_expr()
            order_by: List[A.OrderItem] = []
            limit = None
            if self.at_kw("order"):
                order_by = self._order_by()
            if self.accept_kw("limit"):
                limit = int(self.advance().text)
            return A.Query([], body, order_by, limit)
        return self._query()

    # ------------------------------------------------------- expressions
    def _expr_list(self) -> List[A.ANode]:
        out = [self.expr()]
        while self.accept_op(","):
            out.append(self.expr())
        return out

    def expr(self) -> A.ANode:
        return self._or()

    def _or(self) -> A.ANode:
        left = self._and()
        while self.accept_kw("or"):
            left = A.BinOp("or", left, self._and())
        return left

    def _and(self) -> A.ANode:
        left = self._not()
        while self.accept_kw("and"):
            left = A.BinOp("and", left, self._not())
        return left

    def _not(self) -> A.ANode:
        if self.accept_kw("not"):
            return A.UnOp("not", self._not())
        return self._predicate()

    def _predicate(self) -> A.ANode:
        left = self._additive()
        while True:
            if self.at_op("=", "<>", "!=", "<", "<=", ">", ">="):
                op = self.advance().text
                if op == "!=":
                    op = "<>"
                right = self._additive()
                left = A.BinOp(op, left, right)
                continue
            if self.accept_kw("is"):
                neg = self.accept_kw("not")
                self.expect_kw("null")
                left = A.IsNull(left, neg)
                continue
            neg = False
            save = self.i
            if self.accept_kw("not"):
                neg = True
            if self.accept_kw("between"):
                lo = self._additive()
                self.expect_kw("and")
                hi = self._additive()
                left = A.Between(left, lo, hi, neg)
                continue
            if self.accept_kw("in"):
                self.expect_op("(")
                if self.at_kw("select", "with"):
                    q = self._query()
                    self.expect_op(")")
                    left = A.InSubquery(left, q, neg)
                else:
                    items = self._expr_list()
                    self.expect_op(")")
                    left = A.InList(left, items, neg)
                continue
            if self.accept_kw("like"):
                if self.cur.kind != "str":
                    self.fail("expected LIKE pattern")
                left = A.Like(left, self.advance().text, neg)
                continue
            if neg:
                self.i = save  # NOT belonged to an outer context
            return left

    def _additive(self) -> A.ANode:
        left = self._multiplicative()
        while self.at_op("+", "-") or self.at_op("||"):
            op = self.advance().text
            right = self._multiplicative()
            left = A.BinOp(op, left, right)
        return left

    def _multiplicative(self) -> A.ANode:
        left = self._unary()
        while self.at_op("*", "/", "%"):
            op = self.advance().text
            right = self._unary()
            left = A.BinOp(op, left, right)
        return left

    def _unary(self) -> A.ANode:
        if self.accept_op("-"):
            return A.UnOp("-", self._unary())
        if self.accept_op("+"):
            return self._unary()
        return self._primary()

    def _primary(self) -> A.ANode:
        t = self.cur
        if t.kind == "num":
            self.advance()
            return A.Num(t.text)
        if t.kind == "str":
            self.advance()
            return A.Str(t.text)
        if self.accept_kw("null"):
            return A.Null()
        # typed literals: DATE '2000-01-01' / TIMESTAMP '2000-01-01 12:00:00'
        if (self.cur.kind == "ident" and self.cur.text.lower() in
                ("date", "timestamp") and self.peek().kind == "str"):
            tn = self.advance().text.lower()
            return A.CastE(A.Str(self.advance().text), tn)
        if self.accept_kw("case"):
            return self._case()
        if self.accept_kw("cast"):
            self.expect_op("(")
            e = self.expr()
            self.expect_kw("as")
            tn = self._typename()
            self.expect_op(")")
            return A.CastE(e, tn)
        if self.accept_kw("exists"):
            self.expect_op("(")
            q = self._query()
            self.expect_op(")")
            return A.Exists(q)
        if self.accept_kw("interval"):
            return self._interval()
        if self.accept_kw("substring"):
            # substring(x FROM a [FOR b]) or substring(x, a, b)
            self.expect_op("(")
            e = self.expr()
            if self.accept_kw("from"):
                start = self.expr()
                length = self.expr() if self.accept_kw("for") else None
            else:
                self.expect_op(",")
                start = self.expr()
                length = self.expr() if self.accept_op(",") else None
            self.expect_op(")")
            args = [e, start] + ([length] if length is not None else [])
            return A.FuncCall("substr", args)
        if self.accept_op("("):
            if self.at_kw("select", "with"):
                q = self._query()
                self.expect_op(")")
                return A.ScalarSubquery(q)
            e = self.expr()
            self.expect_op(")")
            return e
        if t.kind == "kw" and t.text in ("grouping", "first", "last", "left", "right") \
                and self.peek().kind == "op" and self.peek().text == "(":
            # keyword-named functions: grouping(c), left(s,n), ...
            self.advance()
            return self._func_call(t.text)
        if t.kind == "ident":
            self.advance()
            if self.at_op("(") :
                return self._func_call(t.text.lower())
            parts = [t.text.lower()]
            while self.at_op(".") and self.peek().kind in ("ident", "kw"):
                self.advance()
                parts.append(self.advance().text.lower())
            return A.Ident(parts)
        self.fail("expected expression")

    def _func_call(self, name: str) -> A.ANode:
        self.expect_op("(")
        distinct = False
        star = False
        args: List[A.ANode] = []
        if self.at_op("*"):
            self.advance()
            star = True
        elif not self.at_op(")"):
            distinct = self.accept_kw("distinct")
            args = self._expr_list()
        self.expect_op(")")
        over = None
        if self.accept_kw("over"):
            over = self._window_spec()
        return A.FuncCall(name, args, distinct, star, over)

    def _window_spec(self) -> A.WindowSpec:
        self.expect_op("(")
        spec = A.WindowSpec()
        if self.accept_kw("partition"):
            self.expect_kw("by")
            spec.partition_by = self._expr_list()
        if self.at_kw("order"):
            spec.order_by = self._order_by()
        if self.at_kw("rows", "range"):
            spec.frame = self.advance().text

            def bound():
                # -> None (unbounded preceding), "U" (unbounded following),
                #    or signed row offset (preceding<0, current=0, following>0)
                if self.accept_kw("unbounded"):
                    if self.accept_kw("preceding"):
                        return None
                    self.expect_kw("following")
                    return "U"
                if self.accept_kw("current"):
                    self.expect_kw("row")
                    return 0
                if self.cur.kind != "num":
                    self.fail("expected frame bound")
                k = int(self.advance().text)
                if self.accept_kw("preceding"):
                    return -k
                self.expect_kw("following")
                return k

            self.expect_kw("between")
            spec.frame_lo = bound()
            self.expect_kw("and")
            spec.frame_hi = bound()
            if spec.frame == "range" and not (
                    spec.frame_lo is None and spec.frame_hi == 0):
                self.fail("RANGE frames support only the default bounds")
        self.expect_op(")")
        return spec

    def _case(self) -> A.ANode:
        operand = None
        if not self.at_kw("when"):
            operand = self.expr()
        whens = []
        while self.accept_kw("when"):
            c = self.expr()
            self.expect_kw("then")
            v = self.expr()
            whens.append((c, v))
        else_ = self.expr() if self.accept_kw("else") else None
        self.expect_kw("end")
        return A.Case(operand, whens, else_)

    def _typename(self) -> str:
        base = self._ident("type name").lower() if self.cur.kind == "ident" else None
        if base is None:
            self.fail("expected type name")
        if self.accept_op("("):
            params = []
            while not self.at_op(")"):
                params.append(self.advance().text)
                self.accept_op(",")
            self.expect_op(")")
            return f"{base}({','.join(params)})"
        return base

    def _interval(self) -> A.ANode:
        # INTERVAL 14 days | INTERVAL '90' day
        if self.cur.kind == "num":
            n = int(self.advance().text)
        elif self.cur.kind == "str":
            n = int(self.advance().text)
        else:
            self.fail("expected interval quantity")
        unit = self.advance().text.lower().rstrip("s")
        if unit not in ("day",):
            raise SqlError(f"unsupported interval unit {unit}")
        return A.Interval(n, unit)


def parse_sql(sql: str) -> A.Query:
    return Parser(sql).parse_query()

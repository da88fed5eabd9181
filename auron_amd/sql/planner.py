"""SQL analyzer + logical planner.

Role parity: the plan Spark's Catalyst hands to AuronConvertStrategy —
since no JVM host exists in-container, this module performs the
resolution/decorrelation/ordering work Catalyst would have done and
produces a logical tree whose expressions are already engine exprs
(auron_amd.exprs). physical.py then plays AuronConverters
(/root/reference/spark-extension/.../AuronConverters.scala:97): it maps
each logical node onto the PhysicalPlanNode set with exchange/broadcast
decisions.

Decorrelation coverage (what the TPC-DS suite needs):
 * correlated scalar-agg subqueries  -> group-by on correlation keys + join
 * [NOT] EXISTS                      -> semi/anti join (+ residual pairs)
 * [NOT] IN (subquery)               -> semi/anti join (equi; NOT IN assumes
                                        non-null keys, as Catalyst's
                                        null-aware anti would be needed)
 * EXISTS/IN under OR                -> existence join (bool mark column)
 * uncorrelated scalar subqueries    -> ScalarSubqueryLit placeholder,
                                        executed at lowering (AQE-style)
"""
from __future__ import annotations

import dataclasses
import datetime as _dt
from dataclasses import dataclass, field
from typing import Dict, List, Optional, Set, Tuple

from .. import dtypes
from ..dtypes import DataType
from ..exprs import (AggFunc, Aliased, CaseWhen, Cast, Coalesce, Col, Expr,
                     InList, IsNull, Like, Literal, Not, Substr, WindowFunc,
                     col, lit)
from .. import functions as F
from ..tpcds.schema import SCHEMAS
from . import ast as A
from .parser import SqlError, parse_sql

EPOCH = _dt.date(1970, 1, 1)


# ----------------------------------------------------------- logical nodes
class LNode:
    pass


@dataclass
class LScan(LNode):
    table: str
    # engine name per source column (identity unless disambiguated)
    colmap: Dict[str, str]
    filters: List[Expr] = field(default_factory=list)


@dataclass
class LCTERef(LNode):
    cte: "MatCTE"
    colmap: Dict[str, str]  # cte output name -> engine name here


@dataclass
class LFilter(LNode):
    child: LNode
    pred: Expr


@dataclass
class LProject(LNode):
    child: LNode
    items: List[Tuple[Expr, str]]


@dataclass
class LJoin(LNode):
    left: LNode
    right: LNode
    kind: str  # inner|left|right|full|semi|anti|existence
    lkeys: List[Expr]
    rkeys: List[Expr]
    residual: Optional[Expr] = None
    existence_name: str = "__exists"
    l_est: float = 0.0
    r_est: float = 0.0
    r_base_dim: bool = False  # right side built only from dimension tables


@dataclass
class LAgg(LNode):
    child: LNode
    keys: List[Tuple[Expr, str]]
    aggs: List[AggFunc]
    # ROLLUP/GROUPING SETS: list of keep-masks over keys, plus the engine
    # name of the grouping-id column (Spark grouping_id bit convention)
    grouping_sets: Optional[List[List[bool]]] = None
    gid_name: str = "__gid"
    key_dtypes: Optional[List[DataType]] = None
    grp_names: Dict[str, str] = field(default_factory=dict)


@dataclass
class LWindow(LNode):
    child: LNode
    partition: List[Expr]
    order: List[Tuple[Expr, bool]]
    funcs: List[Tuple[WindowFunc, str]]
    frame: str = "range"
    frame_lo: object = None
    frame_hi: object = 0


@dataclass
class LReplicate(LNode):
    child: LNode
    count: Expr  # per-row output multiplicity


@dataclass
class LUnionAll(LNode):
    children: List[LNode]
    # per-child projection onto the common output names
    projections: List[List[Tuple[Expr, str]]]


@dataclass
class LSort(LNode):
    child: LNode
    keys: List[Tuple[Expr, bool]]
    limit: Optional[int] = None


@dataclass
class LLimit(LNode):
    child: LNode
    n: int


# ------------------------------------------------------------ helper exprs
@dataclass(eq=False)
class ScalarSubqueryLit(Expr):
    """Placeholder for an uncorrelated scalar subquery; physical lowering
    executes `lquery` and substitutes a Literal (the reference's AQE-time
    subquery materialization on the JVM side)."""
    lquery: "LQuery"

    def eval(self, batch):  # pragma: no cover
        raise RuntimeError("ScalarSubqueryLit must be resolved at lowering")


@dataclass
class MatCTE:
    """A CTE used more than once: materialized exactly once at lowering."""
    name: str
    lquery: "LQuery"
    out_dtypes: Optional[list] = None
    # filled at lowering time
    batches: Optional[list] = None
    est: float = 1e6


@dataclass
class LQuery:
    """A planned query: an unprojected relation + named outputs + sort."""
    rel: "Rel"
    outputs: List[Tuple[Expr, str]]
    order: List[Tuple[Expr, bool]] = field(default_factory=list)
    limit: Optional[int] = None
    distinct: bool = False


# --------------------------------------------------------------- name pool
class NameAlloc:
    def __init__(self):
        self.used: Set[str] = set()
        self.types: Dict[str, DataType] = {}
        self.n = 0

    def fresh(self, base: str, dtype: Optional[DataType]) -> str:
        name = base
        while name in self.used:
            self.n += 1
            name = f"{base}__{self.n}"
        self.used.add(name)
        if dtype is not None:
            self.types[name] = dtype
        return name

    def set_type(self, name: str, dtype: Optional[DataType]):
        if dtype is not None:
            self.types[name] = dtype


@dataclass
class RelCol:
    qual: Optional[str]  # table alias (lower) or None
    name: str  # SQL-visible name (lower)
    engine: str
    dtype: Optional[DataType]


@dataclass
class Rel:
    node: LNode
    cols: List[RelCol]
    est: float = 1e6
    base_dim_only: bool = True  # no fact table below (broadcast candidate)

    def find(self, qual: Optional[str], name: str) -> List[RelCol]:
        name = name.lower()
        out = []
        for c in self.cols:
            if c.name == name and (qual is None or c.qual == qual):
                out.append(c)
        return out


# fact tables scale with SF; everything else is broadcastable dimension
FACT_TABLES = {
    "store_sales", "catalog_sales", "web_sales", "inventory",
    "store_returns", "catalog_returns", "web_returns",
}

def _broadcast_rows() -> float:
    from ..config import BROADCAST_MAX_ROWS, AuronConf

    return float(AuronConf().get(BROADCAST_MAX_ROWS))


BROADCAST_ROWS = _broadcast_rows()


def expr_key(e) -> tuple:
    """Structural key for engine exprs (they are eq=False dataclasses)."""
    if type(e).__name__ == "ScalarSubqueryLit":
        return ("ScalarSubqueryLit", id(e))
    if isinstance(e, Expr) or dataclasses.is_dataclass(e):
        vals = []
        for f in dataclasses.fields(e):
            vals.append(expr_key(getattr(e, f.name)))
        return (type(e).__name__, tuple(vals))
    if isinstance(e, (list, tuple)):
        return tuple(expr_key(x) for x in e)
    return (repr(e),)


def expr_cols(e, out: Optional[Set[str]] = None) -> Set[str]:
    """Engine column names referenced by an engine expr tree."""
    if out is None:
        out = set()
    if isinstance(e, Col):
        out.add(e.name)
        return out
    if type(e).__name__ == "ScalarSubqueryLit":
        return out  # closed scope: resolves to a literal at lowering
    if isinstance(e, Expr) or dataclasses.is_dataclass(e):
        for f in dataclasses.fields(e):
            expr_cols(getattr(e, f.name), out)
        return out
    if isinstance(e, (list, tuple)):
        for x in e:
            expr_cols(x, out)
    return out


def walk_exprs(e, fn):
    """Rewrite an engine expr tree bottom-up with fn(node)->node."""
    if isinstance(e, Expr):
        for f in dataclasses.fields(e):
            v = getattr(e, f.name)
            nv = walk_exprs(v, fn)
            if nv is not v:
                object.__setattr__(e, f.name, nv)
        return fn(e)
    if isinstance(e, list):
        return [walk_exprs(x, fn) for x in e]
    if isinstance(e, tuple):
        return tuple(walk_exprs(x, fn) for x in e)
    return e


# ------------------------------------------------------------------ scopes
class Scope:
    def __init__(self, rels: List[Rel], parent: Optional["Scope"] = None):
        self.rels = rels
        self.parent = parent
        self.outer_used: List[str] = []  # engine cols resolved from parents

    def resolve(self, qual: Optional[str], name: str) -> Optional[RelCol]:
        hits: List[RelCol] = []
        for r in self.rels:
            hits += r.find(qual, name)
        if len(hits) > 1:
            # identical engine col visible via multiple paths is fine
            engines = {h.engine for h in hits}
            if len(engines) > 1:
                raise SqlError(f"ambiguous column {qual or ''}.{name}")
        if hits:
            return hits[0]
        if self.parent is not None:
            hit = self.parent.resolve(qual, name)
            if hit is not None:
                self.outer_used.append(hit.engine)
            return hit
        return None


def infer_dtype(e: Expr, types: Dict[str, DataType]) -> Optional[DataType]:
    """Best-effort dtype inference over engine exprs (used for NULL literal
    typing, date folding and rollup null projections)."""
    if isinstance(e, Col):
        return types.get(e.name)
    if isinstance(e, Literal):
        return e.dtype if e.dtype is not None else _lit_dtype(e.value)
    if isinstance(e, Cast):
        return e.to
    if isinstance(e, CaseWhen):
        for _, v in e.branches:
            dt = infer_dtype(v, types)
            if dt is not None:
                return dt
        return infer_dtype(e.otherwise, types) if e.otherwise is not None else None
    if isinstance(e, Coalesce):
        for a in e.args:
            dt = infer_dtype(a, types)
            if dt is not None:
                return dt
        return None
    if isinstance(e, Substr):
        return dtypes.string
    if type(e).__name__ in ("ConcatStr", "Upper", "Lower", "Trim"):
        return dtypes.string
    if type(e).__name__ in ("Cmp", "BoolOp", "Not", "IsNull", "InList", "Like"):
        return dtypes.bool_ if hasattr(dtypes, "bool_") else None
    if type(e).__name__ == "Arith":
        lt = infer_dtype(e.left, types)
        rt = infer_dtype(e.right, types)
        if lt is None or rt is None:
            return lt or rt
        if dtypes.FLOAT64 in (lt.code, rt.code) or dtypes.FLOAT32 in (lt.code, rt.code):
            return dtypes.float64
        if lt.code == dtypes.DECIMAL64 or rt.code == dtypes.DECIMAL64:
            if e.op in ("/",):
                return dtypes.float64
            # engine Arith keeps max scale for +- and adds for *
            if lt.code == rt.code == dtypes.DECIMAL64:
                sc = lt.scale + rt.scale if e.op == "*" else max(lt.scale, rt.scale)
                return dtypes.decimal64(18, sc)
            dec = lt if lt.code == dtypes.DECIMAL64 else rt
            return dec if e.op != "*" else dtypes.decimal64(18, dec.scale)
        if e.op == "/":
            return dtypes.float64
        return dtypes.int64
    return None


def _lit_dtype(v) -> Optional[DataType]:
    if isinstance(v, bool):
        return dtypes.bool_ if hasattr(dtypes, "bool_") else None
    if isinstance(v, int):
        return dtypes.int64
    if isinstance(v, float):
        return dtypes.float64
    if isinstance(v, str):
        return dtypes.string
    return None


# ---------------------------------------------------------------- planner
class AggCtx:
    """Aggregate-context bookkeeping for one SELECT."""

    def __init__(self):
        self.key_map: Dict[tuple, str] = {}  # expr_key -> engine key name
        self.key_asts: List[Tuple[A.ANode, str]] = []  # group-by AST -> name
        self.key_items: List[Tuple[Expr, str]] = []
        self.aggs: List[AggFunc] = []
        self.agg_map: Dict[tuple, object] = {}  # ast key -> out name | stddev tuple
        self.gid_name = "__gid"
        self.grp_names: Dict[str, str] = {}  # key engine name -> flag col
        self.nkeys = 0
        self.active = False  # replacement enabled (inside post-agg exprs)


class Planner:
    def __init__(self, catalog_schemas: Optional[Dict[str, Dict[str, DataType]]] = None):
        self.schemas = catalog_schemas or SCHEMAS
        self.alloc = NameAlloc()
        self.ctes: Dict[str, object] = {}  # name -> MatCTE | ("inline", AST)
        self.mat_ctes: List[MatCTE] = []
        self.sc_n = 0
        # columns synthesized by subquery joins (__scval/__exists): they
        # exist only AFTER the sub-join attach, so predicates referencing
        # them must ride the residual path, never a pre-attach unit filter
        self._synth_cols: Set[str] = set()

    # ------------------------------------------------------------- entry
    def plan(self, q: A.Query) -> LQuery:
        return self.plan_query(q, None)

    def plan_query(self, q: A.Query, outer: Optional[Scope]) -> LQuery:
        saved = dict(self.ctes)
        try:
            for name, sub in q.ctes:
                uses = _count_cte_uses(q, name)
                if uses > 1:
                    lq = self.plan_query(sub, None)
                    mat = MatCTE(name, lq,
                                 [infer_dtype(e, self.alloc.types)
                                  for e, _ in lq.outputs])
                    self.mat_ctes.append(mat)
                    self.ctes[name] = mat
                else:
                    self.ctes[name] = ("inline", sub)
            body_rel, outputs, ctx = self.plan_body(q.body, outer)
            order: List[Tuple[Expr, bool]] = []
            if q.order_by:
                order = self.plan_order(q.order_by, body_rel, outputs, outer, ctx)
            return LQuery(body_rel, outputs, order, q.limit)
        finally:
            self.ctes = saved

    def plan_body(self, body: A.ANode, outer: Optional[Scope]):
        if isinstance(body, A.Select):
            return self.plan_select(body, outer)
        if isinstance(body, A.SetOp):
            rel, outputs = self.plan_setop(body, outer)
            return rel, outputs, None
        raise SqlError(f"unsupported query body {type(body).__name__}")

    # ------------------------------------------------------------ set ops
    def plan_setop(self, s: A.SetOp, outer: Optional[Scope]):
        sides: List[Tuple[A.ANode, str]] = []

        def flatten(n, op):
            if isinstance(n, A.SetOp) and n.op == op:
                flatten(n.left, op)
                flatten(n.right, op)
            else:
                sides.append(n)

        if s.op in ("union", "union_all"):
            flatten(s, s.op)
            sql_names: Optional[List[str]] = None
            out_names: Optional[List[str]] = None
            projections = []
            children = []
            est = 0.0
            dim_only = True
            for side in sides:
                r, outs = (self.plan_select(side, outer)[:2] if isinstance(side, A.Select)
                           else self.plan_setop(side, outer))
                if out_names is None:
                    sql_names = [n for _, n in outs]
                    out_names = [self.alloc.fresh(n, infer_dtype(e, self.alloc.types))
                                 for e, n in outs]
                assert len(outs) == len(out_names), "UNION arity mismatch"
                projections.append([(e, n2) for (e, _), n2 in zip(outs, out_names)])
                children.append(r.node)
                est += r.est
                dim_only = dim_only and r.base_dim_only
            node = LUnionAll(children, projections)
            cols = [RelCol(None, sn, en, self.alloc.types.get(en))
                    for sn, en in zip(sql_names, out_names)]
            rel = Rel(node, cols, est, dim_only)
            outputs = [(col(en), sn) for sn, en in zip(sql_names, out_names)]
            if s.op == "union":
                rel, outputs = self._distinct(rel, outputs)
            return rel, outputs
        # INTERSECT / EXCEPT via side-tagged group-by (null-safe by
        # construction, the way Catalyst rewrites these too)
        lrel, louts = (self.plan_select(s.left, outer)[:2] if isinstance(s.left, A.Select)
                       else self.plan_setop(s.left, outer))
        rrel, routs = (self.plan_select(s.right, outer)[:2] if isinstance(s.right, A.Select)
                       else self.plan_setop(s.right, outer))
        sql_names = [n for _, n in louts]
        out_names = [self.alloc.fresh(n, infer_dtype(e, self.alloc.types)) for e, n in louts]
        side_name = self.alloc.fresh("__side", dtypes.int64)
        projections = [
            [(e, n2) for (e, _), n2 in zip(louts, out_names)] + [(lit(0), side_name)],
            [(e, n2) for (e, _), n2 in zip(routs, out_names)] + [(lit(1), side_name)],
        ]
        u = LUnionAll([lrel.node, rrel.node], projections)
        cols = [RelCol(None, sn, en, self.alloc.types.get(en))
                for sn, en in zip(sql_names, out_names)]
        if s.op in ("intersect_all", "except_all"):
            # multiset semantics: per-side counts, then replicate each
            # distinct row min(c0,c1) / max(c0-c1,0) times
            c0 = self.alloc.fresh("__c0", dtypes.int64)
            c1 = self.alloc.fresh("__c1", dtypes.int64)
            agg = LAgg(u, [(col(n), n) for n in out_names],
                       [AggFunc("sum", lit(1) - col(side_name), name=c0),
                        AggFunc("sum", col(side_name), name=c1)])
            if s.op == "intersect_all":
                m = CaseWhen([(col(c0) < col(c1), col(c0))], col(c1))
            else:
                m = col(c0) - col(c1)
            self.alloc.set_type(c0, dtypes.int64)
            self.alloc.set_type(c1, dtypes.int64)
            rep = LReplicate(LFilter(agg, m > lit(0)), m)
            rel = Rel(rep, cols, max(lrel.est, rrel.est) / 4,
                      lrel.base_dim_only and rrel.base_dim_only)
            return rel, [(col(en), sn) for sn, en in zip(sql_names, out_names)]
        nmin = self.alloc.fresh("__smin", dtypes.int64)
        nmax = self.alloc.fresh("__smax", dtypes.int64)
        agg = LAgg(u, [(col(n), n) for n in out_names],
                   [AggFunc("min", col(side_name), name=nmin),
                    AggFunc("max", col(side_name), name=nmax)])
        if s.op == "intersect":
            pred = (col(nmin) == 0) & (col(nmax) == 1)
        else:  # except
            pred = (col(nmin) == 0) & (col(nmax) == 0)
        filt = LFilter(agg, pred)
        rel = Rel(filt, cols, max(lrel.est, rrel.est) / 4,
                  lrel.base_dim_only and rrel.base_dim_only)
        return rel, [(col(en), sn) for sn, en in zip(sql_names, out_names)]

    def _distinct(self, rel: Rel, outputs):
        names = []
        proj_items = []
        for e, n in outputs:
            en = self.alloc.fresh(n, infer_dtype(e, self.alloc.types))
            names.append((n, en))
            proj_items.append((e, en))
        node = LProject(rel.node, proj_items)
        agg = LAgg(node, [(col(en), en) for _, en in names], [])
        cols = [RelCol(None, n, en, self.alloc.types.get(en)) for n, en in names]
        return (Rel(agg, cols, rel.est / 4, rel.base_dim_only),
                [(col(en), n) for n, en in names])

    # ------------------------------------------------------------- select
    def plan_select(self, sel: A.Select, outer: Optional[Scope]):
        rel, scope = self.plan_from_where(sel, outer)

        # aggregate context
        has_group = bool(sel.group_by) or sel.grouping_sets is not None
        ctx = AggCtx()
        has_agg = has_group or _has_agg(sel)
        if has_agg:
            rel, scope = self.plan_aggregate(sel, rel, scope, ctx)

        # windows
        win_funcs: List[Tuple[A.FuncCall, str]] = []
        items2 = []
        for it in sel.items:
            e2 = _extract_windows(it.expr, win_funcs, self.alloc)
            items2.append(A.SelectItem(e2, it.alias))
        if win_funcs:
            rel, scope = self.plan_window(win_funcs, rel, scope, ctx)

        # final outputs
        outputs: List[Tuple[Expr, str]] = []
        for it in items2:
            if isinstance(it.expr, A.Star):
                for c in scope.rels[0].cols:
                    if c.name.startswith("__"):
                        continue
                    outputs.append((col(c.engine), c.name))
                continue
            e = self.to_expr(it.expr, scope, ctx)
            name = it.alias
            if name is None:
                name = (it.expr.name if isinstance(it.expr, A.Ident)
                        else _auto_name(it.expr))
            outputs.append((e, name.lower()))
        if sel.distinct:
            # materialize outputs, then group by all
            names = []
            proj = []
            for e, n in outputs:
                en = self.alloc.fresh(n, infer_dtype(e, self.alloc.types))
                names.append((n, en))
                proj.append((e, en))
            node = LProject(rel.node, proj)
            agg = LAgg(node, [(col(en), en) for _, en in names], [])
            cols = [RelCol(None, n, en, self.alloc.types.get(en)) for n, en in names]
            rel = Rel(agg, cols, rel.est / 4, rel.base_dim_only)
            outputs = [(col(en), n) for n, en in names]
        return rel, outputs, ctx

    # --------------------------------------------------------- FROM/WHERE
    def plan_from_where(self, sel: A.Select, outer: Optional[Scope]):
        units: List[Rel] = []
        conds: List[A.ANode] = []
        # trailing outer joins of a left-deep FROM chain: their inner
        # PREFIX flattens into the join graph (so WHERE equi conjuncts
        # like q72's d1.d_week_seq = d2.d_week_seq become graph edges and
        # filters push into scans); each outer join applies afterwards
        pending_outer: List[tuple] = []

        def add_item(item):
            if isinstance(item, A.Join) and item.kind == "inner":
                add_item(item.left)
                add_item(item.right)
                if item.on is not None:
                    conds.extend(_conjuncts(item.on))
            elif isinstance(item, A.Join):
                add_item(item.left)
                rrel = self._join_unit(item.right, outer)
                pending_outer.append((item.kind, rrel, item.on))
            else:
                units.append(self.plan_table(item, outer))

        for item in sel.from_:
            add_item(item)
        if not units:
            # FROM-less select (not used by TPC-DS, but harmless)
            raise SqlError("SELECT without FROM unsupported")
        if sel.where is not None:
            conds.extend(_conjuncts(sel.where))
        scope = Scope(units + [r for _, r, _ in pending_outer], outer)
        post_conds: List[A.ANode] = []
        if pending_outer:
            outer_cols = {c.engine for _, r, _ in pending_outer for c in r.cols}
            unsafe_all = any(k in ("full", "right") for k, _, _ in pending_outer)
            graph_conds = []
            for cnd in conds:
                if unsafe_all:
                    post_conds.append(cnd)
                    continue
                if _contains_subquery(cnd) or _contains_scalar_subquery(cnd):
                    graph_conds.append(cnd)
                    continue
                conv = self._try_convert(cnd, scope)
                if conv is None or (expr_cols(conv) & outer_cols):
                    # references an outer-join right side: apply after it
                    post_conds.append(cnd)
                else:
                    graph_conds.append(cnd)
            conds = graph_conds
        rel = self.join_graph(units, conds, scope)
        for kind, rrel, on in pending_outer:
            rel = self._apply_outer(rel, kind, rrel, on, outer)
        for cnd in post_conds:
            pred = self.to_expr(cnd, Scope([rel], outer), None)
            rel = Rel(LFilter(rel.node, pred), rel.cols,
                      rel.est * _selectivity(pred), rel.base_dim_only)
        return rel, Scope([rel], outer)

    def _try_convert(self, cnd, scope) -> Optional[Expr]:
        try:
            return self.to_expr(cnd, scope, None)
        except SqlError:
            return None

    def _apply_outer(self, lrel: Rel, kind: str, rrel: Rel, on,
                     outer: Optional[Scope]) -> Rel:
        scope = Scope([lrel, rrel], outer)
        lkeys, rkeys, residual, rfilters = [], [], None, []
        lcols = {rc.engine for rc in lrel.cols}
        rcols = {rc.engine for rc in rrel.cols}
        for c in _conjuncts(on) if on is not None else []:
            pair = _equi_pair(c, scope, self, lcols, rcols)
            if pair is not None:
                lkeys.append(pair[0])
                rkeys.append(pair[1])
                continue
            conv = self.to_expr(c, scope, None)
            refs = expr_cols(conv)
            if refs <= rcols and kind == "left":
                rfilters.append(conv)
            else:
                residual = conv if residual is None else (residual & conv)
        rnode = rrel.node
        for f in rfilters:
            rnode = LFilter(rnode, f)
        node = LJoin(lrel.node, rnode, kind, lkeys, rkeys, residual,
                     l_est=lrel.est, r_est=rrel.est,
                     r_base_dim=rrel.base_dim_only)
        return Rel(node, lrel.cols + rrel.cols, max(lrel.est, rrel.est),
                   lrel.base_dim_only and rrel.base_dim_only)

    def plan_table(self, item: A.ANode, outer: Optional[Scope]) -> Rel:
        if isinstance(item, A.Table):
            alias = (item.alias or item.name).lower()
            if item.name in self.ctes:
                return self.plan_cte_ref(item.name, alias, outer)
            schema = self.schemas.get(item.name)
            if schema is None:
                raise SqlError(f"unknown table {item.name}")
            colmap = {}
            cols = []
            for cname, dt in schema.items():
                en = self.alloc.fresh(cname, dt)
                colmap[cname] = en
                cols.append(RelCol(alias, cname, en, dt))
            node = LScan(item.name, colmap)
            from ..tpcds.schema import BASE_ROWS

            est = float(BASE_ROWS.get(item.name, 100_000))
            return Rel(node, cols, est, item.name not in FACT_TABLES)
        if isinstance(item, A.DerivedTable):
            lq = self.plan_query(item.query, outer)
            return self.rel_of_lquery(lq, item.alias.lower())
        raise SqlError(f"unsupported table ref {type(item).__name__}")

    def plan_cte_ref(self, name: str, alias: str, outer: Optional[Scope]) -> Rel:
        entry = self.ctes[name]
        if isinstance(entry, MatCTE):
            colmap = {}
            cols = []
            for (_, n), dt in zip(entry.lquery.outputs, entry.out_dtypes or []):
                en = self.alloc.fresh(n, dt)
                colmap[n] = en
                cols.append(RelCol(alias, n, en, dt))
            return Rel(LCTERef(entry, colmap), cols, entry.est, False)
        _, sub_ast = entry
        lq = self.plan_query(sub_ast, outer)
        return self.rel_of_lquery(lq, alias)

    def rel_of_lquery(self, lq: LQuery, alias: str) -> Rel:
        """Wrap a planned subquery as a relation with its outputs projected.
        Sort keys are exprs over the UNprojected child, so sort runs first."""
        child = lq.rel.node
        if lq.order:
            child = LSort(child, lq.order, lq.limit)
        items = []
        cols = []
        for e, n in lq.outputs:
            dt = infer_dtype(e, self.alloc.types)
            en = self.alloc.fresh(n, dt)
            items.append((e, en))
            cols.append(RelCol(alias, n.lower(), en, dt))
        node: LNode = LProject(child, items)
        if lq.limit is not None and not lq.order:
            node = LLimit(node, lq.limit)
        return Rel(node, cols, lq.rel.est, lq.rel.base_dim_only)

    def plan_outer_join(self, j: A.Join, outer: Optional[Scope]) -> Rel:
        """left/right/full outer ANSI join subtree: structure is fixed."""
        lrel = (self.plan_outer_join(j.left, outer) if isinstance(j.left, A.Join)
                and j.left.kind != "inner" else self._join_unit(j.left, outer))
        rrel = (self.plan_outer_join(j.right, outer) if isinstance(j.right, A.Join)
                and j.right.kind != "inner" else self._join_unit(j.right, outer))
        scope = Scope([lrel, rrel], outer)
        lkeys, rkeys, residual, lfilters, rfilters = [], [], None, [], []
        for c in _conjuncts(j.on) if j.on is not None else []:
            conv = self.to_expr(c, scope, None)
            refs = expr_cols(conv)
            lcols = {rc.engine for rc in lrel.cols}
            rcols = {rc.engine for rc in rrel.cols}
            in_l = refs & lcols
            in_r = refs & rcols
            pair = _equi_pair(c, scope, self, lcols, rcols)
            if pair is not None:
                lkeys.append(pair[0])
                rkeys.append(pair[1])
            elif in_l and not in_r:
                lfilters.append(conv)
            elif in_r and not in_l:
                rfilters.append(conv)
            else:
                residual = conv if residual is None else (residual & conv)
        # ON-side filters: for LEFT join a right-only pred filters the right
        # input; a left-only pred becomes a residual (it must not drop
        # unmatched left rows). Mirror for RIGHT.
        lnode, rnode = lrel.node, rrel.node
        if j.kind in ("left", "full") and lfilters:
            for f in lfilters:
                residual = f if residual is None else (residual & f)
            lfilters = []
        if j.kind in ("right", "full") and rfilters:
            for f in rfilters:
                residual = f if residual is None else (residual & f)
            rfilters = []
        for f in lfilters:
            lnode = LFilter(lnode, f)
        for f in rfilters:
            rnode = LFilter(rnode, f)
        node = LJoin(lnode, rnode, j.kind, lkeys, rkeys, residual,
                     l_est=lrel.est, r_est=rrel.est, r_base_dim=rrel.base_dim_only)
        est = max(lrel.est, rrel.est)
        return Rel(node, lrel.cols + rrel.cols, est,
                   lrel.base_dim_only and rrel.base_dim_only)

    def _join_unit(self, item: A.ANode, outer: Optional[Scope]) -> Rel:
        """An operand of an outer join: possibly an inner-join chain."""
        if isinstance(item, A.Join) and item.kind == "inner":
            units: List[Rel] = []
            conds: List[A.ANode] = []

            def rec(n):
                if isinstance(n, A.Join) and n.kind == "inner":
                    rec(n.left)
                    rec(n.right)
                    if n.on is not None:
                        conds.extend(_conjuncts(n.on))
                elif isinstance(n, A.Join):
                    units.append(self.plan_outer_join(n, outer))
                else:
                    units.append(self.plan_table(n, outer))

            rec(item)
            scope = Scope(units, outer)
            return self.join_graph(units, conds, scope)
        if isinstance(item, A.Join):
            return self.plan_outer_join(item, outer)
        return self.plan_table(item, outer)

    # ---------------------------------------------------------- join graph
    def join_graph(self, units: List[Rel], conds: List[A.ANode],
                   scope: Scope) -> Rel:
        """AuronConvertStrategy-style bottom-up assembly: classify WHERE
        conjuncts into unit-local filters / equi edges / subquery joins /
        residuals, then greedily join (smallest connected unit first)."""
        unit_of: Dict[str, int] = {}
        for i, u in enumerate(units):
            for c in u.cols:
                unit_of[c.engine] = i

        filters: Dict[int, List[Expr]] = {i: [] for i in range(len(units))}
        edges: List[Tuple[int, Expr, int, Expr]] = []
        residuals: List[Tuple[Set[int], Expr]] = []
        sub_joins: List[Tuple[int, Rel, str, List[Expr], List[Expr], Optional[Expr], str]] = []

        def unit_set(refs: Set[str]) -> Set[int]:
            return {unit_of[r] for r in refs if r in unit_of}

        for cnd in conds:
            handled = self._try_subquery_conjunct(cnd, scope, unit_of, units, sub_joins)
            if handled is True:
                continue
            if handled is not None and handled is not False:
                cnd_expr = handled  # rewritten (existence cols substituted)
            else:
                cnd_expr = self.to_expr(cnd, scope, None)
            refs = expr_cols(cnd_expr)
            uset = unit_set(refs)
            # existence-join mark columns resolve later; keep with residuals
            if refs - set(unit_of):
                residuals.append((uset, cnd_expr))
                continue
            if len(uset) <= 1 and not (refs & self._synth_cols):
                tgt = next(iter(uset)) if uset else 0
                filters[tgt].append(cnd_expr)
                continue
            if refs & self._synth_cols:
                residuals.append((uset, cnd_expr))
                continue
            pair = _equi_pair_units(cnd, scope, self, unit_of)
            if pair is not None:
                li, le, ri, re_ = pair
                edges.append((li, le, ri, re_))
            else:
                residuals.append((uset, cnd_expr))

        # apply local filters (push into scans where possible)
        for i, u in enumerate(units):
            for f in filters[i]:
                u.node = _push_filter(u.node, f)
                u.est *= _selectivity(f)

        # attach subquery joins (semi/anti/existence/scalar-agg inner)
        for anchor, sub_rel, kind, lkeys, rkeys, residual, exname in sub_joins:
            u = units[anchor]
            jn = LJoin(u.node, sub_rel.node, kind, lkeys, rkeys, residual,
                       existence_name=exname, l_est=u.est, r_est=sub_rel.est,
                       r_base_dim=sub_rel.base_dim_only)
            newcols = list(u.cols)
            if kind == "inner":
                newcols += sub_rel.cols
            elif kind == "existence":
                newcols.append(RelCol(None, exname, exname,
                                      dtypes.bool_ if hasattr(dtypes, "bool_") else None))
            units[anchor] = Rel(jn, newcols,
                                u.est if kind != "inner" else max(u.est, sub_rel.est) if not sub_rel.base_dim_only else u.est,
                                u.base_dim_only and sub_rel.base_dim_only)
            for c in units[anchor].cols:
                unit_of.setdefault(c.engine, anchor)

        # greedy join: start at the largest unit, join smallest neighbors
        alive = set(range(len(units)))
        merged = {i: i for i in range(len(units))}

        def root(i):
            while merged[i] != i:
                i = merged[i]
            return i

        pending_res = list(residuals)

        def apply_residuals(ri: int):
            nonlocal pending_res
            left = []
            for uset, ex in pending_res:
                roots = {root(x) for x in uset}
                if roots == {ri} or (not roots):
                    units[ri].node = LFilter(units[ri].node, ex)
                    units[ri].est *= _selectivity(ex)
                else:
                    left.append((uset, ex))
            pending_res = left

        for i in alive:
            apply_residuals(i)

        while len({root(i) for i in alive}) > 1:
            roots = sorted({root(i) for i in alive})
            # choose the globally cheapest join pair: attaching a dimension
            # barely grows the fact side, and a fact-fact join gets cheaper
            # with every extra equi edge (q72: cs><inv must wait until the
            # week-linking date dims are attached so BOTH keys apply)
            pair_edges: Dict[Tuple[int, int], int] = {}
            for li, le, ri, re_ in edges:
                lr, rr = root(li), root(ri)
                if lr == rr:
                    continue
                key = (min(lr, rr), max(lr, rr))
                pair_edges[key] = pair_edges.get(key, 0) + 1
            best = None
            for (lr, rr), n_edges in pair_edges.items():
                a, b = units[lr], units[rr]
                if a.base_dim_only or b.base_dim_only:
                    score = max(a.est, b.est) * 0.8
                else:
                    score = max(a.est, b.est) * (2.0 / (1.0 + n_edges))
                if best is None or score < best[0]:
                    best = (score, lr, rr)
            if best is None:
                # cartesian: join the two smallest on a constant key
                small = sorted(roots, key=lambda r: units[r].est)
                lr, rr = small[0], small[1]
                lk, rk = [lit(1)], [lit(1)]
                self._merge(units, merged, lr, rr, lk, rk, edges)
                apply_residuals(root(lr))
                continue
            _, lr, rr = best
            lk, rk = [], []
            for li, le, ri, re_ in edges:
                a, b = root(li), root(ri)
                if (a, b) == (lr, rr):
                    lk.append(le)
                    rk.append(re_)
                elif (a, b) == (rr, lr):
                    lk.append(re_)
                    rk.append(le)
            self._merge(units, merged, lr, rr, lk, rk, edges)
            apply_residuals(root(lr))

        r = units[root(next(iter(alive)))]
        return r

    def _merge(self, units, merged, lr, rr, lkeys, rkeys, edges):
        lu, ru = units[lr], units[rr]
        # keep the big side as the probe (left); build = right — always
        # (a constant-key cross join with the big side as build degrades
        # to one multi-million-entry hash chain, q93)
        if ru.est > lu.est:
            lu, ru = ru, lu
            lkeys, rkeys = rkeys, lkeys
            lr, rr = rr, lr
        node = LJoin(lu.node, ru.node, "inner", lkeys, rkeys,
                     l_est=lu.est, r_est=ru.est, r_base_dim=ru.base_dim_only)
        est = lu.est if not ru.base_dim_only else lu.est * 0.8
        if not lu.base_dim_only and not ru.base_dim_only:
            est = max(lu.est, ru.est)
        nrel = Rel(node, lu.cols + ru.cols, est,
                   lu.base_dim_only and ru.base_dim_only)
        units[lr] = nrel
        merged[rr] = lr

    # ---------------------------------------------------- WHERE subqueries
    def _try_subquery_conjunct(self, cnd, scope, unit_of, units, sub_joins):
        """Handle [NOT] EXISTS / [NOT] IN-subquery / correlated-scalar
        conjuncts. Returns True if fully consumed, False if not a subquery
        conjunct, or a rewritten engine expr (existence-join / scalar)."""
        neg = False
        inner = cnd
        while isinstance(inner, A.UnOp) and inner.op == "not":
            neg = not neg
            inner = inner.operand
        if isinstance(inner, A.Exists):
            if inner.negated:
                neg = not neg
            self._plan_exists(inner.query, "anti" if neg else "semi",
                              None, scope, unit_of, units, sub_joins)
            return True
        if isinstance(inner, A.InSubquery):
            if inner.negated:
                neg = not neg
            self._plan_exists(inner.query, "anti" if neg else "semi",
                              inner.operand, scope, unit_of, units, sub_joins)
            return True
        if _contains_subquery(cnd):
            # EXISTS/IN nested under OR etc: existence joins (mark columns)
            aliases: Dict[str, Expr] = {}
            new_ast = self._rewrite_nested_subqueries(cnd, scope, unit_of,
                                                      units, sub_joins, aliases)
            return self.to_expr(new_ast, scope, None, alias_env=aliases)
        if _contains_scalar_subquery(cnd):
            try:
                return self.to_expr(cnd, scope, None)  # uncorrelated
            except SqlError:
                pass
            # correlated scalar-agg subquery: decorrelate into inner joins
            aliases = {}
            new_ast = self._rewrite_scalar_subqueries(cnd, scope, unit_of,
                                                      units, sub_joins, aliases)
            return self.to_expr(new_ast, scope, None, alias_env=aliases)
        return False

    def _rewrite_nested_subqueries(self, node, scope, unit_of, units,
                                   sub_joins, aliases):
        if isinstance(node, A.Exists):
            exname = self.alloc.fresh("__exists", None)
            anchor = self._plan_exists(node.query, "existence", None, scope,
                                       unit_of, units, sub_joins, exname)
            unit_of[exname] = anchor
            self._synth_cols.add(exname)
            aliases[exname] = col(exname)
            e = A.Ident([exname])
            return A.UnOp("not", e) if node.negated else e
        if isinstance(node, A.InSubquery):
            exname = self.alloc.fresh("__exists", None)
            anchor = self._plan_exists(node.query, "existence", node.operand,
                                       scope, unit_of, units, sub_joins, exname)
            unit_of[exname] = anchor
            self._synth_cols.add(exname)
            aliases[exname] = col(exname)
            e = A.Ident([exname])
            return A.UnOp("not", e) if node.negated else e
        if dataclasses.is_dataclass(node):
            kw = {}
            for f in dataclasses.fields(node):
                v = getattr(node, f.name)
                if isinstance(v, A.ANode):
                    kw[f.name] = self._rewrite_nested_subqueries(
                        v, scope, unit_of, units, sub_joins, aliases)
                elif isinstance(v, list):
                    kw[f.name] = [self._rewrite_nested_subqueries(
                        x, scope, unit_of, units, sub_joins, aliases)
                        if isinstance(x, A.ANode) else x for x in v]
                else:
                    kw[f.name] = v
            return type(node)(**kw)
        return node

    def _rewrite_scalar_subqueries(self, node, scope, unit_of, units,
                                   sub_joins, aliases):
        if isinstance(node, A.ScalarSubquery):
            rel, okeys, ikey_names, vn = self.plan_scalar_agg_subquery(
                node.query, scope)
            outer_refs: Set[str] = set()
            for k in okeys:
                outer_refs |= expr_cols(k)
            anchors = {unit_of[r] for r in outer_refs if r in unit_of}
            if len(anchors) != 1:
                raise SqlError("correlated scalar subquery spans "
                               f"{len(anchors)} relations")
            anchor = next(iter(anchors))
            sub_joins.append((anchor, rel, "inner", okeys,
                              [col(n) for n in ikey_names], None, ""))
            unit_of[vn] = anchor
            self._synth_cols.add(vn)
            aliases[vn] = col(vn)
            return A.Ident([vn])
        if dataclasses.is_dataclass(node):
            kw = {}
            for f in dataclasses.fields(node):
                v = getattr(node, f.name)
                if isinstance(v, A.ANode):
                    kw[f.name] = self._rewrite_scalar_subqueries(
                        v, scope, unit_of, units, sub_joins, aliases)
                elif isinstance(v, list):
                    kw[f.name] = [self._rewrite_scalar_subqueries(
                        x, scope, unit_of, units, sub_joins, aliases)
                        if isinstance(x, A.ANode) else x for x in v]
                else:
                    kw[f.name] = v
            return type(node)(**kw)
        return node

    def _plan_exists(self, q: A.Query, kind: str, in_operand, scope, unit_of,
                     units, sub_joins, exname: str = "__exists"):
        """Plan a predicate subquery into a semi/anti/existence join."""
        if q.ctes:
            raise SqlError("CTE inside predicate subquery unsupported")
        body = q.body
        if not isinstance(body, A.Select):
            raise SqlError("set-op predicate subquery unsupported")
        if body.group_by or _has_agg(body):
            # correlated scalar-agg handled via to_expr path; EXISTS-with-agg
            # not used by TPC-DS
            raise SqlError("aggregated EXISTS subquery unsupported")
        sub_units: List[Rel] = []
        conds: List[A.ANode] = []

        def add_item(item):
            if isinstance(item, A.Join) and item.kind == "inner":
                add_item(item.left)
                add_item(item.right)
                if item.on is not None:
                    conds.extend(_conjuncts(item.on))
            elif isinstance(item, A.Join):
                sub_units.append(self.plan_outer_join(item, scope))
            else:
                sub_units.append(self.plan_table(item, scope))

        for it in body.from_:
            add_item(it)
        if body.where is not None:
            conds.extend(_conjuncts(body.where))
        sub_scope = Scope(sub_units, scope)
        inner_cols: Set[str] = {c.engine for u in sub_units for c in u.cols}

        inner_conds: List[A.ANode] = []
        lkeys: List[Expr] = []
        rkeys: List[Expr] = []
        residual: Optional[Expr] = None
        for cnd in conds:
            if _contains_subquery(cnd) or _contains_scalar_subquery(cnd):
                # nested subquery predicate: assumed uncorrelated with the
                # outermost scope (true for TPC-DS); join_graph handles it
                inner_conds.append(cnd)
                continue
            conv = self.to_expr(cnd, sub_scope, None)
            refs = expr_cols(conv)
            if refs <= inner_cols:
                inner_conds.append(cnd)
                continue
            pair = _equi_pair_io(cnd, sub_scope, self, inner_cols)
            if pair is not None:
                okey, ikey = pair
                lkeys.append(okey)
                rkeys.append(ikey)
            else:
                residual = conv if residual is None else (residual & conv)
        sub_rel = self.join_graph(sub_units, inner_conds, sub_scope)
        if in_operand is not None:
            op_expr = self.to_expr(in_operand, scope, None)
            # subquery select item = membership column
            if len(body.items) != 1:
                raise SqlError("IN subquery must have one select item")
            item_expr = self.to_expr(body.items[0].expr, sub_scope, None)
            lkeys.insert(0, op_expr)
            rkeys.insert(0, item_expr)
        if not lkeys:
            raise SqlError("uncorrelated EXISTS unsupported")
        # anchor: the unit owning the outer keys
        outer_refs: Set[str] = set()
        for k in lkeys:
            outer_refs |= expr_cols(k)
        anchors = {unit_of[r] for r in outer_refs if r in unit_of}
        if len(anchors) != 1:
            raise SqlError(f"predicate subquery spans {len(anchors)} relations")
        anchor = next(iter(anchors))
        sub_joins.append((anchor, sub_rel, kind, lkeys, rkeys,
                          residual, exname))
        return anchor

    # ------------------------------------------------------- aggregation
    def plan_aggregate(self, sel: A.Select, rel: Rel, scope: Scope, ctx: AggCtx):
        keys_ast = list(sel.group_by)
        gsets_ast = sel.grouping_sets
        key_items: List[Tuple[Expr, str]] = []
        key_dts: List[DataType] = []
        if gsets_ast is not None:
            # union of all exprs across sets, order of first appearance
            seen = []
            for gs in gsets_ast:
                for e in gs:
                    if not any(e == s for s in seen):
                        seen.append(e)
            keys_ast = seen
        for k_ast in keys_ast:
            e = self.to_expr(k_ast, scope, None)
            dt = infer_dtype(e, self.alloc.types)
            base = e.name if isinstance(e, Col) else "__key"
            en = self.alloc.fresh(base, dt) if not isinstance(e, Col) else e.name
            ctx.key_map[expr_key(e)] = en
            ctx.key_asts.append((k_ast, en))
            if isinstance(k_ast, A.Ident) and k_ast.qualifier is not None:
                # `GROUP BY t.c` may be spelled bare `c` elsewhere
                ctx.key_asts.append((A.Ident([k_ast.name]), en))
            key_items.append((e, en))
            key_dts.append(dt if dt is not None else dtypes.float64)
        ctx.key_items = key_items
        ctx.nkeys = len(key_items)
        ctx.active = True

        grouping_sets = None
        if sel.group_rollup:
            n = len(key_items)
            grouping_sets = [[i < d for i in range(n)] for d in range(n, -1, -1)]
        elif gsets_ast is not None:
            grouping_sets = []
            for gs in gsets_ast:
                mask = [any(k == g for g in gs) for k in keys_ast]
                grouping_sets.append(mask)
        if grouping_sets is not None:
            ctx.gid_name = self.alloc.fresh("__gid", dtypes.int64)
            for _, kn in key_items:
                ctx.grp_names[kn] = self.alloc.fresh(f"__grp_{kn}", dtypes.int64)

        # collect aggregate calls from select items + having + order
        for it in sel.items:
            _collect_aggs(it.expr, self, scope, ctx)
        if sel.having is not None:
            _collect_aggs(sel.having, self, scope, ctx)

        agg = LAgg(rel.node, key_items, ctx.aggs, grouping_sets,
                   ctx.gid_name, key_dts)
        agg.grp_names = dict(ctx.grp_names)
        cols = []
        for k_ast, (_, n) in zip(keys_ast, key_items):
            if isinstance(k_ast, A.Ident):
                cols.append(RelCol(k_ast.qualifier, k_ast.name, n,
                                   self.alloc.types.get(n)))
            else:
                cols.append(RelCol(None, n, n, self.alloc.types.get(n)))
        for a in ctx.aggs:
            cols.append(RelCol(None, a.name, a.name, self.alloc.types.get(a.name)))
        if grouping_sets is not None:
            cols.append(RelCol(None, ctx.gid_name, ctx.gid_name, dtypes.int64))
            for kn, gn in ctx.grp_names.items():
                cols.append(RelCol(None, gn, gn, dtypes.int64))
        out_rel = Rel(agg, cols, max(rel.est / 10, 1.0), rel.base_dim_only)
        out_scope = Scope([out_rel], scope.parent)
        if sel.having is not None:
            pred = self.to_expr(sel.having, out_scope, ctx)
            out_rel = Rel(LFilter(out_rel.node, pred), out_rel.cols,
                          out_rel.est / 2, out_rel.base_dim_only)
            out_scope = Scope([out_rel], scope.parent)
        return out_rel, out_scope

    # ----------------------------------------------------------- windows
    def plan_window(self, win_funcs: List[Tuple[A.FuncCall, str]], rel: Rel,
                    scope: Scope, ctx: AggCtx):
        # group window funcs by (partition, order, frame) spec
        groups: Dict[tuple, list] = {}
        for fc, name in win_funcs:
            spec = fc.over
            part = [self.to_expr(p, scope, ctx) for p in spec.partition_by]
            order = [(self.to_expr(oi.expr, scope, ctx), oi.ascending)
                     for oi in spec.order_by]
            key = (tuple(expr_key(p) for p in part),
                   tuple((expr_key(e), a) for e, a in order),
                   spec.frame or "range",
                   getattr(spec, "frame_lo", None),
                   getattr(spec, "frame_hi", 0))
            groups.setdefault(key, []).append((fc, name, part, order))
        node = rel.node
        cols = list(rel.cols)
        for key, fns in groups.items():
            part = fns[0][2]
            order = fns[0][3]
            frame = key[2]
            funcs = []
            for fc, name, _, _ in fns:
                wf = self._window_func(fc, scope, ctx)
                funcs.append((wf, name))
                self.alloc.set_type(name, _window_dtype(fc, wf, self.alloc.types))
                cols.append(RelCol(None, name, name, self.alloc.types.get(name)))
            node = LWindow(node, part, order, funcs, frame,
                           key[3], key[4])
        out = Rel(node, cols, rel.est, rel.base_dim_only)
        return out, Scope([out], scope.parent)

    def _window_func(self, fc: A.FuncCall, scope, ctx) -> WindowFunc:
        name = fc.name
        if name in ("rank", "dense_rank", "row_number", "percent_rank",
                    "cume_dist"):
            return WindowFunc(name)
        if name == "ntile":
            n = int(fc.args[0].text)
            return WindowFunc("ntile", None, n)
        if name in ("lead", "lag"):
            arg = self.to_expr(fc.args[0], scope, ctx)
            off = int(fc.args[1].text) if len(fc.args) > 1 else 1
            default = None
            if len(fc.args) > 2:
                d = self.to_expr(fc.args[2], scope, ctx)
                default = d.value if isinstance(d, Literal) else None
            return WindowFunc(name, arg, off, default)
        if name in ("sum", "avg", "min", "max", "count", "first_value",
                    "last_value", "nth_value"):
            arg = self.to_expr(fc.args[0], scope, ctx) if fc.args else None
            fn = {"first_value": "first", "last_value": "last"}.get(name, name)
            return WindowFunc(fn, arg)
        raise SqlError(f"unsupported window function {name}")

    # ------------------------------------------------------------ order by
    def plan_order(self, order_items: List[A.OrderItem], rel: Rel,
                   outputs, outer, ctx: Optional[AggCtx] = None
                   ) -> List[Tuple[Expr, bool]]:
        out: List[Tuple[Expr, bool]] = []
        by_alias = {n: e for e, n in outputs}
        out_keys = {expr_key(e): e for e, _ in outputs}
        scope = Scope([rel], outer)
        def emit(e, oi):
            # explicit NULLS FIRST/LAST that differs from the Spark
            # default (asc->nulls first, desc->nulls last, which the
            # executor implements) desugars into a leading IsNull key —
            # no plan-node change needed
            nf = getattr(oi, "nulls_first", None)
            default_nf = oi.ascending
            if nf is not None and nf != default_nf:
                from ..exprs import IsNull

                out.append((IsNull(e), not nf))
            out.append((e, oi.ascending))

        for oi in order_items:
            e_ast = oi.expr
            if isinstance(e_ast, A.Num) and e_ast.is_int:
                idx = int(e_ast.text) - 1
                if not (0 <= idx < len(outputs)):
                    raise SqlError(f"ORDER BY ordinal {idx+1} out of range")
                emit(outputs[idx][0], oi)
                continue
            if isinstance(e_ast, A.Ident) and e_ast.qualifier is None \
                    and e_ast.name.lower() in by_alias:
                emit(by_alias[e_ast.name.lower()], oi)
                continue
            try:
                e = self.to_expr(e_ast, scope, ctx)
            except SqlError:
                # may reference a select alias inside an expression
                e = self.to_expr(e_ast, scope, ctx, alias_env=by_alias)
            k = expr_key(e)
            if k in out_keys:
                e = out_keys[k]
            emit(e, oi)
        return out

    # ----------------------------------------------------- expr conversion
    def to_expr(self, node: A.ANode, scope: Scope, ctx: Optional[AggCtx],
                alias_env: Optional[Dict[str, Expr]] = None) -> Expr:
        if ctx is not None and ctx.active:
            # AST-level group-key match first: a select/having/order expr
            # that IS a group-by expr references the key output column
            # (children of the key expr are no longer in scope post-agg)
            for ka, kn in ctx.key_asts:
                if node == ka:
                    return col(kn)
        e = self._to_expr(node, scope, ctx, alias_env)
        if ctx is not None and ctx.active:
            k = expr_key(e)
            if k in ctx.key_map:
                return col(ctx.key_map[k])
        return e

    def _to_expr(self, node, scope, ctx, alias_env=None) -> Expr:
        conv = lambda n: self.to_expr(n, scope, ctx, alias_env)
        if isinstance(node, A.Num):
            return lit(int(node.text)) if node.is_int else lit(float(node.text))
        if isinstance(node, A.Str):
            return lit(node.value)
        if isinstance(node, A.Null):
            return Literal(None, dtypes.string)  # retyped by context
        if isinstance(node, A.Ident):
            hit = scope.resolve(node.qualifier and node.qualifier.lower(),
                                node.name)
            if hit is None and alias_env is not None and node.qualifier is None \
                    and node.name.lower() in alias_env:
                return alias_env[node.name.lower()]
            if hit is None:
                raise SqlError(f"unresolved column {'.'.join(node.parts)}")
            return col(hit.engine)
        if isinstance(node, A.BinOp):
            return self._binop(node, scope, ctx, alias_env)
        if isinstance(node, A.UnOp):
            if node.op == "not":
                return ~conv(node.operand)
            if node.op == "-":
                e = conv(node.operand)
                if isinstance(e, Literal) and isinstance(e.value, (int, float)):
                    return lit(-e.value)
                return lit(0) - e
        if isinstance(node, A.IsNull):
            e = conv(node.operand)
            return e.is_not_null() if node.negated else e.is_null()
        if isinstance(node, A.Between):
            e = conv(node.operand)
            lo = self._coerce_lit(_fold_const(conv(node.low)), e)
            hi = self._coerce_lit(_fold_const(conv(node.high)), e)
            b = (e >= lo) & (e <= hi)
            return ~b if node.negated else b
        if isinstance(node, A.InList):
            e = conv(node.operand)
            vals = []
            for it in node.items:
                v = _fold_const(conv(it))
                if not isinstance(v, Literal):
                    raise SqlError("IN list items must be constant")
                cv = self._coerce_lit(v, e)
                vals.append(cv.value if isinstance(cv, Literal) else v.value)
            r = e.isin(vals)
            return ~r if node.negated else r
        if isinstance(node, A.Like):
            e = conv(node.operand)
            r = e.like(node.pattern)
            return ~r if node.negated else r
        if isinstance(node, A.Case):
            return self._case(node, scope, ctx, alias_env)
        if isinstance(node, A.CastE):
            return self._cast(node, scope, ctx, alias_env)
        if isinstance(node, A.FuncCall):
            return self._func(node, scope, ctx, alias_env)
        if isinstance(node, A.ScalarSubquery):
            return self._scalar_subquery(node.query, scope, ctx)
        if isinstance(node, A.Interval):
            raise SqlError("bare INTERVAL outside +/- unsupported")
        if isinstance(node, A.Exists) or isinstance(node, A.InSubquery):
            raise SqlError("predicate subquery in unsupported position")
        raise SqlError(f"unsupported expression {type(node).__name__}")

    def _coerce_lit(self, v: Expr, other: Expr) -> Expr:
        """Fold a string literal to date32 when compared against a date."""
        if isinstance(v, Literal) and isinstance(v.value, str):
            dt = infer_dtype(other, self.alloc.types)
            if dt is not None and dt.code == dtypes.DATE32:
                try:
                    days = (_dt.date.fromisoformat(v.value) - EPOCH).days
                    return Literal(days, dtypes.date32)
                except ValueError:
                    pass
        return v

    def _binop(self, node: A.BinOp, scope, ctx, alias_env) -> Expr:
        conv = lambda n: self.to_expr(n, scope, ctx, alias_env)
        op = node.op
        if op == "and":
            return conv(node.left) & conv(node.right)
        if op == "or":
            return conv(node.left) | conv(node.right)
        if op in ("+", "-") and isinstance(node.right, A.Interval):
            base = conv(node.left)
            n = node.right.n
            if isinstance(base, Literal) and base.dtype is not None \
                    and base.dtype.code == dtypes.DATE32:
                return Literal(base.value + (n if op == "+" else -n), dtypes.date32)
            return F.DateAdd(base, lit(n if op == "+" else -n))
        l = conv(node.left)
        r = conv(node.right)
        if op in ("=", "<>", "<", "<=", ">", ">="):
            l2, r2 = self._coerce_lit(l, r), self._coerce_lit(r, l)
            l, r = l2, r2
            return {"=": l == r, "<>": l != r, "<": l < r, "<=": l <= r,
                    ">": l > r, ">=": l >= r}[op]
        if op == "+":
            # date + int days
            ldt = infer_dtype(l, self.alloc.types)
            if ldt is not None and ldt.code == dtypes.DATE32:
                return F.DateAdd(l, r)
            return l + r
        if op == "-":
            ldt = infer_dtype(l, self.alloc.types)
            if ldt is not None and ldt.code == dtypes.DATE32:
                rdt = infer_dtype(r, self.alloc.types)
                if rdt is not None and rdt.code == dtypes.DATE32:
                    return F.DateDiff(l, r)
                return F.DateSub(l, r)
            return l - r
        if op == "*":
            return l * r
        if op == "/":
            return l / r
        if op == "%":
            return F.Mod(l, r) if hasattr(F, "Mod") else l - (l / r) * r
        if op == "||":
            from ..exprs import ConcatStr

            return ConcatStr([l, r])
        raise SqlError(f"unsupported operator {op}")

    def _case(self, node: A.Case, scope, ctx, alias_env) -> Expr:
        conv = lambda n: self.to_expr(n, scope, ctx, alias_env)
        branches = []
        if node.operand is not None:
            opnd = conv(node.operand)
            for c, v in node.whens:
                branches.append((opnd == conv(c), conv(v)))
        else:
            for c, v in node.whens:
                branches.append((conv(c), conv(v)))
        otherwise = conv(node.else_) if node.else_ is not None else None
        # retype NULL literals from (non-null) sibling branches
        dts = [infer_dtype(v, self.alloc.types) for _, v in branches
               if not _is_null_lit(v)]
        if otherwise is not None and not _is_null_lit(otherwise):
            dts.append(infer_dtype(otherwise, self.alloc.types))
        dt = next((d for d in dts if d is not None), None)
        if dt is not None:
            branches = [(c, Literal(None, dt) if _is_null_lit(v) else v)
                        for c, v in branches]
            if otherwise is not None and _is_null_lit(otherwise):
                otherwise = Literal(None, dt)
        return CaseWhen(branches, otherwise)

    def _cast(self, node: A.CastE, scope, ctx, alias_env) -> Expr:
        e = self.to_expr(node.operand, scope, ctx, alias_env)
        tn = node.typename
        if tn == "date":
            if isinstance(e, Literal) and isinstance(e.value, str):
                return Literal((_dt.date.fromisoformat(e.value) - EPOCH).days,
                               dtypes.date32)
            return Cast(e, dtypes.date32)
        if tn == "timestamp":
            from ..exprs import _parse_ts_micros

            if isinstance(e, Literal) and isinstance(e.value, str):
                return Literal(_parse_ts_micros(e.value), dtypes.timestamp)
            return Cast(e, dtypes.timestamp)
        if tn.startswith("decimal"):
            import re as _re

            m = _re.match(r"decimal\((\d+),(\d+)\)", tn)
            p, s = (int(m.group(1)), int(m.group(2))) if m else (18, 2)
            return Cast(e, dtypes.decimal64(min(p, 18), s))
        if tn in ("int", "integer"):
            return Cast(e, dtypes.int32)
        if tn in ("bigint", "long"):
            return Cast(e, dtypes.int64)
        if tn in ("double", "float"):
            return Cast(e, dtypes.float64)
        if tn in ("string", "varchar", "char"):
            return Cast(e, dtypes.string)
        raise SqlError(f"unsupported cast target {tn}")

    def _func(self, fc: A.FuncCall, scope, ctx, alias_env) -> Expr:
        name = fc.name
        conv = lambda n: self.to_expr(n, scope, ctx, alias_env)
        if fc.over is not None:
            raise SqlError(f"window {name}() outside window planning")
        if name in AGG_NAMES:
            if ctx is None or not ctx.active:
                raise SqlError(f"aggregate {name}() outside GROUP BY context")
            entry = ctx.agg_map.get(_agg_ast_key(fc))
            if entry is None:
                raise SqlError(f"aggregate {name}() was not collected")
            if isinstance(entry, tuple) and entry[0] == "percentile":
                _, en, p = entry
                return F.ListQuantile(col(en), p)
            if isinstance(entry, tuple) and entry[0] == "covar":
                from ..exprs import Sqrt

                _, vname, sx, sy, sxy, sxx, syy, cn = entry
                n_f = Cast(col(cn), dtypes.float64)
                cxy = col(sxy) - col(sx) * col(sy) / n_f
                if vname == "covar_pop":
                    return CaseWhen([(col(cn) > 0, cxy / n_f)],
                                    Literal(None, dtypes.float64))
                if vname == "covar_samp":
                    return CaseWhen([(col(cn) > 1, cxy / (n_f - lit(1.0)))],
                                    Literal(None, dtypes.float64))
                vx = col(sxx) - col(sx) * col(sx) / n_f
                vy = col(syy) - col(sy) * col(sy) / n_f
                return CaseWhen([(col(cn) > 0, cxy / Sqrt(vx * vy))],
                                Literal(None, dtypes.float64))
            if isinstance(entry, tuple) and entry[0] == "stddev":
                from ..exprs import Sqrt

                _, s1, s2, cn, sqrt, pop = entry
                n_f = Cast(col(cn), dtypes.float64)
                denom = n_f if pop else (n_f - lit(1.0))
                var = (col(s2) - col(s1) * col(s1) / n_f) / denom
                out = Sqrt(var) if sqrt else var
                min_n = 0 if pop else 1
                return CaseWhen([(col(cn) > min_n, out)],
                                Literal(None, dtypes.float64))
            return col(entry)
        if name == "grouping":
            if ctx is None or not ctx.grp_names:
                raise SqlError("grouping() outside GROUPING SETS context")
            arg = self.to_expr(fc.args[0], scope, None)
            if not isinstance(arg, Col) or arg.name not in ctx.grp_names:
                # arg may itself be a group key expr replaced by its name
                k = expr_key(arg)
                kn = ctx.key_map.get(k)
                if kn is None or kn not in ctx.grp_names:
                    raise SqlError("grouping() argument is not a group key")
                return col(ctx.grp_names[kn])
            return col(ctx.grp_names[arg.name])
        if name == "substr":
            e = conv(fc.args[0])
            start = _int_lit(conv(fc.args[1]))
            length = _int_lit(conv(fc.args[2])) if len(fc.args) > 2 else 1 << 30
            return Substr(e, start, length)
        if name == "coalesce":
            args = [conv(a) for a in fc.args]
            dt = next((infer_dtype(a, self.alloc.types) for a in args
                       if not _is_null_lit(a)
                       and infer_dtype(a, self.alloc.types) is not None), None)
            if dt is not None:
                args = [Literal(None, dt) if _is_null_lit(a) else a for a in args]
            return Coalesce(args)
        if name == "concat":
            from ..exprs import ConcatStr

            return ConcatStr([conv(a) for a in fc.args])
        if name == "round":
            e = conv(fc.args[0])
            nd = _int_lit(conv(fc.args[1])) if len(fc.args) > 1 else 0
            return F.Round(e, nd)
        if name == "abs":
            from ..exprs import Abs

            return Abs(conv(fc.args[0]))
        if name == "sqrt":
            from ..exprs import Sqrt

            return Sqrt(conv(fc.args[0]))
        if name in ("upper", "ucase"):
            from ..exprs import Upper

            return Upper(conv(fc.args[0]))
        if name in ("lower", "lcase"):
            from ..exprs import Lower

            return Lower(conv(fc.args[0]))
        if name in ("char_length", "length"):
            from ..exprs import Length

            return Length(conv(fc.args[0]))
        if name == "nullif":
            return F.NullIf(conv(fc.args[0]), conv(fc.args[1]))
        if name == "year":
            from ..exprs import DatePart

            return DatePart("year", conv(fc.args[0]))
        if name == "month":
            from ..exprs import DatePart

            return DatePart("month", conv(fc.args[0]))
        if name == "trim":
            return F.Trim(conv(fc.args[0]))
        out = self._fn_extended(name, fc, conv)
        if out is not None:
            return out
        raise SqlError(f"unsupported function {name}")

    def _fn_extended(self, name, fc, conv):
        """Breadth routes beyond the TPC-DS core set (spark_dates.rs /
        spark_strings.rs / spark math function names)."""
        from ..exprs import Cast

        a = fc.args
        one = {"floor": F.Floor, "ceil": F.Ceil, "ceiling": F.Ceil,
               "exp": F.Exp, "ln": F.Ln, "log10": F.Log10,
               "sign": F.Sign, "signum": F.Sign, "isnan": F.IsNan,
               "reverse": F.Reverse, "ascii": F.Ascii, "initcap": F.InitCap,
               "quarter": F.Quarter, "dayofweek": F.DayOfWeek,
               "weekofyear": F.WeekOfYear, "last_day": F.LastDay,
               "hour": F.Hour, "minute": F.Minute, "second": F.Second,
               "unix_timestamp": F.UnixTimestamp,
               "from_unixtime": F.FromUnixtime,
               "to_timestamp": F.ToTimestamp, "to_date": F.ToDate,
               "size": F.ArraySize, "cardinality": F.ArraySize}
        if name in one:
            return one[name](conv(a[0]))
        if name in ("day", "dayofmonth"):
            from ..exprs import DatePart

            return DatePart("day", conv(a[0]))
        if name in ("pow", "power"):
            return F.Pow(conv(a[0]), conv(a[1]))
        if name == "greatest":
            return F.Greatest([conv(x) for x in a])
        if name == "least":
            return F.Least([conv(x) for x in a])
        if name in ("nvl", "ifnull"):
            from ..exprs import Coalesce

            return Coalesce([conv(a[0]), conv(a[1])])
        if name == "nvl2":
            return F.Nvl2(conv(a[0]), conv(a[1]), conv(a[2]))
        if name == "if":
            return F.If(conv(a[0]), conv(a[1]), conv(a[2]))
        if name == "date_add":
            return F.DateAdd(conv(a[0]), conv(a[1]))
        if name == "date_sub":
            return F.DateSub(conv(a[0]), conv(a[1]))
        if name == "datediff":
            return F.DateDiff(conv(a[0]), conv(a[1]))
        if name == "add_months":
            return F.AddMonths(conv(a[0]), _int_lit(conv(a[1])))
        if name == "months_between":
            return F.MonthsBetween(conv(a[0]), conv(a[1]))
        if name == "next_day":
            return F.NextDay(conv(a[0]), _str_lit(conv(a[1])))
        if name == "trunc":
            return F.TruncDate(conv(a[0]), _str_lit(conv(a[1])))
        if name == "date_trunc":
            return F.TruncTimestamp(_str_lit(conv(a[0])), conv(a[1]))
        if name == "date_format":
            return F.DateFormat(conv(a[0]), _str_lit(conv(a[1])))
        if name == "replace":
            return F.Replace(conv(a[0]), _str_lit(conv(a[1])),
                             _str_lit(conv(a[2])) if len(a) > 2 else "")
        if name == "lpad":
            return F.LPad(conv(a[0]), _int_lit(conv(a[1])),
                          _str_lit(conv(a[2])) if len(a) > 2 else " ")
        if name == "rpad":
            return F.RPad(conv(a[0]), _int_lit(conv(a[1])),
                          _str_lit(conv(a[2])) if len(a) > 2 else " ")
        if name == "ltrim":
            return F.Trim(conv(a[0]), mode="leading")
        if name == "rtrim":
            return F.Trim(conv(a[0]), mode="trailing")
        if name == "left":
            return F.Left(conv(a[0]), _int_lit(conv(a[1])))
        if name == "right":
            return F.Right(conv(a[0]), _int_lit(conv(a[1])))
        if name == "repeat":
            return F.Repeat(conv(a[0]), _int_lit(conv(a[1])))  # scalar n
        if name == "space":
            return F.Space(conv(a[0]))
        if name == "translate":
            return F.Translate(conv(a[0]), _str_lit(conv(a[1])),
                               _str_lit(conv(a[2])))
        if name == "find_in_set":
            return F.FindInSet(conv(a[0]), conv(a[1]))
        if name == "split_part":
            return F.SplitPart(conv(a[0]), _str_lit(conv(a[1])),
                               _int_lit(conv(a[2])))
        if name in ("instr", "position"):
            return F.Instr(conv(a[0]), _str_lit(conv(a[1])))
        if name == "locate":
            return F.Instr(conv(a[1]), _str_lit(conv(a[0])))
        if name == "bround":
            return F.Bround(conv(a[0]),
                            _int_lit(conv(a[1])) if len(a) > 1 else 0)
        if name == "concat_ws":
            return F.ConcatWs(_str_lit(conv(a[0])), [conv(x) for x in a[1:]])
        if name == "get_json_object":
            return F.GetJsonObject(conv(a[0]), _str_lit(conv(a[1])))
        if name == "element_at":
            return F.ElementAt(conv(a[0]), _int_lit(conv(a[1])))
        if name == "array":
            return F.MakeArray([conv(x) for x in a])
        if name in ("double", "float"):
            return Cast(conv(a[0]), dtypes.float64)
        if name in ("int", "bigint"):
            return Cast(conv(a[0]), dtypes.int64 if name == "bigint" else dtypes.int32)
        if name == "string":
            return Cast(conv(a[0]), dtypes.string)
        if name == "xxhash64":
            return F.XxHash64([conv(x) for x in a])
        from ..functions import _UNARY_MATH

        if name in _UNARY_MATH:
            return F.UnaryMath(name, conv(a[0]))
        if name == "atan2":
            return F.Atan2(conv(a[0]), conv(a[1]))
        if name == "pi":
            from ..exprs import Literal as _L

            return _L(3.141592653589793)
        if name == "log":
            # log(x) = ln; log(base, x) = ln(x)/ln(base)
            if len(a) == 1:
                return F.Ln(conv(a[0]))
            from ..exprs import Arith as _Ar

            return _Ar("/", F.Ln(conv(a[1])), F.Ln(conv(a[0])))
        if name == "array_contains":
            from ..exprs import Literal as _L2

            v = conv(a[1])
            assert isinstance(v, _L2), "array_contains needs a literal value"
            return F.ArrayContains(conv(a[0]), v.value)
        if name == "substring_index":
            return F.SubstringIndex(conv(a[0]), _str_lit(conv(a[1])),
                                    _int_lit(conv(a[2])))
        if name == "levenshtein":
            return F.Levenshtein(conv(a[0]), conv(a[1]))
        if name == "regexp_extract":
            g = _int_lit(conv(a[2])) if len(a) > 2 else 1
            return F.RegexpExtract(conv(a[0]), _str_lit(conv(a[1])), g)
        if name == "regexp_replace":
            return F.RegexpReplace(conv(a[0]), _str_lit(conv(a[1])),
                                   _str_lit(conv(a[2])))
        if name == "rlike":
            return F.RLike(conv(a[0]), _str_lit(conv(a[1])))
        if name == "rand":
            return F.Rand(_int_lit(conv(a[0])) if a else 42)
        if name == "randn":
            return F.Randn(_int_lit(conv(a[0])) if a else 42)
        if name == "startswith":
            from ..exprs import StringStartsWith

            return StringStartsWith(conv(a[0]), _str_lit(conv(a[1])))
        if name == "endswith":
            from ..exprs import StringEndsWith

            return StringEndsWith(conv(a[0]), _str_lit(conv(a[1])))
        if name == "contains":
            from ..exprs import StringContains

            return StringContains(conv(a[0]), _str_lit(conv(a[1])))
        if name == "unscaled_value":
            return F.UnscaledValue(conv(a[0]))
        if name == "make_decimal":
            p = _int_lit(conv(a[1])) if len(a) > 1 else 18
            sc = _int_lit(conv(a[2])) if len(a) > 2 else 2
            return F.MakeDecimal(conv(a[0]), p, sc)
        return None

    # --------------------------------------------- scalar subqueries
    def _scalar_subquery(self, q: A.Query, scope: Scope, ctx) -> Expr:
        """Uncorrelated scalar subquery -> execute-at-lowering placeholder.
        Correlated ones raise here (outer=None leaves their outer columns
        unresolved) and are decorrelated by _rewrite_scalar_subqueries."""
        lq = self.plan_query(q, None)
        self.sc_n += 1
        return ScalarSubqueryLit(lq)

    def plan_scalar_agg_subquery(self, q: A.Query, outer_scope: Scope):
        """Decorrelate `(SELECT agg_expr FROM ... WHERE corr_conds)`:
        returns (sub_rel, outer_keys, inner_key_names, value_expr_name)."""
        body = q.body
        assert isinstance(body, A.Select)
        sub_units: List[Rel] = []
        conds: List[A.ANode] = []

        def add_item(item):
            if isinstance(item, A.Join) and item.kind == "inner":
                add_item(item.left)
                add_item(item.right)
                if item.on is not None:
                    conds.extend(_conjuncts(item.on))
            else:
                sub_units.append(self.plan_table(item, outer_scope))

        for it in body.from_:
            add_item(it)
        if body.where is not None:
            conds.extend(_conjuncts(body.where))
        sub_scope = Scope(sub_units, outer_scope)
        inner_cols = {c.engine for u in sub_units for c in u.cols}
        inner_conds: List[A.ANode] = []
        okeys: List[Expr] = []
        ikeys: List[Expr] = []
        for cnd in conds:
            conv = self.to_expr(cnd, sub_scope, None)
            refs = expr_cols(conv)
            if refs <= inner_cols:
                inner_conds.append(cnd)
                continue
            pair = _equi_pair_io(cnd, sub_scope, self, inner_cols)
            if pair is None:
                raise SqlError("non-equi correlation in scalar subquery")
            okeys.append(pair[0])
            ikeys.append(pair[1])
        sub_rel = self.join_graph(sub_units, inner_conds, sub_scope)
        # inner group keys: materialize ikeys as named cols
        key_items: List[Tuple[Expr, str]] = []
        for ik in ikeys:
            if isinstance(ik, Col):
                key_items.append((ik, ik.name))
            else:
                en = self.alloc.fresh("__ckey", infer_dtype(ik, self.alloc.types))
                key_items.append((ik, en))
        # aggregate select item
        ctx = AggCtx()
        ctx.key_items = key_items
        ctx.key_map = {expr_key(e): n for e, n in key_items}
        ctx.nkeys = len(key_items)
        sscope = Scope([sub_rel], outer_scope)
        if len(body.items) != 1:
            raise SqlError("scalar subquery must select one item")
        _collect_aggs(body.items[0].expr, self, sscope, ctx)
        ctx.active = True
        val_expr = self.to_expr(body.items[0].expr, sscope, ctx)
        agg = LAgg(sub_rel.node, key_items, ctx.aggs)
        vn = self.alloc.fresh("__scval", infer_dtype(val_expr, self.alloc.types))
        proj = LProject(agg, [(col(n), n) for _, n in key_items] + [(val_expr, vn)])
        cols = [RelCol(None, n, n, self.alloc.types.get(n)) for _, n in key_items]
        cols.append(RelCol(None, vn, vn, self.alloc.types.get(vn)))
        rel = Rel(proj, cols, max(sub_rel.est / 10, 1.0), sub_rel.base_dim_only)
        return rel, okeys, [n for _, n in key_items], vn


# ------------------------------------------------------------- AST helpers
def _conjuncts(node: A.ANode) -> List[A.ANode]:
    if isinstance(node, A.BinOp) and node.op == "and":
        return _conjuncts(node.left) + _conjuncts(node.right)
    if isinstance(node, A.BinOp) and node.op == "or":
        # factor common conjuncts out of a disjunction:
        # (A AND B1) OR (A AND B2) -> A AND (B1 OR B2)  (q41's correlated
        # predicate hides its equi-correlation inside such an OR)
        lcs = _conjuncts(node.left)
        rcs = _conjuncts(node.right)
        common = [c for c in lcs if any(c == r for r in rcs)]
        if common:
            lrest = [c for c in lcs if not any(c == x for x in common)]
            rrest = [c for c in rcs if not any(c == x for x in common)]

            def _and(cs):
                if not cs:
                    return None
                e = cs[0]
                for c in cs[1:]:
                    e = A.BinOp("and", e, c)
                return e

            out = list(common)
            le, re_ = _and(lrest), _and(rrest)
            if le is not None and re_ is not None:
                out.append(A.BinOp("or", le, re_))
            return out
    return [node]


AGG_NAMES = {"sum", "avg", "min", "max", "count", "stddev_samp",
             "stddev", "stddev_pop", "var_samp", "variance", "var_pop",
             "corr", "covar_samp", "covar_pop", "approx_count_distinct",
             "percentile", "percentile_approx", "median"}


def _has_agg(sel: A.Select) -> bool:
    def rec(n) -> bool:
        if isinstance(n, A.FuncCall):
            if n.over is not None:
                # window fn: its args/spec may still contain plain aggs
                return any(rec(a) for a in n.args) or \
                    any(rec(p) for p in n.over.partition_by) or \
                    any(rec(oi.expr) for oi in n.over.order_by)
            if n.name in AGG_NAMES or n.name == "grouping":
                return True
            return any(rec(a) for a in n.args)
        if isinstance(n, (A.ScalarSubquery, A.Exists, A.InSubquery)):
            return False  # separate query scope
        if dataclasses.is_dataclass(n):
            for f in dataclasses.fields(n):
                v = getattr(n, f.name)
                if isinstance(v, A.ANode) and rec(v):
                    return True
                if isinstance(v, list) and any(isinstance(x, A.ANode) and rec(x) for x in v):
                    return True
                if isinstance(v, tuple) and any(isinstance(x, A.ANode) and rec(x) for x in v):
                    return True
        return False

    return any(rec(it.expr) for it in sel.items) or \
        (sel.having is not None and rec(sel.having))


def _contains_scalar_subquery(n) -> bool:
    if isinstance(n, A.ScalarSubquery):
        return True
    if isinstance(n, (A.Exists, A.InSubquery)):
        return False
    if dataclasses.is_dataclass(n):
        for f in dataclasses.fields(n):
            v = getattr(n, f.name)
            if isinstance(v, A.ANode) and _contains_scalar_subquery(v):
                return True
            if isinstance(v, (list, tuple)):
                for x in v:
                    if isinstance(x, A.ANode) and _contains_scalar_subquery(x):
                        return True
    return False


def _contains_subquery(n) -> bool:
    if isinstance(n, (A.Exists, A.InSubquery)):
        return True
    if isinstance(n, A.ScalarSubquery):
        return False
    if dataclasses.is_dataclass(n):
        for f in dataclasses.fields(n):
            v = getattr(n, f.name)
            if isinstance(v, A.ANode) and _contains_subquery(v):
                return True
            if isinstance(v, (list, tuple)):
                for x in v:
                    if isinstance(x, A.ANode) and _contains_subquery(x):
                        return True
    return False


def _auto_name(node: A.ANode) -> str:
    """Spark-style auto-name for an unaliased select item: a compact
    lower-cased rendering of the expression text (NativeConverters keeps
    Catalyst's generated names; this mirrors that convention)."""
    r = _render
    return r(node)


def _render(n) -> str:
    if isinstance(n, A.Num):
        return n.text
    if isinstance(n, A.Str):
        return n.value
    if isinstance(n, A.Null):
        return "null"
    if isinstance(n, A.Ident):
        return n.name.lower()
    if isinstance(n, A.FuncCall):
        if n.star:
            return f"{n.name}(1)" if n.name == "count" else f"{n.name}(*)"
        inner = ", ".join(_render(a) for a in n.args)
        if n.distinct:
            inner = f"distinct {inner}"
        base = f"{n.name}({inner})"
        if n.over is not None:
            return base + " over (...)"
        return base
    if isinstance(n, A.BinOp):
        return f"({_render(n.left)} {n.op} {_render(n.right)})"
    if isinstance(n, A.UnOp):
        return f"({n.op} {_render(n.operand)})"
    if isinstance(n, A.Case):
        return "case when ... end"
    if isinstance(n, A.CastE):
        return f"cast({_render(n.operand)} as {n.typename})"
    if isinstance(n, A.ScalarSubquery):
        return "scalarsubquery()"
    if isinstance(n, A.Interval):
        return f"interval {n.n} day"
    return type(n).__name__.lower()


def _fold_const(e: Expr) -> Expr:
    """Fold literal-only arithmetic (e.g. `1200 + 11` in BETWEEN/IN)."""
    if type(e).__name__ == "Arith":
        l = _fold_const(e.left)
        r = _fold_const(e.right)
        if isinstance(l, Literal) and isinstance(r, Literal) \
                and isinstance(l.value, (int, float)) \
                and isinstance(r.value, (int, float)):
            import operator as _op

            f = {"+": _op.add, "-": _op.sub, "*": _op.mul,
                 "/": _op.truediv}.get(e.op)
            if f is not None:
                v = f(l.value, r.value)
                dt = l.dtype if l.dtype is not None and l.dtype.code == dtypes.DATE32 else None
                return Literal(v, dt)
    return e


def _is_null_lit(e: Expr) -> bool:
    return isinstance(e, Literal) and e.value is None


def _str_lit(e: Expr) -> str:
    if isinstance(e, Literal) and isinstance(e.value, str):
        return e.value
    raise SqlError("expected string literal argument")


def _int_lit(e: Expr) -> int:
    if isinstance(e, Literal) and isinstance(e.value, (int, float)):
        return int(e.value)
    raise SqlError("expected integer literal argument")




def _count_cte_uses(q: A.Query, name: str) -> int:
    count = 0

    def rec(n):
        nonlocal count
        if isinstance(n, A.Table) and n.name == name:
            count += 1
            return
        if isinstance(n, A.Query):
            # a redefinition in a nested WITH shadows; TPC-DS doesn't do it
            rec(n.body)
            for _, sub in n.ctes:
                rec(sub)
            for oi in n.order_by:
                rec(oi.expr)
            return
        if dataclasses.is_dataclass(n):
            for f in dataclasses.fields(n):
                v = getattr(n, f.name)
                if isinstance(v, A.ANode):
                    rec(v)
                elif isinstance(v, (list, tuple)):
                    for x in v:
                        if isinstance(x, A.ANode):
                            rec(x)
                        elif isinstance(x, tuple):
                            for y in x:
                                if isinstance(y, A.ANode):
                                    rec(y)

    rec(q.body)
    for _, sub in q.ctes:
        if sub is not None:
            rec(sub)
    for oi in q.order_by:
        rec(oi.expr)
    return count


def _equi_pair(cnd, scope, planner, lcols: Set[str], rcols: Set[str]):
    """cnd is `a = b` with a from lcols side, b from rcols side?"""
    if not (isinstance(cnd, A.BinOp) and cnd.op == "="):
        return None
    le = planner.to_expr(cnd.left, scope, None)
    re_ = planner.to_expr(cnd.right, scope, None)
    lr, rr = expr_cols(le), expr_cols(re_)
    if lr and rr:
        if lr <= lcols and rr <= rcols:
            return le, re_
        if lr <= rcols and rr <= lcols:
            return re_, le
    return None


def _equi_pair_units(cnd, scope, planner, unit_of: Dict[str, int]):
    if not (isinstance(cnd, A.BinOp) and cnd.op == "="):
        return None
    le = planner.to_expr(cnd.left, scope, None)
    re_ = planner.to_expr(cnd.right, scope, None)
    lr, rr = expr_cols(le), expr_cols(re_)
    lu = {unit_of[c] for c in lr if c in unit_of}
    ru = {unit_of[c] for c in rr if c in unit_of}
    if len(lu) == 1 and len(ru) == 1 and lu != ru:
        return next(iter(lu)), le, next(iter(ru)), re_
    return None


def _equi_pair_io(cnd, sub_scope, planner, inner_cols: Set[str]):
    """`outer_expr = inner_expr` (either order) -> (outer, inner)."""
    if not (isinstance(cnd, A.BinOp) and cnd.op == "="):
        return None
    le = planner.to_expr(cnd.left, sub_scope, None)
    re_ = planner.to_expr(cnd.right, sub_scope, None)
    lr, rr = expr_cols(le), expr_cols(re_)
    if lr and rr:
        if lr <= inner_cols and not (rr & inner_cols):
            return re_, le
        if rr <= inner_cols and not (lr & inner_cols):
            return le, re_
    return None


def _push_filter(node: LNode, pred: Expr) -> LNode:
    if isinstance(node, LScan):
        node.filters.append(pred)
        return node
    return LFilter(node, pred)


def _selectivity(pred: Expr) -> float:
    name = type(pred).__name__
    if name == "Cmp":
        return 0.1 if getattr(pred, "op", "") == "==" else 0.35
    if name == "InList":
        return 0.2
    if name == "Like":
        return 0.3
    if name == "BoolOp":
        return 0.4
    return 0.5


# ------------------------------------------------- aggregate collection
def _collect_aggs(node: A.ANode, planner: Planner, scope: Scope, ctx: AggCtx):
    """Find aggregate FuncCalls (not inside OVER), register AggSpecs."""

    def rec(n):
        if isinstance(n, A.FuncCall):
            if n.over is not None:
                for a in n.args:
                    rec(a)
                for p in n.over.partition_by:
                    rec(p)
                for oi in n.over.order_by:
                    rec(oi.expr)
                return
            if n.name in AGG_NAMES:
                _register_agg(n, planner, scope, ctx)
                return
            for a in n.args:
                rec(a)
            return
        if isinstance(n, (A.ScalarSubquery, A.Exists, A.InSubquery)):
            return
        if dataclasses.is_dataclass(n):
            for f in dataclasses.fields(n):
                v = getattr(n, f.name)
                if isinstance(v, A.ANode):
                    rec(v)
                elif isinstance(v, (list, tuple)):
                    for x in v:
                        if isinstance(x, A.ANode):
                            rec(x)
                        elif isinstance(x, tuple):
                            for y in x:
                                if isinstance(y, A.ANode):
                                    rec(y)

    rec(node)


def _register_agg(fc: A.FuncCall, planner: Planner, scope: Scope, ctx: AggCtx):
    key = _agg_ast_key(fc)
    if key in ctx.agg_map:
        return
    name = fc.name
    alloc = planner.alloc
    if name in ("percentile", "percentile_approx", "median"):
        # exact interpolated percentile over a collected list (Spark
        # percentile semantics; the approx variant is implemented exactly)
        arg = planner.to_expr(fc.args[0], scope, None)
        if name == "median":
            p = 0.5
        else:
            pe = planner.to_expr(fc.args[1], scope, None)
            if not isinstance(pe, Literal):
                raise SqlError("percentile fraction must be a literal")
            p = float(pe.value)
        en = alloc.fresh("__pct", dtypes.list_of(dtypes.float64))
        ctx.aggs.append(AggFunc("collect_list",
                                Cast(arg, dtypes.float64), name=en))
        ctx.agg_map[key] = ("percentile", en, p)
        return
    if name == "approx_count_distinct":
        # exact distinct count is a valid (and here cheap) implementation
        arg = planner.to_expr(fc.args[0], scope, None)
        en = alloc.fresh("__acntd", dtypes.int64)
        ctx.aggs.append(AggFunc("count_distinct", arg, distinct=True, name=en))
        ctx.agg_map[key] = en
        return
    if name in ("corr", "covar_samp", "covar_pop"):
        # two-argument moments: covar = (Σxy - ΣxΣy/n) / (n-1 | n);
        # corr = covar_pop / (stddev_pop(x) * stddev_pop(y))
        x = planner.to_expr(fc.args[0], scope, None)
        y = planner.to_expr(fc.args[1], scope, None)
        fx = Cast(x, dtypes.float64)
        fy = Cast(y, dtypes.float64)
        names = [alloc.fresh(f"__cv_{i}", dtypes.float64) for i in range(5)]
        cn = alloc.fresh("__cv_n", dtypes.int64)
        both = CaseWhen([(Not(IsNull(x)) & Not(IsNull(y)), lit(1))], lit(None))
        for nm, e in zip(names, [fx, fy, fx * fy, fx * fx, fy * fy]):
            # null in EITHER argument drops the pair (SQL semantics)
            ctx.aggs.append(AggFunc(
                "sum", CaseWhen([(Not(IsNull(x)) & Not(IsNull(y)), e)],
                                Literal(None, dtypes.float64)), name=nm))
        ctx.aggs.append(AggFunc("count", both, name=cn))
        ctx.agg_map[key] = ("covar", name, *names, cn)
        return
    if name in ("stddev_samp", "stddev", "stddev_pop", "var_samp",
                "variance", "var_pop"):
        # decomposed moments: var = (sum_sq - sum^2/n) / (n-1 | n);
        # registered as three plain aggs, the replacement in to_expr
        # synthesizes sqrt()/plain per the variant
        arg = planner.to_expr(fc.args[0], scope, None)
        s1 = alloc.fresh("__sd_s1", dtypes.float64)
        s2 = alloc.fresh("__sd_s2", dtypes.float64)
        cn = alloc.fresh("__sd_n", dtypes.int64)
        f = Cast(arg, dtypes.float64)
        ctx.aggs.append(AggFunc("sum", f, name=s1))
        ctx.aggs.append(AggFunc("sum", f * f, name=s2))
        ctx.aggs.append(AggFunc("count", arg, name=cn))
        sqrt = name in ("stddev_samp", "stddev", "stddev_pop")
        pop = name in ("stddev_pop", "var_pop")
        ctx.agg_map[key] = ("stddev", s1, s2, cn, sqrt, pop)
        return
    if fc.star or (name == "count" and not fc.args):
        en = alloc.fresh("__cnt", dtypes.int64)
        ctx.aggs.append(AggFunc("count_star", None, name=en))
        ctx.agg_map[key] = en
        return
    arg = planner.to_expr(fc.args[0], scope, None)
    if name == "count" and fc.distinct:
        en = alloc.fresh("__cntd", dtypes.int64)
        ctx.aggs.append(AggFunc("count_distinct", arg, distinct=True, name=en))
        ctx.agg_map[key] = en
        return
    dt = infer_dtype(arg, alloc.types)
    if name == "count":
        out_dt = dtypes.int64
    elif name == "avg":
        out_dt = dtypes.float64
    elif name in ("min", "max"):
        out_dt = dt
    else:  # sum
        if dt is None:
            out_dt = None
        elif dt.code == dtypes.DECIMAL64:
            out_dt = dtypes.decimal64(18, dt.scale)
        elif dt.code in (dtypes.FLOAT32, dtypes.FLOAT64):
            out_dt = dtypes.float64
        else:
            out_dt = dtypes.int64
    en = alloc.fresh(f"__{name}", out_dt)
    ctx.aggs.append(AggFunc(name, arg, distinct=fc.distinct, name=en))
    ctx.agg_map[key] = en


def _agg_ast_key(fc: A.FuncCall):
    def k(n):
        if isinstance(n, A.Ident):
            return ("id", tuple(p.lower() for p in n.parts[-1:]))
        if isinstance(n, A.Num):
            return ("num", n.text)
        if isinstance(n, A.Str):
            return ("str", n.value)
        if dataclasses.is_dataclass(n):
            vals = []
            for f in dataclasses.fields(n):
                v = getattr(n, f.name)
                if isinstance(v, A.ANode):
                    vals.append(k(v))
                elif isinstance(v, (list, tuple)):
                    vals.append(tuple(k(x) if isinstance(x, A.ANode) else repr(x) for x in v))
                else:
                    vals.append(repr(v))
            return (type(n).__name__, tuple(vals))
        return repr(n)

    return k(fc)


def _extract_windows(node: A.ANode, out: List[Tuple[A.FuncCall, str]],
                     alloc: NameAlloc):
    """Replace window FuncCalls with Ident placeholders (bottom-up)."""
    if isinstance(node, A.FuncCall) and node.over is not None:
        name = alloc.fresh(f"__win_{node.name}", None)
        out.append((node, name))
        return A.Ident([name])
    if isinstance(node, (A.ScalarSubquery, A.Exists, A.InSubquery)):
        return node
    if dataclasses.is_dataclass(node) and isinstance(node, A.ANode):
        kw = {}
        for f in dataclasses.fields(node):
            v = getattr(node, f.name)
            if isinstance(v, A.ANode):
                kw[f.name] = _extract_windows(v, out, alloc)
            elif isinstance(v, list):
                kw[f.name] = [_extract_windows(x, out, alloc) if isinstance(x, A.ANode) else x
                              for x in v]
            elif isinstance(v, tuple):
                kw[f.name] = tuple(_extract_windows(x, out, alloc) if isinstance(x, A.ANode) else x
                                   for x in v)
            else:
                kw[f.name] = v
        return type(node)(**kw)
    return node


def _window_dtype(fc: A.FuncCall, wf: WindowFunc, types) -> Optional[DataType]:
    if wf.fn in ("rank", "dense_rank", "row_number", "ntile", "count"):
        return dtypes.int64
    if wf.fn in ("percent_rank", "cume_dist", "avg"):
        return dtypes.float64
    if wf.arg is not None:
        return infer_dtype(wf.arg, types)
    return None


def plan_query(sql_text: str,
               schemas: Optional[Dict[str, Dict[str, DataType]]] = None):
    """Parse + plan; returns (Planner, LQuery)."""
    ast_q = parse_sql(sql_text)
    p = Planner(schemas)
    lq = p.plan(ast_q)
    return p, lq

"""Logical -> physical convert strategy.

Role parity: AuronConverters.convertSparkPlanRecursively + the Exchange
placement Spark's EnsureRequirements would have done
(/root/reference/spark-extension/src/main/scala/org/apache/spark/sql/auron/
AuronConverters.scala:97, NativeShuffleExchangeBase/NativeBroadcastExchangeBase).
Decides BHJ (broadcast build) vs SHJ (hash exchange both sides), inserts
two-phase aggregation, materializes CTEs used more than once and
uncorrelated scalar subqueries at lowering time (the AQE pattern), and
tracks hash-partitioning properties to elide redundant exchanges.

SPMD caution: every physical decision here must be a pure function of the
logical plan (identical on every rank) — basing a broadcast decision on a
rank-local row count would deadlock the collective schedule.
"""
from __future__ import annotations

from typing import Dict, List, Optional, Set, Tuple

from .. import dtypes
from ..exprs import Aliased, Col, Expr, Literal, col, lit
from ..plan import nodes as P
from .parser import SqlError, parse_sql
from .planner import (BROADCAST_ROWS, LAgg, LCTERef, LFilter, LJoin, LLimit,
                      LNode, LProject, LQuery, LReplicate, LScan, LSort,
                      LUnionAll, LWindow, MatCTE, Planner, ScalarSubqueryLit,
                      expr_cols, expr_key, infer_dtype, walk_exprs)

ANY = ("any",)
SINGLE = ("single",)


def _hash_part(keys: List[Expr]):
    return ("hash", tuple(keys))


def _part_eq(a, b) -> bool:
    if a[0] != b[0]:
        return False
    if a[0] != "hash":
        return True
    if len(a[1]) != len(b[1]):
        return False
    return all(expr_key(x) == expr_key(y) for x, y in zip(a[1], b[1]))


class Lowering:
    def __init__(self, planner: Planner, cat, session):
        self.planner = planner
        self.cat = cat
        self.session = session

    # ------------------------------------------------------------ helpers
    def subst_scalars(self, e: Expr) -> Expr:
        def fn(node):
            if isinstance(node, ScalarSubqueryLit):
                return self._exec_scalar(node.lquery)
            return node

        return walk_exprs(e, fn)

    def _exec_scalar(self, lq: LQuery) -> Literal:
        # scalar-subquery plans are built after the main prune pass ran:
        # prune here or the first subquery scan decodes EVERY column of
        # its fact table (q9 at SF=100: 23 store_sales columns, ~20 s of
        # host-side launch storm)
        prune_lquery(lq)
        plan = self.lower_lquery(lq)
        batch = self.session.collect_all(plan)
        if batch.num_rows == 0:
            c = batch.columns[0]
            return Literal(None, c.dtype)
        c = batch.columns[0]
        vals = c.to_pylist()
        return Literal(vals[0], c.dtype if c.dtype.code != dtypes.DECIMAL64
                       else None)

    def materialize_ctes(self):
        for cte in self.planner.mat_ctes:
            if cte.batches is not None:
                continue
            plan = self.lower_lquery(cte.lquery)
            cte.batches = self.session.execute(plan)

    # ------------------------------------------------------------- entry
    def lower_lquery(self, lq: LQuery) -> P.PlanNode:
        """Outputs projected; ORDER/LIMIT via single exchange (rows land on
        rank 0, the reference's collect contract)."""
        plan, part = self.lower(lq.rel.node)
        if lq.order:
            keys = [(self.subst_scalars(e), asc) for e, asc in lq.order]
            ex = P.Exchange(plan, "single")
            plan = P.Sort(ex, keys, limit=lq.limit)
            if lq.limit is not None:
                plan = P.Limit(plan, lq.limit)
        elif lq.limit is not None:
            plan = P.Limit(P.Exchange(plan, "single"), lq.limit)
        # duplicate output names (legal SQL, e.g. q39's self-join selecting
        # both sides' w_warehouse_sk) get positional suffixes so the
        # columnar batch stays addressable by name
        seen: Dict[str, int] = {}
        items = []
        for e, n in lq.outputs:
            k = seen.get(n, 0)
            seen[n] = k + 1
            items.append(Aliased(self.subst_scalars(e),
                                 n if k == 0 else f"{n}__{k + 1}"))
        return P.Project(plan, items)

    # ---------------------------------------------------------- dispatch
    def lower(self, node: LNode) -> Tuple[P.PlanNode, tuple]:
        m = getattr(self, f"_lower_{type(node).__name__}")
        return m(node)

    def _lower_LScan(self, node: LScan):
        origs = list(node.colmap.keys())
        plan: P.PlanNode = self.cat.scan(node.table, columns=origs)
        engines = [node.colmap[o] for o in origs]
        if engines != origs:
            plan = P.RenameColumns(plan, engines)
        for f in node.filters:
            plan = P.Filter(plan, self.subst_scalars(f))
        return plan, ANY

    def _lower_LCTERef(self, node: LCTERef):
        cte = node.cte
        assert cte.batches is not None, f"CTE {cte.name} not materialized"
        scan = P.MemoryScan(cte.batches)
        out_names = [n for _, n in cte.lquery.outputs]
        items = [Aliased(col(n), node.colmap[n]) for n in out_names
                 if n in node.colmap]
        return P.Project(scan, items), ANY

    def _lower_LFilter(self, node: LFilter):
        child, part = self.lower(node.child)
        return P.Filter(child, self.subst_scalars(node.pred)), part

    def _lower_LProject(self, node: LProject):
        child, part = self.lower(node.child)
        items = [Aliased(self.subst_scalars(e), n) for e, n in node.items]
        # partition property survives when its columns pass through
        # unchanged under the same name
        if part[0] == "hash":
            identity = {n for e, n in node.items
                        if isinstance(e, Col) and e.name == n}
            kept = all(expr_cols(k) <= identity for k in part[1])
            if not kept:
                part = ANY
        return P.Project(child, items), part

    def _lower_LReplicate(self, node: LReplicate):
        # replication is row-local: partitioning survives
        child, part = self.lower(node.child)
        return P.Replicate(child, self.subst_scalars(node.count)), part

    def _lower_LLimit(self, node: LLimit):
        child, part = self.lower(node.child)
        if part != SINGLE:
            child = P.Exchange(child, "single")
        return P.Limit(child, node.n), SINGLE

    def _lower_LSort(self, node: LSort):
        child, part = self.lower(node.child)
        if part != SINGLE:
            child = P.Exchange(child, "single")
        keys = [(self.subst_scalars(e), asc) for e, asc in node.keys]
        plan = P.Sort(child, keys, limit=node.limit)
        if node.limit is not None:
            plan = P.Limit(plan, node.limit)
        return plan, SINGLE

    def _lower_LUnionAll(self, node: LUnionAll):
        outs = []
        for ch, proj in zip(node.children, node.projections):
            cplan, _ = self.lower(ch)
            items = [Aliased(self.subst_scalars(e), n) for e, n in proj]
            outs.append(P.Project(cplan, items))
        return P.Union(outs), ANY

    def _lower_LWindow(self, node: LWindow):
        child, part = self.lower(node.child)
        partition = [self.subst_scalars(e) for e in node.partition]
        if partition:
            want = _hash_part(partition)
            if not _part_eq(part, want) and part != SINGLE:
                child = P.Exchange(child, "hash", partition)
                part = want
        else:
            if part != SINGLE:
                child = P.Exchange(child, "single")
                part = SINGLE
        order = [(self.subst_scalars(e), asc) for e, asc in node.order]
        funcs = [Aliased(wf, n) for wf, n in node.funcs]
        return P.Window(child, partition, order, funcs, frame=node.frame,
                        frame_lo=node.frame_lo, frame_hi=node.frame_hi), part

    def _lower_LJoin(self, node: LJoin):
        lplan, lpart = self.lower(node.left)
        rplan, rpart = self.lower(node.right)
        lkeys = [self.subst_scalars(k) for k in node.lkeys]
        rkeys = [self.subst_scalars(k) for k in node.rkeys]
        residual = self.subst_scalars(node.residual) if node.residual is not None else None
        from ..config import FORCE_SHUFFLED_HASH_JOIN, AuronConf

        const_keys = all(isinstance(k, Literal) for k in rkeys)
        can_broadcast = node.kind in ("inner", "left", "semi", "anti",
                                      "existence")
        if AuronConf().get(FORCE_SHUFFLED_HASH_JOIN):
            can_broadcast = const_keys  # cross joins still broadcast the 1-row side
        small = node.r_base_dim or node.r_est <= BROADCAST_ROWS or const_keys
        if can_broadcast and small:
            plan = P.HashJoin(lplan, rplan, lkeys, rkeys, how=node.kind,
                              build_side="right", broadcast=True,
                              existence_col=node.existence_name,
                              residual=residual)
            return plan, lpart
        lwant = _hash_part(lkeys)
        rwant = _hash_part(rkeys)
        if not _part_eq(lpart, lwant):
            lplan = P.Exchange(lplan, "hash", lkeys)
        if not _part_eq(rpart, rwant):
            rplan = P.Exchange(rplan, "hash", rkeys)
        plan = P.HashJoin(lplan, rplan, lkeys, rkeys, how=node.kind,
                          build_side="right", broadcast=False,
                          existence_col=node.existence_name,
                          residual=residual)
        return plan, lwant

    def _lower_LAgg(self, node: LAgg):
        child, part = self.lower(node.child)
        keys = [(self.subst_scalars(e), n) for e, n in node.keys]
        aggs = []
        for a in node.aggs:
            e = self.subst_scalars(a.expr) if a.expr is not None else None
            aggs.append(type(a)(a.fn, e, a.distinct, a.name))
        gsets = node.grouping_sets
        if gsets is not None:
            child, keys = self._expand_grouping(node, child, keys, aggs)
        key_names = [n for _, n in keys]
        need_complete = any(a.fn in ("count_distinct", "collect_list",
                                     "collect_set") for a in aggs)
        key_aliased = [Aliased(e, n) for e, n in keys]
        key_exprs = [e for e, _ in keys]
        key_cols = [col(n) for n in key_names]
        # already partitioned on the keys (by expr over child, or by output
        # name when keys are pass-through cols) -> one complete pass
        if key_names and (_part_eq(part, _hash_part(key_exprs)) or part == SINGLE):
            return P.HashAgg(child, key_aliased, aggs, mode="complete"), part
        if not key_names and part == SINGLE:
            return P.HashAgg(child, [], aggs, mode="complete"), SINGLE
        if need_complete:
            if key_names:
                ex = P.Exchange(child, "hash", key_exprs)
                plan = P.HashAgg(ex, key_aliased, aggs, mode="complete")
                return plan, _hash_part(key_exprs)
            ex = P.Exchange(child, "single")
            return P.HashAgg(ex, [], aggs, mode="complete"), SINGLE
        partial = P.HashAgg(child, key_aliased, aggs, mode="partial")
        if key_names:
            ex = P.Exchange(partial, "hash", key_cols)
            plan = P.HashAgg(ex, [Aliased(col(n), n) for n in key_names],
                             aggs, mode="final")
            return plan, _hash_part(key_cols)
        ex = P.Exchange(partial, "single")
        return P.HashAgg(ex, [], aggs, mode="final"), SINGLE

    def _expand_grouping(self, node: LAgg, child: P.PlanNode, keys, aggs):
        """ROLLUP/GROUPING SETS: project -> Expand (null-fill + gid + flag
        columns) -> group on keys+gid+flags (expand_exec.rs analogue)."""
        carry: Set[str] = set()
        for a in aggs:
            if a.expr is not None:
                carry |= expr_cols(a.expr)
        carry_l = sorted(carry)
        # a column that is BOTH a grouping key and an aggregate input must
        # be carried under a fresh name, or the nulled key column shadows
        # the aggregate's values in the rolled-up sets (CUBE(x) + sum(x))
        key_names = {n for _, n in keys}
        ren = {c: (f"__cv_{c}" if c in key_names else c) for c in carry_l}
        if any(ren[c] != c for c in carry_l):
            from .planner import walk_exprs

            def _sub(e):
                if isinstance(e, Col) and e.name in ren:
                    return col(ren[e.name])
                return e

            for i, a in enumerate(aggs):
                if a.expr is not None:
                    aggs[i] = type(a)(a.fn, walk_exprs(a.expr, _sub),
                                      a.distinct, a.name)
        pre_items = [Aliased(e, n) for (e, n) in keys] + \
            [Aliased(col(c), ren[c]) for c in carry_l]
        pre = P.Project(child, pre_items)
        nk = len(keys)
        projections = []
        for mask in node.grouping_sets:
            gid = sum((0 if kept else 1) << (nk - 1 - i)
                      for i, kept in enumerate(mask))
            proj = []
            for i, ((e, n), kept) in enumerate(zip(keys, mask)):
                dt = (node.key_dtypes[i] if node.key_dtypes else None) or dtypes.string
                proj.append(Aliased(col(n) if kept else Literal(None, dt), n))
            proj.append(Aliased(lit(gid), node.gid_name))
            for i, (_, n) in enumerate(keys):
                gn = node.grp_names.get(n)
                if gn:
                    proj.append(Aliased(lit(0 if mask[i] else 1), gn))
            proj += [Aliased(col(ren[c]), ren[c]) for c in carry_l]
            projections.append(proj)
        expand = P.Expand(pre, projections)
        new_keys = [(col(n), n) for _, n in keys]
        new_keys.append((col(node.gid_name), node.gid_name))
        for _, n in keys:
            gn = node.grp_names.get(n)
            if gn:
                new_keys.append((col(gn), gn))
        return expand, new_keys




# ------------------------------------------------- logical column pruning
def prune(node: LNode, needed: Set[str]):
    """Trim scans/projects to the columns actually consumed above them."""
    if isinstance(node, LScan):
        keep = {o: e for o, e in node.colmap.items() if e in needed}
        for f in node.filters:
            fcols = expr_cols(f)
            for o, e in node.colmap.items():
                if e in fcols:
                    keep[o] = e
        if not keep:  # keep one column for row counts
            o, e = next(iter(node.colmap.items()))
            keep[o] = e
        node.colmap = keep
        return
    if isinstance(node, LCTERef):
        keep = {n: e for n, e in node.colmap.items() if e in needed}
        if not keep:
            n, e = next(iter(node.colmap.items()))
            keep[n] = e
        node.colmap = keep
        return
    if isinstance(node, LFilter):
        prune(node.child, needed | expr_cols(node.pred))
        return
    if isinstance(node, LReplicate):
        prune(node.child, needed | expr_cols(node.count))
        return
    if isinstance(node, LProject):
        node.items = [(e, n) for e, n in node.items if n in needed] or node.items[:1]
        child_need: Set[str] = set()
        for e, _ in node.items:
            child_need |= expr_cols(e)
        prune(node.child, child_need)
        return
    if isinstance(node, LJoin):
        child_need = set(needed)
        for k in node.lkeys + node.rkeys:
            child_need |= expr_cols(k)
        if node.residual is not None:
            child_need |= expr_cols(node.residual)
        prune(node.left, child_need)
        prune(node.right, child_need)
        return
    if isinstance(node, LAgg):
        child_need: Set[str] = set()
        for e, _ in node.keys:
            child_need |= expr_cols(e)
        for a in node.aggs:
            if a.expr is not None:
                child_need |= expr_cols(a.expr)
        prune(node.child, child_need)
        return
    if isinstance(node, LWindow):
        child_need = set(needed) - {n for _, n in node.funcs}
        for e in node.partition:
            child_need |= expr_cols(e)
        for e, _ in node.order:
            child_need |= expr_cols(e)
        for wf, _ in node.funcs:
            if wf.arg is not None:
                child_need |= expr_cols(wf.arg)
        prune(node.child, child_need)
        return
    if isinstance(node, LUnionAll):
        for ch, proj in zip(node.children, node.projections):
            child_need: Set[str] = set()
            for e, _ in proj:
                child_need |= expr_cols(e)
            prune(ch, child_need)
        return
    if isinstance(node, (LSort,)):
        child_need = set(needed)
        for e, _ in node.keys:
            child_need |= expr_cols(e)
        prune(node.child, child_need)
        return
    if isinstance(node, LLimit):
        prune(node.child, needed)
        return
    raise AssertionError(f"prune: unknown node {type(node).__name__}")


def prune_lquery(lq: LQuery):
    needed: Set[str] = set()
    for e, _ in lq.outputs:
        needed |= expr_cols(e)
    for e, _ in lq.order:
        needed |= expr_cols(e)
    prune(lq.rel.node, needed)


# ----------------------------------------------------------------- entry
def sql_to_plan(sql_text: str, cat, session, schemas=None) -> P.PlanNode:
    """Parse SQL, plan, prune, materialize CTEs/scalars, and return the
    executable physical plan (the engine-side product of the reference's
    TaskDefinition handoff). `schemas` overrides the TPC-DS catalog with
    {table: {column: DataType}} for ad-hoc tables."""
    ast_q = parse_sql(sql_text)
    planner = Planner(catalog_schemas=schemas)
    lq = planner.plan(ast_q)
    prune_lquery(lq)
    for cte in planner.mat_ctes:
        prune_lquery(cte.lquery)
    low = Lowering(planner, cat, session)
    low.materialize_ctes()
    return low.lower_lquery(lq)

"""TPC-DS query registry: every query is planned from the reference's own
SQL text through the SQL front-end (auron_amd.sql) — parse -> logical plan
-> convert strategy -> physical plan. This replaces the round-1 hand-built
plan trees entirely: the engine now ingests plans the way the reference
does (a host hands it a query; AuronConvertStrategy-equivalent lowering
produces the native plan).

SQL sources: auron_amd/tpcds/sql/q*.sql (the TPC-DS specification
queries, same text the reference ships under
dev/auron-it/src/main/resources/tpcds-queries/).
"""
from __future__ import annotations

import os
from typing import List, Optional

from ..plan import nodes as P
from . import datagen

SQL_DIR = os.path.join(os.path.dirname(os.path.abspath(__file__)), "sql")

# two-part queries: the registry's qNN runs the variant the oracle models
VARIANT = {"q14": "q14a", "q23": "q23a", "q24": "q24a", "q39": "q39a"}


class Catalog:
    def __init__(self, root: str, sf: float):
        self.root = root
        self.sf = sf

    def scan(self, table: str, columns: Optional[List[str]] = None) -> P.PlanNode:
        paths = datagen.dataset_paths(self.root, self.sf, table)
        return P.ParquetScan(paths, columns=columns)


def sql_text(qname: str) -> str:
    fn = VARIANT.get(qname, qname)
    with open(os.path.join(SQL_DIR, f"{fn}.sql")) as f:
        return f.read()


def _runner(qname: str):
    text = sql_text(qname)

    def build(cat: Catalog, session) -> P.PlanNode:
        from ..sql import sql_to_plan

        return sql_to_plan(text, cat, session)

    build.__name__ = qname
    return build


QUERIES = {f"q{i}": _runner(f"q{i}") for i in range(1, 100)}

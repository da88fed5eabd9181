"""SQL plan-ingestion front-end.

Role parity: the reference's L1 plan-capture stack — a host engine hands
Auron a fully-optimized physical plan (AuronConvertStrategy.scala:38-299,
NativeConverters.scala:329, auron.proto TaskDefinition). This container
has no JVM, so the front-end that fills that role here is a SQL layer:
parse (lexer/parser) -> resolve + logical plan (planner) -> optimize
(pushdown/pruning) -> physical convert-strategy (physical) emitting the
same PhysicalPlanNode trees a foreign host would ship over the wire
(auron_amd.plan.serde / auron_amd.plan.proto).
"""
from .parser import parse_sql
from .planner import plan_query
from .physical import sql_to_plan

__all__ = ["parse_sql", "plan_query", "sql_to_plan"]

"""SQL AST (the front-end's parse product).

Role parity: Spark's parsed logical plan as consumed by
AuronConvertStrategy (reference: spark-extension/.../AuronConverters.scala
receives Catalyst nodes; here the nodes come from our own parser since no
JVM host exists in-container).
"""
from __future__ import annotations

from dataclasses import dataclass, field
from typing import List, Optional, Tuple


# ----------------------------------------------------------- expressions
class ANode:
    pass


@dataclass
class Num(ANode):
    text: str  # literal text; int vs decimal decided by consumer

    @property
    def is_int(self) -> bool:
        return "." not in self.text and "e" not in self.text.lower()


@dataclass
class Str(ANode):
    value: str


@dataclass
class Null(ANode):
    pass


@dataclass
class Star(ANode):
    qualifier: Optional[str] = None  # t.* (unused in TPC-DS but cheap)


@dataclass
class Ident(ANode):
    parts: List[str]  # ["alias", "col"] or ["col"]

    @property
    def name(self) -> str:
        return self.parts[-1]

    @property
    def qualifier(self) -> Optional[str]:
        return self.parts[0] if len(self.parts) > 1 else None


@dataclass
class FuncCall(ANode):
    name: str  # lower-cased
    args: List[ANode]
    distinct: bool = False
    star: bool = False  # count(*)
    over: Optional["WindowSpec"] = None


@dataclass
class WindowSpec(ANode):
    partition_by: List[ANode] = field(default_factory=list)
    order_by: List["OrderItem"] = field(default_factory=list)
    frame: Optional[str] = None  # "rows" | "range"
    # bounds: None = unbounded preceding, "U" = unbounded following,
    # int = signed row offset (defaults = the SQL default frame)
    frame_lo: object = None
    frame_hi: object = 0


@dataclass
class BinOp(ANode):
    op: str  # + - * / % = <> < <= > >= and or
    left: ANode
    right: ANode


@dataclass
class UnOp(ANode):
    op: str  # - not
    operand: ANode


@dataclass
class IsNull(ANode):
    operand: ANode
    negated: bool = False


@dataclass
class Between(ANode):
    operand: ANode
    low: ANode
    high: ANode
    negated: bool = False


@dataclass
class InList(ANode):
    operand: ANode
    items: List[ANode]
    negated: bool = False


@dataclass
class InSubquery(ANode):
    operand: ANode
    query: "Query"
    negated: bool = False


@dataclass
class Exists(ANode):
    query: "Query"
    negated: bool = False


@dataclass
class Like(ANode):
    operand: ANode
    pattern: str
    negated: bool = False


@dataclass
class Case(ANode):
    operand: Optional[ANode]  # CASE x WHEN v ... (operand form)
    whens: List[Tuple[ANode, ANode]]
    else_: Optional[ANode]


@dataclass
class CastE(ANode):
    operand: ANode
    typename: str  # "date", "decimal(12,2)", "int", ...


@dataclass
class Interval(ANode):
    n: int
    unit: str  # "day"


@dataclass
class ScalarSubquery(ANode):
    query: "Query"


# ----------------------------------------------------------- relations
@dataclass
class Table(ANode):
    name: str
    alias: Optional[str] = None


@dataclass
class DerivedTable(ANode):
    query: "Query"
    alias: str


@dataclass
class Join(ANode):
    left: ANode
    right: ANode
    kind: str  # inner | left | right | full | cross
    on: Optional[ANode] = None


# ----------------------------------------------------------- query shape
@dataclass
class OrderItem(ANode):
    expr: ANode
    ascending: bool = True
    nulls_first: Optional[bool] = None  # None = dialect default


@dataclass
class SelectItem(ANode):
    expr: ANode
    alias: Optional[str] = None


@dataclass
class Select(ANode):
    items: List[SelectItem]
    from_: List[ANode]  # comma-list of Table/DerivedTable/Join trees
    where: Optional[ANode] = None
    group_by: List[ANode] = field(default_factory=list)
    group_rollup: bool = False
    grouping_sets: Optional[List[List[ANode]]] = None
    having: Optional[ANode] = None
    distinct: bool = False


@dataclass
class SetOp(ANode):
    op: str  # union | union_all | intersect | except
    left: ANode  # Select | SetOp
    right: ANode


@dataclass
class Query(ANode):
    ctes: List[Tuple[str, "Query"]]
    body: ANode  # Select | SetOp
    order_by: List[OrderItem] = field(default_factory=list)
    limit: Optional[int] = None

"""Physical plan nodes.

Role parity: the PhysicalPlanNode oneof in the reference's plan protocol
(/root/reference/native-engine/auron-planner/proto/auron.proto:27-57, 27
operator kinds) and the operator set of datafusion-ext-plans. Nodes here
are built by the front-end (query builders / TPC-DS plans), serialized via
auron_amd.plan.serde, and lowered by the executor into device-columnar
pipelines.

Distribution model: one process per GPU; a plan executes
SPMD on every rank; Exchange/Broadcast nodes are the only cross-rank
communication points (RCCL all-to-all / all-gather over xGMI).
"""
from __future__ import annotations

from dataclasses import dataclass, field
from typing import Dict, List, Optional, Tuple

from ..exprs import AggFunc, Aliased, Expr


class PlanNode:
    def children(self) -> List["PlanNode"]:
        return []


@dataclass(eq=False)
class ParquetScan(PlanNode):
    """Host Parquet page read (pyarrow) -> device columns.

    Reference analogue: parquet_exec.rs (scan through JVM FS). Files are
    sharded across ranks round-robin; `columns` prunes projection at the
    reader; `filters` is an optional list of pyarrow-compatible predicates
    pushed into row-group pruning.
    """
    paths: List[str]
    columns: Optional[List[str]] = None
    filters: Optional[list] = None


@dataclass(eq=False)
class MemoryScan(PlanNode):
    """In-memory table scan (tests / broadcast-collected relations)."""
    batches: list  # List[RecordBatch], already rank-local


@dataclass(eq=False)
class Filter(PlanNode):
    child: PlanNode
    predicate: Expr

    def children(self):
        return [self.child]


@dataclass(eq=False)
class Project(PlanNode):
    child: PlanNode
    exprs: List[Aliased]

    def children(self):
        return [self.child]


@dataclass(eq=False)
class HashAgg(PlanNode):
    """Modes mirror the reference (agg/mod.rs:44): partial | final | complete."""
    child: PlanNode
    keys: List[Aliased]
    aggs: List[AggFunc]
    mode: str = "complete"  # partial | final | complete

    def children(self):
        return [self.child]


@dataclass(eq=False)
class HashJoin(PlanNode):
    """BHJ + SHJ in one operator (broadcast_join_exec.rs pattern: the
    `broadcast` flag decides whether the build side is gathered to every
    rank first)."""
    left: PlanNode
    right: PlanNode
    left_keys: List[Expr]
    right_keys: List[Expr]
    how: str = "inner"  # inner|left|right|full|semi|anti|existence
    build_side: str = "right"  # right|left
    broadcast: bool = False  # build side is broadcast (BHJ) vs co-partitioned (SHJ)
    existence_col: str = "exists"
    # extra non-equi join condition evaluated per candidate pair (the
    # reference SMJ inequality-join / Catalyst ExtraCondition analogue);
    # pairs where it is false/null do not count as matches
    residual: Optional[Expr] = None

    def children(self):
        return [self.left, self.right]


@dataclass(eq=False)
class SortMergeJoin(PlanNode):
    """Order-based equi-join (sort_merge_join_exec.rs analogue): both
    sides sorted together, equal-key runs cross-producted — no hash
    table. Same output contract as HashJoin."""
    left: PlanNode
    right: PlanNode
    left_keys: List[Expr]
    right_keys: List[Expr]
    how: str = "inner"
    existence_col: str = "exists"
    residual: Optional[Expr] = None

    def children(self):
        return [self.left, self.right]


@dataclass(eq=False)
class Sort(PlanNode):
    child: PlanNode
    keys: List[Tuple[Expr, bool]]  # (expr, ascending); Spark null ordering default
    limit: Optional[int] = None

    def children(self):
        return [self.child]


@dataclass(eq=False)
class Limit(PlanNode):
    child: PlanNode
    n: int
    offset: int = 0

    def children(self):
        return [self.child]


@dataclass(eq=False)
class Exchange(PlanNode):
    """Shuffle repartition. kind: hash (murmur3 pmod over keys, RCCL
    all-to-all), single (gather everything to rank 0), roundrobin.

    Reference analogue: shuffle_writer_exec.rs + ipc_reader_exec.rs; here
    the write+read pair collapses into one in-flight collective.
    `persist=True` (or AURON_SHUFFLE_PERSIST=1) routes the exchange
    through durable shuffle files instead (auron_amd.shuffle), giving
    the reference's stage-retry contract."""
    child: PlanNode
    kind: str = "hash"  # hash | single | roundrobin
    keys: List[Expr] = field(default_factory=list)
    persist: bool = False

    def children(self):
        return [self.child]


@dataclass(eq=False)
class Broadcast(PlanNode):
    """Collect child result on every rank (rcclAllGather / TorrentBroadcast
    analogue, NativeBroadcastExchangeBase.scala)."""
    child: PlanNode

    def children(self):
        return [self.child]


@dataclass(eq=False)
class Union(PlanNode):
    inputs: List[PlanNode]

    def children(self):
        return list(self.inputs)


@dataclass
class Replicate(PlanNode):
    """Emit each input row `count` times (INTERSECT ALL / EXCEPT ALL
    multiset multiplicities)."""
    child: PlanNode
    count: Expr

    def children(self):
        return [self.child]


@dataclass(eq=False)
class Expand(PlanNode):
    """GROUPING SETS fan-out (expand_exec.rs)."""
    child: PlanNode
    projections: List[List[Aliased]]

    def children(self):
        return [self.child]


@dataclass(eq=False)
class Window(PlanNode):
    """Window functions over (partition_by, order_by) (window_exec.rs).

    `frame` applies to running aggregates when order_by is present:
    "range" (Spark's default — UNBOUNDED PRECEDING..CURRENT ROW with
    peer rows sharing the frame end) or "rows"."""
    child: PlanNode
    partition_by: List[Expr]
    order_by: List[Tuple[Expr, bool]]
    functions: List[Aliased]  # WindowFunc exprs aliased to output names
    frame: str = "range"
    # explicit ROWS bounds: None = unbounded preceding, "U" = unbounded
    # following, int = signed row offset; defaults = the SQL default frame
    frame_lo: object = None
    frame_hi: object = 0

    def children(self):
        return [self.child]


@dataclass(eq=False)
class Generate(PlanNode):
    """Generators (generate_exec.rs): explode_split / posexplode_split
    (delimited-string explode), json_tuple(json, k1..kn), and udtf
    (Python callback mapping an input row to 0..n output rows, the
    spark_udtf_wrapper analogue; udtf_schema = [(name, dtype), ...])."""
    child: PlanNode
    generator: str
    args: List[Expr] = field(default_factory=list)
    udtf: object = None  # callable for generator="udtf" (not serialized)
    udtf_schema: list = field(default_factory=list)

    def children(self):
        return [self.child]


@dataclass(eq=False)
class PyUdaf(PlanNode):
    """Python UDAF (agg/spark_udaf_wrapper.rs analogue): group rows by
    `keys`, call `fn(*value_lists)` once per group with that group's
    input column values as Python lists, emit one row per group with
    the scalar(s) it returns (tuple matching out_schema)."""
    child: PlanNode
    keys: List  # List[Aliased]
    inputs: List[Expr] = field(default_factory=list)
    fn: object = None  # callable (not serialized)
    out_schema: list = field(default_factory=list)  # [(name, dtype)]

    def children(self):
        return [self.child]


@dataclass(eq=False)
class RenameColumns(PlanNode):
    child: PlanNode
    names: List[str]

    def children(self):
        return [self.child]


@dataclass(eq=False)
class EmptyPartitions(PlanNode):
    names: List[str] = field(default_factory=list)


@dataclass(eq=False)
class CoalesceBatches(PlanNode):
    child: PlanNode
    target_rows: int = 1 << 22

    def children(self):
        return [self.child]


@dataclass(eq=False)
class OrcScan(PlanNode):
    """ORC scan via host pyarrow.orc (orc_exec.rs analogue); device decode
    follows the parquet path's trajectory."""
    paths: List[str]
    columns: Optional[List[str]] = None


@dataclass(eq=False)
class ParquetSink(PlanNode):
    """Partitioned parquet writer (parquet_sink_exec.rs analogue). Each
    rank writes part-{rank}-{seq}.parquet under `path`."""
    child: PlanNode
    path: str

    def children(self):
        return [self.child]


@dataclass(eq=False)
class OrcSink(PlanNode):
    child: PlanNode
    path: str

    def children(self):
        return [self.child]


@dataclass(eq=False)
class PyUdf(PlanNode):
    """Host-evaluated UDF projection (spark_udf_wrapper.rs analogue: the
    device->host->device bounce for engine-foreign functions). `fn` maps
    a host RecordBatch to a dict of {name: (values, DataType)}."""
    child: PlanNode
    fn: object
    names: List[str] = field(default_factory=list)

    def children(self):
        return [self.child]


@dataclass(eq=False)
class Debug(PlanNode):
    child: PlanNode
    label: str = ""

    def children(self):
        return [self.child]

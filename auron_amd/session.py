"""AuronSession — the user entry point.

Role parity: AuronSparkSessionExtension + NativeHelper.executeNativePlan
(/root/reference/spark-extension/src/main/scala/org/apache/spark/sql/auron/
 AuronSparkSessionExtension.scala:31, NativeHelper.scala:91). A session
owns the exec context (device, rank/world from torch.distributed) and
runs physical plans on it.
"""
from __future__ import annotations

import torch
from typing import Optional

import torch.distributed as dist

from .column import RecordBatch
from .config import AuronConf
from .engine.executor import ExecContext, Executor
from .plan import nodes as P


class AuronTaskError(RuntimeError):
    """Task failure with (task_id, stage, partition) context."""

    def __init__(self, task_id, stage_id, partition, cause):
        super().__init__(
            f"task {task_id} stage {stage_id} partition {partition} failed: "
            f"{type(cause).__name__}: {cause}")
        self.task_id = task_id
        self.stage_id = stage_id
        self.partition = partition
        self.cause = cause


class AuronSession:
    def __init__(self, conf: Optional[AuronConf] = None, device=None):
        self.conf = conf or AuronConf()
        self.ctx = ExecContext.from_dist(device)
        self.executor = Executor(self.ctx)

    @property
    def rank(self) -> int:
        return self.ctx.rank

    @property
    def world_size(self) -> int:
        return self.ctx.world_size

    def execute(self, plan: P.PlanNode):
        """Run a physical plan; returns this rank's local batches."""
        return self.executor.execute(plan)

    def execute_serialized(self, task_bytes: bytes):
        """JniBridge.callNative analogue: run a serialized TaskDefinition
        (plan/serde.py) and return this rank's batches. Failures surface
        as AuronTaskError carrying (task, stage, partition) — the
        reference's setError-upcall contract (rt.rs:315)."""
        from .plan import serde

        task_id, stage_id, partition, plan = serde.deserialize_task(task_bytes)
        try:
            return self.executor.execute(plan)
        except Exception as e:
            raise AuronTaskError(task_id, stage_id, partition, e) from e

    def collect(self, plan: P.PlanNode) -> RecordBatch:
        """Run and concat this rank's result (driver-side rows analogue).

        On device OOM the query retries once with every cross-query cache
        dropped (staged-bytes HBM cache, broadcast relations) — the
        memmgr's spill tiers handle registered holders, these caches are
        the unregistered residents."""
        try:
            return self.executor.collect(plan)
        except torch.cuda.OutOfMemoryError:
            if self.ctx.device.type != "cuda":
                raise
            from . import parquet_native

            parquet_native.dbuf_cache_clear()
            self.ctx.broadcast_cache.clear()
            self.ctx.broadcast_cache_bytes = 0
            self.executor._scan_cache.clear()
            self.executor._scan_cache_bytes = 0
            torch.cuda.empty_cache()
            return self.executor.collect(plan)

    def collect_all(self, plan: P.PlanNode) -> RecordBatch:
        """Gather the full result on every rank (for result checking)."""
        local = self.collect(plan)
        if self.world_size == 1:
            return local
        from .exchange import all_gather_batch

        parts = all_gather_batch(local, self.ctx.device, self.ctx.group)
        return RecordBatch.concat(parts)

    def metrics(self):
        return dict(self.ctx.metrics)

    def metric_tree(self) -> Optional[dict]:
        """MetricNode tree of the last collect/execute (SQLMetrics
        analogue): {op, time_s, rows, batches, children}."""
        return self.executor.last_metric_tree

    def explain_metrics(self) -> str:
        """Render the last metric tree like the reference's MetricNode
        display (operator, exclusive wall, output rows/batches)."""
        tree = self.metric_tree()
        if tree is None:
            return "<no query executed>"
        lines = []

        def walk(rec, depth):
            lines.append(f"{'  ' * depth}{rec['op']}: "
                         f"{rec['time_s'] * 1e3:.2f} ms, "
                         f"rows={rec['rows']}, batches={rec['batches']}")
            for c in rec["children"]:
                walk(c, depth + 1)

        walk(tree, 0)
        return "\n".join(lines)


def init_distributed(backend: Optional[str] = None) -> ExecContext:
    """Initialize torch.distributed from torchrun env (RCCL on GPU)."""
    import os

    if dist.is_initialized():
        return ExecContext.from_dist()
    if "RANK" not in os.environ:
        return ExecContext.from_dist()
    if backend is None:
        backend = "nccl" if torch.cuda.is_available() else "gloo"
    if backend == "nccl":
        local_rank = int(os.environ.get("LOCAL_RANK", "0"))
        torch.cuda.set_device(local_rank)
    dist.init_process_group(backend=backend)
    return ExecContext.from_dist()

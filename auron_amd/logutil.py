"""Engine logging with rank/task context (logging.rs parity).

The reference's native logger prefixes every line with the Spark
stage/partition/tid thread-locals and takes its level from
`spark.auron.native.log.level`. Here: a standard logging.Logger with a
context filter fed from the executor's EVAL_CONTEXT (rank/partition)
and the same config key (env AURON_LOG_LEVEL)."""
from __future__ import annotations

import logging
import os


class _ContextFilter(logging.Filter):
    def filter(self, record):
        try:
            from . import functions as F

            ctx = F.EVAL_CONTEXT.get()
            record.rank = ctx.get("partition_id", 0)
        except Exception:
            record.rank = 0
        return True


_configured = False


def get_logger(name: str = "auron") -> logging.Logger:
    global _configured
    log = logging.getLogger(name)
    if not _configured:
        level = os.environ.get("AURON_LOG_LEVEL", "WARN").upper()
        h = logging.StreamHandler()
        h.setFormatter(logging.Formatter(
            "%(asctime)s %(levelname)s [rank %(rank)s] %(name)s: %(message)s"))
        h.addFilter(_ContextFilter())
        root = logging.getLogger("auron")
        root.addHandler(h)
        root.setLevel(getattr(logging, level, logging.WARNING))
        _configured = True
    return log

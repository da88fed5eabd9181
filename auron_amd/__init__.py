"""auron_amd — MI355X-native columnar SQL execution engine.

A brand-new implementation of the capability set of Apache Auron
(incubating): an engine-plugin front-end that lowers optimized physical
plans into a native columnar executor. Here the native engine is C++/HIP
for CDNA4 (gfx950): batches live in HBM3E as torch tensors, the hot SQL
kernels (Spark-compatible murmur3, group-by/join hash tables, partition
scatter) are hand-written HIP, and exchange runs over RCCL on xGMI via
torch.distributed. See ARCHITECTURE.md for the component map against the
reference.
"""
import warnings as _warnings

# scans wrap arrow buffers zero-copy; columns are never mutated in place
_warnings.filterwarnings("ignore", message="The given NumPy array is not writable")

from . import dtypes, exprs, ops
from .column import Column, RecordBatch
from .config import AuronConf
from .engine.executor import ExecContext, Executor
from .exprs import AggFunc, Aliased, Col, WindowFunc, col, lit
from .plan import nodes as plan
from .session import AuronSession, init_distributed

__version__ = "0.1.0"

"""Typed configuration (ConfigOption pattern).

Role parity: auron-core ConfigOption/AuronConfiguration +
SparkAuronConfiguration.java (~80 `spark.auron.*` options) and the native
pull-through side (auron-jni-bridge/src/conf.rs:32-65). Keys keep the
`spark.auron.*` shape so reference users find the same knobs; values can
be overridden via environment (AURON_<NAME>) or programmatically.
"""
from __future__ import annotations

import os
from dataclasses import dataclass
from typing import Any, Dict, Optional


@dataclass(frozen=True)
class ConfigOption:
    key: str
    default: Any
    typ: type
    doc: str = ""
    env: Optional[str] = None


_REGISTRY: Dict[str, ConfigOption] = {}


def _opt(key: str, default, typ, doc="", env=None) -> ConfigOption:
    o = ConfigOption(key, default, typ, doc, env)
    _REGISTRY[key] = o
    return o


# conf.rs:32-65 analogues
BATCH_SIZE = _opt("spark.auron.batchSize", 1 << 22, int, "rows per in-flight batch")
MEMORY_FRACTION = _opt("spark.auron.memoryFraction", 0.8, float, "fraction of HBM the memmgr may use")
ENABLE_NATIVE = _opt("spark.auron.enable", True, bool, "master switch")
REQUIRE_NATIVE = _opt("spark.auron.requireNativeKernels", True, bool,
                      "fail loudly if HIP kernels missing on GPU", env="AURON_REQUIRE_NATIVE")
SPILL_COMPRESSION = _opt("spark.auron.spill.compression.codec", "lz4", str)
SHUFFLE_COMPRESSION = _opt("spark.auron.shuffle.compression.codec", "none", str,
                           "xGMI is fast enough that intra-node shuffle ships raw")
SMJ_FALLBACK_ENABLE = _opt("spark.auron.smjfallback.enable", False, bool, env="AURON_SMJ_FALLBACK")
SMJ_FALLBACK_ROWS = _opt("spark.auron.smjfallback.rows.threshold", 10_000_000, int, env="AURON_SMJ_FALLBACK_ROWS")
PARTIAL_AGG_SKIPPING_RATIO = _opt("spark.auron.partialAggSkipping.ratio", 0.999, float, env="AURON_PARTIAL_SKIP_RATIO")
UDF_FALLBACK = _opt("spark.auron.udf.hostFallback.enable", True, bool)
LOG_LEVEL = _opt("spark.auron.native.log.level", "WARN", str, env="AURON_LOG_LEVEL")
IGNORE_CORRUPTED_FILES = _opt("spark.auron.ignoreCorruptedFiles", False, bool,
                              "skip unreadable input files instead of failing the task",
                              env="AURON_IGNORE_CORRUPTED_FILES")
BROADCAST_MAX_ROWS = _opt("spark.auron.broadcast.maxRows", 20_000_000, int,
                          "estimated build-side rows above which the planner "
                          "shuffles both join sides instead of broadcasting",
                          env="AURON_BROADCAST_MAX_ROWS")
BROADCAST_CACHE_BYTES = _opt("spark.auron.broadcast.cache.maxBytes", 4 << 30, int,
                             "device bytes of build-once broadcast relations/"
                             "hash tables kept across queries",
                             env="AURON_BROADCAST_CACHE_BYTES")
FORCE_SHUFFLED_HASH_JOIN = _opt("spark.auron.forceShuffledHashJoin", False, bool,
                                "never broadcast: hash-exchange both sides",
                                env="AURON_FORCE_SHJ")
EXPR_FUSION = _opt("spark.auron.expr.fusion.enable", False, bool,
                   "compile project/filter scalar expression trees into one "
                   "fused interpreter kernel launch (k_expr_exec). Default "
                   "off: the kernel itself measures ~24us/launch but the "
                   "per-batch host dispatch costs more than the at::native "
                   "launches it replaces at TPC-DS shapes (same-box A/B: "
                   "SF=10 suite 8.24s on vs 6.52s off); worthwhile only for "
                   "very wide scalar-heavy projections until the native "
                   "plan walker absorbs the dispatch",
                   env="AURON_EXPR_FUSION")
AGG_STREAMING = _opt("spark.auron.agg.streaming.enable", True, bool,
                     "chunked spill-capable partial aggregation",
                     env="AURON_AGG_STREAMING")
HOST_SPILL_BUDGET = _opt("spark.auron.memory.hostSpillBudget", 64 << 30, int,
                         "host-DRAM bytes of spilled holders before the disk "
                         "tier engages", env="AURON_HOST_SPILL_BUDGET")
SHUFFLE_PERSIST = _opt("spark.auron.shuffle.persist", False, bool,
                       "route exchanges through durable .data/.index files "
                       "(stage-retry contract)", env="AURON_SHUFFLE_PERSIST")
PARQUET_NATIVE = _opt("spark.auron.parquet.native.enable", True, bool,
                      "device parquet decode (else pyarrow host read)",
                      env="AURON_PARQUET_NATIVE")
PARQUET_PAGE_DECOMPRESS = _opt("spark.auron.parquet.pageDecompression.enable",
                               True, bool,
                               "host-decompress snappy/zstd/gzip pages for "
                               "the device decode path")
PARTIAL_AGG_SKIPPING_MINROWS = _opt("spark.auron.partialAggSkipping.minRows",
                                    1 << 14, int,
                                    "below this many rows partial-agg "
                                    "skipping is never sampled")
SORT_LIMIT_TOPK = _opt("spark.auron.sort.topK.enable", True, bool,
                       "ORDER BY+LIMIT keeps only the top K rows per batch")
EXCHANGE_SAMPLE_ROWS = _opt("spark.auron.rangePartition.sampleRows", 4096, int,
                            "per-rank sample size for range-partition bounds")
SCAN_CACHE_BYTES = _opt("spark.auron.scan.cache.maxBytes", 0, int,
                        "per-query decoded-scan cache budget; 0 = auto "
                        "(15% of HBM on GPU, 4 GB on CPU)",
                        env="AURON_SCAN_CACHE_BYTES")
STREAM_BYTES = _opt("spark.auron.stream.minInputBytes", -1, int,
                    "estimated agg-input bytes above which operators "
                    "stream instead of materializing; -1 = auto (8% of "
                    "HBM on GPU, 1 GB on CPU), 0 = always stream",
                    env="AURON_STREAM_BYTES")
SCAN_HBM_CACHE = _opt("spark.auron.scan.hbmStagedCache.maxBytes", 0, int,
                      "staged raw-page bytes kept device-resident across "
                      "reads; 0 = auto (30% of HBM)",
                      env="AURON_SCAN_HBM_CACHE")
HBM_CACHE_RESERVE = _opt("spark.auron.scan.hbmStagedCache.reserveFraction",
                         0.25, float,
                         "free-HBM fraction the staged-bytes cache must "
                         "never consume (evicts itself under pressure)")
SCAN_PREFETCH = _opt("spark.auron.scan.streamPrefetch", 2, int,
                     "files decoded ahead of the consumer on the "
                     "streaming scan path")
SCAN_IO_THREADS = _opt("spark.auron.scan.ioThreads", 8, int,
                       "host page-read/staging thread-pool size",
                       env="AURON_SCAN_IO_THREADS")
STREAM_PEEK_FACTOR = _opt("spark.auron.stream.peekBatches", 8, int,
                          "batch-size multiples buffered before the "
                          "chunked agg commits to streaming")
STREAM_JOIN_PROBE = _opt("spark.auron.stream.joinProbe.enable", True, bool,
                         "probe build-right joins batch-by-batch on the "
                         "streaming path (GPU only)",
                         env="AURON_STREAM_JOIN_PROBE")
SPILL_DIR = _opt("spark.auron.spill.dir", "", str,
                 "directory for disk-tier spill files (empty = system tmp)",
                 env="AURON_SPILL_DIR")
MEM_BUDGET = _opt("spark.auron.memory.budgetBytes", 0, int,
                  "absolute memmgr budget override; 0 = memoryFraction "
                  "of device/host memory", env="AURON_MEM_BUDGET")


class AuronConf:
    def __init__(self, overrides: Optional[Dict[str, Any]] = None):
        self._values: Dict[str, Any] = {}
        if overrides:
            for k, v in overrides.items():
                self.set(k, v)

    def set(self, key: str, value) -> "AuronConf":
        if key not in _REGISTRY:
            raise KeyError(f"unknown config {key}; known: {sorted(_REGISTRY)}")
        self._values[key] = value
        return self

    def get(self, opt: ConfigOption):
        if opt.key in self._values:
            return self._values[opt.key]
        if opt.env and opt.env in os.environ:
            raw = os.environ[opt.env]
            if opt.typ is bool:
                return raw not in ("0", "false", "False")
            return opt.typ(raw)
        return opt.default

    @staticmethod
    def options() -> Dict[str, ConfigOption]:
        return dict(_REGISTRY)

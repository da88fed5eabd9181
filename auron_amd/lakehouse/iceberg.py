"""Apache Iceberg v1/v2 table reader (metadata -> parquet data files).

Role parity: thirdparty/auron-iceberg's AuronConvertProvider — the
reference converts an Iceberg scan into its native parquet scan; here
IcebergTable.scan() resolves {metadata json -> snapshot -> manifest list
(avro) -> manifests (avro) -> live data files} and emits the engine's
ParquetScan over those files, so every downstream operator (and the
device decode path) is shared with plain parquet tables.

Supported: file:// table locations, FORMAT=PARQUET data files,
added/existing manifest entries (status 0/1; deletes=2 are dropped),
v2 delete files are rejected loudly rather than silently mis-read.
"""
from __future__ import annotations

import json
import os
from typing import List, Optional

from ..plan import nodes as P
from . import avro


def _local(p: str) -> str:
    if p.startswith("file://"):
        return p[len("file://"):]
    return p


class IcebergTable:
    def __init__(self, table_path: str):
        self.path = _local(table_path)
        meta_dir = os.path.join(self.path, "metadata")
        hint = os.path.join(meta_dir, "version-hint.text")
        if os.path.exists(hint):
            v = open(hint).read().strip()
            cand = os.path.join(meta_dir, f"v{v}.metadata.json")
        else:
            metas = sorted(f for f in os.listdir(meta_dir)
                           if f.endswith(".metadata.json"))
            assert metas, f"no metadata json under {meta_dir}"
            cand = os.path.join(meta_dir, metas[-1])
        self.metadata = json.load(open(cand))

    def snapshot(self, snapshot_id: Optional[int] = None) -> dict:
        snaps = self.metadata.get("snapshots", [])
        assert snaps, "iceberg table has no snapshots"
        if snapshot_id is None:
            snapshot_id = self.metadata.get("current-snapshot-id",
                                            snaps[-1]["snapshot-id"])
        for s in snaps:
            if s["snapshot-id"] == snapshot_id:
                return s
        raise KeyError(f"snapshot {snapshot_id} not found")

    def data_files(self, snapshot_id: Optional[int] = None) -> List[str]:
        snap = self.snapshot(snapshot_id)
        files: List[str] = []
        if "manifest-list" in snap:
            _, entries = avro.read_file(_local(snap["manifest-list"]))
            manifests = [e["manifest_path"] for e in entries]
        else:  # v1 inline manifest array
            manifests = snap["manifests"]
        for m in manifests:
            _, entries = avro.read_file(_local(m))
            for e in entries:
                if e.get("status", 1) == 2:  # DELETED
                    continue
                df = e["data_file"]
                content = df.get("content", 0)
                if content != 0:
                    raise NotImplementedError(
                        "iceberg v2 delete files are not supported")
                fmt = str(df.get("file_format", "PARQUET")).upper()
                if fmt != "PARQUET":
                    raise NotImplementedError(f"iceberg {fmt} data files")
                files.append(_local(df["file_path"]))
        return files

    def scan(self, columns: Optional[List[str]] = None,
             snapshot_id: Optional[int] = None) -> P.PlanNode:
        return P.ParquetScan(self.data_files(snapshot_id), columns=columns)


class IcebergCatalog:
    """Catalog facade matching tpcds.Catalog's scan() shape, for running
    the suite over Iceberg-registered tables (BASELINE config 5)."""

    def __init__(self, warehouse: str):
        self.warehouse = _local(warehouse)

    def scan(self, table: str, columns: Optional[List[str]] = None) -> P.PlanNode:
        return IcebergTable(os.path.join(self.warehouse, table)).scan(columns)


# ------------------------------------------------------------ test writer
MANIFEST_FILE_SCHEMA = json.dumps({
    "type": "record", "name": "manifest_entry", "fields": [
        {"name": "status", "type": "int"},
        {"name": "snapshot_id", "type": ["null", "long"], "default": None},
        {"name": "data_file", "type": {
            "type": "record", "name": "r2", "fields": [
                {"name": "content", "type": "int"},
                {"name": "file_path", "type": "string"},
                {"name": "file_format", "type": "string"},
                {"name": "record_count", "type": "long"},
                {"name": "file_size_in_bytes", "type": "long"},
            ]}},
    ]})

MANIFEST_LIST_SCHEMA = json.dumps({
    "type": "record", "name": "manifest_file", "fields": [
        {"name": "manifest_path", "type": "string"},
        {"name": "manifest_length", "type": "long"},
        {"name": "partition_spec_id", "type": "int"},
        {"name": "added_snapshot_id", "type": ["null", "long"], "default": None},
    ]})


def write_table(table_path: str, parquet_files: List[str],
                snapshot_id: int = 1) -> IcebergTable:
    """Create a minimal Iceberg table over existing parquet files (test
    fixture writer; the reader is the product)."""
    table_path = _local(table_path)
    meta_dir = os.path.join(table_path, "metadata")
    os.makedirs(meta_dir, exist_ok=True)
    manifest = os.path.join(meta_dir, f"manifest-{snapshot_id}.avro")
    entries = []
    for f in parquet_files:
        entries.append({
            "status": 1, "snapshot_id": snapshot_id,
            "data_file": {
                "content": 0, "file_path": os.path.abspath(f),
                "file_format": "PARQUET",
                "record_count": 0,
                "file_size_in_bytes": os.path.getsize(f),
            }})
    avro.write_file(manifest, MANIFEST_FILE_SCHEMA, entries)
    mlist = os.path.join(meta_dir, f"snap-{snapshot_id}.avro")
    avro.write_file(mlist, MANIFEST_LIST_SCHEMA, [{
        "manifest_path": manifest,
        "manifest_length": os.path.getsize(manifest),
        "partition_spec_id": 0,
        "added_snapshot_id": snapshot_id,
    }])
    meta = {
        "format-version": 2,
        "table-uuid": "00000000-0000-0000-0000-000000000000",
        "location": table_path,
        "current-snapshot-id": snapshot_id,
        "snapshots": [{"snapshot-id": snapshot_id, "manifest-list": mlist}],
    }
    with open(os.path.join(meta_dir, "v1.metadata.json"), "w") as f:
        json.dump(meta, f)
    with open(os.path.join(meta_dir, "version-hint.text"), "w") as f:
        f.write("1")
    return IcebergTable(table_path)

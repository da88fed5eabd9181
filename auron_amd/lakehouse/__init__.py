"""Lakehouse table-format providers.

Role parity: the reference's thirdparty/auron-iceberg (+paimon/hudi)
AuronConvertProvider plugins, which convert table-format scans into
native parquet scans. Here the Iceberg provider resolves a table's
current snapshot down to its parquet data files and hands them to the
same ParquetScan the rest of the engine uses.
"""
from .iceberg import IcebergCatalog, IcebergTable

__all__ = ["IcebergCatalog", "IcebergTable"]

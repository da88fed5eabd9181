"""Run one TPC-DS query once (rocprof target)."""
import sys

from auron_amd import AuronSession
from auron_amd.tpcds import datagen
from auron_amd.tpcds.queries import QUERIES, Catalog

qname = sys.argv[1] if len(sys.argv) > 1 else "q3"
sf = float(sys.argv[2]) if len(sys.argv) > 2 else 1.0
import os
root = os.environ.get("AURON_DATA_ROOT", "data")
datagen.write_dataset(root, sf)
s = AuronSession()
cat = Catalog(root, sf)
plan = QUERIES[qname](cat, s)
out = s.collect(plan)
print(qname, "rows:", out.num_rows)

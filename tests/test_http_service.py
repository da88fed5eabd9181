"""Observability HTTP service (reference http-service parity)."""
import json
import urllib.request

from auron_amd import AuronSession, col, dtypes
from auron_amd import http_service
from auron_amd.column import RecordBatch
from auron_amd.plan import nodes as P


def test_metrics_endpoints():
    s = AuronSession()
    b = RecordBatch.from_pydict({"x": [1, 2, 3]}, {"x": dtypes.int64})
    s.collect(P.Filter(P.MemoryScan([b]), col("x") > 1))
    srv = http_service.start(s)
    try:
        base = f"http://127.0.0.1:{srv.port}"
        health = json.load(urllib.request.urlopen(f"{base}/healthz"))
        assert health == {"ok": True}
        metrics = json.load(urllib.request.urlopen(f"{base}/metrics"))
        assert "op.Filter" in metrics["operators"]
        assert metrics["memory"]["budget_bytes"] > 0
        conf = json.load(urllib.request.urlopen(f"{base}/config"))
        assert "spark.auron.batchSize" in conf
    finally:
        srv.stop()


def test_last_query_endpoint():
    import json
    import urllib.request

    from auron_amd import AuronSession, dtypes
    from auron_amd import http_service
    from auron_amd.column import RecordBatch
    from auron_amd.plan import nodes as P

    s = AuronSession()
    s.collect(P.Limit(P.MemoryScan([RecordBatch.from_pydict(
        {"x": [1, 2, 3]}, {"x": dtypes.int64})]), 2))
    srv = http_service.start(s)
    try:
        url = f"http://127.0.0.1:{srv.port}/last_query"
        d = json.load(urllib.request.urlopen(url))
        assert d["tree"]["op"] == "Limit" and d["tree"]["rows"] == 2
        assert "MemoryScan" in d["rendered"]
    finally:
        srv.stop()

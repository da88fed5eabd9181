"""Optional in-process observability HTTP service.

Role parity: the reference's native HTTP service (feature `http-service`,
/root/reference/native-engine/auron/src/http/mod.rs — pprof CPU profile +
jemalloc heap endpoints, started on first callNative). Here the service
exposes engine metrics and memory-manager state as JSON; profiling hooks
point at rocprofv3 (out-of-process on ROCm).

Endpoints:
  /healthz     liveness
  /metrics     per-operator exclusive wall times + memmgr counters
  /last_query  MetricNode tree of the most recent execute
  /config      resolved configuration
"""
from __future__ import annotations

import json
import threading
from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer
from typing import Optional


class _Handler(BaseHTTPRequestHandler):
    session = None

    def log_message(self, fmt, *args):  # silence request logging
        pass

    def _send(self, code: int, payload):
        body = json.dumps(payload, indent=2, default=str).encode()
        self.send_response(code)
        self.send_header("Content-Type", "application/json")
        self.send_header("Content-Length", str(len(body)))
        self.end_headers()
        self.wfile.write(body)

    def do_GET(self):
        s = type(self).session
        if self.path == "/healthz":
            self._send(200, {"ok": True})
        elif self.path == "/metrics":
            mm = s.ctx.memmgr
            self._send(200, {
                "operators": s.metrics(),
                "memory": {
                    "budget_bytes": mm.budget,
                    "resident_bytes": mm.resident_bytes(),
                    **mm.metrics,
                },
                "rank": s.rank,
                "world_size": s.world_size,
            })
        elif self.path == "/last_query":
            # MetricNode tree of the most recent execute (spark-ui's
            # per-query native-metrics surface)
            self._send(200, {"tree": s.metric_tree(),
                             "rendered": s.explain_metrics()})
        elif self.path == "/config":
            from .config import AuronConf

            opts = {k: s.conf.get(o) for k, o in AuronConf.options().items()}
            self._send(200, opts)
        else:
            self._send(404, {"error": "unknown path"})


class MetricsServer:
    def __init__(self, session, host: str = "127.0.0.1", port: int = 0):
        handler = type("BoundHandler", (_Handler,), {"session": session})
        self.httpd = ThreadingHTTPServer((host, port), handler)
        self.port = self.httpd.server_address[1]
        self._thread = threading.Thread(target=self.httpd.serve_forever,
                                        name="auron-metrics-http", daemon=True)
        self._thread.start()

    def stop(self):
        self.httpd.shutdown()
        self.httpd.server_close()


def start(session, host: str = "127.0.0.1", port: int = 0) -> MetricsServer:
    return MetricsServer(session, host, port)

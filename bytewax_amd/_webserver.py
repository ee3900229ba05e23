"""Ops webserver: ``GET /dataflow`` and ``GET /metrics``.

Parity target: the reference's axum sidecar (reference
src/webserver/mod.rs:22-71): enabled by ``BYTEWAX_DATAFLOW_API_ENABLED``,
port from ``BYTEWAX_DATAFLOW_API_PORT`` (default 3030); ``/dataflow``
serves the rendered graph JSON cached at startup, ``/metrics`` serves
the prometheus exposition.
"""

import os
import threading
from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer
from typing import Optional

from ._metrics import enable_metrics, generate_python_metrics

__all__ = ["maybe_start_webserver"]

_server: Optional[ThreadingHTTPServer] = None


def maybe_start_webserver(flow) -> None:
    """Start the API webserver if BYTEWAX_DATAFLOW_API_ENABLED is set."""
    global _server
    if not os.environ.get("BYTEWAX_DATAFLOW_API_ENABLED"):
        return
    if _server is not None:
        return
    from .visualize import to_json

    enable_metrics()
    flow_json = to_json(flow).encode()
    port = int(os.environ.get("BYTEWAX_DATAFLOW_API_PORT", "3030"))

    class Handler(BaseHTTPRequestHandler):
        def do_GET(self):  # noqa: N802
            if self.path == "/dataflow":
                body = flow_json
                ctype = "application/json"
            elif self.path == "/metrics":
                body = generate_python_metrics()
                ctype = "text/plain; version=0.0.4"
            else:
                self.send_response(404)
                self.end_headers()
                return
            self.send_response(200)
            self.send_header("Content-Type", ctype)
            self.send_header("Content-Length", str(len(body)))
            self.end_headers()
            self.wfile.write(body)

        def log_message(self, fmt, *args):
            pass

    _server = ThreadingHTTPServer(("0.0.0.0", port), Handler)
    t = threading.Thread(target=_server.serve_forever, daemon=True)
    t.start()

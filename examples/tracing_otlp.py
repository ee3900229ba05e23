"""Export per-operator spans to an OTLP collector (reference
examples/tracing.py).

Point BYTEWAX_OTLP_URL at any OTLP/HTTP collector (Jaeger's 4318
ingest works); with no collector the spans are dropped silently and
the flow runs unchanged.  For a self-contained demo this starts a
tiny in-process collector and prints how many span bytes arrived.
"""

import os
import sys
import threading
from http.server import BaseHTTPRequestHandler, HTTPServer
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import bytewax_amd.operators as op
from bytewax_amd.connectors.stdio import StdOutSink
from bytewax_amd.dataflow import Dataflow
from bytewax_amd.testing import TestingSource
from bytewax_amd.tracing import OtlpTracingConfig, setup_tracing


def _local_collector():
    received = []

    class Handler(BaseHTTPRequestHandler):
        def do_POST(self):
            n = int(self.headers.get("Content-Length", 0))
            received.append(self.rfile.read(n))
            self.send_response(200)
            self.end_headers()

        def log_message(self, *args):
            pass

    srv = HTTPServer(("127.0.0.1", 0), Handler)
    threading.Thread(target=srv.serve_forever, daemon=True).start()
    return srv, received


def main():
    srv, received = _local_collector()
    url = os.getenv(
        "BYTEWAX_OTLP_URL", f"http://127.0.0.1:{srv.server_port}"
    )
    tracer = setup_tracing(
        tracing_config=OtlpTracingConfig(
            url=url, service_name="tracing-example"
        ),
        log_level="INFO",
    )

    flow = Dataflow("traced")
    nums = op.input("inp", flow, TestingSource(range(20)))
    doubled = op.map("double", nums, lambda x: x * 2)
    op.output("out", doubled, StdOutSink())

    from bytewax_amd.testing import run_main

    run_main(flow)
    tracer.close()  # flush the span buffer
    srv.shutdown()
    print(
        f"collector received {len(received)} OTLP batch(es), "
        f"{sum(len(b) for b in received)} bytes of spans"
    )


if __name__ == "__main__":
    main()

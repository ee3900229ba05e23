"""Minimal OTLP/HTTP trace exporter (no opentelemetry dependency).

Encodes `ExportTraceServiceRequest` protobuf messages by hand (the
OTLP wire schema is stable and tiny for the span subset we emit) and
POSTs them to an OTLP/HTTP collector endpoint (`/v1/traces`,
`application/x-protobuf`).  This is the transport behind
:func:`bytewax_amd.tracing.setup_tracing` with
:class:`~bytewax_amd.tracing.OtlpTracingConfig` — role parity with
the reference's `src/tracing/otlp_tracing.rs` exporter.

Proto schema (opentelemetry-proto v1, trace service):
  ExportTraceServiceRequest { repeated ResourceSpans resource_spans=1 }
  ResourceSpans { Resource resource=1; repeated ScopeSpans scope_spans=2 }
  Resource { repeated KeyValue attributes=1 }
  KeyValue { string key=1; AnyValue value=2 }
  AnyValue { oneof { string string_value=1; int64 int_value=3; } }
  ScopeSpans { InstrumentationScope scope=1; repeated Span spans=2 }
  InstrumentationScope { string name=1 }
  Span { bytes trace_id=1; bytes span_id=2; string name=5;
         int32 kind=6; fixed64 start_time_unix_nano=7;
         fixed64 end_time_unix_nano=8; repeated KeyValue attributes=9 }
"""

import os
import struct
import threading
import urllib.request
from dataclasses import dataclass, field
from typing import Dict, List, Union

__all__ = ["OtlpSpan", "OtlpHttpExporter"]


def _varint(n: int) -> bytes:
    out = bytearray()
    while True:
        b = n & 0x7F
        n >>= 7
        if n:
            out.append(b | 0x80)
        else:
            out.append(b)
            return bytes(out)


def _tag(f: int, wire: int) -> bytes:
    return _varint((f << 3) | wire)


def _ld(f: int, payload: bytes) -> bytes:
    return _tag(f, 2) + _varint(len(payload)) + payload


def _s(f: int, s: str) -> bytes:
    return _ld(f, s.encode())


def _fixed64(f: int, v: int) -> bytes:
    return _tag(f, 1) + struct.pack("<Q", v)


def _vint(f: int, v: int) -> bytes:
    return _tag(f, 0) + _varint(v)


def _any_value(v: Union[str, int]) -> bytes:
    if isinstance(v, bool) or isinstance(v, int):
        return _vint(3, int(v))
    return _s(1, str(v))


def _kv(key: str, v: Union[str, int]) -> bytes:
    return _s(1, key) + _ld(2, _any_value(v))


@dataclass
class OtlpSpan:
    name: str
    start_ns: int
    end_ns: int
    trace_id: bytes
    span_id: bytes
    attributes: Dict[str, Union[str, int]] = field(default_factory=dict)

    def encode(self) -> bytes:
        out = _ld(1, self.trace_id) + _ld(2, self.span_id)
        out += _s(5, self.name)
        out += _vint(6, 1)  # SPAN_KIND_INTERNAL
        out += _fixed64(7, self.start_ns) + _fixed64(8, self.end_ns)
        for k, v in self.attributes.items():
            out += _ld(9, _kv(k, v))
        return out


def encode_request(service_name: str, spans: List[OtlpSpan]) -> bytes:
    resource = _ld(1, _kv("service.name", service_name))
    scope = _s(1, "bytewax_amd")
    scope_spans = _ld(1, scope) + b"".join(
        _ld(2, sp.encode()) for sp in spans
    )
    resource_spans = _ld(1, resource) + _ld(2, scope_spans)
    return _ld(1, resource_spans)


class OtlpHttpExporter:
    """Buffering OTLP/HTTP span exporter with a background flusher."""

    def __init__(
        self,
        url: str,
        service_name: str,
        flush_interval_s: float = 2.0,
        max_buffer: int = 4096,
    ):
        if url.startswith("grpc://"):
            # The reference config names a gRPC collector; OTLP
            # collectors listen for HTTP on 4318 by convention.
            url = "http://" + url[len("grpc://") :]
        self.url = url.rstrip("/") + "/v1/traces"
        self.service_name = service_name
        self.max_buffer = max_buffer
        self._buf: List[OtlpSpan] = []
        self._lock = threading.Lock()
        self._stop = threading.Event()
        self.trace_id = os.urandom(16)
        self._thread = threading.Thread(
            target=self._loop, args=(flush_interval_s,), daemon=True
        )
        self._thread.start()
        # Final flush even if the guard is never closed explicitly.
        import atexit

        atexit.register(self.flush)

    def record(self, span: OtlpSpan) -> None:
        with self._lock:
            if len(self._buf) < self.max_buffer:
                self._buf.append(span)

    def span(self, name, start_ns, end_ns, attributes=None) -> None:
        self.record(
            OtlpSpan(
                name,
                start_ns,
                end_ns,
                self.trace_id,
                os.urandom(8),
                attributes or {},
            )
        )

    def flush(self) -> None:
        with self._lock:
            spans, self._buf = self._buf, []
        if not spans:
            return
        body = encode_request(self.service_name, spans)
        req = urllib.request.Request(
            self.url,
            data=body,
            headers={"Content-Type": "application/x-protobuf"},
            method="POST",
        )
        try:
            urllib.request.urlopen(req, timeout=5).read()
        except Exception:  # noqa: BLE001
            # Span loss must never break the dataflow; drop the batch
            # (the reference's exporter behaves the same on a dead
            # collector).
            pass

    def _loop(self, interval: float) -> None:
        while not self._stop.wait(interval):
            self.flush()

    def shutdown(self) -> None:
        self._stop.set()
        self.flush()

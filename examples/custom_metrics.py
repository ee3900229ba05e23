"""Export a connector-defined Prometheus metric alongside the engine's
built-ins (reference examples/custom_metrics.py).

Any `prometheus_client` instrument registered by user code shows up
on the webserver's `/metrics` endpoint next to the framework's
per-operator counters (enable with BYTEWAX_DATAFLOW_API_ENABLED=1).
"""

import sys
from datetime import datetime, timedelta, timezone
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import bytewax_amd.operators as op
from bytewax_amd.connectors.stdio import StdOutSink
from bytewax_amd.dataflow import Dataflow
from bytewax_amd.inputs import DynamicSource, StatelessSourcePartition
from prometheus_client import Gauge

NEXT_BATCH_DELAY_GAUGE = Gauge(
    "next_batch_delay",
    "How late the engine polled this partition, in seconds",
    ["step_id", "worker_index"],
    unit="seconds",
)


class PeriodicPartition(StatelessSourcePartition):
    def __init__(self, labels, frequency: timedelta):
        self.frequency = frequency
        self._due = datetime.now(timezone.utc)
        self._ticks = 0
        self._gauge = NEXT_BATCH_DELAY_GAUGE.labels(**labels)

    def next_awake(self):
        return self._due

    def next_batch(self):
        self._ticks += 1
        if self._ticks > 5:
            raise StopIteration()
        lag = datetime.now(timezone.utc) - self._due
        self._due += self.frequency
        self._gauge.set(lag.total_seconds())
        return [self._ticks]


class PeriodicSource(DynamicSource):
    def __init__(self, frequency: timedelta):
        self.frequency = frequency

    def build(self, step_id, worker_index, worker_count):
        labels = {"step_id": step_id, "worker_index": worker_index}
        return PeriodicPartition(labels, self.frequency)


flow = Dataflow("custom_metrics")
ticks = op.input("periodic", flow, PeriodicSource(timedelta(seconds=0.1)))
op.output("out", ticks, StdOutSink())

if __name__ == "__main__":
    from prometheus_client import generate_latest

    from bytewax_amd.testing import run_main

    run_main(flow)
    exposition = generate_latest().decode()
    line = next(
        ln for ln in exposition.splitlines()
        if ln.startswith("next_batch_delay_seconds{")
    )
    print(line)

"""Polling a source on a fixed cadence with `next_awake` (reference
examples/periodic_input.py).

Two variants: a stateless `DynamicSource` and a resumable
`FixedPartitionedSource` whose snapshot carries the schedule so a
restart does not drift.  Each emission reports how late the engine
woke the partition relative to its requested time.
"""

import sys
from datetime import datetime, timedelta, timezone
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import bytewax_amd.operators as op
from bytewax_amd.connectors.stdio import StdOutSink
from bytewax_amd.dataflow import Dataflow
from bytewax_amd.inputs import (
    DynamicSource,
    FixedPartitionedSource,
    StatefulSourcePartition,
    StatelessSourcePartition,
)

TICKS = 5


class PeriodicPartition(StatelessSourcePartition):
    def __init__(self, frequency: timedelta):
        self.frequency = frequency
        self._due = datetime.now(timezone.utc)
        self._ticks = 0

    def next_awake(self):
        return self._due

    def next_batch(self):
        self._ticks += 1
        if self._ticks >= TICKS:
            raise StopIteration()
        lag = datetime.now(timezone.utc) - self._due
        self._due += self.frequency
        return [f"delay (ms): {lag.total_seconds() * 1000:.3f}"]


class PeriodicSource(DynamicSource):
    def __init__(self, frequency: timedelta):
        self.frequency = frequency

    def build(self, step_id, worker_index, worker_count):
        return PeriodicPartition(self.frequency)


stateless_flow = Dataflow("periodic_stateless")
ticks = op.input(
    "periodic", stateless_flow, PeriodicSource(timedelta(seconds=0.1))
)
op.output("out", ticks, StdOutSink())


class ResumablePeriodicPartition(StatefulSourcePartition):
    def __init__(self, frequency: timedelta, due: datetime, ticks: int):
        self.frequency = frequency
        self._due = due
        self._ticks = ticks

    def next_awake(self):
        return self._due

    def next_batch(self):
        self._ticks += 1
        if self._ticks >= TICKS:
            raise StopIteration()
        lag = datetime.now(timezone.utc) - self._due
        self._due += self.frequency
        return [f"delay (ms): {lag.total_seconds() * 1000:.3f}"]

    def snapshot(self):
        # The schedule itself is the state: a resume continues the
        # cadence instead of restarting it.
        return {"due": self._due.isoformat(), "ticks": self._ticks}


class ResumablePeriodicSource(FixedPartitionedSource):
    def __init__(self, frequency: timedelta):
        self.frequency = frequency

    def list_parts(self):
        return ["singleton"]

    def build_part(self, step_id, for_part, resume_state):
        assert for_part == "singleton"
        state = resume_state or {}
        now = datetime.now(timezone.utc).isoformat()
        due = datetime.fromisoformat(state.get("due", now))
        return ResumablePeriodicPartition(
            self.frequency, due, state.get("ticks", 0)
        )


stateful_flow = Dataflow("periodic_stateful")
ticks2 = op.input(
    "periodic", stateful_flow, ResumablePeriodicSource(timedelta(seconds=0.1))
)
op.output("out", ticks2, StdOutSink())

if __name__ == "__main__":
    from bytewax_amd.testing import run_main

    run_main(stateless_flow)
    run_main(stateful_flow)

"""Batch web events into windowed Arrow tables and write a
partitioned Parquet dataset (reference examples/events_to_parquet.py).

The reference simulates web events with the `fake_web_events`
package; this environment has no network, so a small deterministic
simulator of the same event shape feeds the identical flow: JSON
events -> keyed by page -> tumbling-window collect -> one Arrow
Table per (page, window) -> partitioned Parquet sink.
"""

import json
import random
import sys
from datetime import datetime, timedelta, timezone
from pathlib import Path
from typing import Any, List, Optional

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import bytewax_amd.operators as op
import bytewax_amd.operators.windowing as win
from bytewax_amd.dataflow import Dataflow
from bytewax_amd.inputs import FixedPartitionedSource, StatefulSourcePartition
from bytewax_amd.operators.windowing import EventClock, TumblingWindower
from bytewax_amd.outputs import FixedPartitionedSink, StatefulSinkPartition

try:
    from pyarrow import Table, parquet
except ImportError:  # pragma: no cover - pyarrow is in the image
    Table = parquet = None

START = datetime(2024, 3, 1, tzinfo=timezone.utc)
OUT_DIR = Path("parquet_demo_out")
PAGES = ["/home", "/shop", "/about"]


class SimulatedPartition(StatefulSourcePartition):
    """Deterministic stand-in for the fake-web-events simulator."""

    def __init__(self, resume_state: Optional[int]):
        self.i = resume_state or 0
        self.rng = random.Random(77)
        self._events = []
        for j in range(60):
            self._events.append(
                {
                    "event_timestamp": (
                        START + timedelta(seconds=j)
                    ).isoformat(),
                    "page_url_path": self.rng.choice(PAGES),
                    "user_custom_id": f"u{self.rng.randint(1, 9)}",
                }
            )

    def next_batch(self) -> List[str]:
        if self.i >= len(self._events):
            raise StopIteration()
        out = [json.dumps(self._events[self.i])]
        self.i += 1
        return out

    def snapshot(self) -> int:
        return self.i


class FakeWebEventsSource(FixedPartitionedSource):
    def list_parts(self) -> List[str]:
        return ["singleton"]

    def build_part(self, step_id, for_part, resume_state):
        return SimulatedPartition(resume_state)


class ParquetPartition(StatefulSinkPartition):
    def write_batch(self, batch: List[Any]) -> None:
        for table in batch:
            parquet.write_to_dataset(
                table,
                root_path=str(OUT_DIR),
                partition_cols=["year", "month", "day", "page_url_path"],
            )

    def snapshot(self) -> None:
        return None


class ParquetSink(FixedPartitionedSink):
    def list_parts(self) -> List[str]:
        return ["singleton"]

    def part_fn(self, item_key: str) -> int:
        return 0

    def build_part(self, step_id, for_part, resume_state):
        return ParquetPartition()


def add_date_columns(kv):
    key, (win_id, events) = kv
    rows = []
    for e in events:
        ts = datetime.fromisoformat(e["event_timestamp"])
        rows.append(
            {**e, "year": ts.year, "month": ts.month, "day": ts.day}
        )
    return (key, Table.from_pylist(rows))


flow = Dataflow("events_to_parquet")
raw = op.input("inp", flow, FakeWebEventsSource())
events = op.map("parse", raw, json.loads)
keyed = op.key_on("page", events, lambda e: e["page_url_path"])
clock = EventClock(
    ts_getter=lambda e: datetime.fromisoformat(e["event_timestamp"]),
    wait_for_system_duration=timedelta(0),
)
windowed = win.collect_window(
    "window", keyed, clock,
    TumblingWindower(align_to=START, length=timedelta(seconds=20)),
)
tables = op.map("to_arrow", windowed.down, add_date_columns)
op.output("out", tables, ParquetSink())

if __name__ == "__main__":
    from bytewax_amd.testing import run_main

    run_main(flow)
    n = len(list(OUT_DIR.rglob("*.parquet")))
    print(f"wrote {n} parquet files under {OUT_DIR}/")

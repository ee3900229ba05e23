"""Count wiki edits per server over tumbling windows (reference
examples/wikistream.py).

The reference tails Wikimedia's SSE stream with an async client
bridged through `batch_async`; offline, a synthetic async generator
of the same JSON event shape feeds the identical flow: async SSE ->
batch_async -> json -> count_window per server_name.
"""

import json
import random
import sys
from datetime import datetime, timedelta, timezone
from pathlib import Path
from typing import List

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import bytewax_amd.operators as op
import bytewax_amd.operators.windowing as win
from bytewax_amd.connectors.stdio import StdOutSink
from bytewax_amd.dataflow import Dataflow
from bytewax_amd.inputs import (
    FixedPartitionedSource,
    StatefulSourcePartition,
    batch_async,
)
from bytewax_amd.operators.windowing import EventClock, TumblingWindower

SERVERS = ["en.wikipedia.org", "de.wikipedia.org", "commons.wikimedia.org"]
START = datetime(2024, 5, 1, tzinfo=timezone.utc)


async def _fake_sse(n: int = 400):
    """Synthetic recent-changes feed (same JSON shape as the SSE)."""
    rng = random.Random(23)
    for i in range(n):
        yield json.dumps(
            {
                "server_name": rng.choice(SERVERS),
                "type": "edit",
                "timestamp": int(START.timestamp()) + i // 10,
            }
        )


class WikiPartition(StatefulSourcePartition):
    def __init__(self):
        # Gather up to 0.25 s or 1000 items per batch.
        self._batcher = batch_async(
            _fake_sse(), timedelta(seconds=0.25), 1000
        )

    def next_batch(self) -> List[str]:
        return next(self._batcher)

    def snapshot(self) -> None:
        return None


class WikiSource(FixedPartitionedSource):
    def list_parts(self):
        return ["single-part"]

    def build_part(self, step_id, for_part, _resume_state):
        return WikiPartition()


flow = Dataflow("wikistream")
raw = op.input("inp", flow, WikiSource())
events = op.map("parse", raw, json.loads)
clock = EventClock(
    ts_getter=lambda e: datetime.fromtimestamp(
        e["timestamp"], tz=timezone.utc
    ),
    wait_for_system_duration=timedelta(0),
)
counts = win.count_window(
    "count_per_server",
    events,
    clock,
    TumblingWindower(align_to=START, length=timedelta(seconds=10)),
    key=lambda e: e["server_name"],
)
lines = op.map(
    "fmt", counts.down,
    lambda kv: f"{kv[0]} window {kv[1][0]}: {kv[1][1]} edits",
)
op.output("out", lines, StdOutSink())

if __name__ == "__main__":
    from bytewax_amd.testing import run_main

    run_main(flow)

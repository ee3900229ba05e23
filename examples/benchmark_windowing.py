"""The reference's windowing benchmark workload: 1M timestamps into
1-minute tumbling windows on 2 random keys (reference
examples/benchmark_windowing.py shape).

Run: python -m bytewax_amd.run examples.benchmark_windowing:flow
"""

import random
from datetime import datetime, timedelta, timezone

import bytewax_amd.operators as op
import bytewax_amd.operators.windowing as w
from bytewax_amd.connectors.stdio import StdOutSink
from bytewax_amd.dataflow import Dataflow
from bytewax_amd.operators.windowing import EventClock, TumblingWindower
from bytewax_amd.testing import TestingSource

BATCH_SIZE = 100_000
BATCH_COUNT = 10

align_to = datetime(2022, 1, 1, tzinfo=timezone.utc)
inp = [align_to + timedelta(seconds=i) for i in range(BATCH_SIZE)]

clock = EventClock(
    ts_getter=lambda x: x, wait_for_system_duration=timedelta(seconds=0)
)
windower = TumblingWindower(align_to=align_to, length=timedelta(minutes=1))


def add(acc, x):
    acc.append(x)
    return acc


flow = Dataflow("bench")
wo = (
    op.input("in", flow, TestingSource(inp, BATCH_COUNT))
    .then(op.key_on, "key-on", lambda _x: str(random.randrange(0, 2)))
    .then(w.fold_window, "fold-window", clock, windower, list, add, list.__add__)
)
flat = op.flat_map("flatten-window", wo.down, lambda xs: (y for y in xs))
filtered_out = op.filter("filter_all", flat, lambda _x: False)
op.output("stdout", filtered_out, StdOutSink())

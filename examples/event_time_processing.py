"""Event-time windowing with out-of-order and late data (reference
examples/event_time_processing.py).

Sensor readings arrive out of order; an `EventClock` with a lateness
allowance assigns them to their true (event-time) tumbling windows,
and anything behind the watermark lands on the `late` stream instead
of silently corrupting a closed window.
"""

import sys
from datetime import datetime, timedelta, timezone
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import bytewax_amd.operators as op
import bytewax_amd.operators.windowing as win
from bytewax_amd.connectors.stdio import StdOutSink
from bytewax_amd.dataflow import Dataflow
from bytewax_amd.operators.windowing import EventClock, TumblingWindower
from bytewax_amd.testing import TestingSource

START = datetime(2024, 6, 1, tzinfo=timezone.utc)


def at(s):
    return START + timedelta(seconds=s)


# (sensor, event_time, reading) — note 2 arrives after 9, and the
# final 1-second reading is far behind the watermark by then.
READINGS = [
    ("s1", at(1), 17.0),
    ("s1", at(9), 18.5),
    ("s1", at(2), 16.5),  # out of order but inside the allowance
    ("s2", at(3), 20.0),
    ("s1", at(61), 19.0),  # opens the next minute's window
    ("s2", at(62), 21.0),
    ("s1", at(1), 99.0),  # LATE: watermark has passed its window
]

flow = Dataflow("event_time")
readings = op.input("inp", flow, TestingSource(READINGS))
keyed = op.key_on("sensor", readings, lambda r: r[0])
clock = EventClock(
    ts_getter=lambda r: r[1],
    wait_for_system_duration=timedelta(seconds=0),
)
wo = win.fold_window(
    "avg_per_min",
    keyed,
    clock,
    TumblingWindower(align_to=START, length=timedelta(minutes=1)),
    lambda: (0, 0.0),
    lambda acc, r: (acc[0] + 1, acc[1] + r[2]),
    lambda a, b: (a[0] + b[0], a[1] + b[1]),
)
avgs = op.map_value(
    "avg", wo.down, lambda wid_acc: (wid_acc[0], wid_acc[1][1] / wid_acc[1][0])
)
op.output("out", avgs, StdOutSink())
late_tagged = op.map("tag_late", wo.late, lambda kv: ("LATE", kv))
op.output("late_out", late_tagged, StdOutSink())

if __name__ == "__main__":
    from bytewax_amd.testing import run_main

    run_main(flow)

import time, sys
import bytewax_amd.operators as op
from bytewax_amd.dataflow import Dataflow
from bytewax_amd.testing import TestingSink, TestingSource, run_main
from datetime import timedelta
N = 200_000
words = [f"w{i%1000}" for i in range(N)]
best = 0
for rep in range(7):
    out = []
    flow = Dataflow("wc")
    s = op.input("inp", flow, TestingSource(words, batch_size=500))
    counts = op.count_final("count", s, lambda w: w)
    op.output("out", counts, TestingSink(out))
    t0 = time.perf_counter()
    run_main(flow, epoch_interval=timedelta(days=365))
    dt = time.perf_counter() - t0
    best = max(best, N/dt/1e6)
print(f"best {best:.2f} M events/s")

# Reference CI windowing shape (codspeed fixture: 100k events,
# 1 key, 1-min tumbling fold_window).
from datetime import datetime, timezone
import bytewax_amd.operators.windowing as w

ALIGN = datetime(2024, 1, 1, tzinfo=timezone.utc)
NW = 100_000
items = [
    (ALIGN + timedelta(milliseconds=i * 10), i % 100) for i in range(NW)
]
best = 0
for _ in range(7):
    out = []
    flow = Dataflow("fw")
    s = op.input("inp", flow, TestingSource(items, batch_size=500))
    keyed = op.key_on("k", s, lambda it: "ALL")
    clock = w.EventClock(
        ts_getter=lambda it: it[0],
        wait_for_system_duration=timedelta(0),
    )
    wo = w.fold_window(
        "fold", keyed, clock,
        w.TumblingWindower(align_to=ALIGN, length=timedelta(minutes=1)),
        int, lambda acc, it: acc + it[1], lambda a, b: a + b,
    )
    op.output("out", wo.down, TestingSink(out))
    t0 = time.perf_counter()
    run_main(flow, epoch_interval=timedelta(days=365))
    best = max(best, NW / (time.perf_counter() - t0) / 1e6)
print(f"fold_window best {best:.2f} M events/s")

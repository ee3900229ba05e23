"""Size- and time-bounded batching with `op.collect` (reference
examples/batch_operator.py).

A polling source emits ~4 items/s; the first collect fills its size
limit (3 items) before the 1 s timeout, the second (over the derived
averages) hits the timeout first.
"""

import sys
from datetime import timedelta
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import bytewax_amd.operators as op
from bytewax_amd.connectors.stdio import StdOutSink
from bytewax_amd.dataflow import Dataflow
from bytewax_amd.inputs import SimplePollingSource


class CounterSource(SimplePollingSource):
    def __init__(self):
        super().__init__(interval=timedelta(seconds=0.25))
        self._it = iter(range(20))

    def next_item(self):
        return next(self._it)


flow = Dataflow("batching")
nums = op.input("inp", flow, CounterSource())
keyed = op.key_on("key", nums, lambda _x: "ALL")
by_size = op.collect(
    "batch_3_items", keyed, max_size=3, timeout=timedelta(seconds=1)
)
avgs = op.map("avg", by_size, lambda kb: sum(kb[1]) / len(kb[1]))
keyed_avgs = op.key_on("rekey", avgs, lambda _x: "ALL")
by_time = op.collect(
    "batch_avgs", keyed_avgs, max_size=10, timeout=timedelta(seconds=1)
)
lines = op.map("fmt", by_time, lambda kb: f"avg batch: {kb[1]}")
op.output("out", lines, StdOutSink())

if __name__ == "__main__":
    from bytewax_amd.testing import run_main

    run_main(flow)

"""Split one message stream into per-field keyed streams and join
them back (reference examples/split_demo.py)."""

import sys
from dataclasses import dataclass
from datetime import timedelta
from pathlib import Path
from typing import Dict

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import bytewax_amd.operators as op
from bytewax_amd.connectors.stdio import StdOutSink
from bytewax_amd.dataflow import Dataflow
from bytewax_amd.inputs import SimplePollingSource


@dataclass
class Msg:
    key: str
    val: str
    headers: Dict[str, int]
    num: int


class MsgSource(SimplePollingSource):
    def __init__(self):
        super().__init__(interval=timedelta(seconds=0.05))
        self._i = 0

    def next_item(self):
        self._i += 1
        if self._i > 6:
            raise StopIteration()
        key = "abc"[self._i % 3]
        return Msg(key, f"{key}_value", {"seq": self._i}, self._i)


flow = Dataflow("split_demo")
msgs = op.input("inp", flow, MsgSource())
vals = op.map("vals", msgs, lambda m: (m.key, m.val))
heads = op.map("headers", msgs, lambda m: (m.key, m.headers))
nums = op.map("nums", msgs, lambda m: (m.key, m.num))
together = op.join("join", vals, heads, nums)
op.output("out", together, StdOutSink())

if __name__ == "__main__":
    from bytewax_amd.testing import run_main

    run_main(flow)

"""Branch / transform / merge in one small flow (reference
examples/basic.py)."""

import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import bytewax_amd.operators as op
from bytewax_amd.connectors.stdio import StdOutSink
from bytewax_amd.dataflow import Dataflow
from bytewax_amd.testing import TestingSource

flow = Dataflow("basic")
nums = op.input("inp", flow, TestingSource(range(10)))
split = op.branch("evens_odds", nums, lambda x: x % 2 == 0)
halved = op.map("halve", split.trues, lambda x: x // 2)
doubled = op.map("double", split.falses, lambda x: x * 2)
merged = op.merge("merge", halved, doubled)
shifted = op.map("minus_one", merged, lambda x: x - 1)
tagged = op.map("tag", shifted, "<dance>{}</dance>".format)
op.output("out", tagged, StdOutSink())

if __name__ == "__main__":
    from bytewax_amd.testing import run_main

    run_main(flow)

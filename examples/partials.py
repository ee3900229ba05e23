"""Five ways to package a reusable operator step (reference
examples/partials.py): plain lambdas, named functions,
functools.partial, and the `@operator` decorator, all chained with
`Stream.then`."""

import functools
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import bytewax_amd.operators as op
from bytewax_amd.connectors.stdio import StdOutSink
from bytewax_amd.dataflow import Dataflow, Stream, operator
from bytewax_amd.testing import TestingSource


def add_one(x: int) -> int:
    return x + 1


as_lambda = lambda step_id, up: op.map(step_id, up, add_one)  # noqa: E731


def as_function(step_id, up):
    return op.map(step_id, up, add_one)


as_partial = functools.partial(op.map, mapper=add_one)


@operator
def as_operator(step_id: str, up: Stream[int]) -> Stream[int]:
    return op.map("inner", up, add_one)


flow = Dataflow("partials")
nums = op.input("inp", flow, TestingSource(range(5)))
x = nums.then(op.map, "v0", add_one)
x = x.then(as_lambda, "v1")
x = x.then(as_function, "v2")
x = x.then(as_partial, "v3")
x = x.then(as_operator, "v4")
x = op.inspect("insp", x)
op.output("out", x, StdOutSink())

if __name__ == "__main__":
    from bytewax_amd.testing import run_main

    run_main(flow)

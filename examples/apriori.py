"""Frequent item / item-pair counting over market baskets (reference
examples/apriori.py): the candidate-generation step of the Apriori
algorithm as a dataflow."""

import itertools
import sys
from pathlib import Path
from typing import List

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import bytewax_amd.operators as op
from bytewax_amd.connectors.files import FileSource
from bytewax_amd.connectors.stdio import StdOutSink
from bytewax_amd.dataflow import Dataflow

HERE = Path(__file__).resolve().parent

flow = Dataflow("apriori")
lines = op.input("inp", flow, FileSource(HERE / "sample_data" / "baskets.txt"))
baskets = op.map(
    "tokenize", lines, lambda line: [w.strip() for w in line.split(",")]
)
items = op.flatten("items", baskets)
item_counts = op.count_final("count_items", items, lambda item: item)
pairs = op.flat_map(
    "pairs", baskets, lambda basket: itertools.combinations(sorted(basket), 2)
)
pair_counts = op.count_final("count_pairs", pairs, ",".join)
op.output("out_items", item_counts, StdOutSink())
op.output("out_pairs", pair_counts, StdOutSink())

if __name__ == "__main__":
    from bytewax_amd.testing import run_main

    run_main(flow)

"""Stream rows out of a CSV file (reference examples/csv_input.py).

Each row arrives as a dict keyed by the header columns.
"""

import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import bytewax_amd.operators as op
from bytewax_amd.connectors.files import CSVSource
from bytewax_amd.connectors.stdio import StdOutSink
from bytewax_amd.dataflow import Dataflow

HERE = Path(__file__).resolve().parent

flow = Dataflow("csv_input")
stream = op.input(
    "inp", flow, CSVSource(HERE / "sample_data" / "ec2_metrics.csv")
)
op.output("out", stream, StdOutSink())

if __name__ == "__main__":
    from bytewax_amd.testing import run_main

    run_main(flow)

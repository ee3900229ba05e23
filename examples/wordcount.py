"""Classic wordcount: file lines -> tokens -> counts at EOF.

Run: python -m bytewax_amd.run examples.wordcount:flow
"""

import re

import bytewax_amd.operators as op
from bytewax_amd.connectors.files import FileSource
from bytewax_amd.connectors.stdio import StdOutSink
from bytewax_amd.dataflow import Dataflow


def lower(line):
    return line.lower()


def tokenize(line):
    return re.findall(r'[^\s!,.?":;0-9]+', line)


flow = Dataflow("wordcount")
stream = op.input("inp", flow, FileSource("examples/sample_data/wordcount.txt"))
stream = op.map("lower", stream, lower)
stream = op.flat_map("tokenize", stream, tokenize)
count_stream = op.count_final("count", stream, lambda word: word)
op.output("out", count_stream, StdOutSink())

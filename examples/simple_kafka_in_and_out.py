"""Kafka topic in, Kafka topic out (reference
examples/simple_kafka_in_and_out.py).

Requires a reachable broker and the `confluent-kafka` package
(`pip install bytewax-amd[kafka]`); poison messages split onto the
`errs` stream instead of crashing the flow.

Run: python -m bytewax_amd.run examples.simple_kafka_in_and_out:flow
"""

import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import bytewax_amd.operators as op
from bytewax_amd.connectors.kafka import operators as kop
from bytewax_amd.dataflow import Dataflow

BROKERS = ["localhost:19092"]
IN_TOPICS = ["in_topic"]
OUT_TOPIC = "out_topic"

flow = Dataflow("kafka_in_out")
kinp = kop.input("inp", flow, brokers=BROKERS, topics=IN_TOPICS)
op.inspect("inspect_errors", kinp.errs)
op.inspect("inspect_oks", kinp.oks)
kop.output("out", kinp.oks, brokers=BROKERS, topic=OUT_TOPIC)

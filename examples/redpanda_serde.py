"""Schema-registry Avro serde with Redpanda (reference
examples/redpanda_serde.py): deserialize sensor readings with plain
Avro, window per-sensor averages, serialize the aggregates back out.

Subjects expected in the registry: `sensor-key` (identifier, name),
`sensor-value` (timestamp, identifier, value) and `aggregated-value`
(identifier, avg, window_start, window_end).

Requires a reachable broker + registry and the `confluent-kafka` /
`fastavro` packages; run with
`REDPANDA_REGISTRY_URL=http://... python -m bytewax_amd.run
examples.redpanda_serde:flow`.
"""

import os
import sys
from datetime import datetime, timedelta, timezone
from pathlib import Path
from typing import Dict, List

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import bytewax_amd.operators as op
import bytewax_amd.operators.windowing as win
from bytewax_amd.connectors.kafka import KafkaSinkMessage, KafkaSourceMessage
from bytewax_amd.connectors.kafka import operators as kop
from bytewax_amd.connectors.kafka.serde import (
    PlainAvroDeserializer,
    PlainAvroSerializer,
)
from bytewax_amd.dataflow import Dataflow
from bytewax_amd.operators.windowing import SystemClock, TumblingWindower

BROKERS = os.environ.get("KAFKA_SERVER", "localhost:19092").split(";")
IN_TOPICS = os.environ.get("KAFKA_IN_TOPIC", "in-topic").split(";")
OUT_TOPIC = os.environ.get("KAFKA_OUT_TOPIC", "out_topic")
REGISTRY_URL = os.environ["REDPANDA_REGISTRY_URL"]

from confluent_kafka.schema_registry import SchemaRegistryClient

client = SchemaRegistryClient({"url": REGISTRY_URL})
key_schema = client.get_latest_version("sensor-key").schema
val_schema = client.get_latest_version("sensor-value").schema
out_val_schema = client.get_latest_version("aggregated-value").schema

flow = Dataflow("schema_registry")
kinp = kop.input("kafka-in", flow, brokers=BROKERS, topics=IN_TOPICS)
op.inspect("inspect-kafka-errors", kinp.errs).then(op.raises, "kafka-error")

# Plain Avro (no confluent wire framing): the deserializer needs the
# schema up front.
msgs = kop.deserialize(
    "de",
    kinp.oks,
    key_deserializer=PlainAvroDeserializer(schema=key_schema),
    val_deserializer=PlainAvroDeserializer(schema=val_schema),
)
op.inspect("inspect-deser", msgs.errs).then(op.raises, "deser-error")

keyed = op.key_on(
    "key_on_identifier", msgs.oks, lambda msg: msg.key["identifier"]
)


def accumulate(acc: List[int], msg: KafkaSourceMessage) -> List[int]:
    acc.append(msg.value["value"])
    return acc


windows = win.fold_window(
    "calc_avg",
    keyed,
    SystemClock(),
    TumblingWindower(
        timedelta(seconds=1), datetime(2023, 1, 1, tzinfo=timezone.utc)
    ),
    list,
    accumulate,
    list.__add__,
)


def calc_avg(key_wid_batch) -> KafkaSinkMessage[Dict, Dict]:
    key, (_wid, batch) = key_wid_batch
    return KafkaSinkMessage(
        key={"identifier": key, "name": "topic_key"},
        value={
            "identifier": key,
            "avg": sum(batch) // len(batch),
            "window_start": "",
            "window_end": "",
        },
    )


avgs = op.map("avg", windows.down, calc_avg)
op.inspect("inspect-out-data", avgs)

serialized = kop.serialize(
    "ser",
    avgs,
    key_serializer=PlainAvroSerializer(schema=key_schema),
    val_serializer=PlainAvroSerializer(schema=out_val_schema),
)
op.inspect("inspect-serialized", serialized)
kop.output("kafka-out", serialized, brokers=BROKERS, topic=OUT_TOPIC)

"""Confluent schema-registry serde (reference
examples/confluent_serde.py): same pipeline as `redpanda_serde.py`
but with Confluent Cloud auth and confluent-kafka's own Avro
(de)serializers in the Confluent wire format — our
`kop.(de)serialize` operators accept them directly (called with a
`SerializationContext`, like the reference's serde operators).

Env: CONFLUENT_URL, CONFLUENT_USERINFO, CONFLUENT_USERNAME,
CONFLUENT_PASSWORD (plus KAFKA_SERVER / KAFKA_IN_TOPIC /
KAFKA_OUT_TOPIC).  Run:
`python -m bytewax_amd.run examples.confluent_serde:flow`.
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
from bytewax_amd.dataflow import Dataflow
from bytewax_amd.operators.windowing import SystemClock, TumblingWindower

BROKERS = os.environ.get("KAFKA_SERVER", "localhost:19092").split(";")
IN_TOPICS = os.environ.get("KAFKA_IN_TOPIC", "in_topic").split(";")
OUT_TOPIC = os.environ.get("KAFKA_OUT_TOPIC", "out_topic")
CONFLUENT_URL = os.environ["CONFLUENT_URL"]
CONFLUENT_USERINFO = os.environ["CONFLUENT_USERINFO"]

from confluent_kafka.schema_registry import SchemaRegistryClient
from confluent_kafka.schema_registry.avro import (
    AvroDeserializer,
    AvroSerializer,
)

# SASL credentials for both the consumer and the producer.
add_config = {
    "security.protocol": "SASL_SSL",
    "sasl.mechanism": "PLAIN",
    "sasl.username": os.environ["CONFLUENT_USERNAME"],
    "sasl.password": os.environ["CONFLUENT_PASSWORD"],
}

client = SchemaRegistryClient(
    {"url": CONFLUENT_URL, "basic.auth.user.info": CONFLUENT_USERINFO}
)

flow = Dataflow("schema_registry")
kinp = kop.input(
    "kafka-in",
    flow,
    brokers=BROKERS,
    topics=IN_TOPICS,
    add_config=add_config,
)
op.inspect("inspect-kafka-errors", kinp.errs).then(op.raises, "kafka-error")

# Confluent's deserializer fetches the writer schema from the
# registry by the id embedded in the wire format.
msgs = kop.deserialize(
    "de",
    kinp.oks,
    key_deserializer=AvroDeserializer(client),
    val_deserializer=AvroDeserializer(client),
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

key_schema = client.get_latest_version("sensor-key").schema
out_val_schema = client.get_latest_version("aggregated-value").schema
serialized = kop.serialize(
    "ser",
    avgs,
    key_serializer=AvroSerializer(client, key_schema.schema_str),
    val_serializer=AvroSerializer(client, out_val_schema.schema_str),
)
op.inspect("inspect-serialized", serialized)
kop.output("kafka-out", serialized, brokers=BROKERS, topic=OUT_TOPIC)

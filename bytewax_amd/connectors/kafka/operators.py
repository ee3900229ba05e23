"""Stream-native operators for Kafka.

Parity target: ``bytewax.connectors.kafka.operators`` (reference
kafka/operators.py:49-434): ``kop.input`` splitting oks/errs,
``kop.output``, and the (de)serialization operators.
"""

from dataclasses import dataclass
from typing import Dict, List, Optional, TypeVar

import bytewax_amd.operators as op
from ...dataflow import Dataflow, Stream, operator
from . import KafkaError, KafkaSink, KafkaSinkMessage, KafkaSource, KafkaSourceMessage
from .serde import SchemaDeserializer, SchemaSerializer

K = TypeVar("K")
V = TypeVar("V")

__all__ = [
    "KafkaOpOut",
    "deserialize",
    "deserialize_key",
    "deserialize_value",
    "input",
    "output",
    "serialize",
    "serialize_key",
    "serialize_value",
]


@dataclass(frozen=True)
class KafkaOpOut:
    """Streams returned from Kafka input operators."""

    oks: Stream
    """Successfully processed items."""
    errs: Stream
    """Errors."""


@operator
def _kafka_error_split(step_id: str, up: Stream) -> KafkaOpOut:
    """Split a stream of oks and errors."""
    b = op.branch("branch", up, lambda msg: isinstance(msg, KafkaError))
    return KafkaOpOut(oks=b.falses, errs=b.trues)


@operator
def input(  # noqa: A001
    step_id: str,
    flow: Dataflow,
    *,
    brokers: List[str],
    topics: List[str],
    tail: bool = True,
    starting_offset: Optional[int] = None,
    add_config: Optional[Dict[str, str]] = None,
    batch_size: int = 1000,
) -> KafkaOpOut:
    """Consume from Kafka as an input source.

    Errors are routed to the `errs` stream instead of crashing.
    """
    source = KafkaSource(
        brokers,
        topics,
        tail,
        starting_offset,
        add_config,
        batch_size,
        raise_on_errors=False,
    )
    msgs = op.input("kafka_input", flow, source)
    return _kafka_error_split("split_err", msgs)


@operator
def output(
    step_id: str,
    up: Stream,
    *,
    brokers: List[str],
    topic: str,
    add_config: Optional[Dict[str, str]] = None,
) -> None:
    """Produce to Kafka as an output sink."""
    return op.output(
        "kafka_output", up, KafkaSink(brokers, topic, add_config)
    )


@operator
def deserialize_key(
    step_id: str,
    up: Stream,
    deserializer: SchemaDeserializer,
) -> KafkaOpOut:
    """Deserialize message keys; failures go to the `errs` stream."""

    def shim_mapper(msg: KafkaSourceMessage):
        try:
            return msg._with_key(deserializer.de(msg.key))
        except Exception as ex:  # noqa: BLE001
            return KafkaError(ex, msg)

    mapped = op.map("map", up, shim_mapper)
    return _kafka_error_split("split", mapped)


@operator
def deserialize_value(
    step_id: str,
    up: Stream,
    deserializer: SchemaDeserializer,
) -> KafkaOpOut:
    """Deserialize message values; failures go to the `errs` stream."""

    def shim_mapper(msg: KafkaSourceMessage):
        try:
            return msg._with_value(deserializer.de(msg.value))
        except Exception as ex:  # noqa: BLE001
            return KafkaError(ex, msg)

    mapped = op.map("map", up, shim_mapper)
    return _kafka_error_split("split", mapped)


@operator
def deserialize(
    step_id: str,
    up: Stream,
    *,
    key_deserializer: SchemaDeserializer,
    val_deserializer: SchemaDeserializer,
) -> KafkaOpOut:
    """Deserialize both keys and values."""

    def shim_mapper(msg: KafkaSourceMessage):
        try:
            key = key_deserializer.de(msg.key)
            value = val_deserializer.de(msg.value)
            return msg._with_key_and_value(key, value)
        except Exception as ex:  # noqa: BLE001
            return KafkaError(ex, msg)

    mapped = op.map("map", up, shim_mapper)
    return _kafka_error_split("split", mapped)


@operator
def serialize_key(
    step_id: str,
    up: Stream,
    serializer: SchemaSerializer,
) -> Stream:
    """Serialize message keys; crashes on failure."""

    def shim_mapper(msg):
        return msg._with_key(serializer.ser(msg.key))

    return op.map("map", up, shim_mapper)


@operator
def serialize_value(
    step_id: str,
    up: Stream,
    serializer: SchemaSerializer,
) -> Stream:
    """Serialize message values; crashes on failure."""

    def shim_mapper(msg):
        return msg._with_value(serializer.ser(msg.value))

    return op.map("map", up, shim_mapper)


@operator
def serialize(
    step_id: str,
    up: Stream,
    *,
    key_serializer: SchemaSerializer,
    val_serializer: SchemaSerializer,
) -> Stream:
    """Serialize both keys and values."""

    def shim_mapper(msg):
        return msg._with_key_and_value(
            key_serializer.ser(msg.key), val_serializer.ser(msg.value)
        )

    return op.map("map", up, shim_mapper)

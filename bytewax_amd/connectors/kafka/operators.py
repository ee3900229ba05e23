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


def _de_fn(deserializer, field: str):
    """Adapt a deserializer to `fn(msg, payload) -> obj`.

    Accepts our :class:`SchemaDeserializer` or a confluent-kafka
    style callable invoked as ``deserializer(payload, ctx)`` with a
    `SerializationContext` (the reference's serde operators call
    confluent (de)serializers that way; kafka/operators.py:225-429).
    """
    if isinstance(deserializer, SchemaDeserializer):
        return lambda _msg, payload: deserializer.de(payload)
    if callable(deserializer):

        def fn(msg, payload):
            try:
                from confluent_kafka.serialization import (
                    MessageField,
                    SerializationContext,
                )

                ctx = SerializationContext(
                    msg.topic, getattr(MessageField, field)
                )
            except ImportError:
                ctx = None
            return deserializer(payload, ctx)

        return fn
    msg = (
        "expected a SchemaDeserializer or confluent-kafka style "
        f"callable; got {type(deserializer)!r}"
    )
    raise TypeError(msg)


def _ser_fn(serializer, field: str):
    """Adapt a serializer to `fn(msg, obj) -> payload` (see _de_fn)."""
    if isinstance(serializer, SchemaSerializer):
        return lambda _msg, obj: serializer.ser(obj)
    if callable(serializer):

        def fn(msg, obj):
            try:
                from confluent_kafka.serialization import (
                    MessageField,
                    SerializationContext,
                )

                ctx = SerializationContext(
                    msg.topic, getattr(MessageField, field)
                )
            except ImportError:
                ctx = None
            return serializer(obj, ctx)

        return fn
    msg = (
        "expected a SchemaSerializer or confluent-kafka style "
        f"callable; got {type(serializer)!r}"
    )
    raise TypeError(msg)


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

    de = _de_fn(deserializer, "KEY")

    def shim_mapper(msg: KafkaSourceMessage):
        try:
            return msg._with_key(de(msg, msg.key))
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

    de = _de_fn(deserializer, "VALUE")

    def shim_mapper(msg: KafkaSourceMessage):
        try:
            return msg._with_value(de(msg, msg.value))
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

    kde = _de_fn(key_deserializer, "KEY")
    vde = _de_fn(val_deserializer, "VALUE")

    def shim_mapper(msg: KafkaSourceMessage):
        try:
            key = kde(msg, msg.key)
            value = vde(msg, msg.value)
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

    ser = _ser_fn(serializer, "KEY")

    def shim_mapper(msg):
        return msg._with_key(ser(msg, msg.key))

    return op.map("map", up, shim_mapper)


@operator
def serialize_value(
    step_id: str,
    up: Stream,
    serializer: SchemaSerializer,
) -> Stream:
    """Serialize message values; crashes on failure."""

    ser = _ser_fn(serializer, "VALUE")

    def shim_mapper(msg):
        return msg._with_value(ser(msg, msg.value))

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

    kser = _ser_fn(key_serializer, "KEY")
    vser = _ser_fn(val_serializer, "VALUE")

    def shim_mapper(msg):
        return msg._with_key_and_value(
            kser(msg, msg.key), vser(msg, msg.value)
        )

    return op.map("map", up, shim_mapper)

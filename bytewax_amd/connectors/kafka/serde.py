"""Serializers / deserializers for Kafka message payloads.

Parity target: ``bytewax.connectors.kafka.serde`` (reference
kafka/serde.py:15-105).  Avro support requires ``fastavro`` (imported
lazily).
"""

import io
from abc import ABC, abstractmethod
from typing import Any, Dict, Generic, TypeVar

A = TypeVar("A")
B = TypeVar("B")

__all__ = [
    "PlainAvroDeserializer",
    "PlainAvroSerializer",
    "SchemaDeserializer",
    "SchemaSerializer",
]


class SchemaSerializer(ABC, Generic[A, B]):
    """A serializer for a specific schema."""

    @abstractmethod
    def ser(self, obj: A) -> B:
        """Serialize an object."""
        ...


class SchemaDeserializer(ABC, Generic[A, B]):
    """A deserializer for a specific schema."""

    @abstractmethod
    def de(self, data: A) -> B:
        """Deserialize data."""
        ...


class PlainAvroSerializer(SchemaSerializer[Dict, bytes]):
    """Serialize Python dictionaries to plain Avro binary (without the
    confluent wire format's magic byte + schema id header).

    :arg schema: Avro schema (parsed or dict).
    """

    def __init__(self, schema: Any):
        import fastavro

        if isinstance(schema, (dict, list, str)):
            schema = fastavro.parse_schema(schema)
        self.schema = schema

    def ser(self, obj: Dict) -> bytes:
        import fastavro

        bytes_writer = io.BytesIO()
        fastavro.schemaless_writer(bytes_writer, self.schema, obj)
        return bytes_writer.getvalue()


class PlainAvroDeserializer(SchemaDeserializer[bytes, Dict]):
    """Deserialize plain Avro binary to Python dictionaries.

    :arg schema: Avro schema (parsed or dict).
    """

    def __init__(self, schema: Any):
        import fastavro

        if isinstance(schema, (dict, list, str)):
            schema = fastavro.parse_schema(schema)
        self.schema = schema

    def de(self, data: bytes) -> Dict:
        import fastavro

        return fastavro.schemaless_reader(io.BytesIO(data), self.schema)

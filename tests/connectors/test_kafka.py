"""Kafka connector tests (no broker needed; parity with reference
pytests/connectors/test_kafka.py which also runs broker-less)."""

import pytest

import bytewax_amd.operators as op
from bytewax_amd.connectors.kafka import (
    KafkaError,
    KafkaSinkMessage,
    KafkaSourceMessage,
)
from bytewax_amd.connectors.kafka import operators as kop
from bytewax_amd.connectors.kafka.serde import (
    SchemaDeserializer,
    SchemaSerializer,
)
from bytewax_amd.dataflow import Dataflow
from bytewax_amd.testing import TestingSink, TestingSource, run_main


def test_source_message_to_sink():
    src = KafkaSourceMessage(
        key=b"k", value=b"v", topic="t", offset=3, partition=1
    )
    snk = src.to_sink()
    assert isinstance(snk, KafkaSinkMessage)
    assert snk.key == b"k"
    assert snk.value == b"v"


def test_with_key_and_value():
    src = KafkaSourceMessage(key=b"k", value=b"v", topic="t")
    m2 = src._with_key_and_value("K", 42)
    assert m2.key == "K"
    assert m2.value == 42
    assert m2.topic == "t"


class _UpperDe(SchemaDeserializer):
    def de(self, data):
        return data.upper()


class _FailDe(SchemaDeserializer):
    def de(self, data):
        raise ValueError("nope")


class _BangSer(SchemaSerializer):
    def ser(self, obj):
        return obj + b"!"


def test_deserialize_value_ok_and_err():
    msgs = [
        KafkaSourceMessage(key=b"a", value=b"hello"),
        KafkaSourceMessage(key=b"b", value=b"world"),
    ]
    oks, errs = [], []
    flow = Dataflow("f")
    s = op.input("inp", flow, TestingSource(msgs))
    out = kop.deserialize_value("de", s, _UpperDe())
    op.output("ok", out.oks, TestingSink(oks))
    op.output("err", out.errs, TestingSink(errs))
    run_main(flow)
    assert sorted(m.value for m in oks) == [b"HELLO", b"WORLD"]
    assert errs == []


def test_deserialize_value_routes_errors():
    msgs = [KafkaSourceMessage(key=b"a", value=b"x")]
    oks, errs = [], []
    flow = Dataflow("f")
    s = op.input("inp", flow, TestingSource(msgs))
    out = kop.deserialize_value("de", s, _FailDe())
    op.output("ok", out.oks, TestingSink(oks))
    op.output("err", out.errs, TestingSink(errs))
    run_main(flow)
    assert oks == []
    assert len(errs) == 1
    assert isinstance(errs[0], KafkaError)


def test_serialize_key():
    msgs = [KafkaSinkMessage(key=b"a", value=b"1")]
    out = []
    flow = Dataflow("f")
    s = op.input("inp", flow, TestingSource(msgs))
    ser = kop.serialize_key("ser", s, _BangSer())
    op.output("out", ser, TestingSink(out))
    run_main(flow)
    assert out[0].key == b"a!"


def test_kafka_source_requires_confluent():
    try:
        import confluent_kafka  # noqa: F401

        pytest.skip("confluent-kafka installed")
    except ImportError:
        pass
    from bytewax_amd.connectors.kafka import KafkaSource

    with pytest.raises(ImportError):
        KafkaSource(["localhost:9092"], ["topic"])


def test_serde_accepts_confluent_style_callables():
    """The (de)serialize operators accept confluent-kafka style
    callables — invoked as fn(payload, ctx) — in addition to our
    SchemaSerializer/SchemaDeserializer ABCs (reference
    kafka/operators.py:225-429 calls confluent serializers that
    way).  Without confluent-kafka installed ctx is None."""
    msgs = [
        KafkaSourceMessage(key=b"7", value=b"70", topic="t"),
        KafkaSourceMessage(key=b"8", value=b"80", topic="t"),
    ]
    seen_ctx = []

    def de_int(payload, ctx):
        seen_ctx.append(ctx)
        return int(payload)

    out = []
    flow = Dataflow("confluent_style")
    s = op.input("inp", flow, TestingSource(msgs))
    des = kop.deserialize(
        "de", s, key_deserializer=de_int, val_deserializer=de_int
    )
    ser = kop.serialize_value(
        "ser", des.oks, lambda obj, ctx: str(obj * 2).encode()
    )
    op.output("out", ser, TestingSink(out))
    run_main(flow)
    assert [(m.key, m.value) for m in out] == [(7, b"140"), (8, b"160")]
    try:
        from confluent_kafka.serialization import SerializationContext

        assert all(isinstance(c, SerializationContext) for c in seen_ctx)
    except ImportError:
        assert seen_ctx == [None] * 4


def test_serde_rejects_non_callables():
    flow = Dataflow("bad_serde")
    s = op.input("inp", flow, TestingSource([]))
    with pytest.raises(TypeError, match="SchemaDeserializer"):
        kop.deserialize_value("de", s, 42)

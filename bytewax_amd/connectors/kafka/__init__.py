"""Connectors for [Kafka](https://kafka.apache.org).

Parity target: ``bytewax.connectors.kafka`` (reference
connectors/kafka/__init__.py): `KafkaSource` / `KafkaSink` with
manual-offset partition management (consumer groups disabled; the
snapshot is the next offset so recovery is exactly-once),
`KafkaSourceMessage` / `KafkaSinkMessage` dataclasses, a `KafkaError`
passthrough stream, and a consumer-lag prometheus gauge fed by
librdkafka's stats callback.

Requires ``confluent-kafka`` at runtime (imported lazily so the rest
of the framework works without it); the message dataclasses and the
stream-native operators in :mod:`bytewax_amd.connectors.kafka.operators`
are importable and testable without a broker.
"""

from dataclasses import dataclass, field
from typing import Any, Dict, Generic, List, Optional, Tuple, TypeVar

from ...inputs import FixedPartitionedSource, StatefulSourcePartition
from ...outputs import DynamicSink, StatelessSinkPartition

K = TypeVar("K")
V = TypeVar("V")
K2 = TypeVar("K2")
V2 = TypeVar("V2")

__all__ = [
    "KafkaError",
    "KafkaSink",
    "KafkaSinkMessage",
    "KafkaSource",
    "KafkaSourceMessage",
]

_BATCH_SIZE_DEFAULT = 1000


def _require_confluent():
    try:
        import confluent_kafka  # noqa: F401

        return confluent_kafka
    except ImportError as ex:
        msg = (
            "KafkaSource/KafkaSink require the `confluent-kafka` package; "
            "pip install bytewax-amd[kafka]"
        )
        raise ImportError(msg) from ex


@dataclass(frozen=True)
class KafkaSourceMessage(Generic[K, V]):
    """Message read from Kafka."""

    key: K
    value: V
    topic: Optional[str] = field(default=None)
    headers: List[Tuple[str, bytes]] = field(default_factory=list)
    latency: Optional[float] = field(default=None)
    offset: Optional[int] = field(default=None)
    partition: Optional[int] = field(default=None)
    timestamp: Optional[Tuple[int, int]] = field(default=None)

    def to_sink(self) -> "KafkaSinkMessage[K, V]":
        """Convert to a sink message, keeping key, value, headers.

        Consumed metadata (topic, offset, partition, timestamp) is
        dropped: the sink assigns its own.

        Example:
            >>> from bytewax_amd.connectors.kafka import KafkaSourceMessage
            >>> msg = KafkaSourceMessage(
            ...     key=b"k", value=b"v", topic="in", offset=17
            ... )
            >>> msg.to_sink()
            KafkaSinkMessage(key=b'k', value=b'v', topic=None, headers=[], partition=None, timestamp=0)
        """
        return KafkaSinkMessage(
            key=self.key, value=self.value, headers=self.headers
        )

    def _replace_kv(self, key, value) -> "KafkaSourceMessage":
        return KafkaSourceMessage(
            key=key,
            value=value,
            topic=self.topic,
            headers=self.headers,
            latency=self.latency,
            offset=self.offset,
            partition=self.partition,
            timestamp=self.timestamp,
        )

    def _with_key(self, key: K2) -> "KafkaSourceMessage[K2, V]":
        return self._replace_kv(key, self.value)

    def _with_value(self, value: V2) -> "KafkaSourceMessage[K, V2]":
        return self._replace_kv(self.key, value)

    def _with_key_and_value(
        self, key: K2, value: V2
    ) -> "KafkaSourceMessage[K2, V2]":
        return self._replace_kv(key, value)


@dataclass(frozen=True)
class KafkaError(Generic[K, V]):
    """Error from a {py:obj}`KafkaSource`."""

    err: Any
    """Underlying error from the consumer."""
    msg: KafkaSourceMessage[K, V]
    """Message attached to that error."""


@dataclass(frozen=True)
class KafkaSinkMessage(Generic[K, V]):
    """Message to be written to Kafka."""

    key: K
    value: V
    topic: Optional[str] = None
    headers: List[Tuple[str, bytes]] = field(default_factory=list)
    partition: Optional[int] = None
    timestamp: int = 0

    def _replace_kv(self, key, value) -> "KafkaSinkMessage":
        return KafkaSinkMessage(
            key=key,
            value=value,
            topic=self.topic,
            headers=self.headers,
            partition=self.partition,
            timestamp=self.timestamp,
        )

    def _with_key(self, key: K2) -> "KafkaSinkMessage[K2, V]":
        return self._replace_kv(key, self.value)

    def _with_value(self, value: V2) -> "KafkaSinkMessage[K, V2]":
        return self._replace_kv(self.key, value)

    def _with_key_and_value(
        self, key: K2, value: V2
    ) -> "KafkaSinkMessage[K2, V2]":
        return self._replace_kv(key, value)


def _consumer_lag_gauge():
    try:
        from prometheus_client import Gauge

        return Gauge(
            "bytewax_kafka_consumer_lag",
            "Difference between last queued offset and last committed "
            "offset per topic partition",
            ["topic", "partition"],
        )
    except Exception:  # pragma: no cover
        return None


class _KafkaSourcePartition(
    StatefulSourcePartition[Any, Optional[int]]
):
    def __init__(
        self,
        step_id: str,
        consumer,
        topic: str,
        part_idx: int,
        starting_offset: int,
        resume_state: Optional[int],
        batch_size: int,
        raise_on_errors: bool,
    ):
        import confluent_kafka as ck

        self._step_id = step_id
        self._offset = (
            resume_state if resume_state is not None else starting_offset
        )
        self._consumer = consumer
        consumer.assign([ck.TopicPartition(topic, part_idx, self._offset)])
        self._topic = topic
        self._batch_size = batch_size
        self._eof = False
        self._raise_on_errors = raise_on_errors

    def next_batch(self) -> List[Any]:
        import confluent_kafka as ck

        if self._eof:
            raise StopIteration()
        msgs = self._consumer.consume(self._batch_size, 0.001)
        batch: List[Any] = []
        last_offset = None
        for msg in msgs:
            error = msg.error()
            if error is not None:
                if error.code() == ck.KafkaError._PARTITION_EOF:
                    self._eof = True
                    break
                elif self._raise_on_errors:
                    msg_s = (
                        f"error consuming from Kafka topic `{self._topic!r}`: "
                        f"{error!r}"
                    )
                    raise RuntimeError(msg_s)
            ksm = KafkaSourceMessage(
                key=msg.key(),
                value=msg.value(),
                topic=msg.topic(),
                headers=msg.headers() or [],
                latency=msg.latency(),
                offset=msg.offset(),
                partition=msg.partition(),
                timestamp=msg.timestamp(),
            )
            item = ksm if error is None else KafkaError(error, ksm)
            batch.append(item)
            # Error events can report offset -1/None; advancing (or
            # rewinding) from those would re-deliver already-emitted
            # messages after a resume.
            off = msg.offset()
            if off is not None and off >= 0:
                last_offset = off
        if last_offset is not None:
            self._offset = last_offset + 1
        return batch

    def snapshot(self) -> Optional[int]:
        return self._offset

    def close(self) -> None:
        self._consumer.close()


class KafkaSource(FixedPartitionedSource[Any, Optional[int]]):
    """Use a set of Kafka topics as an input source.

    Partitions are the unit of parallelism (keyed
    ``"{partition_idx}-{topic}"``).  Consumer groups are disabled;
    offsets are managed manually and snapshotted for exactly-once
    resume.  Can support exactly-once processing.

    :arg brokers: List of `host:port` broker strings.
    :arg topics: List of topics.
    :arg tail: Whether to wait for new data on EOF (defaults True).
    :arg starting_offset: `OFFSET_BEGINNING` (default) or `OFFSET_END`.
    :arg add_config: Extra `librdkafka` config.
    :arg batch_size: Max messages per batch (default 1000).
    :arg raise_on_errors: Crash on Kafka errors (default) or pass
        {py:obj}`KafkaError` items downstream.
    """

    def __init__(
        self,
        brokers: List[str],
        topics: List[str],
        tail: bool = True,
        starting_offset: Optional[int] = None,
        add_config: Optional[Dict[str, str]] = None,
        batch_size: int = _BATCH_SIZE_DEFAULT,
        raise_on_errors: bool = True,
    ):
        ck = _require_confluent()
        if isinstance(brokers, str):
            msg = "brokers must be an iterable and not a string"
            raise TypeError(msg)
        if isinstance(topics, str):
            msg = "topics must be an iterable and not a string"
            raise TypeError(msg)
        self._brokers = brokers
        self._topics = topics
        self._tail = tail
        self._starting_offset = (
            starting_offset
            if starting_offset is not None
            else ck.OFFSET_BEGINNING
        )
        self._add_config = dict(add_config or {})
        self._batch_size = batch_size
        self._raise_on_errors = raise_on_errors
        self._lag_gauge = _consumer_lag_gauge()

    def list_parts(self) -> List[str]:
        from confluent_kafka.admin import AdminClient

        config = {"bootstrap.servers": ",".join(self._brokers)}
        config.update(self._add_config)
        client = AdminClient(config)
        parts = []
        cluster_metadata = client.list_topics(timeout=10)
        for topic in self._topics:
            topic_metadata = cluster_metadata.topics[topic]
            if topic_metadata.error is not None:
                msg = (
                    f"error listing partitions for Kafka topic `{topic!r}`: "
                    f"{topic_metadata.error.str()}"
                )
                raise RuntimeError(msg)
            for i in topic_metadata.partitions.keys():
                parts.append(f"{i}-{topic}")
        return sorted(parts)

    def build_part(
        self, step_id: str, for_part: str, resume_state: Optional[int]
    ) -> _KafkaSourcePartition:
        import json

        from confluent_kafka import Consumer

        idx, topic = for_part.split("-", 1)
        gauge = self._lag_gauge

        def stats_cb(json_stats: str):
            if gauge is None:
                return
            stats = json.loads(json_stats)
            for t in stats.get("topics", {}).values():
                for p_id, p in t.get("partitions", {}).items():
                    if int(p_id) >= 0:
                        gauge.labels(t["topic"], p_id).set(
                            p.get("consumer_lag", 0)
                        )

        config = {
            "bootstrap.servers": ",".join(self._brokers),
            # Consumer group magic is disabled: assignment is manual
            # and offsets are snapshotted by the recovery engine.
            "group.id": f"BYTEWAX_IGNORED-{step_id}",
            "enable.auto.commit": "false",
            "enable.partition.eof": str(not self._tail),
            "statistics.interval.ms": "5000",
            "stats_cb": stats_cb,
        }
        config.update(self._add_config)
        consumer = Consumer(config)
        return _KafkaSourcePartition(
            step_id,
            consumer,
            topic,
            int(idx),
            self._starting_offset,
            resume_state,
            self._batch_size,
            self._raise_on_errors,
        )


class _KafkaSinkPartition(StatelessSinkPartition[Any]):
    def __init__(self, producer, topic: Optional[str]):
        self._producer = producer
        self._topic = topic

    def write_batch(self, items: List[KafkaSinkMessage]) -> None:
        for msg in items:
            topic = msg.topic if msg.topic is not None else self._topic
            if topic is None:
                err = f"no topic to write to for message {msg!r}"
                raise RuntimeError(err)
            kwargs = {}
            if msg.partition is not None:
                kwargs["partition"] = msg.partition
            self._producer.produce(
                topic,
                msg.value,
                msg.key,
                headers=msg.headers,
                **kwargs,
            )
            self._producer.poll(0)
        self._producer.flush()

    def close(self) -> None:
        self._producer.flush()


class KafkaSink(DynamicSink[Any]):
    """Use a single Kafka topic as an output sink.

    Items consumed from the dataflow must be
    {py:obj}`KafkaSinkMessage` with both `key` and `value` as bytes.
    Workers are the unit of parallelism; can support at-least-once
    processing.

    :arg brokers: List of `host:port` broker strings.
    :arg topic: Topic to produce to; `None` requires each message to
        carry its own topic.
    :arg add_config: Extra `librdkafka` config.
    """

    def __init__(
        self,
        brokers: List[str],
        topic: Optional[str],
        add_config: Optional[Dict[str, str]] = None,
    ):
        _require_confluent()
        self._brokers = brokers
        self._topic = topic
        self._add_config = dict(add_config or {})

    def build(
        self, step_id: str, worker_index: int, worker_count: int
    ) -> _KafkaSinkPartition:
        from confluent_kafka import Producer

        config = {"bootstrap.servers": ",".join(self._brokers)}
        config.update(self._add_config)
        return _KafkaSinkPartition(Producer(config), self._topic)

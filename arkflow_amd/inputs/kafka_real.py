"""Real Kafka driver: confluent_kafka consumer + transactional producer.

Mirrors the reference's rdkafka usage:
  - input (input/kafka.rs:183-296): per-message poll; auto OFFSET STORE
    disabled so only acked messages are ever committed (crash safety); ack =
    store_offsets(offset+1). The group commit timer then commits stored
    offsets (enable.auto.commit stays on, exactly rdkafka's store_offset
    pattern).
  - output (output/kafka.rs:348-446): exactly-once via a transactional
    producer — ``write_batch`` = init'd transaction → begin → produce all →
    commit, with fencing-aware error mapping: fatal/fenced errors raise
    ConnectionError_ (a newer producer with the same transactional.id owns
    the log); abortable errors abort the txn and raise ProcessError so the
    ack is withheld and the batch replays.

confluent_kafka (librdkafka) is not installed in the build image; the driver
activates when it is importable AND the config names real brokers. The
test suite runs the same contract body against this driver when
``KAFKA_BOOTSTRAP`` is set (tests/test_brokers.py), the way the reference
gates its testcontainers suites (kafka_eos.rs:29-33).
"""
from __future__ import annotations

import asyncio
import time
from typing import List, Optional, Tuple

from ..batch import Column, MessageBatch
from ..errors import ConnectionError_, EOFError_, ProcessError
from ..spi import Ack


def _import_client():
    try:
        import confluent_kafka  # type: ignore
        return confluent_kafka
    except ImportError as e:  # pragma: no cover - env without librdkafka
        raise ConnectionError_(
            "kafka real driver requires confluent_kafka; install it or use "
            "driver: memory for the in-process bus") from e


class RealKafkaAck(Ack):
    """store_offsets(offset+1): marks the message consumed for the next
    auto-commit tick. Kafka's offset model cannot express ack gaps —
    identical caveat to the fake driver's KafkaAck."""

    def __init__(self, consumer, topic: str, partition: int, offset: int):
        self.consumer = consumer
        self.topic, self.partition, self.offset = topic, partition, offset

    async def ack(self) -> None:
        ck = _import_client()
        self.consumer.store_offsets(offsets=[
            ck.TopicPartition(self.topic, self.partition, self.offset + 1)])


class RealKafkaConsumer:
    """Driver object behind KafkaInput when ``driver != memory``."""

    def __init__(self, brokers: List[str], topics: List[str], group: str,
                 config: Optional[dict] = None):
        self.brokers = brokers
        self.topics = topics
        self.group = group
        self.extra = dict(config or {})
        self.consumer = None
        self._closed = False

    def connect(self) -> None:
        ck = _import_client()
        conf = {
            "bootstrap.servers": ",".join(self.brokers),
            "group.id": self.group,
            # the reference disables automatic offset STORE: only explicitly
            # acked messages are committed (input/kafka.rs crash safety)
            "enable.auto.commit": True,
            "enable.auto.offset.store": False,
            "auto.offset.reset": "earliest",
            # see only committed rows of EOS producers
            "isolation.level": "read_committed",
            **self.extra,
        }
        try:
            self.consumer = ck.Consumer(conf)
            self.consumer.subscribe(self.topics)
        except Exception as e:  # noqa: BLE001
            raise ConnectionError_(f"kafka connect failed: {e}") from e

    async def read(self, codec=None) -> Tuple[MessageBatch, Ack]:
        ck = _import_client()
        loop = asyncio.get_running_loop()
        while True:
            if self._closed:
                raise EOFError_("kafka input closed")
            msg = await loop.run_in_executor(None, self.consumer.poll, 0.25)
            if msg is None:
                continue
            err = msg.error()
            if err is not None:
                if err.code() == ck.KafkaError._PARTITION_EOF:
                    continue
                if err.fatal():
                    raise ConnectionError_(f"kafka consumer fatal: {err}")
                continue  # transient errors: keep polling
            ts_type, ts = msg.timestamp()
            batch = MessageBatch(
                {
                    "__value__": Column.from_bytes([msg.value() or b""]),
                    "__meta_source": Column.from_strings([msg.topic()]),
                    "__meta_partition": Column.from_numeric(
                        [msg.partition()]),
                    "__meta_offset": Column.from_numeric([msg.offset()]),
                    "__meta_key": Column.from_bytes([msg.key() or b""]),
                    "__meta_timestamp": Column.from_numeric(
                        [ts / 1000.0 if ts_type != 0 else time.time()]),
                },
                input_name=msg.topic(),
            )
            if codec is not None:
                from ..codecs.helper import apply_codec
                batch = apply_codec(batch, codec)
            return batch, RealKafkaAck(self.consumer, msg.topic(),
                                       msg.partition(), msg.offset())

    def close(self) -> None:
        self._closed = True
        if self.consumer is not None:
            try:
                self.consumer.commit(asynchronous=False)  # flush stored acks
            except Exception:  # noqa: BLE001
                pass
            self.consumer.close()


class RealKafkaProducer:
    """Driver object behind KafkaOutput when ``driver != memory``.

    exactly_once=True → transactional producer; ``write_txn`` is one
    begin/produce-all/commit unit with the reference's fencing-aware error
    mapping (output/kafka.rs:348-446).
    """

    def __init__(self, brokers: List[str], exactly_once: bool,
                 transactional_id: Optional[str] = None,
                 compression: Optional[str] = None,
                 config: Optional[dict] = None):
        self.brokers = brokers
        self.exactly_once = exactly_once
        self.transactional_id = transactional_id
        self.compression = compression
        self.extra = dict(config or {})
        self.producer = None

    def connect(self) -> None:
        ck = _import_client()
        conf = {"bootstrap.servers": ",".join(self.brokers), **self.extra}
        if self.compression:
            conf["compression.type"] = self.compression
        if self.exactly_once:
            if not self.transactional_id:
                raise ConnectionError_(
                    "exactly_once kafka output requires transactional_id")
            conf["transactional.id"] = self.transactional_id
            conf["enable.idempotence"] = True
        try:
            self.producer = ck.Producer(conf)
            if self.exactly_once:
                self.producer.init_transactions(30.0)
        except Exception as e:  # noqa: BLE001
            raise self._map_error(e, "init_transactions")

    def _map_error(self, e, where: str):
        """Reference error taxonomy: fenced/fatal → ConnectionError_
        (unrecoverable — a newer producer owns this transactional.id);
        everything else → ProcessError (ack withheld, batch replays)."""
        ck = None
        try:
            ck = _import_client()
        except ConnectionError_:
            pass
        kerr = getattr(e, "args", [None])[0]
        if ck is not None and isinstance(kerr, ck.KafkaError):
            if kerr.fatal() or kerr.code() == ck.KafkaError._FENCED:
                return ConnectionError_(
                    f"kafka producer fenced/fatal during {where}: {kerr}")
        return ProcessError(f"kafka {where} failed: {e}")

    def produce_rows(self, topic: str, rows) -> None:
        """rows: (key, value) pairs for the fixed topic, or
        (topic, key, value) triples when the topic is a per-row expr."""
        for row in rows:
            t, key, value = row if len(row) == 3 else (topic, *row)
            self.producer.produce(t, value=value, key=key)
            self.producer.poll(0)

    def write_plain(self, topic: str, rows) -> None:
        try:
            self.produce_rows(topic, rows)
            self.producer.flush(30.0)  # delivery before the engine acks
        except Exception as e:  # noqa: BLE001
            raise self._map_error(e, "produce")

    def write_txn(self, topic: str, rows) -> None:
        ck = _import_client()
        try:
            self.producer.begin_transaction()
        except Exception as e:  # noqa: BLE001
            raise self._map_error(e, "begin_transaction")
        try:
            self.produce_rows(topic, rows)
            self.producer.commit_transaction(30.0)
        except ck.KafkaException as e:
            kerr = e.args[0]
            if kerr.fatal() or kerr.code() == ck.KafkaError._FENCED:
                raise ConnectionError_(
                    f"kafka producer fenced/fatal: {kerr}") from e
            if kerr.txn_requires_abort():
                try:
                    self.producer.abort_transaction(30.0)
                except Exception:  # noqa: BLE001
                    pass
                raise ProcessError(
                    f"kafka transaction aborted (will replay): {kerr}") from e
            if kerr.retriable():
                # retriable commit error: librdkafka retries internally;
                # surface as retryable so the batch replays
                raise ProcessError(
                    f"kafka commit retriable error: {kerr}") from e
            raise ProcessError(f"kafka transaction failed: {kerr}") from e

    def close(self) -> None:
        if self.producer is not None:
            try:
                self.producer.flush(10.0)
            except Exception:  # noqa: BLE001
                pass

"""Broker inputs/outputs: kafka / mqtt / nats / pulsar / redis.

The reference implements these over rdkafka, rumqttc, async-nats, pulsar and
redis crates (crates/arkflow-plugin/src/{input,output}/*.rs). This
environment has no network and no broker client libraries, so each component
has two drivers:

  - ``driver: memory`` — an in-process fake bus with real broker semantics
    (partitions, offsets, consumer groups, transactions) so delivery
    guarantees are TESTABLE offline, the way the reference gates its
    testcontainers suites.
  - real drivers — kafka: full confluent_kafka consumer/transactional-
    producer (inputs/kafka_real.py), exercised by the same contract test
    bodies when ``KAFKA_BOOTSTRAP`` is set; redis (redis-py: pubsub/BLPOP/
    XREAD), mqtt (paho) and nats (nats-py core + JetStream w/ per-message
    acks) in inputs/pubsub_real.py, env-gated the same way. pulsar has NO
    real driver — its connect() raises ConnectionError_ and only the fake
    bus carries its semantics.

Kafka semantics mirrored from the reference:
  input: per-message read, ``__meta_*`` metadata columns, ack = commit offset
  (store_offset, auto-commit off — input/kafka.rs:183-296);
  output: exactly-once via transactional producer — write_batch =
  begin / send-all / commit (output/kafka.rs:348-446).
"""
from __future__ import annotations

import asyncio
import time
from collections import defaultdict
from typing import Dict, List, Optional, Tuple

from ..batch import Column, MessageBatch
from ..errors import ConfigError, ConnectionError_, EOFError_
from ..registry import register
from ..spi import Ack, Input, NoopAck, Output


class FakeBus:
    """Process-global fake broker: topics → partitions → append-only logs."""

    _instances: Dict[str, "FakeBus"] = {}

    def __init__(self):
        self.topics: Dict[str, List[List[Tuple[bytes, bytes, float]]]] = {}
        self.commits: Dict[Tuple[str, str, int], int] = {}  # (grp,topic,p)→off
        self.subscribers: Dict[str, List[asyncio.Queue]] = defaultdict(list)
        self.notify: asyncio.Event = asyncio.Event()

    @classmethod
    def get(cls, name: str = "default") -> "FakeBus":
        if name not in cls._instances:
            cls._instances[name] = FakeBus()
        return cls._instances[name]

    @classmethod
    def reset(cls, name: str = "default") -> None:
        cls._instances.pop(name, None)

    def ensure_topic(self, topic: str, partitions: int = 1) -> None:
        if topic not in self.topics:
            self.topics[topic] = [[] for _ in range(partitions)]

    def produce(self, topic: str, key: Optional[bytes], value: bytes,
                partition: Optional[int] = None) -> Tuple[int, int]:
        self.ensure_topic(topic)
        parts = self.topics[topic]
        if partition is None:
            partition = (hash(key) % len(parts)) if key else 0
        parts[partition].append((key or b"", value, time.time()))
        offset = len(parts[partition]) - 1
        for q in self.subscribers.get(topic, []):
            try:
                q.put_nowait((partition, offset, key or b"", value))
            except asyncio.QueueFull:
                pass
        self.notify.set()
        return partition, offset

    def fetch(self, group: str, topic: str) -> Optional[Tuple[int, int,
                                                              bytes, bytes]]:
        self.ensure_topic(topic)
        for p, log in enumerate(self.topics[topic]):
            committed = self.commits.get((group, topic, p), 0)
            if committed < len(log):
                key, value, ts = log[committed]
                return p, committed, key, value
        return None

    def commit(self, group: str, topic: str, partition: int,
               offset: int) -> None:
        cur = self.commits.get((group, topic, partition), 0)
        self.commits[(group, topic, partition)] = max(cur, offset + 1)


# ---------------------------------------------------------------------- kafka
class KafkaAck(Ack):
    """Commit the consumed offset on ack (input/kafka.rs:277-296).

    Kafka's offset model cannot express ack gaps (commit N ⇒ everything
    below N is consumed) — identical to the reference's store_offset
    semantics. For gap-safe recovery enable the WAL, whose cursor is a
    contiguous-ack low watermark (wal/wal.py advance())."""

    def __init__(self, bus: FakeBus, group: str, topic: str, partition: int,
                 offset: int):
        self.bus, self.group = bus, group
        self.topic, self.partition, self.offset = topic, partition, offset

    async def ack(self) -> None:
        self.bus.commit(self.group, self.topic, self.partition, self.offset)


class KafkaInput(Input):
    def __init__(self, config: dict, resource=None):
        self.brokers = config.get("brokers", ["memory://default"])
        if isinstance(self.brokers, str):
            self.brokers = [self.brokers]
        self.topics = config.get("topics") or [config.get("topic")]
        if not self.topics or self.topics[0] is None:
            raise ConfigError("kafka input requires 'topic' or 'topics'")
        self.group = config.get("consumer_group", "arkflow")
        self.driver = config.get("driver") or (
            "memory" if str(self.brokers[0]).startswith("memory://")
            else "real")
        from ..codecs.helper import build_codec
        self.codec = build_codec(config, resource)
        self.client_config = config.get("client_config") or {}
        self.bus: Optional[FakeBus] = None
        self._real = None
        self._closed = False
        # in-memory read positions (per consumer instance): a read advances
        # the position; only ack() commits the offset durably. A NEW consumer
        # in the same group resumes from the committed offset — at-least-once.
        self._positions: Dict[Tuple[str, int], int] = {}

    async def connect(self) -> None:
        if self.driver == "memory":
            name = str(self.brokers[0]).removeprefix("memory://") or "default"
            self.bus = FakeBus.get(name)
            for t in self.topics:
                self.bus.ensure_topic(t)
                for p in range(len(self.bus.topics[t])):
                    self._positions[(t, p)] = self.bus.commits.get(
                        (self.group, t, p), 0)
            return
        from .kafka_real import RealKafkaConsumer
        self._real = RealKafkaConsumer(
            [str(b) for b in self.brokers], list(self.topics), self.group,
            self.client_config)
        self._real.connect()

    async def read(self) -> Tuple[MessageBatch, Ack]:
        if self._closed:
            raise EOFError_("kafka input closed")
        if self.driver != "memory":
            return await self._real.read(self.codec)
        while True:
            for topic in self.topics:
                item = None
                for p, log in enumerate(self.bus.topics[topic]):
                    pos = self._positions.get((topic, p),
                                              self.bus.commits.get(
                                                  (self.group, topic, p), 0))
                    if pos < len(log):
                        key, value, _ts = log[pos]
                        self._positions[(topic, p)] = pos + 1
                        item = (p, pos, key, value)
                        break
                if item is not None:
                    p, off, key, value = item
                    batch = MessageBatch(
                        {
                            "__value__": Column.from_bytes([value]),
                            "__meta_source": Column.from_strings([topic]),
                            "__meta_partition": Column.from_numeric([p]),
                            "__meta_offset": Column.from_numeric([off]),
                            "__meta_key": Column.from_bytes([key]),
                            "__meta_timestamp": Column.from_numeric(
                                [time.time()]),
                        },
                        input_name=topic,
                    )
                    if self.codec is not None:
                        from ..codecs.helper import apply_codec
                        batch = apply_codec(batch, self.codec)
                    return batch, KafkaAck(self.bus, self.group, topic, p, off)
            self.bus.notify.clear()
            from ..aio import event_wait
            if not await event_wait(self.bus.notify, 0.5) and self._closed:
                raise EOFError_("kafka input closed")

    async def close(self) -> None:
        self._closed = True
        if self.bus:
            self.bus.notify.set()
        if self._real is not None:
            self._real.close()


def _expr_cfg(v):
    """Reference Expr<T> config shape (expr/mod.rs:29-210): a plain value,
    or {"expr": "<sql expression>"} evaluated per row against the batch."""
    if isinstance(v, dict) and "expr" in v:
        from ..sql.parser import parse_expression
        return ("expr", parse_expression(str(v["expr"])))
    return ("value", None if v is None else str(v))


def _eval_expr_rows(cfg, batch: MessageBatch) -> List[Optional[str]]:
    kind, v = cfg
    n = batch.num_rows
    if kind == "value":
        return [v] * n
    import torch

    from ..sql.eval import Env, eval_expr
    out = eval_expr(v, Env(batch.columns, n, batch.device))
    if isinstance(out, Column):
        return [x.decode("utf-8", "replace")
                if isinstance(x, (bytes, bytearray)) else str(x)
                for x in out.to_pylist()]
    if isinstance(out, torch.Tensor):
        vals = out.cpu().tolist()
        return [str(x) for x in vals]
    return [str(out)] * n


class KafkaOutput(Output):
    """Exactly-once: write_batch = one transaction (output/kafka.rs:348-446).

    The fake bus stages rows and appends atomically on commit; consumers only
    ever see committed rows (read_committed). ``topic`` and ``key`` accept
    either a constant or {"expr": "<sql>"} evaluated per row — the
    reference's Expr<T> dynamic routing (output/kafka.rs topic/key)."""

    def __init__(self, config: dict, resource=None):
        self.brokers = config.get("brokers", ["memory://default"])
        if isinstance(self.brokers, str):
            self.brokers = [self.brokers]
        topic = config.get("topic")
        if not topic:
            raise ConfigError("kafka output requires 'topic'")
        self.topic_cfg = _expr_cfg(topic)
        self.topic_expr = topic if isinstance(topic, str) else None
        self.key_cfg = (_expr_cfg(config["key"])
                        if isinstance(config.get("key"), dict) else None)
        self.key_column = config.get("key_column")
        if self.key_column is None and isinstance(config.get("key"), str):
            self.key_column = config["key"]
        self.value_column = config.get("value_column", "__value__")
        self.exactly_once = bool(config.get("exactly_once", False))
        self.transactional_id = config.get("transactional_id")
        self.compression = config.get("compression")
        self.client_config = config.get("client_config") or {}
        self.driver = config.get("driver") or (
            "memory" if str(self.brokers[0]).startswith("memory://")
            else "real")
        self.bus: Optional[FakeBus] = None
        self._real = None

    async def connect(self) -> None:
        if self.driver == "memory":
            name = str(self.brokers[0]).removeprefix("memory://") or "default"
            self.bus = FakeBus.get(name)
            if self.topic_expr:  # dynamic topics materialize on first write
                self.bus.ensure_topic(self.topic_expr)
            return
        from .kafka_real import RealKafkaProducer
        self._real = RealKafkaProducer(
            [str(b) for b in self.brokers], self.exactly_once,
            transactional_id=self.transactional_id,
            compression=self.compression, config=self.client_config)
        self._real.connect()

    def _rows(self, batch: MessageBatch
              ) -> List[Tuple[str, Optional[bytes], bytes]]:
        """(topic, key, value) per row — topic/key evaluated per row when
        configured as expressions."""
        col = batch.columns.get(self.value_column)
        if col is not None and col.kind == "binary":
            values = col.to_pylist()
        else:
            values = batch.to_json_lines()
        topics = _eval_expr_rows(self.topic_cfg, batch)
        keys: List[Optional[bytes]] = [None] * len(values)
        if self.key_cfg is not None:
            keys = [None if k is None else k.encode()
                    for k in _eval_expr_rows(self.key_cfg, batch)]
        elif self.key_column and self.key_column in batch.columns:
            kc = batch.column(self.key_column)
            keys = [str(v).encode() if not isinstance(v, (bytes, bytearray))
                    else bytes(v) for v in kc.to_pylist()]
        return list(zip(topics, keys, values))

    async def write(self, batch: MessageBatch) -> None:
        rows = self._rows(batch)
        if self._real is not None:
            loop = asyncio.get_running_loop()
            await loop.run_in_executor(
                None, self._real.write_plain, self.topic_expr, rows)
            return
        for topic, key, value in rows:
            self.bus.ensure_topic(topic)
            self.bus.produce(topic, key, value)

    async def write_batch(self, batches) -> None:
        if not self.exactly_once:
            for b in batches:
                await self.write(b)
            return
        # transaction: stage everything, then append atomically
        staged = []
        for b in batches:
            staged.extend(self._rows(b))
        if self._real is not None:
            # real driver: one begin/produce-all/commit transaction with
            # fencing-aware error mapping (kafka_real.write_txn)
            loop = asyncio.get_running_loop()
            await loop.run_in_executor(
                None, self._real.write_txn, self.topic_expr, staged)
            return
        # commit point — a failure above leaves the log untouched
        for topic, key, value in staged:
            self.bus.ensure_topic(topic)
            self.bus.produce(topic, key, value)

    async def close(self) -> None:
        if self._real is not None:
            self._real.close()


# ------------------------------------------------------------- pub/sub family
class _PubSubInput(Input):
    """Shared fake pub/sub input (mqtt / nats / redis pubsub / pulsar)."""

    kind = "pubsub"

    def __init__(self, config: dict, resource=None):
        self.topic = config.get("topic") or config.get("subject") \
            or config.get("channel")
        if not self.topic:
            raise ConfigError(f"{self.kind} input requires a topic/subject")
        self.url = str(config.get("url", "memory://default"))
        self.driver = config.get("driver") or (
            "memory" if self.url.startswith("memory://") else "real")
        self.jetstream = bool(config.get("jetstream", False))
        self.subscription = config.get("subscription", "arkflow")
        from ..codecs.helper import build_codec
        self.codec = build_codec(config, resource)
        self.bus: Optional[FakeBus] = None
        self._q: Optional[asyncio.Queue] = None
        self._closed = False

    def _make_real(self):
        """Real client for this kind, or raise ConnectionError_.
        redis → redis-py; mqtt → paho-mqtt; nats → nats-py (core +
        JetStream). pulsar has no real driver (fake bus only)."""
        from urllib.parse import urlparse

        from .pubsub_real import (RealMqttClient, RealNatsClient,
                                  RealPulsarClient, RealRedisClient)
        if self.kind == "redis":
            return RealRedisClient(self.url, getattr(self, "mode", "pubsub"))
        if self.kind == "nats":
            return RealNatsClient(self.url,
                                  jetstream=bool(getattr(self, "jetstream",
                                                         False)))
        if self.kind == "pulsar":
            return RealPulsarClient(self.url,
                                    getattr(self, "subscription", "arkflow"))
        if self.kind == "mqtt":
            u = urlparse(self.url if "://" in self.url
                         else f"mqtt://{self.url}")
            return RealMqttClient(u.hostname or "127.0.0.1",
                                  u.port or 1883,
                                  username=u.username, password=u.password)
        raise ConnectionError_(
            f"no {self.kind} client library in this environment; "
            "use driver: memory")

    async def connect(self) -> None:
        if self.driver != "memory":
            self._real = self._make_real()
            self._real.connect(subscribe=self.topic)
            return
        self.bus = FakeBus.get(self.url.removeprefix("memory://") or "default")
        self._q = asyncio.Queue(maxsize=4096)
        self.bus.subscribers[self.topic].append(self._q)

    async def read(self) -> Tuple[MessageBatch, Ack]:
        if self._closed:
            raise EOFError_(f"{self.kind} input closed")
        if getattr(self, "_real", None) is not None:
            from .pubsub_real import make_batch
            if self.kind == "redis":
                value, off = await self._real.read(self.topic)
                return make_batch(self.topic, value, self.codec, off)
            if self.kind in ("nats", "pulsar"):
                topic, value, ack = await self._real.read()
                batch, noop = make_batch(topic, value, self.codec)
                return batch, (ack or noop)
            topic, value = await self._real.read()
            return make_batch(topic, value, self.codec)
        item = await self._q.get()
        if item is None:
            raise EOFError_(f"{self.kind} input closed")
        p, off, key, value = item
        batch = MessageBatch(
            {
                "__value__": Column.from_bytes([value]),
                "__meta_source": Column.from_strings([self.topic]),
                "__meta_timestamp": Column.from_numeric([time.time()]),
            },
            input_name=self.topic,
        )
        if self.codec is not None:
            from ..codecs.helper import apply_codec
            batch = apply_codec(batch, self.codec)
        return batch, NoopAck()

    async def close(self) -> None:
        self._closed = True
        if getattr(self, "_real", None) is not None:
            self._real.close()
        if self._q is not None:
            self._q.put_nowait(None)
            if self.bus and self._q in self.bus.subscribers.get(self.topic, []):
                self.bus.subscribers[self.topic].remove(self._q)


class _PubSubOutput(Output):
    kind = "pubsub"

    def __init__(self, config: dict, resource=None):
        self.topic = config.get("topic") or config.get("subject") \
            or config.get("channel")
        if not self.topic:
            raise ConfigError(f"{self.kind} output requires a topic/subject")
        self.url = str(config.get("url", "memory://default"))
        self.driver = config.get("driver") or (
            "memory" if self.url.startswith("memory://") else "real")
        self.raw_value = bool(config.get("raw_value", True))
        self.mode = config.get("mode", "pubsub")  # redis: pubsub|list|stream
        self.bus: Optional[FakeBus] = None

    async def connect(self) -> None:
        if self.driver != "memory":
            self._real = _PubSubInput._make_real(self)
            self._real.connect()
            return
        self.bus = FakeBus.get(self.url.removeprefix("memory://") or "default")
        self.bus.ensure_topic(self.topic)

    async def write(self, batch: MessageBatch) -> None:
        from ..batch import DEFAULT_BINARY_VALUE_FIELD
        if self.raw_value and DEFAULT_BINARY_VALUE_FIELD in batch.columns:
            values = batch.binary_values()
        else:
            values = batch.to_json_lines()
        if getattr(self, "_real", None) is not None:
            if self.kind == "nats":
                for v in values:
                    await self._real.aproduce(self.topic, v)
                return
            loop = asyncio.get_running_loop()
            for v in values:
                await loop.run_in_executor(
                    None, self._real.produce, self.topic, v)
            return
        for v in values:
            self.bus.produce(self.topic, None, v)

    async def close(self) -> None:
        if getattr(self, "_real", None) is not None:
            self._real.close()


def _mk_pubsub(kind: str):
    class In(_PubSubInput):
        pass

    class Out(_PubSubOutput):
        pass

    In.kind = kind
    Out.kind = kind
    return In, Out


MqttInput, MqttOutput = _mk_pubsub("mqtt")
NatsInput, NatsOutput = _mk_pubsub("nats")
PulsarInput, PulsarOutput = _mk_pubsub("pulsar")
_RedisPubSubIn, RedisOutput = _mk_pubsub("redis")


class RedisInput(_RedisPubSubIn):
    """redis input modes (reference input/redis.rs): pubsub (default),
    list (BRPOP-style: consume from the log, at-least-once like kafka) and
    stream (offset-tracked) — over the fake bus, list/stream use the
    append-only topic log with a consumer position."""

    def __init__(self, config: dict, resource=None):
        super().__init__(config, resource)
        self.mode = config.get("mode", "pubsub")
        self._pos = 0

    async def read(self):
        if self.mode == "pubsub":
            return await super().read()
        # list/stream: consume the topic log in order
        import time as _t
        while True:
            if self._closed:
                raise EOFError_("redis input closed")
            self.bus.ensure_topic(self.topic)
            log = self.bus.topics[self.topic][0]
            if self._pos < len(log):
                key, value, _ts = log[self._pos]
                self._pos += 1
                batch = MessageBatch(
                    {"__value__": Column.from_bytes([value]),
                     "__meta_source": Column.from_strings([self.topic]),
                     "__meta_offset": Column.from_numeric([self._pos - 1]),
                     "__meta_timestamp": Column.from_numeric([_t.time()])},
                    input_name=self.topic)
                if self.codec is not None:
                    from ..codecs.helper import apply_codec
                    batch = apply_codec(batch, self.codec)
                return batch, NoopAck()
            self.bus.notify.clear()
            from ..aio import event_wait
            await event_wait(self.bus.notify, 0.5)


# ---- registrations ------------------------------------------------------------
@register("input", "kafka",
          description="Kafka consumer: per-message read, __meta_* columns, "
                      "ack = offset commit (driver: memory for in-process bus)",
          example={"type": "kafka", "brokers": ["memory://default"],
                   "topic": "events", "consumer_group": "g1"})
def _build_kafka_in(config, resource=None):
    return KafkaInput(config, resource)


@register("output", "kafka",
          description="Kafka producer; exactly_once = transactional "
                      "write_batch",
          example={"type": "kafka", "brokers": ["memory://default"],
                   "topic": "out", "exactly_once": True})
def _build_kafka_out(config, resource=None):
    return KafkaOutput(config, resource)


for _name, _in, _out in (("mqtt", MqttInput, MqttOutput),
                         ("nats", NatsInput, NatsOutput),
                         ("pulsar", PulsarInput, PulsarOutput)):
    register("input", _name,
             description=f"{_name} subscriber (driver: memory offline)",
             example={"type": _name, "url": "memory://default",
                      "topic": "t"})(
        (lambda cls: lambda config, resource=None: cls(config, resource))(_in))
    register("output", _name,
             description=f"{_name} publisher (driver: memory offline)",
             example={"type": _name, "url": "memory://default",
                      "topic": "t"})(
        (lambda cls: lambda config, resource=None: cls(config, resource))(_out))


@register("input", "redis",
          description="redis subscriber: pubsub/list/stream modes "
                      "(driver: memory offline)",
          example={"type": "redis", "url": "memory://default", "topic": "t",
                   "mode": "list"})
def _build_redis_in(config, resource=None):
    return RedisInput(config, resource)


@register("output", "redis",
          description="redis publisher (driver: memory offline)",
          example={"type": "redis", "url": "memory://default", "topic": "t"})
def _build_redis_out(config, resource=None):
    return RedisOutput(config, resource)

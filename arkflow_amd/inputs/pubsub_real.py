"""Real redis / MQTT / NATS drivers behind the pub-sub driver switch.

Like inputs/kafka_real.py: complete client implementations against the
public redis-py, paho-mqtt and nats-py APIs, active when the library is
importable and the config names a real endpoint; the fake bus keeps
carrying the semantics offline. Env-gated tests: REDIS_URL / MQTT_HOST
(tests/test_brokers.py), the way the reference gates its testcontainers
suites.

Reference mapping:
  redis (input/redis.rs): pubsub / list (BRPOP) / stream (XREAD) modes;
  output publishes / RPUSHes / XADDs.
  mqtt (input/mqtt.rs, rumqttc): subscriber with QoS, publisher.
  nats (input/nats.rs): core subscribe + JetStream durable consumer whose
  per-message ack is the engine Ack.
"""
from __future__ import annotations

import asyncio
import queue as _queue
import time
from typing import Optional, Tuple

from ..batch import Column, MessageBatch
from ..errors import ConnectionError_, EOFError_
from ..spi import Ack, NoopAck


# ------------------------------------------------------------------- redis
class RealRedisClient:
    """redis-py driver: pubsub / list / stream consume + produce."""

    def __init__(self, url: str, mode: str = "pubsub"):
        self.url = url
        self.mode = mode
        self.client = None
        self._pubsub = None
        self._stream_last = "$"
        self._closed = False

    def connect(self, subscribe: Optional[str] = None) -> None:
        try:
            import redis  # type: ignore
        except ImportError as e:
            raise ConnectionError_(
                "redis real driver requires redis-py; use driver: memory"
            ) from e
        try:
            self.client = redis.Redis.from_url(self.url)
            self.client.ping()
            if subscribe is not None and self.mode == "pubsub":
                self._pubsub = self.client.pubsub(
                    ignore_subscribe_messages=True)
                self._pubsub.subscribe(subscribe)
        except Exception as e:  # noqa: BLE001
            raise ConnectionError_(f"redis connect failed: {e}") from e

    async def read(self, topic: str) -> Tuple[bytes, Optional[int]]:
        """Returns (payload, offset-or-None); blocks via executor."""
        loop = asyncio.get_running_loop()
        while not self._closed:
            if self.mode == "pubsub":
                msg = await loop.run_in_executor(
                    None, lambda: self._pubsub.get_message(timeout=0.25))
                if msg and msg.get("type") == "message":
                    return msg["data"], None
            elif self.mode == "list":
                item = await loop.run_in_executor(
                    None, lambda: self.client.blpop(topic, timeout=1))
                if item is not None:
                    return item[1], None
            else:  # stream
                got = await loop.run_in_executor(
                    None, lambda: self.client.xread(
                        {topic: self._stream_last}, count=1, block=250))
                if got:
                    _, entries = got[0]
                    eid, fields = entries[0]
                    self._stream_last = eid
                    payload = fields.get(b"value") or next(
                        iter(fields.values()), b"")
                    return payload, None
        raise EOFError_("redis input closed")

    def produce(self, topic: str, payload: bytes) -> None:
        if self.mode == "pubsub":
            self.client.publish(topic, payload)
        elif self.mode == "list":
            self.client.rpush(topic, payload)
        else:
            self.client.xadd(topic, {"value": payload})

    def close(self) -> None:
        self._closed = True
        try:
            if self._pubsub is not None:
                self._pubsub.close()
            if self.client is not None:
                self.client.close()
        except Exception:  # noqa: BLE001
            pass


# -------------------------------------------------------------------- mqtt
class RealMqttClient:
    """paho-mqtt driver: background network loop feeding a local queue."""

    def __init__(self, host: str, port: int = 1883, qos: int = 1,
                 client_id: str = "", username: Optional[str] = None,
                 password: Optional[str] = None):
        self.host, self.port, self.qos = host, port, qos
        self.client_id = client_id
        self.username, self.password = username, password
        self.client = None
        self._q: _queue.Queue = _queue.Queue(maxsize=4096)
        self._closed = False

    def connect(self, subscribe: Optional[str] = None) -> None:
        try:
            import paho.mqtt.client as mqtt  # type: ignore
        except ImportError as e:
            raise ConnectionError_(
                "mqtt real driver requires paho-mqtt; use driver: memory"
            ) from e
        try:
            try:  # paho 2.x requires an API version; 1.x has no such arg
                self.client = mqtt.Client(
                    mqtt.CallbackAPIVersion.VERSION1,
                    client_id=self.client_id)
            except AttributeError:
                self.client = mqtt.Client(client_id=self.client_id)
            if self.username:
                self.client.username_pw_set(self.username, self.password)

            def on_message(cli, userdata, msg):
                try:
                    self._q.put_nowait((msg.topic, msg.payload))
                except _queue.Full:
                    pass

            self.client.on_message = on_message
            self.client.connect(self.host, self.port, keepalive=30)
            if subscribe is not None:
                self.client.subscribe(subscribe, qos=self.qos)
            self.client.loop_start()
        except ConnectionError_:
            raise
        except Exception as e:  # noqa: BLE001
            raise ConnectionError_(f"mqtt connect failed: {e}") from e

    async def read(self) -> Tuple[str, bytes]:
        loop = asyncio.get_running_loop()
        while not self._closed:
            try:
                return await loop.run_in_executor(
                    None, lambda: self._q.get(timeout=0.25))
            except _queue.Empty:
                continue
        raise EOFError_("mqtt input closed")

    def produce(self, topic: str, payload: bytes) -> None:
        info = self.client.publish(topic, payload, qos=self.qos)
        info.wait_for_publish(timeout=10.0)

    def close(self) -> None:
        self._closed = True
        if self.client is not None:
            try:
                self.client.loop_stop()
                self.client.disconnect()
            except Exception:  # noqa: BLE001
                pass


def make_batch(topic: str, payload: bytes, codec=None,
               offset: Optional[int] = None) -> Tuple[MessageBatch, Ack]:
    cols = {
        "__value__": Column.from_bytes([payload]),
        "__meta_source": Column.from_strings([topic]),
        "__meta_timestamp": Column.from_numeric([time.time()]),
    }
    if offset is not None:
        cols["__meta_offset"] = Column.from_numeric([offset])
    batch = MessageBatch(cols, input_name=topic)
    if codec is not None:
        from ..codecs.helper import apply_codec
        batch = apply_codec(batch, codec)
    return batch, NoopAck()


# -------------------------------------------------------------------- nats
class _JsAck(Ack):
    def __init__(self, msg):
        self.msg = msg

    async def ack(self) -> None:
        await self.msg.ack()


class RealNatsClient:
    """nats-py driver: core subscribe/publish; ``jetstream: true`` uses a
    durable JetStream consumer whose per-message ack becomes the engine's
    Ack (reference input/nats.rs covers both)."""

    def __init__(self, url: str, jetstream: bool = False,
                 durable: str = "arkflow"):
        self.url = url
        self.jetstream = jetstream
        self.durable = durable
        self.nc = None
        self.sub = None
        self._closed = False

    async def aconnect(self, subscribe: Optional[str] = None) -> None:
        try:
            import nats  # type: ignore
        except ImportError as e:
            raise ConnectionError_(
                "nats real driver requires nats-py; use driver: memory"
            ) from e
        try:
            self.nc = await nats.connect(self.url, connect_timeout=10)
            if subscribe is not None:
                if self.jetstream:
                    js = self.nc.jetstream()
                    self.sub = await js.subscribe(subscribe,
                                                  durable=self.durable,
                                                  manual_ack=True)
                else:
                    self.sub = await self.nc.subscribe(subscribe)
        except ConnectionError_:
            raise
        except Exception as e:  # noqa: BLE001
            raise ConnectionError_(f"nats connect failed: {e}") from e

    def connect(self, subscribe: Optional[str] = None) -> None:
        # the pub/sub family's connect() is async already in our SPI, but
        # this wrapper is invoked from sync context in _make_real; defer to
        # the first read/produce via a lazy ensure
        self._pending_subscribe = subscribe

    async def _ensure(self) -> None:
        if self.nc is None:
            await self.aconnect(getattr(self, "_pending_subscribe", None))

    async def read(self):
        await self._ensure()
        while not self._closed:
            try:
                msg = await self.sub.next_msg(timeout=0.25)
            except Exception:  # noqa: BLE001  (nats TimeoutError)
                continue
            ack = _JsAck(msg) if self.jetstream else None
            return msg.subject, msg.data, ack
        raise EOFError_("nats input closed")

    async def aproduce(self, topic: str, payload: bytes) -> None:
        await self._ensure()
        if self.jetstream:
            await self.nc.jetstream().publish(topic, payload)
        else:
            await self.nc.publish(topic, payload)
            await self.nc.flush(timeout=10)

    def close(self) -> None:
        self._closed = True


# ------------------------------------------------------------------ pulsar
class _PulsarAck(Ack):
    def __init__(self, consumer, msg):
        self.consumer, self.msg = consumer, msg

    async def ack(self) -> None:
        import asyncio as _a
        await _a.get_running_loop().run_in_executor(
            None, self.consumer.acknowledge, self.msg)


class RealPulsarClient:
    """pulsar-client driver: shared-subscription consumer with per-message
    acknowledge (the engine Ack), blocking calls on executor threads
    (reference input/pulsar.rs + pulsar/common.rs)."""

    def __init__(self, url: str, subscription: str = "arkflow"):
        self.url = url
        self.subscription = subscription
        self.client = None
        self.consumer = None
        self._producers = {}
        self._closed = False

    def connect(self, subscribe: Optional[str] = None) -> None:
        try:
            import pulsar  # type: ignore
        except ImportError as e:
            raise ConnectionError_(
                "pulsar real driver requires pulsar-client; use "
                "driver: memory") from e
        try:
            self.client = pulsar.Client(self.url)
            if subscribe is not None:
                self.consumer = self.client.subscribe(
                    subscribe, self.subscription,
                    consumer_type=pulsar.ConsumerType.Shared)
        except ConnectionError_:
            raise
        except Exception as e:  # noqa: BLE001
            raise ConnectionError_(f"pulsar connect failed: {e}") from e

    async def read(self):
        import asyncio as _a
        loop = _a.get_running_loop()
        while not self._closed:
            try:
                msg = await loop.run_in_executor(
                    None, lambda: self.consumer.receive(timeout_millis=250))
            except Exception:  # noqa: BLE001  (pulsar Timeout)
                continue
            return (msg.topic_name(), msg.data(),
                    _PulsarAck(self.consumer, msg))
        raise EOFError_("pulsar input closed")

    def produce(self, topic: str, payload: bytes) -> None:
        prod = self._producers.get(topic)
        if prod is None:
            prod = self.client.create_producer(topic)
            self._producers[topic] = prod
        prod.send(payload)

    def close(self) -> None:
        self._closed = True
        try:
            if self.consumer is not None:
                self.consumer.close()
            for p in self._producers.values():
                p.close()
            if self.client is not None:
                self.client.close()
        except Exception:  # noqa: BLE001
            pass

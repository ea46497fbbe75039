"""Kafka driver contract: ONE test body runs against the in-process fake bus
always, and against a REAL broker via confluent_kafka when KAFKA_BOOTSTRAP
is set — the way the reference gates its testcontainers EOS suite
(kafka_eos.rs:29-33). `pytest tests/test_kafka_contract.py` with
`KAFKA_BOOTSTRAP=host:9092` exercises inputs/kafka_real.py end to end."""
import asyncio
import os
import uuid

import pytest

from arkflow_amd.inputs.brokers import FakeBus, KafkaInput, KafkaOutput

REAL = os.environ.get("KAFKA_BOOTSTRAP")


@pytest.fixture(params=["memory", "real"])
def kafka_cfg(request):
    """Returns (make_input_cfg, make_output_cfg, unique_topic)."""
    if request.param == "real":
        if not REAL:
            pytest.skip("KAFKA_BOOTSTRAP not set")
        brokers = [REAL]
    else:
        FakeBus.reset("contract")
        brokers = ["memory://contract"]
    topic = f"t-{uuid.uuid4().hex[:8]}"

    def in_cfg(**kw):
        return {"brokers": brokers, "topic": topic,
                "consumer_group": f"g-{uuid.uuid4().hex[:6]}", **kw}

    def out_cfg(**kw):
        return {"brokers": brokers, "topic": topic, **kw}

    yield in_cfg, out_cfg, topic
    if request.param == "memory":
        FakeBus.reset("contract")


def _run(coro, timeout=60):
    loop = asyncio.new_event_loop()
    try:
        return loop.run_until_complete(asyncio.wait_for(coro, timeout))
    finally:
        loop.close()


@pytest.mark.timeout(120)
def test_produce_consume_ack_roundtrip(kafka_cfg):
    in_cfg, out_cfg, topic = kafka_cfg

    async def main():
        out = KafkaOutput(out_cfg())
        await out.connect()
        from arkflow_amd.batch import MessageBatch
        await out.write(MessageBatch.from_binary([b"m1", b"m2", b"m3"]))
        await out.close()

        inp = KafkaInput(in_cfg())
        await inp.connect()
        got = []
        for _ in range(3):
            batch, ack = await inp.read()
            got.extend(batch.binary_values())
            assert batch.column("__meta_offset") is not None
            assert batch.column("__meta_partition") is not None
            await ack.ack()
        await inp.close()
        assert sorted(got) == [b"m1", b"m2", b"m3"]

    _run(main())


@pytest.mark.timeout(120)
def test_ack_commits_offset_new_consumer_resumes(kafka_cfg):
    in_cfg, out_cfg, topic = kafka_cfg
    group = f"g-{uuid.uuid4().hex[:6]}"

    async def main():
        out = KafkaOutput(out_cfg())
        await out.connect()
        from arkflow_amd.batch import MessageBatch
        await out.write(MessageBatch.from_binary([b"a", b"b"]))
        await out.close()

        inp = KafkaInput(in_cfg(consumer_group=group))
        await inp.connect()
        b1, ack1 = await inp.read()
        await ack1.ack()  # commit first message only
        first = b1.binary_values()[0]
        await inp.close()

        # a NEW consumer in the same group resumes after the commit
        inp2 = KafkaInput(in_cfg(consumer_group=group))
        await inp2.connect()
        b2, ack2 = await inp2.read()
        await ack2.ack()
        await inp2.close()
        assert b2.binary_values()[0] != first

    _run(main())


@pytest.mark.timeout(120)
def test_eos_write_batch_transactional(kafka_cfg):
    in_cfg, out_cfg, topic = kafka_cfg

    async def main():
        out = KafkaOutput(out_cfg(
            exactly_once=True,
            transactional_id=f"tx-{uuid.uuid4().hex[:8]}"))
        await out.connect()
        from arkflow_amd.batch import MessageBatch
        await out.write_batch([MessageBatch.from_binary([b"x1"]),
                               MessageBatch.from_binary([b"x2", b"x3"])])
        await out.close()

        inp = KafkaInput(in_cfg())  # read_committed
        await inp.connect()
        got = []
        for _ in range(3):
            batch, ack = await inp.read()
            got.extend(batch.binary_values())
            await ack.ack()
        await inp.close()
        assert sorted(got) == [b"x1", b"x2", b"x3"]

    _run(main())

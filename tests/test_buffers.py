"""Window/buffer semantics tests (reference buffer/{memory,tumbling_window,
sliding_window,session_window}.rs inline tests)."""
import asyncio

import pytest

from arkflow_amd.batch import MessageBatch
from arkflow_amd.buffers.memory_buffer import MemoryBuffer
from arkflow_amd.buffers.windows import (
    SessionWindowBuffer,
    SlidingWindowBuffer,
    TumblingWindowBuffer,
)
from arkflow_amd.spi import Ack


class TAck(Ack):
    def __init__(self, log, tag):
        self.log, self.tag = log, tag

    async def ack(self):
        self.log.append(self.tag)


def _mk(vals):
    return MessageBatch.from_dict({"v": vals}, input_name="in")


def test_memory_buffer_capacity(run):
    async def main():
        buf = MemoryBuffer({"capacity": 5})
        log = []
        for i in range(5):
            await buf.write(_mk([i]), TAck(log, i))
        batch, ack = await asyncio.wait_for(buf.read(), 2)
        assert batch.num_rows == 5
        assert batch.column("v").to_pylist() == [0, 1, 2, 3, 4]
        await ack.ack()
        assert log == [0, 1, 2, 3, 4]

    run(main())


def test_memory_buffer_timeout(run):
    async def main():
        buf = MemoryBuffer({"capacity": 1000, "timeout": "50ms"})
        log = []
        await buf.write(_mk([1, 2]), TAck(log, "a"))
        batch, ack = await asyncio.wait_for(buf.read(), 2)
        assert batch.num_rows == 2
        await ack.ack()
        assert log == ["a"]

    run(main())


def test_memory_buffer_drain_on_flush(run):
    async def main():
        buf = MemoryBuffer({"capacity": 1000})
        log = []
        await buf.write(_mk([1]), TAck(log, "a"))
        await buf.flush()
        item = await asyncio.wait_for(buf.read(), 2)
        assert item is not None and item[0].num_rows == 1
        assert await asyncio.wait_for(buf.read(), 2) is None

    run(main())


def test_tumbling_window(run):
    async def main():
        buf = TumblingWindowBuffer({"interval": "60ms"})
        log = []
        await buf.write(_mk([1]), TAck(log, 1))
        await buf.write(_mk([2]), TAck(log, 2))
        batch, ack = await asyncio.wait_for(buf.read(), 2)
        assert batch.column("v").to_pylist() == [1, 2]
        await ack.ack()
        assert sorted(log) == [1, 2]

    run(main())


def test_sliding_window_ack_on_leave(run):
    """sliding_window.rs:148-163 — acks released only when batches leave."""
    async def main():
        buf = SlidingWindowBuffer({"window_size": 3, "slide_size": 2})
        log = []
        for i in range(2):
            await buf.write(_mk([i]), TAck(log, i))
        batch, ack = await asyncio.wait_for(buf.read(), 2)
        assert batch.column("v").to_pylist() == [0, 1]
        await ack.ack()
        assert log == []  # nothing left the window yet
        for i in range(2, 4):
            await buf.write(_mk([i]), TAck(log, i))
        batch2, ack2 = await asyncio.wait_for(buf.read(), 2)
        # after 4 writes window holds last 3 → batch 0 left
        await ack2.ack()
        assert log == [0]
        assert batch2.column("v").to_pylist() == [1, 2, 3]
        # drain: remaining window acked
        await buf.flush()
        item = await asyncio.wait_for(buf.read(), 2)
        _, ack3 = item
        await ack3.ack()
        assert sorted(log) == [0, 1, 2, 3]
        assert await asyncio.wait_for(buf.read(), 2) is None

    run(main())


def test_session_window_gap(run):
    async def main():
        buf = SessionWindowBuffer({"gap": "40ms"})
        log = []
        await buf.write(_mk([1]), TAck(log, 1))
        await asyncio.sleep(0.01)
        await buf.write(_mk([2]), TAck(log, 2))
        t0 = asyncio.get_event_loop().time()
        batch, ack = await asyncio.wait_for(buf.read(), 2)
        waited = asyncio.get_event_loop().time() - t0
        assert batch.column("v").to_pylist() == [1, 2]
        assert waited >= 0.02  # closed only after the gap
        await ack.ack()
        assert sorted(log) == [1, 2]

    run(main())


def test_window_join_two_inputs(run):
    """BaseWindow SQL join across named inputs (reference buffer/join.rs)."""
    async def main():
        buf = TumblingWindowBuffer({
            "interval": "40ms",
            "join": {
                "query": "SELECT orders.id, orders.amount, users.name "
                         "FROM orders JOIN users ON orders.uid = users.uid",
                "inputs": ["orders", "users"],
            },
        })
        log = []
        orders = MessageBatch.from_dict(
            {"id": [1, 2], "amount": [10.0, 20.0], "uid": [7, 8]},
            input_name="orders")
        users = MessageBatch.from_dict(
            {"uid": [7, 8], "name": ["ann", "bob"]}, input_name="users")
        await buf.write(orders, TAck(log, "o"))
        await buf.write(users, TAck(log, "u"))
        batch, ack = await asyncio.wait_for(buf.read(), 2)
        assert batch.num_rows == 2
        assert batch.column("name").to_strlist() == ["ann", "bob"]
        await ack.ack()
        assert sorted(log) == ["o", "u"]

    run(main())


def test_window_join_waits_for_all_inputs(run):
    """join.rs:62-130 — no emit until all expected inputs are present."""
    async def main():
        buf = TumblingWindowBuffer({
            "interval": "30ms",
            "join": {
                "query": "SELECT a.v FROM a JOIN b ON a.v = b.v",
                "inputs": ["a", "b"],
            },
        })
        log = []
        await buf.write(MessageBatch.from_dict({"v": [1]}, input_name="a"),
                        TAck(log, "a"))
        # only input a present: first window tick must NOT emit
        with pytest.raises(asyncio.TimeoutError):
            await asyncio.wait_for(buf.read(), 0.15)
        await buf.write(MessageBatch.from_dict({"v": [1]}, input_name="b"),
                        TAck(log, "b"))
        batch, ack = await asyncio.wait_for(buf.read(), 2)
        assert batch.num_rows == 1

    run(main())


def test_device_ring_buffer_views():
    import torch
    from arkflow_amd.buffers.ring import DeviceRingBuffer
    ring = DeviceRingBuffer(capacity=8)
    b1 = MessageBatch.from_dict({"v": torch.tensor([1., 2., 3.])})
    b2 = MessageBatch.from_dict({"v": torch.tensor([4., 5.])})
    r1 = ring.append(b1)
    r2 = ring.append(b2)
    assert r1 == (0, 3) and r2 == (3, 5)
    s = ring.slice_many([r1, r2])
    assert s.column("v").to_pylist() == [1., 2., 3., 4., 5.]
    # zero-copy: contiguous range shares storage with the ring
    assert s.column("v").data.data_ptr() == ring.cols["v"].data_ptr()
    # wrap-around after release
    ring.release_before(5)
    r3 = ring.append(MessageBatch.from_dict(
        {"v": torch.tensor([6., 7., 8., 9., 10.])}))
    assert r3 == (5, 10)
    assert ring.slice(*r3).column("v").to_pylist() == [6., 7., 8., 9., 10.]
    # growth with live (unreleased) data preserved
    big = MessageBatch.from_dict({"v": torch.arange(20, dtype=torch.float32)})
    r4 = ring.append(big)
    assert ring.slice(*r4).column("v").to_pylist() == list(range(20))
    assert ring.slice(*r3).column("v").to_pylist() == [6., 7., 8., 9., 10.]
    # binary columns are not ring-able
    assert ring.append(MessageBatch.from_dict({"s": ["x"]})) is None


def test_window_ring_release_on_ack(run):
    """Ring rows are reusable only after the emitted window is ACKED
    (the emitted batch is a view over the ring)."""
    async def main():
        buf = MemoryBuffer({"capacity": 2, "device_ring": True})
        log = []
        import torch
        await buf.write(MessageBatch.from_dict(
            {"v": torch.tensor([1., 2.])}, input_name="in"), TAck(log, 1))
        batch, ack = await asyncio.wait_for(buf.read(), 2)
        assert batch.column("v").to_pylist() == [1., 2.]
        ring = buf.rings["in"]
        assert ring.head == 0  # not yet released
        await ack.ack()
        assert ring.head == 2  # released after ack
        assert log == [1]

    run(main())


def test_no_lost_notify_write_before_read(run):
    """Reference stream/mod.rs:384-395 regression: a write that lands BEFORE
    the reader awaits must still wake it (event set before wait)."""
    from arkflow_amd.buffers.windows import SlidingWindowBuffer
    from arkflow_amd.spi import NoopAck

    async def main():
        buf = SlidingWindowBuffer({"window_size": 2, "slide_size": 1})
        # writes complete before any read() is pending
        await buf.write(MessageBatch.from_dict({"v": [1.0]}), NoopAck())
        batch, ack = await asyncio.wait_for(buf.read(), 2)
        assert batch.num_rows == 1
        await ack.ack()
        # again, with the notify event already consumed once
        await buf.write(MessageBatch.from_dict({"v": [2.0]}), NoopAck())
        batch, ack = await asyncio.wait_for(buf.read(), 2)
        assert batch.num_rows == 2  # sliding window holds both rows
        await ack.ack()

    run(main())


def test_memory_buffer_concurrent_writers(run):
    """Many concurrent writers against one reader: all rows arrive, no
    deadlock, capacity triggers respected."""
    from arkflow_amd.buffers.memory_buffer import MemoryBuffer
    from arkflow_amd.spi import NoopAck

    async def main():
        buf = MemoryBuffer({"capacity": 64, "device_ring": False})
        total = 0

        async def writer(w):
            for i in range(50):
                await buf.write(MessageBatch.from_dict(
                    {"v": [float(w * 100 + i)] * 3}), NoopAck())

        async def reader():
            nonlocal total
            while total < 8 * 50 * 3:
                batch, ack = await asyncio.wait_for(buf.read(), 5)
                total += batch.num_rows
                await ack.ack()

        await asyncio.gather(reader(), *(writer(w) for w in range(8)))
        assert total == 8 * 50 * 3

    run(main(), timeout=30)


def test_windowed_join_example_runs_e2e(run):
    """examples/windowed_join.yaml (multiple_inputs + window join) runs
    through the live engine and emits joined rows."""
    import os
    import yaml
    import arkflow_amd as af
    from arkflow_amd.config import EngineConfig

    path = os.path.join(os.path.dirname(__file__), "..", "examples",
                        "windowed_join.yaml")
    raw = yaml.safe_load(open(path))
    raw["streams"][0]["buffer"]["interval"] = "150ms"  # speed up for CI
    raw["streams"][0]["output"] = {"type": "memory"}
    cfg = EngineConfig.from_dict(raw)
    assert not cfg.validate()
    eng = af.Engine(cfg)

    async def main():
        cancel = asyncio.Event()
        task = asyncio.ensure_future(eng.run_with_cancellation(cancel))
        for _ in range(100):
            await asyncio.sleep(0.1)
            try:
                e = eng.runtime.get("joiner")
                if e.metrics.output_messages > 0:
                    break
            except Exception:
                pass
        cancel.set()
        try:
            await asyncio.wait_for(task, 25)
        except asyncio.TimeoutError:
            lines = []
            for t in asyncio.all_tasks():
                if t is asyncio.current_task():
                    continue
                st = t.get_stack(limit=6)
                lines.append(" <- ".join(
                    f"{f.f_code.co_qualname}:{f.f_lineno}" for f in st))
            raise AssertionError("engine shutdown hang; tasks:\n"
                                 + "\n".join(lines))
        out = eng.runtime.get("joiner").stream.output
        rows = [r for b in out.batches for r in b.to_rows()]
        assert rows, "join emitted nothing"
        assert {"uid", "amount", "score"} <= set(rows[0].keys())
        assert eng.runtime.get("joiner").metrics.processing_errors == 0

    run(main(), timeout=60)


def test_session_window_deterministic_clock(run):
    """Session gap semantics under an injected clock: bursts separated by
    more than `gap` close a session; activity inside the gap keeps it
    open (session_window.rs:107-143) — no sleeps, fully deterministic."""
    import torch

    from arkflow_amd.batch import Column, MessageBatch
    from arkflow_amd.buffers.windows import SessionWindowBuffer
    from arkflow_amd.spi import NoopAck

    clock = [0.0]
    buf = SessionWindowBuffer({"gap": "5s"})
    buf._now = lambda: clock[0]

    def b(i):
        return MessageBatch({"id": Column(
            "numeric", torch.tensor([i], dtype=torch.int64))})

    # burst 1: t=0,1,2 — activity within the gap keeps the session open
    for t, i in [(0.0, 0), (1.0, 1), (2.0, 2)]:
        clock[0] = t
        run(buf.write(b(i), NoopAck()))
        assert buf.try_emit() is None
    clock[0] = 6.9  # 4.9s after last message: still open
    assert buf.try_emit() is None
    clock[0] = 7.1  # gap exceeded: session closes with the whole burst
    out = buf.try_emit()
    assert out is not None
    batch, _ack = out
    assert batch.column("id").data.tolist() == [0, 1, 2]
    # idle: nothing more to emit
    assert buf.try_emit() is None
    # burst 2 starts a fresh session
    clock[0] = 100.0
    run(buf.write(b(7), NoopAck()))
    assert buf.try_emit() is None
    clock[0] = 106.0
    out2 = buf.try_emit()
    assert out2 is not None and out2[0].column("id").data.tolist() == [7]


def test_tumbling_window_deterministic_clock(run):
    """Tumbling intervals under an injected clock: each window emits once
    per interval with exactly the batches that arrived inside it."""
    import torch

    from arkflow_amd.batch import Column, MessageBatch
    from arkflow_amd.buffers.windows import TumblingWindowBuffer
    from arkflow_amd.spi import NoopAck

    clock = [0.0]
    buf = TumblingWindowBuffer({"interval": "10s"})
    buf._now = lambda: clock[0]

    def b(i):
        return MessageBatch({"id": Column(
            "numeric", torch.tensor([i], dtype=torch.int64))})

    for t, i in [(0.0, 0), (3.0, 1), (9.0, 2)]:
        clock[0] = t
        run(buf.write(b(i), NoopAck()))
        assert buf.try_emit() is None
    clock[0] = 10.5
    out = buf.try_emit()
    assert out is not None
    assert out[0].column("id").data.tolist() == [0, 1, 2]
    # next window: one batch at t=12, fires after t=20.5
    clock[0] = 12.0
    run(buf.write(b(9), NoopAck()))
    clock[0] = 20.0
    assert buf.try_emit() is None
    clock[0] = 22.6
    out2 = buf.try_emit()
    assert out2 is not None and out2[0].column("id").data.tolist() == [9]

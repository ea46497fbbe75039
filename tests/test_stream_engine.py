"""Engine-level tests with stub components — the reference's dominant test
pattern (stream/mod.rs:594-1448: StubInput, StubOutput/always-fails,
CountingOutput, ordering, backpressure)."""
import asyncio

import pytest

import arkflow_amd as af
from arkflow_amd.batch import MessageBatch
from arkflow_amd.config import EngineConfig, PipelineConfig, StreamConfig
from arkflow_amd.errors import EOFError_
from arkflow_amd.pipeline import Pipeline
from arkflow_amd.spi import Ack, Input, NoopAck, Output, Processor
from arkflow_amd.stream import Stream, build_stream


class StubInput(Input):
    """VecDeque-backed input, EOF at end (reference stream/mod.rs:594+)."""

    def __init__(self, batches):
        self.batches = list(batches)
        self.acked = []

    async def read(self):
        if not self.batches:
            raise EOFError_()
        b = self.batches.pop(0)

        class A(Ack):
            def __init__(self, sink, tag):
                self.sink, self.tag = sink, tag

            async def ack(self):
                self.sink.append(self.tag)

        return b, A(self.acked, b.num_rows)


class CountingOutput(Output):
    def __init__(self):
        self.rows = 0
        self.batches = []

    async def write(self, batch):
        self.rows += batch.num_rows
        self.batches.append(batch)


class FailingOutput(Output):
    """Always fails → acks withheld (reference StubOutput)."""

    async def write(self, batch):
        raise RuntimeError("sink down")


class AddOneProcessor(Processor):
    async def process(self, batch):
        import torch
        col = batch.column("v")
        from arkflow_amd.batch import Column
        return [batch.with_columns({"v": Column("numeric", col.data + 1)})]


class ExplodeProcessor(Processor):
    """Returns multiple batches (ProcessResult::Multiple)."""

    async def process(self, batch):
        return [batch, batch]


class DropAllProcessor(Processor):
    async def process(self, batch):
        return []


def _mk(v):
    return MessageBatch.from_dict({"v": v})


def _stream(inp, procs, out, thread_num=2, error_output=None, buffer=None):
    cfg = StreamConfig(
        id="t", input={"type": "memory"}, output={"type": "drop"},
        pipeline=PipelineConfig(thread_num=thread_num),
    )
    return Stream(cfg, inp, Pipeline(procs), out,
                  error_output=error_output, buffer=buffer)


def test_end_to_end_order_and_acks(run):
    inp = StubInput([_mk([i]) for i in range(50)])
    out = CountingOutput()
    s = _stream(inp, [AddOneProcessor()], out, thread_num=4)
    run(s.run(asyncio.Event()))
    assert out.rows == 50
    # ordered output: values arrive in input order despite 4 workers
    vals = [b.column("v").to_pylist()[0] for b in out.batches]
    assert vals == [i + 1 for i in range(50)]
    assert len(inp.acked) == 50


def test_multiple_fanout(run):
    inp = StubInput([_mk([1, 2])])
    out = CountingOutput()
    s = _stream(inp, [ExplodeProcessor(), AddOneProcessor()], out)
    run(s.run(asyncio.Event()))
    assert out.rows == 4  # exploded to 2 batches × 2 rows


def test_none_result_still_acks(run):
    inp = StubInput([_mk([1]), _mk([2])])
    out = CountingOutput()
    s = _stream(inp, [DropAllProcessor()], out)
    run(s.run(asyncio.Event()))
    assert out.rows == 0
    assert len(inp.acked) == 2


def test_failing_output_withholds_acks(run):
    inp = StubInput([_mk([1]), _mk([2])])
    out = FailingOutput()
    s = _stream(inp, [], out)
    run(s.run(asyncio.Event()))
    assert inp.acked == []  # ack withheld on sink failure
    assert s.metrics.output_errors == 2


def test_error_routed_to_error_output(run):
    class Boom(Processor):
        async def process(self, batch):
            raise ValueError("boom")

    inp = StubInput([_mk([7])])
    out = CountingOutput()
    err_out = CountingOutput()
    s = _stream(inp, [Boom()], out, error_output=err_out)
    run(s.run(asyncio.Event()))
    assert out.rows == 0
    assert err_out.rows == 1
    assert s.metrics.processing_errors == 1
    assert len(inp.acked) == 1  # error path still acks after error_output


def test_cancellation_stops_stream(run):
    class Endless(Input):
        async def read(self):
            await asyncio.sleep(0.01)
            return _mk([1]), NoopAck()

    out = CountingOutput()
    s = _stream(Endless(), [], out)

    async def main():
        cancel = asyncio.Event()
        task = asyncio.ensure_future(s.run(cancel))
        await asyncio.sleep(0.15)
        cancel.set()
        await asyncio.wait_for(task, 5)

    run(main())
    assert out.rows > 0


def test_backpressure_bounded(run):
    """1024-in-flight cap: a slow output must not let in-flight grow
    unboundedly (reference stream/mod.rs:1345 notify wakeup test)."""
    N = 300
    inp = StubInput([_mk([i]) for i in range(N)])

    class SlowOutput(CountingOutput):
        async def write(self, batch):
            await asyncio.sleep(0)
            await super().write(batch)

    out = SlowOutput()
    s = _stream(inp, [], out, thread_num=8)
    run(s.run(asyncio.Event()))
    assert out.rows == N
    vals = [b.column("v").to_pylist()[0] for b in out.batches]
    assert vals == list(range(N))


def test_engine_runs_config_to_eof(run):
    cfg = EngineConfig.from_dict({
        "streams": [{
            "id": "s1",
            "input": {"type": "generate", "count": 100, "batch_size": 10,
                      "interval": "0ms",
                      "fields": {"v": {"dtype": "float32"}}},
            "pipeline": {"thread_num": 2, "processors": []},
            "output": {"type": "memory"},
        }]
    })
    eng = af.Engine(cfg)
    run(eng.run_with_cancellation())
    entry = eng.runtime.entries["s1"]
    assert entry.metrics.input_messages == 100
    assert entry.metrics.output_messages == 100
    assert entry.state.value == "stopped"


def test_config_validation():
    cfg = EngineConfig.from_dict({
        "streams": [{
            "id": "bad",
            "input": {"type": "nope"},
            "output": {"type": "drop"},
        }]
    })
    errs = cfg.validate()
    assert any("unknown input type" in e for e in errs)


def test_runtime_lifecycle(run):
    async def main():
        cfg = EngineConfig.from_dict({
            "streams": [{
                "id": "s1",
                "input": {"type": "generate", "batch_size": 5,
                          "interval": "5ms",
                          "fields": {"v": {"dtype": "float32"}}},
                "output": {"type": "drop"},
            }]
        })
        eng = af.Engine(cfg)
        for sc in cfg.streams:
            eng.runtime.register(sc)
        await eng.runtime.start("s1")
        assert eng.runtime.get("s1").state.value == "running"
        await asyncio.sleep(0.05)
        await eng.runtime.stop("s1")
        assert eng.runtime.get("s1").state.value == "stopped"
        await eng.runtime.restart("s1")
        assert eng.runtime.get("s1").state.value == "running"
        assert eng.runtime.get("s1").metrics.restarts == 1
        await eng.runtime.stop("s1")
        ops_kinds = {e.kind for e in eng.runtime.events.list()}
        assert {"registered", "starting", "running", "stopping"} <= ops_kinds

    run(main())


def test_replace_config_and_rollback(run):
    """reference runtime.rs:554-632 replace_config with rollback-on-failure."""
    async def main():
        cfg = EngineConfig.from_dict({
            "streams": [{
                "id": "s1",
                "input": {"type": "generate", "batch_size": 2,
                          "interval": "10ms",
                          "fields": {"v": {"dtype": "float32"}}},
                "output": {"type": "drop"},
            }]
        })
        eng = af.Engine(cfg)
        eng.runtime.register(cfg.streams[0])
        await eng.runtime.start("s1")
        # replace with a valid config
        from arkflow_amd.config import _parse_stream
        new_sc = _parse_stream({
            "id": "s1",
            "input": {"type": "generate", "batch_size": 5, "interval": "10ms",
                      "fields": {"v": {"dtype": "float32"}}},
            "output": {"type": "memory"},
        }, 0)
        await eng.runtime.replace_config("s1", new_sc)
        assert eng.runtime.get("s1").state.value == "running"
        assert eng.runtime.get("s1").config.output == {"type": "memory"}
        # replace with a BROKEN config → rollback to previous, still running
        bad_sc = _parse_stream({
            "id": "s1",
            "input": {"type": "generate", "batch_size": 1},
            "output": {"type": "sql"},  # missing 'table' → build error
        }, 0)
        import pytest as _pt
        with _pt.raises(Exception):
            await eng.runtime.replace_config("s1", bad_sc)
        assert eng.runtime.get("s1").state.value == "running"
        assert eng.runtime.get("s1").config.output == {"type": "memory"}
        await eng.runtime.stop_all()

    run(main(), timeout=30)


def test_control_plane_apply_and_rollback(run):
    async def main():
        cfg = EngineConfig.from_dict({
            "streams": [{
                "id": "s1",
                "input": {"type": "generate", "batch_size": 1,
                          "interval": "20ms",
                          "fields": {"v": {"dtype": "float32"}}},
                "output": {"type": "drop"},
            }]
        })
        eng = af.Engine(cfg)
        for sc in cfg.streams:
            eng.runtime.register(sc)
        await eng.runtime.start("s1")
        cp = eng.control_plane
        v1 = {"streams": [{"id": "s1",
                           "input": {"type": "generate", "batch_size": 2,
                                     "interval": "20ms",
                                     "fields": {"v": {"dtype": "float32"}}},
                           "output": {"type": "drop"}}]}
        r1 = await cp.apply_configuration(v1, note="v1")
        assert r1["applied"] and r1["version"] == 1
        v2 = {"streams": [{"id": "s2",
                           "input": {"type": "generate", "batch_size": 1,
                                     "interval": "20ms",
                                     "fields": {"v": {"dtype": "float32"}}},
                           "output": {"type": "drop"}}]}
        r2 = await cp.apply_configuration(v2, note="v2")
        assert r2["applied"] and set(eng.runtime.entries) == {"s2"}
        rb = await cp.rollback(1)
        assert rb["applied"] and set(eng.runtime.entries) == {"s1"}
        assert len(cp.versions.list()) == 3  # v1, v2, rollback-apply
        await eng.runtime.stop_all()

    run(main(), timeout=30)


def test_input_disconnect_reconnect(run, monkeypatch):
    """DisconnectionError → backoff, reconnect, resume reading
    (reference stream/mod.rs:289-306)."""
    from arkflow_amd import stream as stream_mod
    from arkflow_amd.errors import DisconnectionError
    from arkflow_amd.pipeline import Pipeline
    from arkflow_amd.spi import Input, NoopAck
    from arkflow_amd.stream import Stream

    monkeypatch.setattr(stream_mod, "RECONNECT_SECS", 0.05)

    class FlakyInput(Input):
        def __init__(self):
            self.reads = 0
            self.connects = 0

        async def connect(self):
            self.connects += 1

        async def read(self):
            self.reads += 1
            if self.reads == 2:
                raise DisconnectionError("broker gone")
            if self.reads > 4:
                from arkflow_amd.errors import EOFError_
                raise EOFError_("done")
            return _mk([float(self.reads)]), NoopAck()

        async def close(self):
            pass

    async def main():
        inp = FlakyInput()
        out = CountingOutput()
        sc = StreamConfig(id="r", input={"type": "memory"},
                          output={"type": "drop"},
                          pipeline=PipelineConfig(thread_num=1))
        s = Stream(sc, inp, Pipeline([]), out)
        await asyncio.wait_for(s.run(asyncio.Event()), 15)
        assert s.metrics.input_reconnects == 1
        assert inp.connects >= 2  # initial + reconnect
        assert out.rows == 3  # reads 1, 3, 4 delivered

    run(main())


def test_many_streams_one_engine(run, tmp_path):
    """Four concurrent streams (different buffers/durability) share one
    engine: all progress, stop cleanly, no cross-talk."""
    async def main():
        cfg = EngineConfig.from_dict({"streams": [
            {"id": "plain",
             "input": {"type": "generate", "batch_size": 64,
                       "interval": "2ms",
                       "fields": {"v": {"dtype": "float32"}}},
             "output": {"type": "memory"}},
            {"id": "durable",
             "input": {"type": "generate", "batch_size": 64,
                       "interval": "2ms",
                       "fields": {"v": {"dtype": "float32"}}},
             "durability": {"enabled": True, "path": str(tmp_path),
                            "sync_policy": "group_commit"},
             "output": {"type": "drop"}},
            {"id": "windowed",
             "input": {"type": "generate", "batch_size": 64,
                       "interval": "2ms",
                       "fields": {"k": {"dtype": "int64", "low": 0,
                                        "high": 8},
                                  "v": {"dtype": "float32"}}},
             "buffer": {"type": "tumbling_window", "interval": "100ms"},
             "pipeline": {"processors": [
                 {"type": "sql",
                  "query": "SELECT k, count(*) AS n FROM flow GROUP BY k"}]},
             "output": {"type": "drop"}},
            {"id": "exprs",
             "input": {"type": "generate", "batch_size": 64,
                       "interval": "2ms",
                       "fields": {"v": {"dtype": "float32"}}},
             "pipeline": {"processors": [
                 {"type": "vrl", "statement": ".v2 = .v * 2"}]},
             "output": {"type": "drop"}},
        ]})
        eng = af.Engine(cfg)
        cancel = asyncio.Event()
        task = asyncio.ensure_future(eng.run_with_cancellation(cancel))
        await asyncio.sleep(2.0)
        cancel.set()
        await asyncio.wait_for(task, 60)
        for sid in ("plain", "durable", "windowed", "exprs"):
            e = eng.runtime.get(sid)
            assert e.state.value == "stopped", sid
            assert e.metrics.input_messages > 0, sid
            assert e.metrics.processing_errors == 0, sid
            assert e.metrics.output_errors == 0, sid

    run(main(), timeout=90)


def test_lifecycle_timeout_marks_operation(run):
    """A lifecycle op that exceeds its timeout is recorded as TIMEOUT
    (reference control_plane.rs op bookkeeping)."""
    async def main():
        cfg = EngineConfig.from_dict({"streams": [{
            "id": "s1",
            "input": {"type": "generate", "batch_size": 1, "interval": "50ms",
                      "fields": {"v": {"dtype": "float32"}}},
            "output": {"type": "drop"}}]})
        eng = af.Engine(cfg)
        for sc in cfg.streams:
            eng.runtime.register(sc)

        async def hang(stream_id):
            await asyncio.sleep(60)

        eng.runtime.start = hang  # simulate a stuck start
        r = await eng.control_plane.lifecycle("s1", "start", timeout=0.1)
        assert r["state"] == "timed_out"
        op = eng.runtime.operations.get(r["id"])
        assert op.state.value == "timed_out"

    run(main(), timeout=30)


def test_fusable_chain_detection():
    """Whole-step graph fusion triggers exactly on the generate→filter→mlp
    chain (GPU device) and nothing else."""
    import torch

    from arkflow_amd.inputs.generate import GenerateInput
    from arkflow_amd.processors.inference import InferenceProcessor
    from arkflow_amd.processors.sql import SqlProcessor
    from arkflow_amd.stream import fusable_chain

    class R:
        device = torch.device("cuda")

    class CfgIn:
        def __init__(self, **kw):
            self.input = kw

    fields = {f"f{i}": {"dtype": "float32"} for i in range(4)}
    fields["k"] = {"dtype": "int64", "low": 0, "high": 10}
    gen = GenerateInput({"batch_size": 64, "interval": "0ms",
                         "fields": fields})
    sql = SqlProcessor({"query": "SELECT * FROM flow WHERE f0 >= 0.5"})
    mlp = InferenceProcessor({"model": "mlp_anomaly",
                              "columns": [f"f{i}" for i in range(4)]})
    cfg = CfgIn()
    assert fusable_chain(cfg, gen, [sql, mlp], R())
    # opt-out
    assert not fusable_chain(CfgIn(fuse=False), gen, [sql, mlp], R())
    # CPU device → no
    class RC:
        device = torch.device("cpu")
    assert not fusable_chain(cfg, gen, [sql, mlp], RC())
    # non-trivial sql → no
    agg = SqlProcessor({"query": "SELECT k, count(*) c FROM flow GROUP BY k"})
    assert not fusable_chain(cfg, gen, [agg, mlp], R())
    # count-limited generate → no (EOF semantics differ)
    gen2 = GenerateInput({"batch_size": 64, "interval": "0ms", "count": 100,
                          "fields": fields})
    assert not fusable_chain(cfg, gen2, [sql, mlp], R())
    # mismatched inference columns → no
    mlp2 = InferenceProcessor({"model": "mlp_anomaly", "columns": ["f0"]})
    assert not fusable_chain(cfg, gen, [sql, mlp2], R())


def test_direct_mode_passthrough(run):
    """Single-stage streams (empty pipeline, no buffer/WAL) take the direct
    read→write→ack loop; semantics (EOF, metrics, acks) match the graph."""
    import arkflow_amd as af
    from arkflow_amd.config import EngineConfig

    class _Cap:
        pass

    cfg = EngineConfig.from_dict({"streams": [{
        "id": "direct",
        "input": {"type": "generate", "batch_size": 8, "interval": "0ms",
                  "count": 64, "fields": {"v": {"dtype": "float32"}}},
        "output": {"type": "memory"},
    }]})
    from arkflow_amd.stream import build_stream
    stream = build_stream(cfg.streams[0])
    assert not stream.pipeline.processors

    async def main():
        import asyncio
        cancel = asyncio.Event()
        await asyncio.wait_for(stream.run(cancel), 30)

    run(main())
    assert stream.metrics.input_messages == 64
    assert stream.metrics.output_messages == 64
    assert sum(b.num_rows for b in stream.output.batches) == 64


def test_dedicated_thread_stream_lifecycle(run):
    """A dedicated_thread stream runs in its own event loop thread;
    start/stop/metrics behave identically to the in-loop path."""
    import arkflow_amd as af
    from arkflow_amd.config import EngineConfig

    cfg = EngineConfig.from_dict({"streams": [{
        "id": "th",
        "dedicated_thread": True,
        "input": {"type": "generate", "batch_size": 16, "interval": "5ms",
                  "fields": {"v": {"dtype": "float32"}}},
        "pipeline": {"thread_num": 1, "processors": [
            {"type": "sql", "query": "SELECT * FROM flow WHERE v >= 0"}]},
        "output": {"type": "drop"},
    }]})
    eng = af.Engine(cfg)
    for sc in cfg.streams:
        eng.runtime.register(sc)

    async def main():
        import asyncio
        await eng.runtime.start("th")
        entry = eng.runtime.get("th")
        assert entry.thread is not None and entry.thread.is_alive()
        for _ in range(100):
            if entry.metrics.output_messages > 0:
                break
            await asyncio.sleep(0.05)
        assert entry.metrics.output_messages > 0
        await eng.runtime.stop("th")
        assert entry.state.value == "stopped"
        assert not entry.thread.is_alive()
        # restartable
        await eng.runtime.start("th")
        await asyncio.sleep(0.2)
        await eng.runtime.stop_all()

    run(main(), timeout=60)


def test_fusable_agg_chain_detection():
    """The GROUP BY step-graph fusion triggers exactly on generate→sql
    (simple filter + single-key aggregates) on GPU, with sane table sizing."""
    import torch

    from arkflow_amd.inputs.generate import GenerateInput
    from arkflow_amd.processors.sql import SqlProcessor
    from arkflow_amd.stream import fusable_agg_chain

    class R:
        device = torch.device("cuda")

    class RC:
        device = torch.device("cpu")

    class CfgIn:
        def __init__(self, **kw):
            self.input = kw

    fields = {"f0": {"dtype": "float32"}, "f1": {"dtype": "float32"},
              "key": {"dtype": "int64", "low": 0, "high": 300}}
    gen = GenerateInput({"batch_size": 64, "interval": "0ms",
                         "fields": fields})
    agg = SqlProcessor({
        "query": "SELECT key, count(*) AS c, sum(f0) AS s, min(f1) AS m "
                 "FROM flow WHERE f0 >= 0.2 GROUP BY key"})
    cfg = CfgIn()
    spec = fusable_agg_chain(cfg, gen, [agg], R())
    assert spec is not None
    key, filt, plan, g_cap, table_size = spec
    assert key == "key" and filt == ("f0", 3, 0.2)
    assert g_cap == 300 and table_size == 2048
    assert [p[0] for p in plan] == ["key", "count", "sum", "min"]
    # opt-outs / mismatches
    assert fusable_agg_chain(CfgIn(fuse=False), gen, [agg], R()) is None
    assert fusable_agg_chain(cfg, gen, [agg], RC()) is None
    plain = SqlProcessor({"query": "SELECT * FROM flow WHERE f0 >= 0.5"})
    assert fusable_agg_chain(cfg, gen, [plain], R()) is None
    # key range too large for the LDS group tile → eager path
    wide = dict(fields, key={"dtype": "int64", "low": 0, "high": 1 << 20})
    genw = GenerateInput({"batch_size": 64, "interval": "0ms",
                          "fields": wide})
    assert fusable_agg_chain(cfg, genw, [agg], R()) is None
    # no WHERE is still fusable (keep-all filter)
    nowhere = SqlProcessor(
        {"query": "SELECT key, count(*) AS c FROM flow GROUP BY key"})
    spec2 = fusable_agg_chain(cfg, gen, [nowhere], R())
    assert spec2 is not None and spec2[1] is None


def test_lazy_step_batch_semantics():
    """_LazyStepBatch materializes columns exactly once, on first access;
    num_rows/device never trigger materialization (the fused source hands
    these out per step — building views eagerly cost more than the GPU
    step, profiles r2-21)."""
    import torch

    from arkflow_amd.batch import Column
    from arkflow_amd.ops.stepgraph import _LazyStepBatch

    calls = []

    def build():
        calls.append(1)
        return {"a": Column("numeric", torch.arange(4)),
                "b": Column("numeric", torch.ones(4))}

    b = _LazyStepBatch(build, 4, torch.device("cpu"), input_name="generate")
    assert b.num_rows == 4 and len(b) == 4
    assert b.device.type == "cpu"
    assert b.input_name == "generate"
    assert not calls  # nothing materialized yet
    assert b.column("a").data.tolist() == [0, 1, 2, 3]
    assert calls == [1]
    assert set(b.column_names) == {"a", "b"}
    assert calls == [1]  # cached — built once


def test_engine_dynamic_topic_routing_e2e():
    """YAML stream with expr topic/key routes rows into per-key-computed
    fake-bus topics through the full engine."""
    import asyncio

    from arkflow_amd.config import EngineConfig
    from arkflow_amd.inputs.brokers import FakeBus
    from arkflow_amd.stream import build_stream

    FakeBus.reset() if hasattr(FakeBus, "reset") else None
    cfg = EngineConfig.from_dict({"streams": [{
        "id": "router",
        "input": {"type": "generate", "batch_size": 64, "interval": "0ms",
                  "count": 192,
                  "fields": {"key": {"dtype": "int64", "low": 0, "high": 8},
                             "v": {"dtype": "float32"}}},
        "pipeline": {"thread_num": 1, "processors": []},
        "output": {"type": "kafka", "brokers": ["memory://router_e2e"],
                   "topic": {"expr": "'shard-' || (key % 2)"},
                   "key": {"expr": "key"}},
    }]})
    stream = build_stream(cfg.streams[0])

    async def run():
        cancel = asyncio.Event()
        await asyncio.wait_for(stream.run(cancel), 30)

    asyncio.new_event_loop().run_until_complete(run())
    bus = FakeBus.get("router_e2e")
    msgs = {t: [m for part in parts for m in part]
            for t, parts in bus.topics.items() if t.startswith("shard-")}
    assert set(msgs) == {"shard-0", "shard-1"}
    assert sum(len(v) for v in msgs.values()) == 192
    for t, rows in msgs.items():
        want = t.removeprefix("shard-")
        assert all(int(m[0]) % 2 == int(want) for m in rows)


def test_stream_sql_with_subquery_runs(run):
    """A pipeline sql processor using a scalar subquery executes per batch
    (the rewrite happens per execute; pre-parsed statements are shared)."""
    import torch

    from arkflow_amd.batch import MessageBatch
    from arkflow_amd.processors.sql import SqlProcessor

    proc = SqlProcessor({
        "query": "SELECT a FROM flow WHERE a > (SELECT avg(a) FROM flow) "
                 "ORDER BY a"})
    for base in (0, 100):
        batch = MessageBatch.from_dict(
            {"a": torch.arange(base, base + 10, dtype=torch.int64)})
        out = run(proc.process(batch))[0]
        # avg of base..base+9 is base+4.5 → keeps the top 5
        assert out.column("a").data.tolist() == list(
            range(base + 5, base + 10))

"""Stream: the per-stream task graph (THE hot loop).

MI355X-native re-design of reference crates/arkflow-core/src/stream/mod.rs:
  input → [WAL] → [buffer] → N processor workers → sequence-ordered output
with bounded queues, 1024-in-flight backpressure (stream/mod.rs:37,388-395),
at-least-once ack chain, EOF/reconnect semantics (:282-306), WAL replay before
ingest (:190-210) and the close order input→buffer→pipeline→output→
error_output→WAL (:542-591).

Host orchestration is asyncio (the Tokio analog); per-batch compute runs in
HIP kernels / C++ that release the GIL, so `thread_num` workers here give
overlap of GPU compute with ingest + output exactly like the reference's
worker tasks.
"""
from __future__ import annotations

import asyncio
import logging
import time
from typing import List, Optional

from .batch import MessageBatch
from .config import StreamConfig
from .errors import DisconnectionError, EOFError_
from .metrics import RuntimeMetrics
from .pipeline import Pipeline
from .registry import build_component
from .spi import Ack, Buffer, Input, NoopAck, Output, Resource

log = logging.getLogger("arkflow_amd.stream")

MAX_IN_FLIGHT = 1024       # reference stream/mod.rs:37
RECONNECT_SECS = 5.0       # reference stream/mod.rs:300 (patchable in tests)

_SENTINEL = object()


async def _race(coro, cancel: asyncio.Event):
    """Run *coro* but abort (returning _SENTINEL) if *cancel* fires first."""
    task = asyncio.ensure_future(coro)
    waiter = asyncio.ensure_future(cancel.wait())
    done, _ = await asyncio.wait(
        {task, waiter}, return_when=asyncio.FIRST_COMPLETED
    )
    if task in done:
        waiter.cancel()
        return task.result()
    task.cancel()
    try:
        await task
    except (asyncio.CancelledError, Exception):  # noqa: BLE001
        pass
    return _SENTINEL


class Stream:
    def __init__(
        self,
        config: StreamConfig,
        input_: Input,
        pipeline: Pipeline,
        output: Output,
        error_output: Optional[Output] = None,
        buffer: Optional[Buffer] = None,
        wal=None,
        temporaries: Optional[dict] = None,
        metrics: Optional[RuntimeMetrics] = None,
    ):
        self.config = config
        self.input = input_
        self.pipeline = pipeline
        self.output = output
        self.error_output = error_output
        self.buffer = buffer
        self.wal = wal
        self.temporaries = temporaries or {}
        self.metrics = metrics or RuntimeMetrics()
        self.device = _resolve_device(config.device)
        self.thread_num = config.pipeline.resolved_thread_num()
        # sequencing / backpressure state
        self._seq = 0
        self._next_seq = 0
        self._bp_event = asyncio.Event()

    # ------------------------------------------------------------------- run
    async def run(self, cancel: asyncio.Event) -> None:
        await self.input.connect()
        await self.output.connect()
        if self.error_output is not None:
            await self.error_output.connect()
        for t in self.temporaries.values():
            await t.connect()

        # single-stage passthrough (empty pipeline, no buffer/WAL/error
        # output — e.g. a whole-step-graph fused source feeding drop): skip
        # the worker/reorder task graph entirely; queue hops and sequencing
        # cost ~0.1 ms/step of pure asyncio overhead at batch 8192
        if (not self.pipeline.processors and self.buffer is None
                and self.wal is None and self.error_output is None):
            try:
                await self._run_direct(cancel)
            finally:
                await self._close_all()
            return

        qsize = self.thread_num * 4
        input_q: asyncio.Queue = asyncio.Queue(maxsize=qsize)
        output_q: asyncio.Queue = asyncio.Queue(maxsize=qsize)

        tasks: List[asyncio.Task] = []
        if self.buffer is not None:
            tasks.append(asyncio.ensure_future(
                self._do_buffer(input_q)))
        for i in range(self.thread_num):
            tasks.append(asyncio.ensure_future(
                self._do_processor(input_q, output_q, cancel)))
        out_task = asyncio.ensure_future(
            self._do_output(output_q))
        tasks.append(out_task)

        # WAL recovery replay BEFORE live ingest (stream/mod.rs:190-210):
        # consumers are already running so the bounded queue drains.
        if self.wal is not None:
            async for seq, batch in self.wal.read_after_cursor():
                from .wal.wal import WalAck
                if self.device.type == "cuda":
                    batch = batch.to(self.device)  # H2D: restore residency
                await self._forward(input_q, batch,
                                    WalAck(self.wal, seq, NoopAck()))

        in_task = asyncio.ensure_future(
            self._do_input(input_q, cancel))

        try:
            await in_task
            # input done (EOF or cancel): drain the buffer's final window
            if self.buffer is not None:
                await self.buffer.flush()
                await tasks[0]  # buffer task exits after drain
            # stop processors
            for _ in range(self.thread_num):
                await input_q.put(_SENTINEL)
            await asyncio.gather(
                *[t for t in tasks if t is not out_task and not t.done()])
            await out_task
        finally:
            for t in tasks + [in_task]:
                if not t.done():
                    t.cancel()
            # bounded: cleanup must terminate even if a sub-task wedges
            # (intermittent cancel-delivery stall; NOTES.md item 12).
            # asyncio.wait (not wait_for+gather) so a timeout leaves the
            # wedged tasks inspectable before we force-cancel them.
            t_w0 = time.monotonic()
            done, pending = await asyncio.wait(
                set(tasks) | {in_task}, timeout=10)
            if time.monotonic() - t_w0 > 5:
                log.warning("stream %s: cleanup wait took %.2fs; done=%s "
                            "pending=%s", self.config.id,
                            time.monotonic() - t_w0,
                            [repr(t)[:120] for t in done],
                            [repr(t)[:160] for t in pending])
            if pending:
                stuck = []
                for t in pending:
                    frames = t.get_stack(limit=5)
                    stuck.append(" <- ".join(
                        f"{f.f_code.co_qualname}:{f.f_lineno}"
                        for f in frames))
                log.warning("stream %s: cleanup tasks stalled; forcing "
                            "close. stuck: %s", self.config.id, stuck)
                for t in pending:
                    t.cancel()
                await asyncio.wait(pending, timeout=5)
            await self._close_all()

    async def _run_direct(self, cancel: asyncio.Event) -> None:
        """Tight read→write→ack loop with the same EOF / reconnect / error
        semantics as the full graph (_do_input + _do_output)."""
        # sources that never block indefinitely (fused step graphs,
        # interval-0 generators) skip the per-read cancellation race —
        # ~20-30 us of task machinery per step; cancel is still honored
        # at every loop iteration
        fast = getattr(self.input, "nonblocking", False)
        while not cancel.is_set():
            try:
                t0 = time.perf_counter_ns()
                if fast:
                    # a nonblocking read may complete without ever yielding;
                    # yield explicitly so siblings (and the cancel setter)
                    # run — ~2 us vs the race's ~25 us
                    await asyncio.sleep(0)
                    batch, ack = await self.input.read()
                else:
                    item = await _race(self.input.read(), cancel)
                    if item is _SENTINEL:
                        return
                    batch, ack = item
            except EOFError_:
                return
            except DisconnectionError:
                self.metrics.input_reconnects += 1
                if await _race(asyncio.sleep(RECONNECT_SECS),
                               cancel) is _SENTINEL:
                    return
                try:
                    await self.input.connect()
                except Exception:  # noqa: BLE001
                    self.metrics.input_errors += 1
                continue
            except Exception:  # noqa: BLE001
                self.metrics.input_errors += 1
                log.exception("input read error in stream %s",
                              self.config.id)
                continue
            self.metrics.input_batches += 1
            self.metrics.input_messages += batch.num_rows
            self.metrics.stage_ns["input"] += time.perf_counter_ns() - t0
            try:
                t1 = time.perf_counter_ns()
                await self.output.write_batch([batch])
                self.metrics.output_batches += 1
                self.metrics.output_messages += batch.num_rows
                self.metrics.stage_ns["output"] += \
                    time.perf_counter_ns() - t1
                await ack.ack()
            except Exception:  # noqa: BLE001
                self.metrics.output_errors += 1
                log.exception("output error in stream %s", self.config.id)
                # ack withheld → at-least-once replay upstream

    async def _close_all(self) -> None:
        # close order per reference stream/mod.rs:542-591
        for closer in (
            self.input.close,
            (self.buffer.close if self.buffer else None),
            self.pipeline.close,
            self.output.close,
            (self.error_output.close if self.error_output else None),
            (self.wal.close if self.wal else None),
        ):
            if closer is None:
                continue
            try:
                await closer()
            except Exception:  # noqa: BLE001
                log.exception("close failure in stream %s", self.config.id)
        for t in self.temporaries.values():
            try:
                await t.close()
            except Exception:  # noqa: BLE001
                pass

    # ----------------------------------------------------------------- input
    async def _forward(self, input_q: asyncio.Queue, batch: MessageBatch,
                       ack: Ack) -> None:
        """Route to buffer or straight to the processor queue
        (stream/mod.rs:229-243)."""
        if self.buffer is not None:
            await self.buffer.write(batch, ack)
        else:
            await input_q.put((batch, ack))

    async def _do_input(self, input_q: asyncio.Queue, cancel: asyncio.Event
                        ) -> None:
        while not cancel.is_set():
            try:
                self._t_read0 = time.perf_counter_ns()
                item = await _race(self.input.read(), cancel)
                if item is _SENTINEL:
                    break
                batch, ack = item
            except EOFError_:
                break
            except DisconnectionError:
                self.metrics.input_reconnects += 1
                slept = await _race(asyncio.sleep(RECONNECT_SECS), cancel)
                if slept is _SENTINEL:
                    break
                try:
                    await self.input.connect()
                except Exception:  # noqa: BLE001
                    self.metrics.input_errors += 1
                continue
            except Exception:  # noqa: BLE001
                self.metrics.input_errors += 1
                log.exception("input read error in stream %s", self.config.id)
                continue
            self.metrics.input_batches += 1
            self.metrics.input_messages += batch.num_rows
            self.metrics.stage_ns["input"] += time.perf_counter_ns() - (
                getattr(self, "_t_read0", time.perf_counter_ns()))
            if self.wal is not None:
                from .wal.wal import WalAck
                seq = await self.wal.append(batch)
                self.metrics.wal_lag = max(0, seq - self.wal.store.cursor)
                ack = WalAck(self.wal, seq, ack)
            await self._forward(input_q, batch, ack)

    # ---------------------------------------------------------------- buffer
    async def _do_buffer(self, input_q: asyncio.Queue) -> None:
        while True:
            item = await self.buffer.read()
            if item is None:
                break
            batch, ack = item
            await input_q.put((batch, ack))

    # ------------------------------------------------------------- processor
    async def _do_processor(self, input_q: asyncio.Queue,
                            output_q: asyncio.Queue,
                            cancel: asyncio.Event) -> None:
        while True:
            # backpressure (stream/mod.rs:388-395): bound un-acked in-flight
            while self._seq - self._next_seq > MAX_IN_FLIGHT:
                self._bp_event.clear()
                await self._bp_event.wait()
            item = await input_q.get()
            if item is _SENTINEL:
                await output_q.put(_SENTINEL)
                return
            batch, ack = item
            seq = self._seq
            self._seq += 1
            try:
                t0 = time.perf_counter_ns()
                results = await self.pipeline.process(batch)
                self.metrics.stage_ns["process"] += time.perf_counter_ns() - t0
                await output_q.put((seq, results, None, ack))
            except Exception as e:  # noqa: BLE001
                self.metrics.processing_errors += 1
                await output_q.put((seq, None, (batch, e), ack))

    # ---------------------------------------------------------------- output
    async def _do_output(self, output_q: asyncio.Queue) -> None:
        pending = {}
        sentinels = 0
        while sentinels < self.thread_num or pending:
            item = await output_q.get()
            if item is _SENTINEL:
                sentinels += 1
                continue
            seq, results, err, ack = item
            pending[seq] = (results, err, ack)
            while self._next_seq in pending:
                results, err, ack = pending.pop(self._next_seq)
                await self._emit(results, err, ack)
                self._next_seq += 1
                self._bp_event.set()  # wake backpressured processors

    async def _emit(self, results, err, ack: Ack) -> None:
        if err is not None:
            batch, exc = err
            if self.error_output is not None:
                try:
                    await self.error_output.write_batch([batch])
                    await ack.ack()
                except Exception:  # noqa: BLE001
                    self.metrics.output_errors += 1
                    log.exception("error-output failure in %s", self.config.id)
            else:
                log.error("stream %s processing error (no error_output): %s",
                          self.config.id, exc)
                await ack.ack()
            return
        if not results:
            await ack.ack()
            return
        try:
            t0 = time.perf_counter_ns()
            await self.output.write_batch(results)
            self.metrics.stage_ns["output"] += time.perf_counter_ns() - t0
            self.metrics.output_batches += len(results)
            self.metrics.output_messages += sum(b.num_rows for b in results)
            await ack.ack()
        except Exception:  # noqa: BLE001
            # ack withheld → WAL / source offset replay (stream/mod.rs:517-537)
            self.metrics.output_errors += 1
            log.exception("output failure in stream %s (ack withheld)",
                          self.config.id)


def build_stream(config: StreamConfig) -> Stream:
    """StreamConfig → wired Stream (reference stream/mod.rs:1495 build()).

    Build order: temporaries → input → pipeline → output → error_output →
    buffer → WAL.
    """
    resource = Resource()
    resource.device = _resolve_device(config.device)

    temporaries = {}
    for t_spec in config.temporary:
        spec = dict(t_spec)
        name = spec.pop("name", spec.get("type"))
        temporaries[name] = build_component("temporary", spec, resource)
    resource.temporaries = temporaries

    input_ = build_component("input", config.input, resource)
    processors = [
        build_component("processor", p, resource)
        for p in config.pipeline.processors
    ]
    output = build_component("output", config.output, resource)
    error_output = (
        build_component("output", config.error_output, resource)
        if config.error_output else None
    )
    buffer = (
        build_component("buffer", config.buffer, resource)
        if config.buffer else None
    )
    wal = None
    if config.durability and config.durability.enabled:
        from .parallel import dist as afdist
        from .wal.wal import Wal
        # per-rank WAL identity: N engine shards sharing one config must not
        # interleave frames in one log file
        sid = config.id if afdist.world_size() <= 1 \
            else f"{config.id}.rank{afdist.rank()}"
        wal = Wal.open(config.durability, stream_id=sid)

    # whole-step hipGraph fusion: a generate → sql(simple filter) →
    # mlp-inference chain on GPU replays as ONE graph per step
    # (ops/stepgraph.py — the same path bench.py's flagship uses)
    if wal is None and fusable_chain(config, input_, processors, resource):
        from .models.mlp import MlpAnomalyDetector
        from .ops.stepgraph import FusedGenerateFilterInfer, FusedStepSource
        sqlp, infp = processors
        col, opi, scalar = sqlp._fast_filter
        op = {0: "<", 1: "<=", 2: ">", 3: ">=", 4: "==", 5: "!="}[opi]
        float_fields = [f for f, spec in input_.fields.items()
                        if str(spec.get("dtype", "float32"))
                        not in ("int32", "int64")]
        mlp = MlpAnomalyDetector(len(float_fields), infp._mlp_hidden,
                                 resource.device, infp.seed)

        def mk(seed):
            return FusedGenerateFilterInfer(
                input_.fields, input_.batch_size, col, op, scalar, mlp,
                resource.device, seed=seed)

        log.info("stream %s: fused generate→filter→mlp into one hipGraph",
                 config.id)
        # direct-mode streams (no buffer) feeding a non-retaining output
        # (drop) never hold a batch past the loop iteration — zero-copy
        # views are safe and skip ~17 clone kernels per step
        clone = not (buffer is None
                     and getattr(output, "retains", True) is False)
        seeds = iter(range(1, 16))
        input_ = FusedStepSource(
            mk(input_.seed), ninstances=2, clone=clone,
            make_instance=lambda: mk(input_.seed + next(seeds) * 7919))
        processors = []

    # same fusion for generate → sql(filter + GROUP BY aggregates): the
    # capture-safe hash-agg chain replays inside the step graph
    agg_spec = (fusable_agg_chain(config, input_, processors, resource)
                if wal is None else None)
    if agg_spec is not None:
        from .ops.stepgraph import FusedGenerateAgg, FusedStepSource
        key, filt, plan, g_cap, table_size = agg_spec
        if filt is None:
            f0 = next(f for f, s in input_.fields.items()
                      if str(s.get("dtype", "float32"))
                      not in ("int32", "int64"))
            fcol, op, scalar = f0, "!=", float("nan")  # x != nan ⇒ keep all
        else:
            fcol, opi, scalar = filt
            op = {0: "<", 1: "<=", 2: ">", 3: ">=", 4: "==", 5: "!="}[opi]

        def mka(seed):
            return FusedGenerateAgg(
                input_.fields, input_.batch_size, fcol, op, scalar, key,
                plan, resource.device, seed=seed, g_cap=g_cap,
                table_size=table_size)

        log.info("stream %s: fused generate→filter→group-by into one "
                 "hipGraph", config.id)
        clone = not (buffer is None
                     and getattr(output, "retains", True) is False)
        seeds = iter(range(1, 16))
        input_ = FusedStepSource(
            mka(input_.seed), ninstances=2, clone=clone,
            make_instance=lambda: mka(input_.seed + next(seeds) * 7919))
        processors = []

    return Stream(
        config, input_, Pipeline(processors), output, error_output,
        buffer, wal, temporaries,
    )


def fusable_chain(config, input_, processors, resource) -> bool:
    """True when the stream is exactly the fusable hot chain:
    GPU device; generate input (typed fields, no count/interval, ≤1 int64
    key, f32 floats); pipeline = [sql simple scalar filter on a float
    field, mlp_anomaly inference over the float fields with the default
    score column]. ``fuse: false`` in the stream config opts out."""
    import torch

    from .inputs.generate import GenerateInput
    from .processors.inference import InferenceProcessor
    from .processors.sql import SqlProcessor

    if config.input.get("fuse") is False or \
            getattr(resource, "device", torch.device("cpu")).type != "cuda":
        return False
    if not isinstance(input_, GenerateInput) or not input_.fields:
        return False
    if input_.count is not None or input_.interval_secs > 0:
        return False
    floats, ints = [], []
    for f, spec in input_.fields.items():
        dt = str(spec.get("dtype", "float32"))
        if dt == "float32":
            floats.append(f)
        elif dt == "int64":
            ints.append(f)
        else:
            return False
    if not floats or len(ints) > 1:
        return False
    if len(processors) != 2:
        return False
    sqlp, infp = processors
    if not isinstance(sqlp, SqlProcessor) or sqlp._fast_filter is None:
        return False
    if sqlp._fast_filter[0] not in floats:
        return False
    if not isinstance(infp, InferenceProcessor) or \
            infp.model_name not in ("mlp", "mlp_anomaly"):
        return False
    if infp.output_column != "score":
        return False
    if infp.in_features and infp.in_features != len(floats):
        return False
    cols = infp.columns
    if cols is not None and list(cols) != floats:
        return False
    return True


def fusable_agg_chain(config, input_, processors, resource):
    """When the stream is exactly generate → sql(simple filter + single-key
    GROUP BY aggregates) on GPU, return (key, filt, plan, g_cap, table_size)
    for FusedGenerateAgg; else None. The key field's configured range bounds
    the group count (must fit the 2048-group LDS tile). ``fuse: false``
    opts out."""
    import torch

    from .inputs.generate import GenerateInput
    from .processors.sql import SqlProcessor

    if config.input.get("fuse") is False or \
            getattr(resource, "device", torch.device("cpu")).type != "cuda":
        return None
    if not isinstance(input_, GenerateInput) or not input_.fields:
        return None
    if input_.count is not None or input_.interval_secs > 0:
        return None
    if len(processors) != 1 or not isinstance(processors[0], SqlProcessor):
        return None
    fast = processors[0]._fast_agg
    if fast is None:
        return None
    key, filt, plan = fast
    spec = input_.fields.get(key)
    if spec is None or str(spec.get("dtype", "")) != "int64":
        return None
    for f, s in input_.fields.items():
        if f == key:
            continue
        if str(s.get("dtype", "float32")) != "float32":
            return None
    if filt is not None and filt[0] not in input_.fields:
        return None
    for fn, col, _alias in plan:
        if fn in ("sum", "min", "max", "avg") and col not in input_.fields:
            return None
    key_range = int(float(spec.get("high", 100.0))) - \
        int(float(spec.get("low", 0.0)))
    if key_range < 1 or key_range > 2048:
        return None  # group table must fit the LDS-tiled reduction
    g_cap = min(max(key_range, 1), 2048)
    table_size = 256
    while table_size < 4 * g_cap:
        table_size <<= 1
    return key, filt, plan, g_cap, table_size


def _resolve_device(device: Optional[str]):
    import torch
    if device is not None:
        return torch.device(device)
    if torch.cuda.is_available():
        return torch.device("cuda", torch.cuda.current_device())
    return torch.device("cpu")

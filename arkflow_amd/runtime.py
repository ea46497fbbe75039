"""Process control plane: RuntimeManager + operation/event stores.

Mirrors reference crates/arkflow-core/src/{runtime.rs,control.rs}:
per-stream supervision with the Created/Starting/Running/Stopping/Stopped/
Failed/Restarting state machine (control.rs:57-65), desired/observed
generation + convergence (runtime.rs:250-272), a bounded idempotent-terminal
OperationStore (runtime.rs:33-185) and a 128-event ring EventStore
(runtime.rs:26-30,189-201).
"""
from __future__ import annotations

import asyncio
import enum
import logging
import time
import uuid
from collections import deque
from typing import Dict, List, Optional

from .config import StreamConfig
from .errors import ArkError, ConfigError
from .metrics import ControlEvent, RuntimeMetrics

log = logging.getLogger("arkflow_amd.runtime")

SHUTDOWN_TIMEOUT_SECS = 30.0  # reference runtime.rs stop timeout
EVENT_RING_CAPACITY = 128     # reference runtime.rs:21
OPERATION_STORE_CAPACITY = 256


class StreamState(str, enum.Enum):
    CREATED = "created"
    STARTING = "starting"
    RUNNING = "running"
    STOPPING = "stopping"
    STOPPED = "stopped"
    FAILED = "failed"
    RESTARTING = "restarting"


class DesiredState(str, enum.Enum):
    RUNNING = "running"
    STOPPED = "stopped"


class ConvergenceState(str, enum.Enum):
    CONVERGED = "converged"
    PENDING = "pending"
    DIVERGED = "diverged"


class OperationState(str, enum.Enum):
    PENDING = "pending"
    RUNNING = "running"
    SUCCEEDED = "succeeded"
    FAILED = "failed"
    TIMED_OUT = "timed_out"


class FailureClass(str, enum.Enum):
    """reference control.rs:179-189."""
    CONFIG = "config"
    CONNECTIVITY = "connectivity"
    RUNTIME = "runtime"
    TIMEOUT = "timeout"
    UNKNOWN = "unknown"


class EventStore:
    def __init__(self, capacity: int = EVENT_RING_CAPACITY):
        self._ring: deque = deque(maxlen=capacity)
        self._seq = 0
        self._subscribers: List[asyncio.Queue] = []
        self._sub_loops: dict = {}  # id(queue) → owning loop

    def push(self, stream_id: str, kind: str, message: str = "") -> ControlEvent:
        self._seq += 1
        ev = ControlEvent(self._seq, stream_id, kind, message)
        self._ring.append(ev)
        for q in list(self._subscribers):
            # dedicated-thread streams push from foreign threads; asyncio
            # queues are loop-affine, so route through the subscriber's loop
            loop = self._sub_loops.get(id(q))
            try:
                running = asyncio.get_running_loop()
            except RuntimeError:
                running = None
            try:
                if loop is not None and loop is not running:
                    loop.call_soon_threadsafe(self._safe_put, q, ev)
                else:
                    self._safe_put(q, ev)
            except RuntimeError:
                pass  # subscriber loop already closed
        return ev

    @staticmethod
    def _safe_put(q: asyncio.Queue, ev) -> None:
        try:
            q.put_nowait(ev)
        except asyncio.QueueFull:
            pass

    def list(self, after_seq: int = 0, limit: int = 100) -> List[ControlEvent]:
        return [e for e in self._ring if e.seq > after_seq][:limit]

    def subscribe(self) -> asyncio.Queue:
        q: asyncio.Queue = asyncio.Queue(maxsize=EVENT_RING_CAPACITY)
        self._subscribers.append(q)
        self._sub_loops[id(q)] = asyncio.get_event_loop()
        return q

    def unsubscribe(self, q: asyncio.Queue) -> None:
        if q in self._subscribers:
            self._subscribers.remove(q)
        self._sub_loops.pop(id(q), None)


class Operation:
    def __init__(self, op_id: str, stream_id: str, kind: str):
        self.id = op_id
        self.stream_id = stream_id
        self.kind = kind
        self.state = OperationState.PENDING
        self.error: Optional[str] = None
        self.created_at = time.time()
        self.finished_at: Optional[float] = None

    def to_dict(self) -> dict:
        return {
            "id": self.id, "stream_id": self.stream_id, "kind": self.kind,
            "state": self.state.value, "error": self.error,
            "created_at": self.created_at, "finished_at": self.finished_at,
        }


class OperationStore:
    """Bounded, idempotent-terminal (reference runtime.rs:33-185)."""

    def __init__(self, capacity: int = OPERATION_STORE_CAPACITY):
        self._ops: Dict[str, Operation] = {}
        self._order: deque = deque()
        self.capacity = capacity

    def create(self, stream_id: str, kind: str) -> Operation:
        op = Operation(uuid.uuid4().hex[:12], stream_id, kind)
        self._ops[op.id] = op
        self._order.append(op.id)
        while len(self._order) > self.capacity:
            old = self._order.popleft()
            self._ops.pop(old, None)
        return op

    def get(self, op_id: str) -> Optional[Operation]:
        return self._ops.get(op_id)

    def finish(self, op_id: str, state: OperationState,
               error: Optional[str] = None) -> None:
        op = self._ops.get(op_id)
        if op is None:
            return
        if op.state in (OperationState.SUCCEEDED, OperationState.FAILED,
                        OperationState.TIMED_OUT):
            return  # terminal states are idempotent
        op.state = state
        op.error = error
        op.finished_at = time.time()

    def list(self, limit: int = 100) -> List[Operation]:
        return [self._ops[i] for i in list(self._order)[-limit:] if i in self._ops]


class RuntimeEntry:
    def __init__(self, stream_id: str, config: StreamConfig):
        self.stream_id = stream_id
        self.config = config
        self.state = StreamState.CREATED
        self.desired = DesiredState.STOPPED
        self.desired_generation = 0
        self.observed_generation = 0
        self.metrics = RuntimeMetrics()
        self.task: Optional[asyncio.Task] = None
        self.cancel: Optional[asyncio.Event] = None
        self.thread = None            # dedicated-thread mode
        self.thread_loop = None
        self.last_error: Optional[str] = None
        self.failure_class: Optional[FailureClass] = None
        self.stream = None  # live Stream while running (embedding/test access)

    @property
    def convergence(self) -> ConvergenceState:
        if self.desired_generation != self.observed_generation:
            return ConvergenceState.PENDING
        want_running = self.desired == DesiredState.RUNNING
        is_running = self.state == StreamState.RUNNING
        if want_running == is_running:
            return ConvergenceState.CONVERGED
        if self.state == StreamState.FAILED:
            return ConvergenceState.DIVERGED
        return ConvergenceState.PENDING

    def snapshot(self) -> dict:
        return {
            "id": self.stream_id,
            "state": self.state.value,
            "desired": self.desired.value,
            "desired_generation": self.desired_generation,
            "observed_generation": self.observed_generation,
            "convergence": self.convergence.value,
            "last_error": self.last_error,
            "failure_class": self.failure_class.value if self.failure_class else None,
            "metrics": self.metrics.snapshot(),
        }


class RuntimeManager:
    """Supervises N streams (reference runtime.rs:440-738)."""

    def __init__(self):
        self.entries: Dict[str, RuntimeEntry] = {}
        self.events = EventStore()
        self.operations = OperationStore()

    # -------------------------------------------------------------- registry
    def register(self, config: StreamConfig) -> RuntimeEntry:
        if config.id in self.entries:
            raise ConfigError(f"stream {config.id!r} already registered")
        entry = RuntimeEntry(config.id, config)
        self.entries[config.id] = entry
        self.events.push(config.id, "registered")
        return entry

    def get(self, stream_id: str) -> RuntimeEntry:
        e = self.entries.get(stream_id)
        if e is None:
            raise ArkError(f"unknown stream {stream_id!r}")
        return e

    def list_streams(self) -> List[dict]:
        return [e.snapshot() for e in self.entries.values()]

    # ------------------------------------------------------------- lifecycle
    async def start(self, stream_id: str) -> None:
        entry = self.get(stream_id)
        if entry.state in (StreamState.RUNNING, StreamState.STARTING):
            return
        entry.desired = DesiredState.RUNNING
        entry.desired_generation += 1
        entry.state = StreamState.STARTING
        self.events.push(stream_id, "starting")
        from .stream import build_stream
        try:
            stream = build_stream(entry.config)
        except Exception as e:  # noqa: BLE001
            entry.state = StreamState.FAILED
            entry.last_error = str(e)
            entry.failure_class = FailureClass.CONFIG
            self.events.push(stream_id, "failed", str(e))
            raise
        entry.metrics = stream.metrics
        entry.stream = stream
        if getattr(entry.config, "dedicated_thread", False):
            # own event loop in a thread: N streams scale past one loop's
            # Python ceiling (kernels release the GIL); cancel crosses via
            # call_soon_threadsafe, stop() awaits the thread join
            import threading
            loop = asyncio.new_event_loop()
            entry.thread_loop = loop
            entry.cancel = asyncio.Event()

            def runner():
                asyncio.set_event_loop(loop)
                try:
                    loop.run_until_complete(self._supervise(entry, stream))
                finally:
                    loop.close()

            entry.thread = threading.Thread(
                target=runner, name=f"stream-{stream_id}", daemon=True)
            entry.thread.start()
            entry.task = asyncio.ensure_future(
                asyncio.get_running_loop().run_in_executor(
                    None, entry.thread.join))
        else:
            entry.cancel = asyncio.Event()
            entry.task = asyncio.ensure_future(
                self._supervise(entry, stream))
        entry.state = StreamState.RUNNING
        entry.observed_generation = entry.desired_generation
        self.events.push(stream_id, "running")

    async def _supervise(self, entry: RuntimeEntry, stream) -> None:
        """Supervised spawn: on exit record Stopped/Failed + event
        (reference runtime.rs:489-542)."""
        try:
            await stream.run(entry.cancel)
            if entry.state != StreamState.STOPPING:
                entry.state = StreamState.STOPPED
            else:
                entry.state = StreamState.STOPPED
            self.events.push(entry.stream_id, "stopped")
        except asyncio.CancelledError:
            entry.state = StreamState.STOPPED
            self.events.push(entry.stream_id, "stopped", "cancelled")
        except Exception as e:  # noqa: BLE001
            entry.state = StreamState.FAILED
            entry.last_error = str(e)
            entry.failure_class = FailureClass.RUNTIME
            self.events.push(entry.stream_id, "failed", str(e))
            log.exception("stream %s failed", entry.stream_id)

    async def stop(self, stream_id: str,
                   timeout: float = SHUTDOWN_TIMEOUT_SECS) -> None:
        entry = self.get(stream_id)
        entry.desired = DesiredState.STOPPED
        entry.desired_generation += 1
        if entry.task is None or entry.task.done():
            entry.state = StreamState.STOPPED
            entry.observed_generation = entry.desired_generation
            return
        entry.state = StreamState.STOPPING
        self.events.push(stream_id, "stopping")
        if entry.thread_loop is not None:
            try:
                entry.thread_loop.call_soon_threadsafe(entry.cancel.set)
            except RuntimeError:
                pass  # loop already closed
        else:
            entry.cancel.set()
        try:
            await asyncio.wait_for(asyncio.shield(entry.task), timeout)
        except asyncio.TimeoutError:
            # surfaced as an event: a stream that needed the force-stop is a
            # shutdown-liveness bug worth noticing in production
            self.events.push(stream_id, "stop_timeout",
                             f"force-cancelled after {timeout}s")
            log.warning("stream %s did not stop within %ss; force-cancelling",
                        stream_id, timeout)
            entry.task.cancel()
            try:
                await entry.task
            except (asyncio.CancelledError, Exception):  # noqa: BLE001
                pass
        except Exception:  # noqa: BLE001
            pass
        entry.observed_generation = entry.desired_generation

    async def restart(self, stream_id: str) -> None:
        entry = self.get(stream_id)
        await self.stop(stream_id)
        entry.metrics.restarts += 1
        restarts = entry.metrics.restarts
        await self.start(stream_id)
        entry.metrics.restarts = restarts
        self.events.push(stream_id, "restarted")

    async def replace_config(self, stream_id: str, new_config: StreamConfig
                             ) -> None:
        """Reconcile + rollback-on-failure (reference runtime.rs:554-632)."""
        entry = self.get(stream_id)
        old = entry.config
        was_running = entry.state == StreamState.RUNNING
        if was_running:
            await self.stop(stream_id)
        entry.config = new_config
        if was_running:
            try:
                await self.start(stream_id)
            except Exception:  # noqa: BLE001
                entry.config = old  # rollback
                self.events.push(stream_id, "config_rollback")
                await self.start(stream_id)
                raise

    async def start_all(self) -> None:
        for sid in list(self.entries):
            await self.start(sid)

    async def stop_all(self) -> None:
        # signal all first, then await CONCURRENTLY — a stream that needs
        # the force-stop timeout must not serialize behind the others
        for e in self.entries.values():
            if e.cancel is not None and e.task is not None and not e.task.done():
                e.desired = DesiredState.STOPPED
                e.state = StreamState.STOPPING
                e.cancel.set()
        await asyncio.gather(*(self.stop(sid) for sid in list(self.entries)),
                             return_exceptions=True)

    async def wait_all(self) -> None:
        tasks = [e.task for e in self.entries.values()
                 if e.task is not None and not e.task.done()]
        if tasks:
            await asyncio.gather(*tasks, return_exceptions=True)

    def all_stopped(self) -> bool:
        return all(
            e.task is None or e.task.done() for e in self.entries.values()
        )

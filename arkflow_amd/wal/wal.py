"""Wal coordinator: seq assignment, sync policies, group-commit flusher,
cursor advance, replay.

Mirrors reference crates/arkflow-core/src/wal/mod.rs: `append` (:352) assigns
a monotonically increasing seq and either commits synchronously (per_entry)
or stages + notifies a background flusher (group_commit / periodic,
:312-338,393-402); `WalAck` (:432) advances the cursor THEN acks the inner
source ack; recovery replays everything after the cursor (§3.4).
"""
from __future__ import annotations

import asyncio
import heapq
from typing import AsyncIterator, List, Optional, Tuple

from ..config import DurabilityConfig
from ..registry import build_component
from ..spi import Ack
from .store import _nwal, batch_parts, deserialize_batch, serialize_batch


class WalAck(Ack):
    """Ack decorator: advance WAL cursor, then the source ack
    (reference wal/mod.rs:432-454)."""

    def __init__(self, wal: "Wal", seq: int, inner: Ack):
        self.wal = wal
        self.seq = seq
        self.inner = inner

    async def ack(self) -> None:
        await self.wal.advance(self.seq)
        await self.inner.ack()


MAX_PENDING_BATCHES = 256  # staged-append backpressure (reference bounds its
                           # PUT channel at 16 segments, s3.rs flume channel)


class Wal:
    def __init__(self, store, sync_policy: str = "group_commit",
                 group_window_ms: int = 5, periodic_interval_ms: int = 200,
                 cursor_flush_every: int = 64):
        self.store = store
        self.sync_policy = sync_policy
        self.group_window = group_window_ms / 1000.0
        self.periodic_interval = periodic_interval_ms / 1000.0
        self._seq = store.max_seq
        self._pending: List[Tuple[int, bytes]] = []
        self._pending_event: Optional[asyncio.Event] = None
        self._flusher_task: Optional[asyncio.Task] = None
        self._closed = False
        self._cursor_dirty = 0
        self.cursor_flush_every = cursor_flush_every
        # low-watermark ack tracking: the cursor only advances past a seq when
        # EVERY issued seq at or below it is acked — out-of-order acks (e.g. a
        # failed output write for batch k while k+1 succeeded) must not leap
        # the cursor over the un-acked entry or it would be lost on replay
        self._out_heap: List[int] = []  # issued (appended/replayed) seqs
        self._acked: set = set()
        self._max_issued = store.max_seq

    @staticmethod
    def open(config: DurabilityConfig, stream_id: str = "stream") -> "Wal":
        store = build_component("wal_store", {
            "type": config.backend,
            "path": config.path,
            "stream_id": stream_id,
            **config.extra,
        })
        return Wal(store, config.sync_policy, config.group_window_ms,
                   config.periodic_interval_ms)

    # ------------------------------------------------------------------ append
    async def append(self, batch) -> int:
        self._seq += 1
        seq = self._seq
        self._issue(seq)
        if self.sync_policy == "per_entry":
            loop = asyncio.get_running_loop()
            fast = getattr(self.store, "append_framed", None) \
                if _nwal is not None and not getattr(self.store, "compress",
                                                     False) else None
            if fast is not None:
                await loop.run_in_executor(
                    None, lambda: fast(
                        [_nwal.encode_frame_parts(seq, batch_parts(batch))],
                        True))
            else:
                payload = serialize_batch(batch)
                await loop.run_in_executor(
                    None, self.store.append_batch, [(seq, payload)], True)
            return seq
        # staged: serialization (incl. any D2H) happens in the flusher
        # thread, keeping the ingest loop free (reference stages + notifies,
        # wal/mod.rs:312-338 — its ~1-50 µs staged append defers the work)
        self._pending.append((seq, batch))
        self._ensure_flusher()
        self._pending_event.set()
        if len(self._pending) >= MAX_PENDING_BATCHES:
            # backpressure: the disk can't keep up — make the ingest path
            # pay for the flush instead of growing the staged queue
            await self.flush_pending()
        return seq

    def _ensure_flusher(self) -> None:
        if self._flusher_task is None or self._flusher_task.done():
            self._pending_event = asyncio.Event()
            self._flusher_task = asyncio.ensure_future(self._flusher())

    async def _flusher(self) -> None:
        """Background group-commit/periodic flusher
        (reference wal/mod.rs flush_pending)."""
        while not self._closed:
            if self.sync_policy == "group_commit":
                await self._pending_event.wait()
                await asyncio.sleep(self.group_window)
            else:  # periodic
                await asyncio.sleep(self.periodic_interval)
            await self.flush_pending()
            self._pending_event.clear()
            if self._closed:
                return

    async def flush_pending(self) -> None:
        if not self._pending:
            return
        staged, self._pending = self._pending, []
        loop = asyncio.get_running_loop()
        fast = getattr(self.store, "append_framed", None) \
            if _nwal is not None and not getattr(self.store, "compress",
                                                 False) else None

        def work():
            if fast is not None:
                # single-pass native framing straight from serialization
                # parts — no intermediate payload join
                fast([_nwal.encode_frame_parts(seq, batch_parts(b))
                      for seq, b in staged], True)
                return
            entries = [(seq, serialize_batch(b)) for seq, b in staged]
            self.store.append_batch(entries, True)

        await loop.run_in_executor(None, work)

    # ----------------------------------------------------------------- advance
    def _issue(self, seq: int) -> None:
        heapq.heappush(self._out_heap, seq)
        if seq > self._max_issued:
            self._max_issued = seq

    def _frontier(self) -> int:
        while self._out_heap and self._out_heap[0] in self._acked:
            self._acked.discard(heapq.heappop(self._out_heap))
        if self._out_heap:
            return self._out_heap[0] - 1  # smallest un-acked outstanding
        return self._max_issued

    async def advance(self, seq: int) -> None:
        """Low-watermark cursor advance (contiguous-ack frontier); flushed
        every N acks + on close."""
        self._acked.add(seq)
        frontier = self._frontier()
        if frontier <= self.store.cursor:
            return
        self._cursor_dirty += 1
        if self._cursor_dirty >= self.cursor_flush_every:
            self._cursor_dirty = 0
            loop = asyncio.get_running_loop()
            await loop.run_in_executor(None, self.store.write_cursor, frontier)
        else:
            # in-memory advance only; persisted on the next flush/close
            self.store._cursor = max(self.store._cursor, frontier)

    # ------------------------------------------------------------------ replay
    async def read_after_cursor(self) -> AsyncIterator:
        await self.flush_pending()
        loop = asyncio.get_running_loop()
        entries = await loop.run_in_executor(
            None, lambda: list(self.store.read_after(self.store.cursor)))
        for seq, payload in entries:
            self._issue(seq)
            yield seq, deserialize_batch(payload)

    # ------------------------------------------------------------------- close
    async def close(self) -> None:
        self._closed = True
        await self.flush_pending()
        if self._flusher_task is not None:
            if self._pending_event:
                self._pending_event.set()
            self._flusher_task.cancel()
            try:
                await self._flusher_task
            except (asyncio.CancelledError, Exception):  # noqa: BLE001
                pass
        loop = asyncio.get_running_loop()
        await loop.run_in_executor(
            None, self.store.write_cursor, self.store.cursor)
        await loop.run_in_executor(None, self.store.close)

"""BaseWindow: shared machinery for the windowing buffers.

Mirrors reference crates/arkflow-plugin/src/buffer/window.rs (per-input-name
queues, per-input concat on emit, optional SQL join across named inputs) and
buffer/join.rs (join emitted only when all expected inputs present).

Window contents are device-resident MessageBatches held by reference
(zero-copy); emit concatenation is a device kernel. Ack semantics follow the
reference: acks are released when their rows leave the window.
"""
from __future__ import annotations

import asyncio
import time
from collections import defaultdict, deque
from typing import Deque, Dict, List, Optional, Tuple

from ..batch import MessageBatch, concat_batches
from ..spi import Ack, Buffer, VecAck


class BaseWindowBuffer(Buffer):
    def __init__(self, config: dict, resource=None):
        # per-input-name FIFO of (item, ack) where item is a MessageBatch or
        # an absolute (start, end) range into that input's device ring
        # (reference window.rs DashMap of batches; SURVEY §2.9 ring mapping)
        self.queues: Dict[str, Deque[Tuple[object, Ack]]] = \
            defaultdict(deque)
        self.use_ring = bool(config.get("device_ring", True))
        self.rings: Dict[str, "DeviceRingBuffer"] = {}
        self._notify = asyncio.Event()
        self._draining = False
        self._closed = False
        self.resource = resource
        join_cfg = config.get("join")
        self.join_query: Optional[str] = None
        self.expected_inputs: List[str] = []
        if join_cfg:
            self.join_query = join_cfg.get("query")
            self.expected_inputs = list(
                join_cfg.get("inputs")
                or (resource.input_names if resource else [])
            )

    # ------------------------------------------------------------------ write
    async def write(self, batch: MessageBatch, ack: Ack) -> None:
        name = batch.input_name or "default"
        item: object = batch
        if self.use_ring:
            from .ring import DeviceRingBuffer
            ring = self.rings.get(name)
            if ring is None:
                ring = self.rings[name] = DeviceRingBuffer()
            rng = ring.append(batch)
            if rng is not None:
                item = rng  # the window holds a view range, not a copy
        self.queues[name].append((item, ack))
        self.on_write(batch)
        self._notify.set()

    def _materialize(self, name: str, items: list) -> MessageBatch:
        """items: ordered [(item, ack)] for one input → one batch.
        Consecutive ring ranges coalesce into zero-copy slices."""
        ring = self.rings.get(name)
        parts: List[MessageBatch] = []
        pending_ranges: List[Tuple[int, int]] = []

        def flush_ranges():
            if pending_ranges:
                parts.append(ring.slice_many(list(pending_ranges)))
                pending_ranges.clear()

        for it, _ in items:
            if isinstance(it, tuple):
                pending_ranges.append(it)
            else:
                flush_ranges()
                parts.append(it)
        flush_ranges()
        out = parts[0] if len(parts) == 1 else concat_batches(parts)
        return out

    def _release_ack(self, inner: Ack, items_by_name: dict) -> Ack:
        """Ring rows may be overwritten only after downstream ACKS the emitted
        window (the emitted batch is a VIEW over the ring). Wrap the combined
        ack so release happens post-ack; un-acked windows keep their rows
        (the ring grows instead of overwriting)."""
        rings = self.rings

        class _RingReleaseAck(Ack):
            async def ack(self_inner) -> None:
                await inner.ack()
                for name, lst in items_by_name.items():
                    ring = rings.get(name)
                    if ring is None:
                        continue
                    ends = [it[1] for it, _ in lst if isinstance(it, tuple)]
                    if ends:
                        ring.release_before(max(ends))

        return _RingReleaseAck()

    def on_write(self, batch: MessageBatch) -> None:
        """Subclass hook (e.g. session gap tracking)."""

    # ------------------------------------------------------------------- read
    async def read(self) -> Optional[Tuple[MessageBatch, Ack]]:
        while True:
            emitted = self.try_emit(draining=self._draining)
            if emitted is not None:
                return emitted
            if self._draining:
                final = self.drain_remaining()
                if final is not None:
                    return final
                return None
            self._notify.clear()
            timeout = self.next_deadline()
            if timeout is None:
                await self._notify.wait()
            else:
                # NOT wait_for: in py3.10, external cancellation landing on
                # the same tick as the timeout is converted to TimeoutError —
                # swallowing it here left a cancelled-but-running buffer task
                # waiting forever (intermittent shutdown stall, NOTES #12).
                # asyncio.wait's timeout never masks cancellation.
                waiter = asyncio.ensure_future(self._notify.wait())
                try:
                    await asyncio.wait({waiter}, timeout=max(timeout, 1e-4))
                finally:
                    waiter.cancel()

    async def flush(self) -> None:
        """End-of-input: emit everything left, then read() returns None."""
        self._draining = True
        self._notify.set()

    # --------------------------------------------------------------- triggers
    def try_emit(self, draining: bool = False
                 ) -> Optional[Tuple[MessageBatch, Ack]]:
        """Subclass: emit a window if its trigger fired, else None."""
        raise NotImplementedError

    def drain_remaining(self) -> Optional[Tuple[MessageBatch, Ack]]:
        """Default final drain: one combined window of whatever is left."""
        return self._emit_all()

    def next_deadline(self) -> Optional[float]:
        """Seconds until the next timer trigger (None = no timer)."""
        return None

    # ------------------------------------------------------------------- emit
    def _total_buffered(self) -> int:
        return sum(len(q) for q in self.queues.values())

    def _emit_all(self) -> Optional[Tuple[MessageBatch, Ack]]:
        """Pop everything; single input → concat; multi-input → join SQL
        (reference window.rs:100-178 process_window + join.rs)."""
        items: Dict[str, List[Tuple[MessageBatch, Ack]]] = {}
        for name, q in self.queues.items():
            if q:
                items[name] = list(q)
                q.clear()
        if not items:
            return None
        acks = [a for lst in items.values() for _, a in lst]
        if self.join_query and len(self.expected_inputs) > 1:
            # join across named inputs; skip (re-buffer) until all present
            missing = [n for n in self.expected_inputs if n not in items]
            if missing and not self._draining:
                for name, lst in items.items():
                    self.queues[name].extendleft(reversed(lst))
                return None
            tables = {
                name: self._materialize(name, lst)
                for name, lst in items.items()
            }
            from ..sql.engine import SqlExecutor
            first = self.expected_inputs[0]
            if first in tables:
                tables.setdefault("flow", tables[first])
            result = SqlExecutor(self.join_query).execute(tables)
            return result, self._release_ack(VecAck(acks), items)
        try:
            per_input = [self._materialize(name, lst)
                         for name, lst in items.items()]
            combined = per_input[0] if len(per_input) == 1 \
                else concat_batches(per_input)
        except ValueError:
            # heterogeneous schemas across inputs: emit the first input now,
            # re-buffer the rest (reference concats per input)
            name, lst = next(iter(items.items()))
            for other, olst in list(items.items())[1:]:
                self.queues[other].extendleft(reversed(olst))
            combined = self._materialize(name, lst)
            acks = [a for _, a in lst]
            return combined, self._release_ack(VecAck(acks), {name: lst})
        return combined, self._release_ack(VecAck(acks), items)

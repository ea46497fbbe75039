"""Windowing buffers: tumbling / sliding / session.

Mirrors reference crates/arkflow-plugin/src/buffer/{tumbling_window.rs,
sliding_window.rs,session_window.rs}:
  - tumbling: fixed non-overlapping interval; emits all buffered content each
    tick.
  - sliding: count-based window_size/slide_size (+ optional interval);
    overlapping — a batch is acked only when it leaves the window
    (sliding_window.rs:148-163).
  - session: gap-based close tracking last_message_time
    (session_window.rs:107-143).
All support the window-join config of BaseWindowBuffer.
"""
from __future__ import annotations

import time
from collections import deque
from typing import Deque, Optional, Tuple

from ..batch import MessageBatch
from ..inputs.generate import _parse_duration
from ..registry import register
from ..spi import Ack, NoopAck, VecAck
from .window import BaseWindowBuffer


class TumblingWindowBuffer(BaseWindowBuffer):
    def __init__(self, config: dict, resource=None):
        super().__init__(config, resource)
        self.interval = _parse_duration(config.get("interval", "1s"))
        self._deadline = None
        self._now = time.monotonic  # injectable clock (deterministic tests)

    def on_write(self, batch: MessageBatch) -> None:
        if self._deadline is None:
            self._deadline = self._now() + self.interval

    def try_emit(self, draining: bool = False):
        if self._deadline is None or self._now() < self._deadline:
            return None
        self._deadline = (self._now() + self.interval
                          if self._total_buffered() else None)
        out = self._emit_all()
        if out is None:
            self._deadline = None
        return out

    def next_deadline(self) -> Optional[float]:
        if self._deadline is None:
            return None
        return self._deadline - self._now()


class SlidingWindowBuffer(BaseWindowBuffer):
    """Count-based overlapping window over batches. On device, batches land
    in the shared ring once and every overlapping emit is slice views —
    the reference re-concats the whole window per slide
    (sliding_window.rs process_slide)."""

    def __init__(self, config: dict, resource=None):
        super().__init__(config, resource)
        self.window_size = int(config.get("window_size", 10))
        self.slide_size = int(config.get("slide_size", 5))
        self.interval = _parse_duration(config.get("interval", "0s"))
        self.window: Deque[Tuple[object, Ack]] = deque()  # batch | ring range
        self._new_since_emit = 0
        self._deadline = None

    async def write(self, batch: MessageBatch, ack: Ack) -> None:
        item: object = batch
        if self.use_ring:
            from .ring import DeviceRingBuffer
            ring = self.rings.get("window")
            if ring is None:
                ring = self.rings["window"] = DeviceRingBuffer()
            rng = ring.append(batch)
            if rng is not None:
                item = rng
        self.window.append((item, ack))
        self._new_since_emit += 1
        if self.interval > 0 and self._deadline is None:
            self._deadline = time.monotonic() + self.interval
        self._notify.set()

    def _window_batch(self) -> MessageBatch:
        return self._materialize("window", list(self.window))

    def try_emit(self, draining: bool = False):
        timer_fired = (self._deadline is not None
                       and time.monotonic() >= self._deadline)
        if self._new_since_emit < self.slide_size and not timer_fired:
            return None
        if not self.window:
            self._deadline = None
            return None
        self._new_since_emit = 0
        self._deadline = (time.monotonic() + self.interval
                          if self.interval > 0 else None)
        # batches leaving the window are the only ones acked now; their ring
        # rows free on ack (the emitted views no longer cover them)
        leaving = []
        left_items = []
        while len(self.window) > self.window_size:
            it, a = self.window.popleft()
            leaving.append(a)
            left_items.append((it, a))
        combined = self._window_batch()
        if leaving:
            return combined, self._release_ack(VecAck(leaving),
                                               {"window": left_items})
        return combined, NoopAck()

    def drain_remaining(self):
        if not self.window:
            return None
        items = list(self.window)
        acks = [a for _, a in items]
        combined = self._window_batch()
        self.window.clear()
        return combined, self._release_ack(VecAck(acks), {"window": items})

    def next_deadline(self) -> Optional[float]:
        if self._deadline is None:
            return None
        return self._deadline - time.monotonic()


class SessionWindowBuffer(BaseWindowBuffer):
    def __init__(self, config: dict, resource=None):
        super().__init__(config, resource)
        self.gap = _parse_duration(config.get("gap", "1s"))
        self._last_message: Optional[float] = None
        self._now = time.monotonic  # injectable clock (deterministic tests)

    def on_write(self, batch: MessageBatch) -> None:
        self._last_message = self._now()

    def try_emit(self, draining: bool = False):
        if self._last_message is None:
            return None
        if self._now() - self._last_message < self.gap:
            return None
        self._last_message = None
        return self._emit_all()

    def next_deadline(self) -> Optional[float]:
        if self._last_message is None:
            return None
        return self._last_message + self.gap - self._now()


@register("buffer", "tumbling_window",
          description="Fixed non-overlapping time window",
          example={"type": "tumbling_window", "interval": "10s"})
def _build_tumbling(config: dict, resource=None) -> TumblingWindowBuffer:
    return TumblingWindowBuffer(config, resource)


@register("buffer", "sliding_window",
          description="Count-based overlapping window (window_size/slide_size);"
                      " batches ack when they leave the window",
          example={"type": "sliding_window", "window_size": 10,
                   "slide_size": 5})
def _build_sliding(config: dict, resource=None) -> SlidingWindowBuffer:
    return SlidingWindowBuffer(config, resource)


@register("buffer", "session_window",
          description="Gap-based session window",
          example={"type": "session_window", "gap": "5s"})
def _build_session(config: dict, resource=None) -> SessionWindowBuffer:
    return SessionWindowBuffer(config, resource)

"""`memory` buffer: capacity-or-timeout accumulator.

Mirrors reference crates/arkflow-plugin/src/buffer/memory.rs: accumulate until
`capacity` messages or `timeout`, then pop ALL, concat, emit with a combined
ack (memory.rs:70-132).
"""
from __future__ import annotations

import time
from typing import Optional, Tuple

from ..batch import MessageBatch
from ..inputs.generate import _parse_duration
from ..registry import register
from ..spi import Ack
from .window import BaseWindowBuffer


class MemoryBuffer(BaseWindowBuffer):
    def __init__(self, config: dict, resource=None):
        super().__init__(config, resource)
        self.capacity = int(config.get("capacity", 0))  # 0 = unbounded
        self.timeout_secs = _parse_duration(config.get("timeout", "0s"))
        self._deadline: Optional[float] = None
        self._rows = 0

    def on_write(self, batch: MessageBatch) -> None:
        self._rows += batch.num_rows
        if self.timeout_secs > 0 and self._deadline is None:
            self._deadline = time.monotonic() + self.timeout_secs

    def try_emit(self, draining: bool = False
                 ) -> Optional[Tuple[MessageBatch, Ack]]:
        trigger = False
        if self.capacity and self._rows >= self.capacity:
            trigger = True
        if self._deadline is not None and time.monotonic() >= self._deadline:
            trigger = True
        if not trigger:
            return None
        out = self._emit_all()
        self._rows = 0
        self._deadline = None
        return out

    def drain_remaining(self):
        self._rows = 0
        self._deadline = None
        return self._emit_all()

    def next_deadline(self) -> Optional[float]:
        if self._deadline is None:
            return None
        return self._deadline - time.monotonic()


@register("buffer", "memory",
          description="Accumulate until capacity rows or timeout, emit one "
                      "concatenated batch with a combined ack",
          example={"type": "memory", "capacity": 8192, "timeout": "100ms"})
def _build_memory_buffer(config: dict, resource=None) -> MemoryBuffer:
    return MemoryBuffer(config, resource)

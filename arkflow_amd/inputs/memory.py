"""`memory` input: in-process queue for tests and embedding
(reference crates/arkflow-plugin/src/input/memory.rs:41-120)."""
from __future__ import annotations

import asyncio
from typing import Tuple

from ..batch import MessageBatch
from ..errors import EOFError_
from ..registry import register
from ..spi import Ack, Input, NoopAck


class MemoryInput(Input):
    def __init__(self, config: dict, resource=None):
        self._q: asyncio.Queue = asyncio.Queue()
        self._closed = False
        init = config.get("messages") or []
        for m in init:
            if isinstance(m, str):
                m = m.encode()
            self._q.put_nowait(MessageBatch.from_binary([m], input_name="memory"))
        if config.get("eof_after_init", bool(init)):
            self._q.put_nowait(None)

    # test/embedding API (reference memory.rs push/push_bytes)
    def push(self, batch: MessageBatch) -> None:
        self._q.put_nowait(batch)

    def push_bytes(self, payload: bytes) -> None:
        self._q.put_nowait(MessageBatch.from_binary([payload], input_name="memory"))

    def finish(self) -> None:
        """Mark EOF after the queued items."""
        self._q.put_nowait(None)

    async def read(self) -> Tuple[MessageBatch, Ack]:
        if self._closed:
            raise EOFError_("memory input closed")
        item = await self._q.get()
        if item is None:
            self._closed = True
            raise EOFError_("memory input drained")
        return item, NoopAck()


@register("input", "memory",
          description="In-memory queue input (testing/embedding)",
          example={"type": "memory", "messages": ['{"a": 1}']})
def _build_memory(config: dict, resource=None) -> MemoryInput:
    return MemoryInput(config, resource)

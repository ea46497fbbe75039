"""`batch` processor: accumulate N batches or timeout_ms, concat, emit
(reference crates/arkflow-plugin/src/processor/batch.rs:39-126)."""
from __future__ import annotations

import time
from typing import List

from ..batch import MessageBatch, concat_batches
from ..registry import register
from ..spi import Processor


class BatchProcessor(Processor):
    def __init__(self, config: dict, resource=None):
        self.count = int(config.get("count", 10))
        self.timeout_ms = float(config.get("timeout_ms", 1000))
        self._acc: List[MessageBatch] = []
        self._first_at = None

    async def process(self, batch: MessageBatch) -> List[MessageBatch]:
        if batch.num_rows:
            if not self._acc:
                self._first_at = time.monotonic()
            self._acc.append(batch)
        if not self._acc:
            return []
        expired = (time.monotonic() - self._first_at) * 1000 >= self.timeout_ms
        if len(self._acc) >= self.count or expired:
            out, self._acc = self._acc, []
            self._first_at = None
            return [concat_batches(out)]
        return []

    async def close(self) -> None:
        self._acc = []


@register("processor", "batch",
          description="Accumulate N batches or timeout_ms, concat, emit",
          example={"type": "batch", "count": 10, "timeout_ms": 1000})
def _build_batch(config: dict, resource=None) -> BatchProcessor:
    return BatchProcessor(config, resource)

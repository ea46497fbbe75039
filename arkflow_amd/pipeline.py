"""Pipeline: ordered processor chain with Single/Multiple/None fan-out
semantics (reference crates/arkflow-core/src/pipeline/mod.rs:24-94)."""
from __future__ import annotations

from typing import List, Sequence

from .batch import MessageBatch
from .spi import Processor


class Pipeline:
    def __init__(self, processors: Sequence[Processor]):
        self.processors = list(processors)

    async def process(self, batch: MessageBatch) -> List[MessageBatch]:
        """Fold the batch through processors; a processor returning multiple
        batches has each re-applied to the remaining chain
        (pipeline/mod.rs:57-85)."""
        batches = [batch]
        for proc in self.processors:
            nxt: List[MessageBatch] = []
            for b in batches:
                nxt.extend(await proc.process(b))
            batches = nxt
            if not batches:
                break
        return batches

    async def close(self) -> None:
        for proc in self.processors:
            await proc.close()

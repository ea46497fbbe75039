"""Basic outputs: stdout / drop / memory.

stdout mirrors reference output/stdout.rs (batch → JSON lines); drop mirrors
output/drop.rs; memory is the test-capture analog of the reference's
RecordingOutput stub (stream/mod.rs tests).
"""
from __future__ import annotations

import sys
from typing import List

from ..batch import DEFAULT_BINARY_VALUE_FIELD, MessageBatch
from ..registry import register
from ..spi import Output


class StdoutOutput(Output):
    def __init__(self, config: dict, resource=None):
        self.raw_value = bool(config.get("raw_value", False))

    async def write(self, batch: MessageBatch) -> None:
        out = sys.stdout
        if self.raw_value and DEFAULT_BINARY_VALUE_FIELD in batch.columns:
            for payload in batch.binary_values():
                out.write(payload.decode("utf-8", "replace") + "\n")
        else:
            for line in batch.to_json_lines():
                out.write(line.decode() + "\n")
        out.flush()


class DropOutput(Output):
    retains = False  # discards immediately: zero-copy fused batches are safe

    def __init__(self, config: dict, resource=None):
        pass

    async def write(self, batch: MessageBatch) -> None:
        return None


class MemoryOutput(Output):
    """Captures written batches for assertions / embedding."""

    def __init__(self, config: dict, resource=None):
        self.batches: List[MessageBatch] = []
        if resource is not None and isinstance(config.get("capture_key"), str):
            # allow tests to retrieve the instance through the resource
            setattr(resource, config["capture_key"], self)

    async def write(self, batch: MessageBatch) -> None:
        self.batches.append(batch)

    @property
    def total_rows(self) -> int:
        return sum(b.num_rows for b in self.batches)


@register("output", "stdout",
          description="Print batches as JSON lines to stdout",
          example={"type": "stdout"})
def _build_stdout(config: dict, resource=None) -> StdoutOutput:
    return StdoutOutput(config, resource)


@register("output", "drop",
          description="Discard all batches",
          example={"type": "drop"})
def _build_drop(config: dict, resource=None) -> DropOutput:
    return DropOutput(config, resource)


@register("output", "memory",
          description="Capture batches in memory (testing/embedding)",
          example={"type": "memory"})
def _build_memory_out(config: dict, resource=None) -> MemoryOutput:
    return MemoryOutput(config, resource)

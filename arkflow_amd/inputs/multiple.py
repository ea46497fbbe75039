"""`multiple_inputs`: fan-in of named child inputs.

Mirrors reference crates/arkflow-plugin/src/input/multiple_inputs.rs: child
inputs run concurrently, batches carry their child's name as input_name, and
the names are registered into Resource.input_names so window join buffers
know the expected inputs (:179-186).
"""
from __future__ import annotations

import asyncio
from typing import Dict, Tuple

from ..batch import MessageBatch
from ..errors import ConfigError, EOFError_
from ..registry import build_component, register
from ..spi import Ack, Input


class MultipleInputs(Input):
    def __init__(self, config: dict, resource=None):
        inputs_cfg = config.get("inputs")
        if not isinstance(inputs_cfg, dict) or not inputs_cfg:
            raise ConfigError(
                "multiple_inputs requires 'inputs': {name: {type: ...}}")
        self.children: Dict[str, Input] = {}
        for name, spec in inputs_cfg.items():
            self.children[name] = build_component("input", spec, resource)
        if resource is not None:
            resource.input_names = list(self.children)
        self._q: asyncio.Queue = asyncio.Queue(maxsize=64)
        self._tasks = []
        self._live = 0

    async def connect(self) -> None:
        for child in self.children.values():
            await child.connect()
        self._live = len(self.children)
        for name, child in self.children.items():
            self._tasks.append(
                asyncio.ensure_future(self._pump(name, child)))

    async def _pump(self, name: str, child: Input) -> None:
        try:
            while True:
                batch, ack = await child.read()
                renamed = MessageBatch(batch.columns, input_name=name)
                await self._q.put((renamed, ack))
        except EOFError_:
            pass
        except asyncio.CancelledError:
            return
        finally:
            self._live -= 1
            if self._live <= 0:
                try:
                    # never await here: on shutdown the reader may be gone
                    # and a full queue would block close() forever
                    self._q.put_nowait(None)
                except asyncio.QueueFull:
                    pass

    async def read(self) -> Tuple[MessageBatch, Ack]:
        item = await self._q.get()
        if item is None:
            raise EOFError_("all child inputs exhausted")
        return item

    async def close(self) -> None:
        for t in self._tasks:
            if not t.done():
                t.cancel()
        try:
            # bounded: a pump wedged in a queue handoff must not stall
            # engine shutdown (intermittent; see NOTES.md item 12)
            await asyncio.wait_for(
                asyncio.gather(*self._tasks, return_exceptions=True), 5)
        except asyncio.TimeoutError:
            pass
        for child in self.children.values():
            await child.close()


@register("input", "multiple_inputs",
          description="Fan-in of named child inputs (names drive window joins)",
          example={"type": "multiple_inputs",
                   "inputs": {"a": {"type": "generate"},
                              "b": {"type": "generate"}}})
def _build_multiple(config: dict, resource=None) -> MultipleInputs:
    return MultipleInputs(config, resource)

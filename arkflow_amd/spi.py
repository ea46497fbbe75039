"""Component SPI: the async traits every plugin implements.

Mirrors the reference's per-kind trait + builder + registry pattern
(crates/arkflow-core/src/{input,output,processor,buffer,codec,temporary}/mod.rs)
as Python ABCs. Builders are plain callables ``(config: dict, resource) -> obj``
registered in :mod:`arkflow_amd.registry`.
"""
from __future__ import annotations

import abc
from typing import Iterable, List, Optional, Sequence, Tuple

from .batch import MessageBatch


class Ack(abc.ABC):
    """Acknowledgement gate for at-least-once delivery
    (reference input/mod.rs:43-52)."""

    @abc.abstractmethod
    async def ack(self) -> None: ...


class NoopAck(Ack):
    async def ack(self) -> None:
        return None


class VecAck(Ack):
    """Combined ack over many children (reference input/mod.rs:66-105)."""

    def __init__(self, acks: Iterable[Ack]):
        self.acks = list(acks)

    async def ack(self) -> None:
        for a in self.acks:
            await a.ack()


class Input(abc.ABC):
    """reference input/mod.rs:55-64."""

    async def connect(self) -> None:
        return None

    @abc.abstractmethod
    async def read(self) -> Tuple[MessageBatch, Ack]:
        """Next batch + its ack. Raises EOFError_ when exhausted,
        DisconnectionError on transient failure."""

    async def close(self) -> None:
        return None


class Output(abc.ABC):
    """reference output/mod.rs:26-122. ``write_batch`` is one ack-range /
    transaction unit; the default loops ``write`` (output/mod.rs:49)."""

    async def connect(self) -> None:
        return None

    @abc.abstractmethod
    async def write(self, batch: MessageBatch) -> None: ...

    async def write_batch(self, batches: Sequence[MessageBatch]) -> None:
        for b in batches:
            await self.write(b)

    async def close(self) -> None:
        return None


class Processor(abc.ABC):
    """reference processor/mod.rs:32-79. Returns 0..N batches
    (``ProcessResult::{None,Single,Multiple}`` collapse to a list)."""

    @abc.abstractmethod
    async def process(self, batch: MessageBatch) -> List[MessageBatch]: ...

    async def close(self) -> None:
        return None


class Buffer(abc.ABC):
    """reference buffer/mod.rs:27-37."""

    @abc.abstractmethod
    async def write(self, batch: MessageBatch, ack: Ack) -> None: ...

    @abc.abstractmethod
    async def read(self) -> Optional[Tuple[MessageBatch, Ack]]:
        """Blocks until a window/capacity trigger emits; None = closed+drained."""

    async def flush(self) -> None:
        return None

    async def close(self) -> None:
        return None


class Encoder(abc.ABC):
    @abc.abstractmethod
    def encode(self, batch: MessageBatch) -> List[bytes]: ...


class Decoder(abc.ABC):
    @abc.abstractmethod
    def decode(self, payloads: Sequence[bytes]) -> MessageBatch: ...


class Codec(Encoder, Decoder):
    """reference codec/mod.rs:25-37."""


class Temporary(abc.ABC):
    """Keyed external lookup table joinable from SQL
    (reference temporary/mod.rs:40-44)."""

    async def connect(self) -> None:
        return None

    @abc.abstractmethod
    async def get(self, keys: list) -> Optional[MessageBatch]: ...

    async def close(self) -> None:
        return None


class Resource:
    """Build-time shared context (reference core Resource): carries the
    temporary tables by name and the input names registered by fan-in inputs
    (input/multiple_inputs.rs:179-186) for window joins."""

    def __init__(self):
        self.temporaries: dict = {}
        self.input_names: List[str] = []

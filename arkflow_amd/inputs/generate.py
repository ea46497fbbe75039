"""`generate` input: synthetic batches on an interval.

Mirrors reference crates/arkflow-plugin/src/input/generate.rs (318 LoC): emits
a payload every `interval`, `batch_size` rows per batch, optional `count` →
EOF. Drives all benchmarks/examples.

Extensions for the GPU engine:
  - ``fields``: generate typed numeric columns directly (bypasses JSON parse)
    so benches can exercise the device path without the ingest codec;
    ``context: "<json>"`` keeps the reference behavior (same JSON payload
    replicated per row into ``__value__``).
  - ``rate``: target rows/sec; interval derived if not given.
"""
from __future__ import annotations

import asyncio
import json
import time
from typing import Optional, Tuple

import torch

from ..batch import Column, MessageBatch
from ..errors import EOFError_
from ..registry import register
from ..spi import Ack, Input, NoopAck


class GenerateInput(Input):
    def __init__(self, config: dict, resource=None):
        self.context: Optional[str] = config.get("context")
        self.batch_size = int(config.get("batch_size", 1))
        self.count: Optional[int] = (
            int(config["count"]) if config.get("count") is not None else None
        )
        interval = config.get("interval", "0ms")
        self.interval_secs = _parse_duration(interval)
        # interval-0 generators never wait: the engine's direct loop may
        # skip its per-read cancellation race
        self.nonblocking = self.interval_secs <= 0
        self.fields = config.get("fields")  # {name: {dtype, low, high}}
        self.seed = int(config.get("seed", 0x5EED))
        self.device = torch.device(config.get("device")) if config.get("device") \
            else getattr(resource, "device", torch.device("cpu"))
        self._emitted = 0
        self._gen: Optional[torch.Generator] = None
        self._next_deadline: Optional[float] = None
        self._payload_batch: Optional[MessageBatch] = None

    async def read(self) -> Tuple[MessageBatch, Ack]:
        if self.count is not None and self._emitted >= self.count:
            raise EOFError_("generate count reached")
        if self.interval_secs > 0:
            now = time.monotonic()
            if self._next_deadline is None:
                self._next_deadline = now
            delay = self._next_deadline - now
            if delay > 0:
                await asyncio.sleep(delay)
            self._next_deadline += self.interval_secs
        n = self.batch_size
        if self.count is not None:
            n = min(n, self.count - self._emitted)
        self._emitted += n
        return self._make_batch(n), NoopAck()

    def _make_batch(self, n: int) -> MessageBatch:
        if self.fields:
            if self._gen is None:
                self._gen = torch.Generator(device=self.device)
                self._gen.manual_seed(self.seed)
                self._float_fields = [
                    (name, spec) for name, spec in self.fields.items()
                    if str(spec.get("dtype", "float32")) not in
                    ("int32", "int64")
                ]
                # one fused rand*scale+offset launch for ALL float fields:
                # rows of a [nf, n] tensor are contiguous zero-copy columns
                self._scale = torch.tensor(
                    [[float(s.get("high", 100.0)) - float(s.get("low", 0.0))]
                     for _, s in self._float_fields],
                    device=self.device, dtype=torch.float32)
                self._offset = torch.tensor(
                    [[float(s.get("low", 0.0))] for _, s in
                     self._float_fields],
                    device=self.device, dtype=torch.float32)
            cols = {}
            nf = len(self._float_fields)
            if nf:
                block = torch.rand((nf, n), generator=self._gen,
                                   device=self.device, dtype=torch.float32)
                block = torch.addcmul(self._offset, block, self._scale)
                for i, (name, spec) in enumerate(self._float_fields):
                    t = block[i]
                    dtype = str(spec.get("dtype", "float32"))
                    if dtype != "float32":
                        t = t.to(getattr(torch, dtype))
                    cols[name] = Column("numeric", t)
            for name, spec in self.fields.items():
                dtype = str(spec.get("dtype", "float32"))
                if dtype not in ("int32", "int64"):
                    continue
                low = float(spec.get("low", 0.0))
                high = float(spec.get("high", 100.0))
                t = torch.randint(
                    int(low), max(int(high), int(low) + 1), (n,),
                    generator=self._gen, device=self.device,
                    dtype=getattr(torch, dtype),
                )
                cols[name] = Column("numeric", t)
            cols = {name: cols[name] for name in self.fields}  # declared order
            return MessageBatch(cols, input_name="generate")
        payload = (self.context or '{"timestamp": 0, "value": 1}').encode()
        if self._payload_batch is None or self._payload_batch.num_rows != n:
            self._payload_batch = MessageBatch.from_binary(
                [payload] * n, input_name="generate")
        return self._payload_batch


def _parse_duration(v) -> float:
    """humantime-style '1s' / '10ms' / '500us' / numbers = seconds."""
    if isinstance(v, (int, float)):
        return float(v)
    s = str(v).strip().lower()
    for suffix, mult in (("ms", 1e-3), ("us", 1e-6), ("µs", 1e-6),
                         ("ns", 1e-9), ("m", 60.0), ("h", 3600.0), ("s", 1.0)):
        if s.endswith(suffix):
            return float(s[: -len(suffix)] or 0) * mult
    return float(s)


@register("input", "generate",
          description="Synthetic data generator (interval/batch_size/count; "
                      "'fields' for typed columns, 'context' for JSON payloads)",
          example={"type": "generate", "context": '{"value": 10}',
                   "interval": "10ms", "batch_size": 1000, "count": 10000})
def _build_generate(config: dict, resource=None) -> GenerateInput:
    return GenerateInput(config, resource)

"""`modbus` input (reference input/modbus.rs — tokio-modbus TCP register
reads). Offline env: a `driver: memory` register file supports tests; real
TCP activates when pymodbus is importable."""
from __future__ import annotations

import asyncio
import time
from typing import Dict, List, Tuple

from ..batch import Column, MessageBatch
from ..errors import ConnectionError_, EOFError_
from ..registry import register
from ..spi import Ack, Input, NoopAck

_MEMORY_REGISTERS: Dict[str, List[int]] = {}


def set_memory_registers(unit: str, values: List[int]) -> None:
    """Test hook: back the fake modbus device."""
    _MEMORY_REGISTERS[unit] = list(values)


class ModbusInput(Input):
    def __init__(self, config: dict, resource=None):
        self.address = config.get("address", "memory://dev0")
        self.start = int(config.get("start_register", 0))
        self.count_regs = int(config.get("register_count", 8))
        self.interval = float(config.get("interval_secs", 1.0))
        self.count = config.get("count")
        self.driver = config.get("driver") or (
            "memory" if str(self.address).startswith("memory://") else "real")
        self._reads = 0

    async def connect(self) -> None:
        if self.driver == "memory":
            return
        try:
            import pymodbus  # type: ignore  # noqa: F401
        except ImportError as e:
            raise ConnectionError_(
                "no modbus client library; use driver: memory") from e

    async def read(self) -> Tuple[MessageBatch, Ack]:
        if self.count is not None and self._reads >= int(self.count):
            raise EOFError_("modbus poll count reached")
        if self._reads > 0:
            await asyncio.sleep(self.interval)
        self._reads += 1
        unit = str(self.address).removeprefix("memory://")
        regs = _MEMORY_REGISTERS.get(unit, [0] * (self.start + self.count_regs))
        window = regs[self.start:self.start + self.count_regs]
        batch = MessageBatch({
            "register": Column.from_numeric(
                list(range(self.start, self.start + len(window)))),
            "value": Column.from_numeric(window),
            "__meta_timestamp": Column.from_numeric(
                [time.time()] * len(window)),
        }, input_name="modbus")
        return batch, NoopAck()


@register("input", "modbus",
          description="Modbus TCP register poller (driver: memory offline)",
          example={"type": "modbus", "address": "memory://dev0",
                   "start_register": 0, "register_count": 8})
def _build_modbus(config, resource=None):
    return ModbusInput(config, resource)

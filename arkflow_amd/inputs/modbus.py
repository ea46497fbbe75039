"""`modbus` input (reference input/modbus.rs — tokio-modbus TCP register
reads). Offline env: a `driver: memory` register file supports tests; the
real driver polls holding registers over TCP via pymodbus (3.x API with a
2.x fallback), env-gated like the other broker clients."""
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
        self.unit_id = int(config.get("unit_id", 1))  # modbus slave id
        self.driver = config.get("driver") or (
            "memory" if str(self.address).startswith("memory://") else "real")
        self._reads = 0
        self._client = None

    async def connect(self) -> None:
        if self.driver == "memory":
            return
        try:
            from pymodbus.client import ModbusTcpClient  # type: ignore
        except ImportError:
            try:  # pymodbus 2.x layout
                from pymodbus.client.sync import (  # type: ignore
                    ModbusTcpClient)
            except ImportError as e:
                raise ConnectionError_(
                    "modbus real driver requires pymodbus; use "
                    "driver: memory") from e
        host, _, port = str(self.address).partition(":")
        self._client = ModbusTcpClient(host, port=int(port or 502))
        ok = await asyncio.get_running_loop().run_in_executor(
            None, self._client.connect)
        if not ok:
            raise ConnectionError_(f"modbus connect failed: {self.address}")

    def _poll_real(self) -> List[int]:
        try:  # pymodbus 3.x keyword; 2.x uses `unit`
            rr = self._client.read_holding_registers(
                self.start, count=self.count_regs, slave=self.unit_id)
        except TypeError:
            rr = self._client.read_holding_registers(
                self.start, self.count_regs, unit=self.unit_id)
        if rr.isError():
            from ..errors import DisconnectionError
            raise DisconnectionError(f"modbus read error: {rr}")
        return list(rr.registers)

    async def read(self) -> Tuple[MessageBatch, Ack]:
        if self.count is not None and self._reads >= int(self.count):
            raise EOFError_("modbus poll count reached")
        if self._reads > 0:
            await asyncio.sleep(self.interval)
        self._reads += 1
        if self._client is not None:
            window = await asyncio.get_running_loop().run_in_executor(
                None, self._poll_real)
        else:
            unit = str(self.address).removeprefix("memory://")
            regs = _MEMORY_REGISTERS.get(
                unit, [0] * (self.start + self.count_regs))
            window = regs[self.start:self.start + self.count_regs]
        batch = MessageBatch({
            "register": Column.from_numeric(
                list(range(self.start, self.start + len(window)))),
            "value": Column.from_numeric(window),
            "__meta_timestamp": Column.from_numeric(
                [time.time()] * len(window)),
        }, input_name="modbus")
        return batch, NoopAck()

    async def close(self) -> None:
        if self._client is not None:
            try:
                self._client.close()
            except Exception:  # noqa: BLE001
                pass


@register("input", "modbus",
          description="Modbus TCP register poller (driver: memory offline)",
          example={"type": "modbus", "address": "memory://dev0",
                   "start_register": 0, "register_count": 8})
def _build_modbus(config, resource=None):
    return ModbusInput(config, resource)

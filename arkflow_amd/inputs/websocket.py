"""`websocket` input/output (reference input/websocket.rs — tungstenite
client). aiohttp client; testable against a local aiohttp ws server."""
from __future__ import annotations

from typing import Tuple

from ..batch import DEFAULT_BINARY_VALUE_FIELD, MessageBatch
from ..errors import ConfigError, DisconnectionError, EOFError_
from ..registry import register
from ..spi import Ack, Input, NoopAck, Output


class WebSocketInput(Input):
    def __init__(self, config: dict, resource=None):
        self.url = config.get("url")
        if not self.url:
            raise ConfigError("websocket input requires 'url'")
        self._session = None
        self._ws = None

    async def connect(self) -> None:
        import aiohttp
        self._session = aiohttp.ClientSession()
        try:
            self._ws = await self._session.ws_connect(self.url)
        except Exception as e:  # noqa: BLE001
            await self._session.close()
            self._session = None
            raise DisconnectionError(str(e)) from e

    async def read(self) -> Tuple[MessageBatch, Ack]:
        import aiohttp
        if self._ws is None:
            raise DisconnectionError("websocket not connected")
        msg = await self._ws.receive()
        if msg.type == aiohttp.WSMsgType.TEXT:
            payload = msg.data.encode()
        elif msg.type == aiohttp.WSMsgType.BINARY:
            payload = msg.data
        elif msg.type in (aiohttp.WSMsgType.CLOSE, aiohttp.WSMsgType.CLOSED,
                          aiohttp.WSMsgType.CLOSING):
            raise EOFError_("websocket closed")
        else:
            raise DisconnectionError(f"ws error {msg.type}")
        return (MessageBatch.from_binary([payload], input_name="websocket"),
                NoopAck())

    async def close(self) -> None:
        if self._ws is not None:
            await self._ws.close()
        if self._session is not None:
            await self._session.close()


class WebSocketOutput(Output):
    def __init__(self, config: dict, resource=None):
        self.url = config.get("url")
        if not self.url:
            raise ConfigError("websocket output requires 'url'")
        self.raw_value = bool(config.get("raw_value", True))
        self._session = None
        self._ws = None

    async def connect(self) -> None:
        import aiohttp
        self._session = aiohttp.ClientSession()
        self._ws = await self._session.ws_connect(self.url)

    async def write(self, batch: MessageBatch) -> None:
        if self.raw_value and DEFAULT_BINARY_VALUE_FIELD in batch.columns:
            for payload in batch.binary_values():
                await self._ws.send_bytes(payload)
        else:
            for line in batch.to_json_lines():
                await self._ws.send_bytes(line)

    async def close(self) -> None:
        if self._ws is not None:
            await self._ws.close()
        if self._session is not None:
            await self._session.close()


@register("input", "websocket",
          description="WebSocket client subscriber",
          example={"type": "websocket", "url": "ws://127.0.0.1:9001/feed"})
def _build_ws_in(config, resource=None):
    return WebSocketInput(config, resource)


@register("output", "websocket",
          description="WebSocket client publisher",
          example={"type": "websocket", "url": "ws://127.0.0.1:9001/sink"})
def _build_ws_out(config, resource=None):
    return WebSocketOutput(config, resource)

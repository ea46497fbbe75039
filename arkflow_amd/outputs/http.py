"""`http` output: POST batches to an endpoint (reference output/http.rs)."""
from __future__ import annotations

from ..batch import DEFAULT_BINARY_VALUE_FIELD, MessageBatch
from ..errors import ConfigError
from ..registry import register
from ..spi import Output


class HttpOutput(Output):
    def __init__(self, config: dict, resource=None):
        self.url = config.get("url")
        if not self.url:
            raise ConfigError("http output requires 'url'")
        self.token = config.get("token")
        self.content_type = config.get("content_type", "application/json")
        self.raw_value = bool(config.get("raw_value", False))
        self._session = None

    async def connect(self) -> None:
        import aiohttp
        headers = {}
        if self.token:
            headers["Authorization"] = f"Bearer {self.token}"
        self._session = aiohttp.ClientSession(headers=headers)

    async def write(self, batch: MessageBatch) -> None:
        if self.raw_value and DEFAULT_BINARY_VALUE_FIELD in batch.columns:
            payload = b"\n".join(batch.binary_values())
        else:
            payload = b"\n".join(batch.to_json_lines())
        async with self._session.post(
                self.url, data=payload,
                headers={"Content-Type": self.content_type}) as resp:
            if resp.status >= 400:
                raise RuntimeError(f"http output: {resp.status}")

    async def close(self) -> None:
        if self._session is not None:
            await self._session.close()


@register("output", "http",
          description="POST batches (JSON lines or raw __value__) to a URL",
          example={"type": "http", "url": "http://127.0.0.1:8086/sink"})
def _build_http_out(config: dict, resource=None) -> HttpOutput:
    return HttpOutput(config, resource)

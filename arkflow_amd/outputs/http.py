"""`http` output: POST batches to an endpoint (reference output/http.rs:
retry loop w/ exponential backoff :181-216, timeout + custom headers)."""
from __future__ import annotations

import asyncio

from ..batch import DEFAULT_BINARY_VALUE_FIELD, MessageBatch
from ..errors import ConfigError, ProcessError
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
        self.retry_count = int(config.get("retry_count", 0))
        self.timeout_ms = int(config.get("timeout_ms", 30_000))
        self.headers = dict(config.get("headers") or {})
        self._session = None

    async def connect(self) -> None:
        import aiohttp
        headers = dict(self.headers)
        if self.token:
            headers["Authorization"] = f"Bearer {self.token}"
        self._session = aiohttp.ClientSession(
            headers=headers,
            timeout=aiohttp.ClientTimeout(total=self.timeout_ms / 1000.0))

    async def write(self, batch: MessageBatch) -> None:
        if self.raw_value and DEFAULT_BINARY_VALUE_FIELD in batch.columns:
            payload = b"\n".join(batch.binary_values())
        else:
            payload = b"\n".join(batch.to_json_lines())
        last = None
        for attempt in range(self.retry_count + 1):
            try:
                async with self._session.post(
                        self.url, data=payload,
                        headers={"Content-Type": self.content_type}) as resp:
                    if resp.status < 400:
                        return
                    body = (await resp.text())[:200]
                    last = ProcessError(
                        f"http output: status {resp.status}, {body}")
            except ProcessError:
                raise
            except Exception as e:  # noqa: BLE001 - connect/timeout errors
                last = ProcessError(f"http output: {e}")
            if attempt < self.retry_count:
                # exponential backoff, 100 ms * 2^attempt (output/http.rs:208)
                await asyncio.sleep(0.1 * (2 ** attempt))
        raise last

    async def close(self) -> None:
        if self._session is not None:
            await self._session.close()


@register("output", "http",
          description="POST batches (JSON lines or raw __value__) to a URL",
          example={"type": "http", "url": "http://127.0.0.1:8086/sink"})
def _build_http_out(config: dict, resource=None) -> HttpOutput:
    return HttpOutput(config, resource)

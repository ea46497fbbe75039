"""`http` input: HTTP server endpoint ingesting request bodies.

Mirrors reference crates/arkflow-plugin/src/input/http.rs (:455): an embedded
server receives POST bodies into a bounded queue; optional bearer-token auth
(auth_middleware.rs) and a token-bucket rate limiter (rate_limiter.rs).
"""
from __future__ import annotations

import asyncio
import time
from typing import Optional, Tuple

from ..batch import MessageBatch
from ..errors import EOFError_
from ..registry import register
from ..spi import Ack, Input, NoopAck


class TokenBucket:
    """Simple token-bucket limiter (reference rate_limiter.rs, governor)."""

    def __init__(self, rate: float, burst: int):
        self.rate = rate
        self.burst = burst
        self.tokens = float(burst)
        self.last = time.monotonic()

    def allow(self) -> bool:
        now = time.monotonic()
        self.tokens = min(self.burst, self.tokens + (now - self.last) * self.rate)
        self.last = now
        if self.tokens >= 1.0:
            self.tokens -= 1.0
            return True
        return False


class Lockout:
    """Failed-auth lockout per client (reference auth_middleware.rs account
    lockout): after `max_failures` bad tokens, the client is rejected with
    429 for `lockout_secs`; a success clears the counter."""

    def __init__(self, max_failures: int = 5, lockout_secs: float = 60.0):
        self.max_failures = max_failures
        self.lockout_secs = lockout_secs
        self._failures = {}
        self._locked_until = {}

    def locked(self, client: str) -> bool:
        until = self._locked_until.get(client, 0.0)
        if until and time.monotonic() < until:
            return True
        if until:
            del self._locked_until[client]
            self._failures.pop(client, None)
        return False

    def failure(self, client: str) -> None:
        n = self._failures.get(client, 0) + 1
        self._failures[client] = n
        if n >= self.max_failures:
            self._locked_until[client] = time.monotonic() + self.lockout_secs

    def success(self, client: str) -> None:
        self._failures.pop(client, None)


class HttpInput(Input):
    def __init__(self, config: dict, resource=None):
        self.address = config.get("address", "127.0.0.1:0")
        self.path = config.get("path", "/ingest")
        self.token = config.get("token")
        self.queue_size = int(config.get("queue_size", 1024))
        rate = config.get("rate_limit")
        self.bucket = TokenBucket(float(rate), int(config.get("burst", rate)))\
            if rate else None
        self.lockout = Lockout(
            int(config.get("max_auth_failures", 5)),
            float(config.get("lockout_secs", 60.0))) if self.token else None
        from ..codecs.helper import build_codec
        self.codec = build_codec(config, resource)
        self._q: asyncio.Queue = asyncio.Queue(maxsize=self.queue_size)
        self._runner = None
        self._site = None
        self.port: Optional[int] = None
        self._closed = False

    async def connect(self) -> None:
        from aiohttp import web

        async def handler(request):
            if self.token:
                client = request.remote or "?"
                if self.lockout.locked(client):
                    return web.Response(status=429, text="locked out")
                auth = request.headers.get("Authorization", "")
                import hmac
                if not hmac.compare_digest(auth, f"Bearer {self.token}"):
                    self.lockout.failure(client)
                    return web.Response(status=401, text="unauthorized")
                self.lockout.success(client)
            if self.bucket and not self.bucket.allow():
                return web.Response(status=429, text="rate limited")
            body = await request.read()
            if not body:
                return web.Response(status=400, text="empty body")
            batch = MessageBatch.from_binary([body], input_name="http")
            try:
                self._q.put_nowait(batch)
            except asyncio.QueueFull:
                return web.Response(status=503, text="queue full")
            return web.Response(status=200, text="ok")

        app = web.Application()
        app.router.add_post(self.path, handler)
        self._runner = web.AppRunner(app)
        await self._runner.setup()
        host, _, port = self.address.partition(":")
        self._site = web.TCPSite(self._runner, host, int(port or 0))
        await self._site.start()
        self.port = self._runner.addresses[0][1] if self._runner.addresses \
            else None

    async def read(self) -> Tuple[MessageBatch, Ack]:
        if self._closed:
            raise EOFError_("http input closed")
        batch = await self._q.get()
        if batch is None:
            raise EOFError_("http input closed")
        if self.codec is not None:
            from ..codecs.helper import apply_codec
            batch = apply_codec(batch, self.codec)
        return batch, NoopAck()

    async def close(self) -> None:
        self._closed = True
        self._q.put_nowait(None)
        if self._runner is not None:
            await self._runner.cleanup()


@register("input", "http",
          description="HTTP server ingest endpoint (bearer auth + rate limit)",
          example={"type": "http", "address": "127.0.0.1:8085",
                   "path": "/ingest"})
def _build_http(config: dict, resource=None) -> HttpInput:
    return HttpInput(config, resource)

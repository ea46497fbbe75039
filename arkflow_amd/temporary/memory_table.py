"""Temporary lookup tables (SQL-joinable external tables).

`memory` is the native keyed table; `redis` mirrors reference
crates/arkflow-plugin/src/temporary/redis.rs and activates when a redis
client library is importable (offline env: connect() raises).
"""
from __future__ import annotations

from typing import Optional

from ..batch import MessageBatch
from ..errors import ConnectionError_
from ..registry import register
from ..spi import Temporary


class MemoryTemporary(Temporary):
    """Keyed in-memory table: config rows + programmatic put()."""

    def __init__(self, config: dict, resource=None):
        self.key_column = config.get("key_column", "key")
        self._rows = {}
        for row in config.get("rows") or []:
            self._rows[row[self.key_column]] = row

    def put(self, key, row: dict) -> None:
        self._rows[key] = row

    async def get(self, keys: Optional[list]) -> Optional[MessageBatch]:
        if keys is None:
            rows = list(self._rows.values())
        else:
            want = set()
            for k in keys:
                if isinstance(k, (bytes, bytearray)):
                    k = k.decode("utf-8", "replace")
                want.add(k)
            rows = [self._rows[k] for k in want if k in self._rows]
        if not rows:
            return None
        cols = {}
        names = list(rows[0].keys())
        return MessageBatch.from_dict(
            {n: [r.get(n) for r in rows] for n in names})


class RedisTemporary(Temporary):
    def __init__(self, config: dict, resource=None):
        self.url = config.get("url", "redis://127.0.0.1:6379")
        self.key_prefix = config.get("key_prefix", "")
        self._client = None

    async def connect(self) -> None:
        try:
            import redis.asyncio as redis  # type: ignore
        except ImportError as e:
            raise ConnectionError_(
                "redis client library not installed in this environment"
            ) from e
        self._client = redis.from_url(self.url)

    async def get(self, keys: Optional[list]) -> Optional[MessageBatch]:
        if self._client is None:
            raise ConnectionError_("redis temporary not connected")
        import json
        vals = await self._client.mget(
            [f"{self.key_prefix}{k}" for k in (keys or [])])
        rows = [json.loads(v) for v in vals if v]
        if not rows:
            return None
        names = list(rows[0].keys())
        return MessageBatch.from_dict(
            {n: [r.get(n) for r in rows] for n in names})


@register("temporary", "memory",
          description="In-memory keyed lookup table for SQL joins",
          example={"type": "memory", "key_column": "id",
                   "rows": [{"id": 1, "label": "x"}]})
def _build_memory_temp(config: dict, resource=None) -> MemoryTemporary:
    return MemoryTemporary(config, resource)


@register("temporary", "redis",
          description="Redis-backed lookup table (requires redis client)",
          example={"type": "redis", "url": "redis://127.0.0.1:6379"})
def _build_redis_temp(config: dict, resource=None) -> RedisTemporary:
    return RedisTemporary(config, resource)

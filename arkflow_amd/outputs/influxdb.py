"""`influxdb` output: InfluxDB 2.x line protocol over HTTP
(reference crates/arkflow-plugin/src/output/influxdb.rs, 818 LoC).

The line-protocol formatter is fully offline-testable; the HTTP write path
targets /api/v2/write with org/bucket/token query+header auth.
"""
from __future__ import annotations

import time
from typing import List, Optional

from ..batch import MessageBatch
from ..errors import ConfigError
from ..registry import register
from ..spi import Output


def _escape_tag(v: str) -> str:
    return v.replace("\\", "\\\\").replace(",", "\\,").replace(
        " ", "\\ ").replace("=", "\\=")


def _escape_field_str(v: str) -> str:
    return v.replace("\\", "\\\\").replace('"', '\\"')


def format_line_protocol(batch: MessageBatch, measurement: str,
                         tag_columns: List[str],
                         field_columns: Optional[List[str]] = None,
                         time_column: Optional[str] = None) -> List[bytes]:
    rows = batch.to_rows()
    lines = []
    for r in rows:
        tags = "".join(
            f",{_escape_tag(t)}={_escape_tag(str(_dec(r[t]))) }"
            for t in tag_columns if t in r
        )
        names = field_columns or [
            n for n in r
            if n not in tag_columns and n != time_column
            and not n.startswith("__meta_")
        ]
        fields = []
        for n in names:
            v = _dec(r.get(n))
            if isinstance(v, bool):
                fields.append(f"{_escape_tag(n)}={'t' if v else 'f'}")
            elif isinstance(v, int):
                fields.append(f"{_escape_tag(n)}={v}i")
            elif isinstance(v, float):
                fields.append(f"{_escape_tag(n)}={v}")
            elif v is not None:
                fields.append(f'{_escape_tag(n)}="{_escape_field_str(str(v))}"')
        if not fields:
            continue
        ts = r.get(time_column) if time_column else None
        ts_ns = int(float(ts) * 1e9) if ts is not None else time.time_ns()
        lines.append(
            f"{_escape_tag(measurement)}{tags} {','.join(fields)} {ts_ns}"
            .encode())
    return lines


def _dec(v):
    if isinstance(v, (bytes, bytearray)):
        return v.decode("utf-8", "replace")
    return v


class InfluxDbOutput(Output):
    def __init__(self, config: dict, resource=None):
        self.url = config.get("url")
        if not self.url:
            raise ConfigError("influxdb output requires 'url'")
        self.org = config.get("org", "")
        self.bucket = config.get("bucket", "")
        self.token = config.get("token")
        self.measurement = config.get("measurement", "arkflow")
        self.tag_columns = list(config.get("tags") or [])
        self.field_columns = config.get("fields")
        self.time_column = config.get("time_column")
        self._session = None

    async def connect(self) -> None:
        import aiohttp
        headers = {"Content-Type": "text/plain; charset=utf-8"}
        if self.token:
            headers["Authorization"] = f"Token {self.token}"
        self._session = aiohttp.ClientSession(headers=headers)

    async def write(self, batch: MessageBatch) -> None:
        lines = format_line_protocol(batch, self.measurement,
                                     self.tag_columns, self.field_columns,
                                     self.time_column)
        if not lines:
            return
        url = (f"{self.url.rstrip('/')}/api/v2/write"
               f"?org={self.org}&bucket={self.bucket}&precision=ns")
        async with self._session.post(url, data=b"\n".join(lines)) as resp:
            if resp.status >= 300:
                raise RuntimeError(f"influxdb write: {resp.status}")

    async def close(self) -> None:
        if self._session is not None:
            await self._session.close()


@register("output", "influxdb",
          description="InfluxDB 2.x line-protocol writer",
          example={"type": "influxdb", "url": "http://127.0.0.1:8086",
                   "org": "o", "bucket": "b", "measurement": "m",
                   "tags": ["sensor"]})
def _build_influx(config, resource=None):
    return InfluxDbOutput(config, resource)


class _FakeMongoStore:
    """Process-global fake document store (driver: memory) so the
    row→document path is testable offline, like the broker fake bus."""

    collections = {}  # (db, coll) → list of docs

    @classmethod
    def get(cls, db: str, coll: str):
        return cls.collections.setdefault((db, coll), [])

    @classmethod
    def reset(cls):
        cls.collections.clear()


class MongoDbOutput(Output):
    """`mongodb` output (reference output/mongodb.rs) — document inserts.

    drivers: ``memory`` (url memory://…, in-process fake store), real via
    motor (async) or pymongo (executor thread) when importable."""

    def __init__(self, config: dict, resource=None):
        self.url = config.get("url", "mongodb://127.0.0.1:27017")
        self.database = config.get("database", "arkflow")
        self.collection = config.get("collection", "events")
        self.driver = config.get("driver") or (
            "memory" if self.url.startswith("memory://") else "real")
        self._coll = None
        self._fake = None
        self._pym = None

    async def connect(self) -> None:
        if self.driver == "memory":
            self._fake = _FakeMongoStore.get(self.database, self.collection)
            return
        try:
            import motor.motor_asyncio as motor  # type: ignore
            client = motor.AsyncIOMotorClient(self.url)
            self._coll = client[self.database][self.collection]
            return
        except ImportError:
            pass
        try:
            import pymongo  # type: ignore
            client = pymongo.MongoClient(self.url)
            self._pym = client[self.database][self.collection]
        except ImportError:
            from ..errors import ConnectionError_
            raise ConnectionError_(
                "no mongodb client library (motor/pymongo); use "
                "driver: memory for the in-process store") from None

    async def write(self, batch: MessageBatch) -> None:
        docs = []
        for r in batch.to_rows():
            docs.append({k: _dec(v) for k, v in r.items()})
        if self._fake is not None:
            self._fake.extend(docs)
        elif self._pym is not None:
            import asyncio
            await asyncio.get_running_loop().run_in_executor(
                None, self._pym.insert_many, docs)
        else:
            await self._coll.insert_many(docs)


@register("output", "mongodb",
          description="MongoDB document inserts (requires client library)",
          example={"type": "mongodb", "url": "mongodb://127.0.0.1:27017",
                   "database": "d", "collection": "c"})
def _build_mongo(config, resource=None):
    return MongoDbOutput(config, resource)

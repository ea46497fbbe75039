"""`sql` input and output: relational DB source/sink.

Mirrors reference crates/arkflow-plugin/src/{input,output}/sql.rs: input runs
a query against MySQL/Postgres/SQLite/DuckDB and streams the result as
batches; output does batched INSERT with optional UPSERT. SQLite is fully
native here (stdlib); other engines activate when their client library is
importable.
"""
from __future__ import annotations

import sqlite3
from typing import List, Optional, Tuple

from ..batch import Column, DEFAULT_RECORD_BATCH, MessageBatch
from ..errors import ConfigError, ConnectionError_, EOFError_
from ..registry import register
from ..spi import Ack, Input, NoopAck, Output


def _connect(config: dict):
    engine = config.get("engine", "sqlite")
    if engine == "sqlite":
        path = config.get("path") or config.get("database")
        if not path:
            raise ConfigError("sqlite requires 'path'")
        return sqlite3.connect(path)
    if engine == "duckdb":
        try:
            import duckdb  # type: ignore
        except ImportError as e:
            raise ConnectionError_("duckdb not installed") from e
        return duckdb.connect(config.get("path", ":memory:"))
    raise ConnectionError_(
        f"no {engine} client library in this environment (sqlite/duckdb only)")


class SqlInput(Input):
    def __init__(self, config: dict, resource=None):
        self.config = config
        self.query = config.get("query")
        if not self.query:
            raise ConfigError("sql input requires 'query'")
        self.batch_size = int(config.get("batch_size", DEFAULT_RECORD_BATCH))
        self._rows: Optional[List[tuple]] = None
        self._names: List[str] = []
        self._pos = 0

    async def connect(self) -> None:
        conn = _connect(self.config)
        cur = conn.execute(self.query)
        self._names = [d[0] for d in cur.description]
        self._rows = cur.fetchall()
        conn.close()

    async def read(self) -> Tuple[MessageBatch, Ack]:
        if self._rows is None:
            await self.connect()
        if self._pos >= len(self._rows):
            raise EOFError_("sql input exhausted")
        chunk = self._rows[self._pos:self._pos + self.batch_size]
        self._pos += len(chunk)
        cols = {}
        for i, name in enumerate(self._names):
            vals = [r[i] for r in chunk]
            if vals and isinstance(vals[0], (bytes, bytearray)):
                cols[name] = Column.from_bytes(vals)
            elif vals and isinstance(vals[0], str):
                cols[name] = Column.from_strings(vals)
            else:
                cols[name] = Column.from_numeric(
                    [0 if v is None else v for v in vals])
        return MessageBatch(cols, input_name="sql"), NoopAck()


class SqlOutput(Output):
    def __init__(self, config: dict, resource=None):
        self.config = config
        self.table = config.get("table")
        if not self.table:
            raise ConfigError("sql output requires 'table'")
        self.upsert_keys = config.get("upsert_keys") or []
        self.create = bool(config.get("create_table", True))
        self._conn = None

    async def connect(self) -> None:
        self._conn = _connect(self.config)

    async def write(self, batch: MessageBatch) -> None:
        rows = batch.to_rows()
        if not rows:
            return
        names = batch.column_names
        if self.create:
            cols_sql = ", ".join(f'"{n}"' for n in names)
            if self.upsert_keys:
                keys = ", ".join(f'"{k}"' for k in self.upsert_keys)
                cols_sql += f", PRIMARY KEY ({keys})"
            self._conn.execute(
                f'CREATE TABLE IF NOT EXISTS "{self.table}" ({cols_sql})')
            self.create = False
        placeholders = ", ".join("?" for _ in names)
        cols_sql = ", ".join(f'"{n}"' for n in names)
        stmt = f'INSERT INTO "{self.table}" ({cols_sql}) VALUES ({placeholders})'
        if self.upsert_keys:
            keys = ", ".join(f'"{k}"' for k in self.upsert_keys)
            updates = ", ".join(
                f'"{n}"=excluded."{n}"' for n in names
                if n not in self.upsert_keys)
            stmt += f" ON CONFLICT({keys}) DO UPDATE SET {updates}"
        data = []
        for r in rows:
            data.append(tuple(
                v.decode("utf-8", "replace")
                if isinstance(v, (bytes, bytearray)) else v
                for v in (r[n] for n in names)))
        self._conn.executemany(stmt, data)
        self._conn.commit()

    async def close(self) -> None:
        if self._conn is not None:
            self._conn.close()


@register("input", "sql",
          description="Query a relational DB (sqlite/duckdb native) into "
                      "batches",
          example={"type": "sql", "engine": "sqlite", "path": "db.sqlite",
                   "query": "SELECT * FROM t"})
def _build_sql_in(config, resource=None):
    return SqlInput(config, resource)


@register("output", "sql",
          description="Batched INSERT (optional UPSERT) into a relational DB",
          example={"type": "sql", "engine": "sqlite", "path": "db.sqlite",
                   "table": "out", "upsert_keys": ["id"]})
def _build_sql_out(config, resource=None):
    return SqlOutput(config, resource)

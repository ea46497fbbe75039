"""`sql` input and output: relational DB source/sink.

Mirrors reference crates/arkflow-plugin/src/{input,output}/sql.rs: input runs
a query against MySQL/Postgres/SQLite/DuckDB and streams the result as
batches; output does batched INSERT with optional UPSERT. SQLite is fully
native (stdlib); duckdb, mysql (pymysql) and postgres (psycopg2) are real
DB-API drivers that activate when the client library is importable —
engine-correct placeholders (%s vs ?) and upsert dialects (ON CONFLICT vs
ON DUPLICATE KEY UPDATE) included.
"""
from __future__ import annotations

import sqlite3
from typing import List, Optional, Tuple

from ..batch import Column, DEFAULT_RECORD_BATCH, MessageBatch
from ..errors import ConfigError, ConnectionError_, EOFError_
from ..registry import register
from ..spi import Ack, Input, NoopAck, Output


class _DbApi:
    """Adapter giving pymysql/psycopg2 connections the conn.execute shape
    the sqlite path uses; %s paramstyle; explicit commit."""

    paramstyle = "%s"

    def __init__(self, conn, dialect: str):
        self._c = conn
        self.dialect = dialect

    def execute(self, sql, params=None):
        cur = self._c.cursor()
        cur.execute(sql, params or ())
        self._c.commit()
        return cur

    def executemany(self, sql, data):
        cur = self._c.cursor()
        cur.executemany(sql, data)
        return cur

    def commit(self):
        self._c.commit()

    def close(self):
        self._c.close()


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
    if engine == "mysql":
        try:
            import pymysql  # type: ignore
        except ImportError as e:
            raise ConnectionError_("mysql engine requires pymysql") from e
        return _DbApi(pymysql.connect(
            host=config.get("host", "127.0.0.1"),
            port=int(config.get("port", 3306)),
            user=config.get("user", "root"),
            password=config.get("password", ""),
            database=config.get("database")), "mysql")
    if engine in ("postgres", "postgresql"):
        try:
            import psycopg2  # type: ignore
        except ImportError as e:
            raise ConnectionError_(
                "postgres engine requires psycopg2") from e
        if config.get("dsn"):
            conn = psycopg2.connect(config["dsn"])
        else:
            conn = psycopg2.connect(
                host=config.get("host", "127.0.0.1"),
                port=int(config.get("port", 5432)),
                user=config.get("user", "postgres"),
                password=config.get("password", ""),
                dbname=config.get("database"))
        return _DbApi(conn, "postgres")
    raise ConnectionError_(f"unknown sql engine {engine!r} "
                           "(sqlite|duckdb|mysql|postgres)")


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
            qd = "`" if getattr(self._conn, "dialect", "") == "mysql" \
                else '"'
            if getattr(self._conn, "dialect", "") in ("mysql", "postgres"):
                cols_sql = ", ".join(f"{qd}{n}{qd} TEXT" for n in names)
            else:
                cols_sql = ", ".join(f"{qd}{n}{qd}" for n in names)
            if self.upsert_keys:
                keys = ", ".join(f"{qd}{k}{qd}" for k in self.upsert_keys)
                cols_sql += f", PRIMARY KEY ({keys})"
            self._conn.execute(
                f"CREATE TABLE IF NOT EXISTS {qd}{self.table}{qd} "
                f"({cols_sql})")
            self.create = False
        ph = getattr(self._conn, "paramstyle", "?")
        dialect = getattr(self._conn, "dialect", "sqlite")
        q = "`" if dialect == "mysql" else '"'
        placeholders = ", ".join(ph for _ in names)
        cols_sql = ", ".join(f"{q}{n}{q}" for n in names)
        stmt = (f"INSERT INTO {q}{self.table}{q} ({cols_sql}) "
                f"VALUES ({placeholders})")
        if self.upsert_keys:
            if dialect == "mysql":
                updates = ", ".join(
                    f"{q}{n}{q}=VALUES({q}{n}{q})" for n in names
                    if n not in self.upsert_keys)
                stmt += f" ON DUPLICATE KEY UPDATE {updates}"
            else:
                keys = ", ".join(f"{q}{k}{q}" for k in self.upsert_keys)
                updates = ", ".join(
                    f"{q}{n}{q}=excluded.{q}{n}{q}" for n in names
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
          description="Query a relational DB (sqlite/duckdb native; "
                      "mysql/postgres via pymysql/psycopg2) into batches",
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

"""`file` input: CSV/JSON/Parquet/Arrow files → batches.

Mirrors reference crates/arkflow-plugin/src/input/file.rs (:46-90): format
selection, streaming record batches (8192 rows), optional SQL query applied
to each batch, and remote sources: ``http(s)://`` URLs (requests) and
``s3://bucket/key`` objects (the SigV4 client from wal/object_store.py,
endpoint from config or MINIO_ENDPOINT/S3_ENDPOINT). Ballista offload is
deliberately descoped (RCCL sharding replaces remote offload);
local paths support globs.
"""
from __future__ import annotations

import glob as globmod
import os
from typing import List, Tuple

import numpy as np
import torch

from ..batch import Column, DEFAULT_RECORD_BATCH, MessageBatch
from ..errors import ConfigError, EOFError_
from ..registry import register
from ..spi import Ack, Input, NoopAck


def _table_to_batch(table, device=None) -> MessageBatch:
    import pyarrow as pa
    cols = {}
    for name in table.column_names:
        arr = table.column(name).combine_chunks()
        if pa.types.is_integer(arr.type):
            cols[name] = Column.from_numeric(torch.from_numpy(
                arr.cast(pa.int64()).to_numpy(zero_copy_only=False).copy()))
        elif pa.types.is_floating(arr.type):
            cols[name] = Column.from_numeric(torch.from_numpy(
                arr.cast(pa.float64()).to_numpy(zero_copy_only=False).copy()))
        elif pa.types.is_boolean(arr.type):
            cols[name] = Column.from_numeric(torch.from_numpy(
                arr.to_numpy(zero_copy_only=False).astype(np.bool_)))
        elif pa.types.is_binary(arr.type) or pa.types.is_large_binary(arr.type):
            cols[name] = Column.from_bytes(
                [v if v is not None else b"" for v in arr.to_pylist()])
        else:
            cols[name] = Column.from_strings(
                [str(v) if v is not None else "" for v in arr.to_pylist()])
    b = MessageBatch(cols, input_name="file")
    if device is not None:
        b = b.to(device)
    return b


class FileInput(Input):
    def __init__(self, config: dict, resource=None):
        path = config.get("path")
        if not path:
            raise ConfigError("file input requires 'path'")
        self.endpoint = config.get("endpoint")  # s3:// endpoint override
        remote = path.startswith(("http://", "https://", "s3://"))
        self.paths = sorted(globmod.glob(path)) if not remote and any(
            c in path for c in "*?[") else [path]
        self.format = config.get("format")  # csv|json|parquet|arrow (by ext)
        self.batch_size = int(config.get("batch_size", DEFAULT_RECORD_BATCH))
        self.query = config.get("query")
        self._executor = None
        if self.query:
            from ..sql.engine import SqlExecutor
            self._executor = SqlExecutor(self.query)
        self.device = getattr(resource, "device", None)
        self._chunks: List[MessageBatch] = []
        self._loaded = False

    def _fmt(self, path: str) -> str:
        if self.format:
            return self.format
        ext = os.path.splitext(path.split("?")[0])[1].lower().lstrip(".")
        return {"jsonl": "json", "ndjson": "json", "pq": "parquet",
                "feather": "arrow", "ipc": "arrow"}.get(ext, ext or "csv")

    def _fetch_remote(self, path: str):
        """http(s):// and s3://bucket/key sources → pyarrow buffer
        (reference input/file.rs:46-90 object-store URLs)."""
        import pyarrow as pa
        if path.startswith(("http://", "https://")):
            import requests
            r = requests.get(path, timeout=60)
            r.raise_for_status()
            return pa.BufferReader(r.content)
        import os as _os
        rest = path[len("s3://"):]
        bucket, _, key = rest.partition("/")
        if not bucket or not key:
            raise ConfigError(f"bad s3 url {path!r} (s3://bucket/key)")
        from ..wal.object_store import S3ObjectStore
        ep = self.endpoint or _os.environ.get("MINIO_ENDPOINT") \
            or _os.environ.get("S3_ENDPOINT")
        if not ep:
            raise ConfigError("s3:// paths need 'endpoint' or MINIO_ENDPOINT")
        got = S3ObjectStore(ep, bucket).get(key)
        if got is None:
            raise ConfigError(f"s3 object not found: {path}")
        return pa.BufferReader(got[0])

    def _load_all(self) -> None:
        import pyarrow as pa
        for path in self.paths:
            fmt = self._fmt(path)
            src_buf = None
            if path.startswith(("http://", "https://", "s3://")):
                src_buf = self._fetch_remote(path)
            if fmt == "csv":
                import pyarrow.csv as pacsv
                table = pacsv.read_csv(src_buf or path)
            elif fmt == "json":
                import pyarrow.json as pajson
                table = pajson.read_json(src_buf or path)
            elif fmt == "parquet":
                import pyarrow.parquet as pq
                table = pq.read_table(src_buf or path)
            elif fmt == "arrow":
                if src_buf is not None:
                    table = pa.ipc.open_file(src_buf).read_all()
                else:
                    with pa.memory_map(path) as src:
                        table = pa.ipc.open_file(src).read_all()
            else:
                raise ConfigError(f"unknown file format {fmt!r}")
            for start in range(0, table.num_rows, self.batch_size):
                chunk = table.slice(start, self.batch_size)
                batch = _table_to_batch(chunk, self.device)
                if self._executor is not None:
                    batch = self._executor.execute({"flow": batch})
                    batch.input_name = "file"
                if batch.num_rows:
                    self._chunks.append(batch)
        self._loaded = True

    async def read(self) -> Tuple[MessageBatch, Ack]:
        if not self._loaded:
            self._load_all()
        if not self._chunks:
            raise EOFError_("file input exhausted")
        return self._chunks.pop(0), NoopAck()


@register("input", "file",
          description="CSV/JSON/Parquet/Arrow file reader (glob paths, "
                      "optional SQL query per chunk)",
          example={"type": "file", "path": "data/*.parquet"})
def _build_file(config: dict, resource=None) -> FileInput:
    return FileInput(config, resource)

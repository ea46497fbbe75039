"""`file` output: append batches to jsonl/csv/parquet files."""
from __future__ import annotations

import os
from typing import List

from ..batch import MessageBatch
from ..errors import ConfigError
from ..registry import register
from ..spi import Output


class FileOutput(Output):
    def __init__(self, config: dict, resource=None):
        self.path = config.get("path")
        if not self.path:
            raise ConfigError("file output requires 'path'")
        self.format = config.get("format")
        self._f = None
        self._tables: List = []  # parquet: buffered until close

    def _fmt(self) -> str:
        if self.format:
            return self.format
        ext = os.path.splitext(self.path)[1].lower().lstrip(".")
        return {"jsonl": "json", "ndjson": "json", "pq": "parquet"}.get(
            ext, ext or "json")

    async def connect(self) -> None:
        os.makedirs(os.path.dirname(self.path) or ".", exist_ok=True)
        if self._fmt() in ("json", "csv"):
            self._f = open(self.path, "ab")

    async def write(self, batch: MessageBatch) -> None:
        fmt = self._fmt()
        if fmt == "json":
            for line in batch.to_json_lines():
                self._f.write(line + b"\n")
            self._f.flush()
        elif fmt == "csv":
            rows = batch.to_rows()
            names = batch.column_names
            if self._f.tell() == 0:
                self._f.write((",".join(names) + "\n").encode())
            for r in rows:
                vals = []
                for n in names:
                    v = r[n]
                    if isinstance(v, (bytes, bytearray)):
                        v = v.decode("utf-8", "replace")
                    vals.append(str(v))
                self._f.write((",".join(vals) + "\n").encode())
            self._f.flush()
        elif fmt == "parquet":
            import pyarrow as pa
            d = {}
            for name, col in batch.columns.items():
                vals = col.to_pylist()
                if col.kind == "binary":
                    vals = [v.decode("utf-8", "replace")
                            if v is not None else None for v in vals]
                d[name] = vals
            self._tables.append(pa.table(d))
        else:
            raise ConfigError(f"unknown file output format {fmt!r}")

    async def close(self) -> None:
        if self._f is not None:
            self._f.close()
        if self._tables:
            import pyarrow as pa
            import pyarrow.parquet as pq
            pq.write_table(pa.concat_tables(self._tables), self.path)
            self._tables = []


@register("output", "file",
          description="Append batches to a jsonl/csv/parquet file",
          example={"type": "file", "path": "out.jsonl"})
def _build_file_out(config: dict, resource=None) -> FileOutput:
    return FileOutput(config, resource)

"""`debezium_json` codec: Debezium CDC envelope → flattened columns.

Mirrors reference crates/arkflow-plugin/src/codec/debezium.rs: the `after`
(or `before` on deletes) document is flattened into columns, with __op,
__ts_ms and __source_* metadata columns appended.
"""
from __future__ import annotations

import json
from typing import List, Sequence

from ..batch import MessageBatch
from ..registry import register
from ..spi import Codec


class DebeziumJsonCodec(Codec):
    def __init__(self, config: dict, resource=None):
        self.source_fields = config.get("source_fields",
                                        ["db", "table", "lsn"])

    def decode(self, payloads: Sequence[bytes]) -> MessageBatch:
        rows = []
        for p in payloads:
            env = json.loads(p)
            payload = env.get("payload", env)
            op = payload.get("op", "c")
            doc = payload.get("after") or payload.get("before") or {}
            row = dict(doc)
            row["__op"] = op
            row["__ts_ms"] = payload.get("ts_ms", 0)
            src = payload.get("source") or {}
            for f in self.source_fields:
                if f in src:
                    row[f"__source_{f}"] = src[f]
            rows.append(row)
        if not rows:
            return MessageBatch({})
        names: List[str] = []
        for r in rows:
            for k in r:
                if k not in names:
                    names.append(k)
        return MessageBatch.from_dict(
            {n: [r.get(n) if r.get(n) is not None else
                 ("" if any(isinstance(x.get(n), str) for x in rows) else 0)
                 for r in rows] for n in names})

    def encode(self, batch: MessageBatch) -> List[bytes]:
        out = []
        for row in batch.to_rows():
            doc = {}
            meta = {}
            for k, v in row.items():
                if isinstance(v, (bytes, bytearray)):
                    v = v.decode("utf-8", "replace")
                if k.startswith("__"):
                    meta[k] = v
                else:
                    doc[k] = v
            env = {"payload": {
                "op": meta.get("__op", "c"),
                "ts_ms": meta.get("__ts_ms", 0),
                "after": doc,
            }}
            out.append(json.dumps(env, separators=(",", ":")).encode())
        return out


@register("codec", "debezium_json",
          description="Debezium CDC envelope → flattened columns "
                      "(+__op/__ts_ms/__source_*)",
          example={"type": "debezium_json"})
def _build_debezium(config, resource=None):
    return DebeziumJsonCodec(config, resource)

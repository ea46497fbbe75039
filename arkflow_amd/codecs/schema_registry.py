"""`schema_registry` codec: Confluent wire format
``[0x00][schema-id u32 BE][payload]``.

Mirrors reference crates/arkflow-plugin/src/codec/schema_registry.rs: a
per-schema-id descriptor cache with a pluggable resolver. The REST resolver
activates when a registry URL is reachable; offline, schemas come from the
``schemas: {id: proto_src}`` config map (the cache layer is identical).
"""
from __future__ import annotations

import struct
from typing import Dict, List, Optional, Sequence

from ..batch import MessageBatch
from ..errors import ConfigError, ProcessError
from ..processors.proto_wire import ProtoSchema, decode_message
from ..registry import register
from ..spi import Codec

MAGIC = 0x00


class SchemaRegistryCodec(Codec):
    def __init__(self, config: dict, resource=None):
        self.url: Optional[str] = config.get("url")
        self._cache: Dict[int, ProtoSchema] = {}
        self.default_id = int(config.get("default_schema_id", 1))
        for sid, src in (config.get("schemas") or {}).items():
            self._cache[int(sid)] = ProtoSchema.parse(src)
        if not self._cache and not self.url:
            raise ConfigError(
                "schema_registry codec requires 'schemas' (offline) or 'url'")

    def _resolve(self, schema_id: int) -> ProtoSchema:
        if schema_id in self._cache:
            return self._cache[schema_id]
        if self.url:
            import json
            import urllib.request
            with urllib.request.urlopen(
                    f"{self.url}/schemas/ids/{schema_id}", timeout=5) as r:
                body = json.load(r)
            schema = ProtoSchema.parse(body["schema"])
            self._cache[schema_id] = schema
            return schema
        raise ProcessError(f"unknown schema id {schema_id}")

    def decode(self, payloads: Sequence[bytes]) -> MessageBatch:
        rows = []
        schema = None
        for p in payloads:
            if len(p) < 5 or p[0] != MAGIC:
                raise ProcessError("bad confluent wire header")
            (schema_id,) = struct.unpack(">I", p[1:5])
            schema = self._resolve(schema_id)
            rows.append(decode_message(p[5:], schema))
        if not rows:
            return MessageBatch({})
        names = [schema.fields[no][0] for no in sorted(schema.fields)]
        return MessageBatch.from_dict(
            {n: [r.get(n) for r in rows] for n in names})

    def encode(self, batch: MessageBatch) -> List[bytes]:
        from ..processors.proto_wire import encode_message
        schema = self._resolve(self.default_id)
        header = bytes([MAGIC]) + struct.pack(">I", self.default_id)
        out = []
        for row in batch.to_rows():
            clean = {k: (v.decode("utf-8", "replace")
                         if isinstance(v, (bytes, bytearray)) else v)
                     for k, v in row.items()}
            out.append(header + encode_message(clean, schema))
        return out


@register("codec", "schema_registry",
          description="Confluent wire format [0x00][schema-id][payload] with "
                      "per-id schema cache",
          example={"type": "schema_registry",
                   "schemas": {"1": "message M { double v = 1; }"}})
def _build_schema_registry(config, resource=None):
    return SchemaRegistryCodec(config, resource)

"""`protobuf` codec: batch ↔ scalar proto3 wire format
(reference codec/protobuf.rs)."""
from __future__ import annotations

from typing import List, Sequence

import torch

from ..batch import Column, MessageBatch
from ..processors.proto_wire import decode_message, encode_message
from ..processors.protobuf_proc import _load_schema
from ..registry import register
from ..spi import Codec


class ProtobufCodec(Codec):
    def __init__(self, config: dict, resource=None):
        self.schema = _load_schema(config)
        self.device = getattr(resource, "device", None)

    def encode(self, batch: MessageBatch) -> List[bytes]:
        rows = batch.to_rows()
        out = []
        for r in rows:
            clean = {}
            for name, (no, t) in self.schema.by_name.items():
                v = r.get(name)
                if isinstance(v, (bytes, bytearray)) and t == "string":
                    v = v.decode("utf-8", "replace")
                clean[name] = v
            out.append(encode_message(clean, self.schema))
        return out

    def decode(self, payloads: Sequence[bytes]) -> MessageBatch:
        rows = [decode_message(p, self.schema) for p in payloads]
        cols = {}
        for no in sorted(self.schema.fields):
            name, t = self.schema.fields[no]
            vals = [r[name] for r in rows]
            if t in ("string", "bytes"):
                cols[name] = Column.from_bytes(
                    [v.encode() if isinstance(v, str) else v for v in vals])
            elif t in ("double", "float"):
                cols[name] = Column.from_numeric(
                    torch.tensor(vals, dtype=torch.float64))
            elif t == "bool":
                cols[name] = Column.from_numeric(
                    torch.tensor(vals, dtype=torch.bool))
            else:
                cols[name] = Column.from_numeric(
                    torch.tensor(vals, dtype=torch.int64))
        return MessageBatch(cols)


@register("codec", "protobuf",
          description="batch ↔ scalar proto3 wire format",
          example={"type": "protobuf",
                   "proto": "message M { double v = 1; }"})
def _build_protobuf_codec(config: dict, resource=None) -> ProtobufCodec:
    return ProtobufCodec(config, resource)

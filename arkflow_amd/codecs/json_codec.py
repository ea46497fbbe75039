"""`json` codec: batch ↔ line-JSON bytes (reference codec/json.rs)."""
from __future__ import annotations

from typing import List, Sequence

from ..batch import MessageBatch
from ..processors.json_proc import json_payloads_to_columns
from ..registry import register
from ..spi import Codec


class JsonCodec(Codec):
    def __init__(self, config: dict, resource=None):
        self.projection = config.get("columns")
        self.device = getattr(resource, "device", None)

    def encode(self, batch: MessageBatch) -> List[bytes]:
        return batch.to_json_lines()

    def decode(self, payloads: Sequence[bytes]) -> MessageBatch:
        return json_payloads_to_columns(list(payloads), self.projection,
                                        self.device)


@register("codec", "json",
          description="batch ↔ line-delimited JSON",
          example={"type": "json"})
def _build_json_codec(config: dict, resource=None) -> JsonCodec:
    return JsonCodec(config, resource)

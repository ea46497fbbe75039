"""Codec-on-input helper (reference input/codec_helper.rs): inputs with a
`codec:` config decode raw __value__ payloads into typed columns as the batch
enters the stream, preserving __meta_* columns when row counts align."""
from __future__ import annotations



from ..batch import DEFAULT_BINARY_VALUE_FIELD, MessageBatch
from ..registry import build_component


def build_codec(config: dict, resource=None):
    spec = config.get("codec")
    if not spec:
        return None
    return build_component("codec", spec, resource)


def apply_codec(batch: MessageBatch, codec) -> MessageBatch:
    if codec is None:
        return batch
    col = batch.columns.get(DEFAULT_BINARY_VALUE_FIELD)
    if col is None or col.kind != "binary":
        return batch
    decoded = codec.decode(col.to_pylist())
    if decoded.num_rows == batch.num_rows:
        meta = {k: v for k, v in batch.columns.items()
                if k.startswith("__meta_")}
        decoded = decoded.with_columns(meta)
    decoded.input_name = batch.input_name
    return decoded

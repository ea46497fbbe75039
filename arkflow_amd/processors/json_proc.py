"""JSON ↔ columnar processors.

Mirrors reference crates/arkflow-plugin/src/processor/json.rs +
component/json.rs: `json_to_arrow` parses the ``__value__`` JSON payloads into
typed columns with the schema inferred from the FIRST record
(infer_json_schema(..., Some(1)), component/json.rs:27) and optional column
projection; `arrow_to_json` renders rows as line-delimited JSON bytes in
``__value__``.

Parse path is the survey's "host-side simdjson-style parse → staging →
device" mapping: pyarrow's C++ line-JSON reader does the parse, columns are
then moved to the stream's device in one copy per column.
"""
from __future__ import annotations

import json
from typing import List, Optional

import numpy as np
import torch

from ..batch import Column, DEFAULT_BINARY_VALUE_FIELD, MessageBatch
from ..errors import ProcessError
from ..registry import register
from ..spi import Processor


def json_payloads_to_columns(payloads: List[bytes],
                             projection: Optional[List[str]] = None,
                             device=None) -> MessageBatch:
    if not payloads:
        return MessageBatch({})
    try:
        import pyarrow as pa
        import pyarrow.json as pajson
        blob = b"\n".join(p.strip() for p in payloads if p.strip())
        table = pajson.read_json(
            pa.BufferReader(blob),
            parse_options=pajson.ParseOptions(newlines_in_values=False),
        )
    except Exception as e:  # noqa: BLE001
        raise ProcessError(f"JSON parse failed: {e}") from e
    cols = {}
    names = projection or table.column_names
    for name in names:
        if name not in table.column_names:
            continue
        arr = table.column(name).combine_chunks()
        import pyarrow as pa
        if pa.types.is_integer(arr.type):
            t = torch.from_numpy(
                arr.cast(pa.int64()).to_numpy(zero_copy_only=False).copy())
            col = Column("numeric", t)
        elif pa.types.is_floating(arr.type):
            t = torch.from_numpy(
                arr.cast(pa.float64()).to_numpy(zero_copy_only=False).copy())
            col = Column("numeric", t)
        elif pa.types.is_boolean(arr.type):
            t = torch.from_numpy(
                arr.to_numpy(zero_copy_only=False).astype(np.bool_))
            col = Column("numeric", t)
        elif pa.types.is_string(arr.type) or pa.types.is_large_string(arr.type):
            col = Column.from_strings(arr.to_pylist())
        else:
            # nested / other types → JSON-encoded strings
            col = Column.from_strings(
                [json.dumps(v, separators=(",", ":")) if v is not None else ""
                 for v in arr.to_pylist()])
        if device is not None:
            col = col.to(device)
        cols[name] = col
    return MessageBatch(cols)


class JsonToArrowProcessor(Processor):
    def __init__(self, config: dict, resource=None):
        self.projection = config.get("columns") or config.get("projection")
        self.value_field = config.get("value_field",
                                      DEFAULT_BINARY_VALUE_FIELD)
        dev = config.get("device")
        self.device = torch.device(dev) if dev else getattr(
            resource, "device", None)
        self.keep_meta = bool(config.get("keep_meta", True))
        # fixed-schema GPU fast path: {name: float|int|bool|str} decodes;
        # dotted names ("user.id") extract one level of nesting on-device
        # on-device (csrc/json_decode.hip: scalar extraction + two-pass string
        # copy-out with escape/\uXXXX handling); no schema → host pyarrow
        # parse + inference
        self.schema = config.get("schema")
        if self.schema:
            for t in self.schema.values():
                if t not in ("float", "int", "bool", "str", "string"):
                    from ..errors import ConfigError
                    raise ConfigError(f"json schema type {t!r} not supported "
                                      "(float|int|bool|str)")
        # schemaless: infer from the FIRST record (reference
        # component/json.rs:27 infer_json_schema(..., Some(1))) and cache —
        # repeated batches then hit the same GPU decode kernel as the
        # fixed-schema path instead of the host pyarrow parser. False =
        # inference found GPU-unsupported shapes (arrays, >1 nesting level).
        self._inferred = None

    def _infer_schema(self, payload: bytes):
        """First-record schema: scalars + one dict level (dotted names)."""
        try:
            doc = json.loads(payload)
        except Exception:  # noqa: BLE001
            return False
        if not isinstance(doc, dict) or not doc:
            return False
        schema = {}

        def typ(v):
            if isinstance(v, bool):
                return "bool"
            if isinstance(v, int):
                return "int"
            if isinstance(v, float):
                return "float"
            if isinstance(v, str):
                return "str"
            return None

        for k, v in doc.items():
            if "." in k:
                return False  # literal dots collide with nested paths
            if isinstance(v, dict):
                for k2, v2 in v.items():
                    t = typ(v2)
                    if t is None or "." in k2:
                        return False
                    schema[f"{k}.{k2}"] = t
                continue
            t = typ(v)
            if t is None:
                return False
            schema[k] = t
        if self.projection:
            schema = {k: t for k, t in schema.items()
                      if k in self.projection}
        return schema or False

    def _decode_gpu(self, col, schema=None) -> MessageBatch:
        schema = schema or self.schema
        from .. import ops
        nat = ops.require_native()
        names, kinds, slot = [], [], []
        fcols, icols, scols = [], [], []
        for name, t in schema.items():
            names.append(name)
            if t == "float":
                kinds.append(1)
                slot.append(len(fcols))
                fcols.append(name)
            elif t in ("str", "string"):
                kinds.append(2)
                slot.append(len(scols))
                scols.append(name)
            else:
                kinds.append(0)
                slot.append(len(icols))
                icols.append(name)
        out_f, out_i, found, err, strings, summary = nat.json_decode(
            col.data, col.offsets, names, kinds, slot,
            len(icols), len(fcols), len(scols))
        # summary = [err, all_valid×nf, string totals×ns] in ONE readback
        summary = summary.tolist()
        if summary[0] != 0:
            raise ProcessError("json decode error (malformed document)")
        all_valid = summary[1:1 + max(len(names), 1)]
        cols = {}
        fbool = found.to(torch.bool)
        for f, name in enumerate(names):
            v = None if bool(all_valid[f]) else fbool[f]
            if schema[name] == "float":
                data = out_f[fcols.index(name)].contiguous()
            elif schema[name] == "bool":
                data = out_i[icols.index(name)].to(torch.bool)
            elif schema[name] in ("str", "string"):
                sdata, soffs = strings[scols.index(name)]
                cols[name] = Column("binary", sdata.contiguous(),
                                    soffs.contiguous(), validity=v)
                continue
            else:
                data = out_i[icols.index(name)].contiguous()
            cols[name] = Column("numeric", data, validity=v)
        return MessageBatch(cols)

    def _decode_host_schema(self, col, schema=None) -> MessageBatch:
        """CPU fallback for the fixed-schema path: same output columns
        (incl. dotted nested paths) as the GPU kernel."""
        schema = schema or self.schema
        import json as _json
        vals = {name: [] for name in schema}
        for payload in col.to_pylist():
            try:
                doc = _json.loads(payload)
            except Exception:  # noqa: BLE001
                raise ProcessError("json decode error (malformed document)")
            for name in schema:
                cur = doc
                for part in name.split("."):
                    cur = cur.get(part) if isinstance(cur, dict) else None
                    if cur is None:
                        break
                vals[name].append(cur)
        cols = {}
        for name, t in schema.items():
            v = vals[name]
            if t in ("str", "string"):
                miss = [not isinstance(x, str) for x in v]
            else:  # numeric schema: non-scalar values → absent (GPU parity)
                miss = [x is None or isinstance(x, (str, dict, list))
                        for x in v]
            if t in ("str", "string"):
                c = Column.from_strings(
                    ["" if m else x for m, x in zip(miss, v)])
            elif t == "float":
                c = Column.from_numeric(torch.tensor(
                    [0.0 if m else float(x) for m, x in zip(miss, v)],
                    dtype=torch.float64))
            elif t == "bool":
                c = Column.from_numeric(torch.tensor(
                    [False if m else bool(x) for m, x in zip(miss, v)]))
            else:
                c = Column.from_numeric(torch.tensor(
                    [0 if m else int(x) for m, x in zip(miss, v)],
                    dtype=torch.int64))
            if any(miss):
                c.validity = torch.tensor([not m for m in miss])
            cols[name] = c
        return MessageBatch(cols)

    async def process(self, batch: MessageBatch) -> List[MessageBatch]:
        if batch.num_rows == 0:
            return []
        col = batch.columns.get(self.value_field)
        if col is None or col.kind != "binary":
            raise ProcessError(
                f"json_to_arrow: no binary column {self.value_field!r}")
        schema = self.schema
        if schema is None:
            if self._inferred is None:
                lo = int(col.offsets[0].item())
                hi = int(col.offsets[1].item())
                first = bytes(col.data[lo:hi].cpu().numpy().tobytes())
                self._inferred = self._infer_schema(first)
            if self._inferred:
                schema = self._inferred
        if schema and col.data.is_cuda:
            out = self._decode_gpu(col, schema)
        elif schema:
            out = self._decode_host_schema(col, schema)
            if self.device is not None:
                out = out.to(self.device)
        else:
            out = json_payloads_to_columns(col.to_pylist(), self.projection,
                                           self.device)
        if self.keep_meta:
            meta = {k: (v.to(self.device) if self.device is not None else v)
                    for k, v in batch.columns.items()
                    if k.startswith("__meta_") and len(v) == out.num_rows}
            out = out.with_columns(meta)
        out.input_name = batch.input_name
        return [out]


class ArrowToJsonProcessor(Processor):
    def __init__(self, config: dict, resource=None):
        self.drop_meta = bool(config.get("drop_meta", False))

    async def process(self, batch: MessageBatch) -> List[MessageBatch]:
        if batch.num_rows == 0:
            return []
        b = batch.drop_meta() if self.drop_meta else batch
        lines = b.to_json_lines()
        out = MessageBatch.from_binary(lines, input_name=batch.input_name)
        return [out]


@register("processor", "json_to_arrow",
          description="Parse __value__ JSON payloads into typed device "
                      "columns (schema inferred from the first record)",
          example={"type": "json_to_arrow"})
def _build_j2a(config: dict, resource=None) -> JsonToArrowProcessor:
    return JsonToArrowProcessor(config, resource)


@register("processor", "arrow_to_json",
          description="Render rows as line-delimited JSON into __value__",
          example={"type": "arrow_to_json"})
def _build_a2j(config: dict, resource=None) -> ArrowToJsonProcessor:
    return ArrowToJsonProcessor(config, resource)

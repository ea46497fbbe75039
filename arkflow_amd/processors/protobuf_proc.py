"""`protobuf_to_arrow` / `arrow_to_protobuf` processors.

Mirrors reference crates/arkflow-plugin/src/processor/protobuf.rs +
component/protobuf.rs: dynamic decode/encode of scalar proto3 fields from a
.proto source (no nested/repeated/map/oneof — header :14-25). The decode hot
path is a GPU kernel parsing the device-resident binary column directly
(csrc/proto_decode.hip) — numeric fields into typed columns, string/bytes
fields via a two-pass span-record + copy-out into binary columns.
"""
from __future__ import annotations

from typing import List

import torch

from ..batch import Column, DEFAULT_BINARY_VALUE_FIELD, MessageBatch
from ..errors import ConfigError, ProcessError
from ..registry import register
from ..spi import Processor
from .proto_wire import ProtoSchema, decode_message, encode_message

_KIND_ENUM = {
    "varint": 0, "zigzag": 1, "f64": 2, "f32": 3,
    "u64": 4, "i64": 5, "u32": 6, "i32": 7,
}
_FLOAT_TYPES = {"double", "float"}
_INT_OUT = {"int32", "int64", "sint32", "sint64", "sfixed32", "sfixed64",
            "bool"}


def _load_schema(config: dict) -> ProtoSchema:
    src = config.get("proto")
    if not src and config.get("proto_path"):
        with open(config["proto_path"]) as f:
            src = f.read()
    if not src:
        raise ConfigError("protobuf processor requires 'proto' or 'proto_path'")
    return ProtoSchema.parse(src, config.get("message"))


def build_gpu_spec(schema):
    """Kernel field spec from a ProtoSchema: (fno, kind, isf, slot,
    int_fields, float_fields, str_fields) — shared by the processor and the
    fused bench graph (ops/stepgraph.FusedProtoMlp)."""
    fno, kind, isf, slot = [], [], [], []
    int_fields, float_fields, str_fields = [], [], []
    for no in sorted(schema.fields):
        name, t = schema.fields[no]
        fno.append(no)
        if t in ("string", "bytes"):
            kind.append(9)
            isf.append(0)
            slot.append(len(str_fields))
            str_fields.append(name)
        elif t in _FLOAT_TYPES:
            kind.append(_KIND_ENUM["f64" if t == "double" else "f32"])
            isf.append(1)
            slot.append(len(float_fields))
            float_fields.append(name)
        else:
            # uint32/uint64/fixed stay in int64 (values < 2^63 in
            # practice); only true floats go to the f64 output
            k = {"sint32": 1, "sint64": 1, "fixed64": 4, "sfixed64": 5,
                 "fixed32": 6, "sfixed32": 7}.get(t, 0)
            kind.append(k)
            isf.append(0)
            slot.append(len(int_fields))
            int_fields.append(name)
    return fno, kind, isf, slot, int_fields, float_fields, str_fields


class ProtobufToArrowProcessor(Processor):
    def __init__(self, config: dict, resource=None):
        self.schema = _load_schema(config)
        self.value_field = config.get("value_field",
                                      DEFAULT_BINARY_VALUE_FIELD)
        self.device = getattr(resource, "device", None)

    async def process(self, batch: MessageBatch) -> List[MessageBatch]:
        if batch.num_rows == 0:
            return []
        col = batch.columns.get(self.value_field)
        if col is None or col.kind != "binary":
            raise ProcessError(
                f"protobuf_to_arrow: no binary column {self.value_field!r}")
        if col.data.is_cuda:
            out = self._decode_gpu(col)
        else:
            out = self._decode_cpu(col)
        out.input_name = batch.input_name
        return [out]

    # ---------------------------------------------------------------- gpu path
    def _decode_gpu(self, col: Column) -> MessageBatch:
        from .. import ops
        nat = ops.require_native()
        (fno, kind, isf, slot,
         int_fields, float_fields, str_fields) = build_gpu_spec(self.schema)
        out_i, out_f, err, strings, summary = nat.proto_decode(
            col.data, col.offsets, fno, kind, isf, slot,
            len(int_fields), len(float_fields), len(str_fields))
        if int(summary[0]) != 0:  # err rode the one consolidated readback
            raise ProcessError("protobuf decode error (malformed message)")
        cols = {}
        for i, name in enumerate(int_fields):
            t = self.schema.by_name[name][1]
            data = out_i[i]
            if t == "bool":
                data = data.to(torch.bool)
            cols[name] = Column("numeric", data.contiguous())
        for i, name in enumerate(float_fields):
            cols[name] = Column("numeric", out_f[i].contiguous())
        for i, name in enumerate(str_fields):
            sdata, soffs = strings[i]
            cols[name] = Column("binary", sdata.contiguous(),
                                soffs.contiguous())
        # preserve declared field order
        ordered = {self.schema.fields[no][0]: cols[self.schema.fields[no][0]]
                   for no in sorted(self.schema.fields)
                   if self.schema.fields[no][0] in cols}
        return MessageBatch(ordered)

    # ---------------------------------------------------------------- cpu path
    def _decode_cpu(self, col: Column) -> MessageBatch:
        rows = [decode_message(p, self.schema) for p in col.to_pylist()]
        cols = {}
        for no in sorted(self.schema.fields):
            name, t = self.schema.fields[no]
            vals = [r[name] for r in rows]
            if t in ("string", "bytes"):
                cols[name] = Column.from_bytes(
                    [v.encode() if isinstance(v, str) else v for v in vals])
            elif t in _FLOAT_TYPES:
                cols[name] = Column.from_numeric(
                    torch.tensor(vals, dtype=torch.float64))
            elif t == "bool":
                cols[name] = Column.from_numeric(
                    torch.tensor(vals, dtype=torch.bool))
            else:
                cols[name] = Column.from_numeric(
                    torch.tensor(vals, dtype=torch.int64))
        out = MessageBatch(cols)
        if self.device is not None and self.device.type == "cuda":
            out = out.to(self.device)
        return out


class ArrowToProtobufProcessor(Processor):
    def __init__(self, config: dict, resource=None):
        self.schema = _load_schema(config)

    async def process(self, batch: MessageBatch) -> List[MessageBatch]:
        if batch.num_rows == 0:
            return []
        rows = batch.to_rows()
        payloads = []
        for r in rows:
            clean = {}
            for name, (no, t) in self.schema.by_name.items():
                v = r.get(name)
                if isinstance(v, (bytes, bytearray)) and t == "string":
                    v = v.decode("utf-8", "replace")
                clean[name] = v
            payloads.append(encode_message(clean, self.schema))
        return [MessageBatch.from_binary(payloads,
                                         input_name=batch.input_name)]


@register("processor", "protobuf_to_arrow",
          description="Decode scalar proto3 messages from __value__ into "
                      "typed columns (GPU varint kernel for numeric schemas)",
          example={"type": "protobuf_to_arrow",
                   "proto": "message M { double v = 1; }"})
def _build_p2a(config: dict, resource=None) -> ProtobufToArrowProcessor:
    return ProtobufToArrowProcessor(config, resource)


@register("processor", "arrow_to_protobuf",
          description="Encode rows as scalar proto3 messages into __value__",
          example={"type": "arrow_to_protobuf",
                   "proto": "message M { double v = 1; }"})
def _build_a2p(config: dict, resource=None) -> ArrowToProtobufProcessor:
    return ArrowToProtobufProcessor(config, resource)

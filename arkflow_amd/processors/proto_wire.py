"""Minimal proto3 schema parser + wire codec for SCALAR fields.

The reference's protobuf processors support scalar proto3 fields only — no
nested/repeated/map/oneof (crates/arkflow-plugin/src/processor/protobuf.rs
header :14-25). This module parses that subset out of a .proto source
directly (no protoc in this environment) and en/decodes the wire format.
The decode hot path for numeric-only schemas runs on GPU
(csrc/proto_decode.hip, one thread per message); this host codec is the
CPU path and the oracle.
"""
from __future__ import annotations

import re
import struct
from typing import Dict, Optional, Tuple

from ..errors import ConfigError

SCALARS = {
    "double": ("f64", 1), "float": ("f32", 5),
    "int32": ("varint", 0), "int64": ("varint", 0),
    "uint32": ("varint", 0), "uint64": ("varint", 0),
    "sint32": ("zigzag", 0), "sint64": ("zigzag", 0),
    "bool": ("varint", 0),
    "fixed64": ("u64", 1), "sfixed64": ("i64", 1),
    "fixed32": ("u32", 5), "sfixed32": ("i32", 5),
    "string": ("bytes", 2), "bytes": ("bytes", 2),
}

_FIELD_RE = re.compile(
    r"\b(double|float|u?int32|u?int64|sint32|sint64|bool|"
    r"s?fixed32|s?fixed64|string|bytes)\s+(\w+)\s*=\s*(\d+)\s*;")
_MSG_RE = re.compile(r"message\s+(\w+)\s*\{([^{}]*)\}", re.S)


class ProtoSchema:
    """fields: field_no → (name, proto_type)."""

    def __init__(self, fields: Dict[int, Tuple[str, str]], name: str = "Msg"):
        self.name = name
        self.fields = fields
        self.by_name = {n: (no, t) for no, (n, t) in fields.items()}

    @staticmethod
    def parse(proto_src: str, message: Optional[str] = None) -> "ProtoSchema":
        msgs = _MSG_RE.findall(proto_src)
        if not msgs:
            raise ConfigError("no message found in .proto source")
        if message:
            body = dict(msgs).get(message)
            if body is None:
                raise ConfigError(f"message {message!r} not in .proto")
            name = message
        else:
            name, body = msgs[0]
        fields = {}
        for m in _FIELD_RE.finditer(body):
            ptype, fname, fno = m.group(1), m.group(2), int(m.group(3))
            fields[fno] = (fname, ptype)
        if not fields:
            raise ConfigError(f"message {name!r} has no scalar fields")
        return ProtoSchema(fields, name)

    def numeric_only(self) -> bool:
        return all(t not in ("string", "bytes") for _, t in
                   self.fields.values())


# ------------------------------------------------------------------ wire codec
def _read_varint(buf: bytes, pos: int) -> Tuple[int, int]:
    result = 0
    shift = 0
    while True:
        b = buf[pos]
        pos += 1
        result |= (b & 0x7F) << shift
        if not (b & 0x80):
            return result, pos
        shift += 7
        if shift > 63:
            raise ValueError("varint too long")


def _write_varint(v: int) -> bytes:
    if v < 0:
        v &= (1 << 64) - 1
    out = bytearray()
    while True:
        b = v & 0x7F
        v >>= 7
        if v:
            out.append(b | 0x80)
        else:
            out.append(b)
            return bytes(out)


def _zigzag_enc(v: int) -> int:
    return (v << 1) ^ (v >> 63)


def _zigzag_dec(v: int) -> int:
    return (v >> 1) ^ -(v & 1)


def decode_message(buf: bytes, schema: ProtoSchema) -> Dict[str, object]:
    out: Dict[str, object] = {}
    pos = 0
    n = len(buf)
    while pos < n:
        tag, pos = _read_varint(buf, pos)
        fno, wt = tag >> 3, tag & 7
        spec = schema.fields.get(fno)
        if wt == 0:
            v, pos = _read_varint(buf, pos)
            if spec:
                name, t = spec
                if t in ("sint32", "sint64"):
                    out[name] = _zigzag_dec(v)
                elif t == "bool":
                    out[name] = bool(v)
                elif t in ("int32", "int64"):
                    out[name] = v - (1 << 64) if v >= (1 << 63) else v
                else:
                    out[name] = v
        elif wt == 1:
            raw = buf[pos:pos + 8]
            if len(raw) < 8:
                raise ValueError("truncated fixed64 field")
            pos += 8
            if spec:
                name, t = spec
                if t == "double":
                    out[name] = struct.unpack("<d", raw)[0]
                elif t == "sfixed64":
                    out[name] = struct.unpack("<q", raw)[0]
                else:
                    out[name] = struct.unpack("<Q", raw)[0]
        elif wt == 2:
            ln, pos = _read_varint(buf, pos)
            raw = buf[pos:pos + ln]
            if len(raw) < ln:
                raise ValueError("truncated length-delimited field")
            pos += ln
            if spec:
                name, t = spec
                out[name] = raw.decode("utf-8", "replace") \
                    if t == "string" else raw
        elif wt == 5:
            raw = buf[pos:pos + 4]
            if len(raw) < 4:
                raise ValueError("truncated fixed32 field")
            pos += 4
            if spec:
                name, t = spec
                if t == "float":
                    out[name] = struct.unpack("<f", raw)[0]
                elif t == "sfixed32":
                    out[name] = struct.unpack("<i", raw)[0]
                else:
                    out[name] = struct.unpack("<I", raw)[0]
        else:
            raise ValueError(f"unsupported wire type {wt}")
    # defaults for absent fields (proto3 semantics)
    for fno, (name, t) in schema.fields.items():
        if name not in out:
            if t == "string":
                out[name] = ""
            elif t == "bytes":
                out[name] = b""
            elif t == "bool":
                out[name] = False
            elif t in ("double", "float"):
                out[name] = 0.0
            else:
                out[name] = 0
    return out


def encode_message(row: Dict[str, object], schema: ProtoSchema) -> bytes:
    parts = []
    for fno in sorted(schema.fields):
        name, t = schema.fields[fno]
        v = row.get(name)
        if v is None:
            continue
        kind, wt = SCALARS[t]
        # proto3 default elision
        if (kind in ("varint", "zigzag", "u64", "i64", "u32", "i32")
                and int(v) == 0) or \
           (kind in ("f64", "f32") and float(v) == 0.0) or \
           (kind == "bytes" and not v):
            continue
        parts.append(_write_varint((fno << 3) | wt))
        if kind == "varint":
            parts.append(_write_varint(int(v)))
        elif kind == "zigzag":
            parts.append(_write_varint(_zigzag_enc(int(v))))
        elif kind == "f64":
            parts.append(struct.pack("<d", float(v)))
        elif kind == "f32":
            parts.append(struct.pack("<f", float(v)))
        elif kind == "u64":
            parts.append(struct.pack("<Q", int(v)))
        elif kind == "i64":
            parts.append(struct.pack("<q", int(v)))
        elif kind == "u32":
            parts.append(struct.pack("<I", int(v)))
        elif kind == "i32":
            parts.append(struct.pack("<i", int(v)))
        else:
            raw = v.encode() if isinstance(v, str) else bytes(v)
            parts.append(_write_varint(len(raw)))
            parts.append(raw)
    return b"".join(parts)

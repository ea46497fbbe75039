"""Columnar data model: GPU-resident message batches.

MI355X-native analog of the reference's ``MessageBatch(RecordBatch)`` wrapper
(see reference crates/arkflow-core/src/lib.rs:243-467). Columns are Arrow-style
(data / offsets / validity) but live as torch tensors so a batch is
device-resident in HBM3E on a GPU and plain host memory on CPU — one code path
for both. Binary payloads live in a column named ``__value__``
(lib.rs:52 DEFAULT_BINARY_VALUE_FIELD); source metadata is carried in
``__meta_*`` columns (lib.rs:56-69) so SQL can query provenance directly.
"""
from __future__ import annotations

import json
from typing import Dict, Iterable, List, Optional, Sequence, Union

import numpy as np
import torch

DEFAULT_BINARY_VALUE_FIELD = "__value__"
DEFAULT_RECORD_BATCH = 8192  # reference lib.rs:53

META_SOURCE = "__meta_source"
META_PARTITION = "__meta_partition"
META_OFFSET = "__meta_offset"
META_KEY = "__meta_key"
META_TIMESTAMP = "__meta_timestamp"
META_INGEST_TIME = "__meta_ingest_time"
META_EXT_PREFIX = "__meta_ext_"
META_COLUMNS = (
    META_SOURCE,
    META_PARTITION,
    META_OFFSET,
    META_KEY,
    META_TIMESTAMP,
    META_INGEST_TIME,
)

_TORCH_NUMERIC = {
    torch.float64,
    torch.float32,
    torch.float16,
    torch.bfloat16,
    torch.int64,
    torch.int32,
    torch.int16,
    torch.int8,
    torch.uint8,
    torch.bool,
}


class Column:
    """One Arrow-style column backed by torch tensors.

    kind:
      - "numeric": ``data`` is a 1-D torch tensor of any numeric/bool dtype.
      - "binary":  ``data`` is uint8 bytes, ``offsets`` int64 of length n+1
                   (also used for utf8 strings; semantics are caller-level).
    ``validity`` is an optional bool tensor (True = valid). None = all valid.
    """

    __slots__ = ("kind", "data", "offsets", "validity")

    def __init__(
        self,
        kind: str,
        data: torch.Tensor,
        offsets: Optional[torch.Tensor] = None,
        validity: Optional[torch.Tensor] = None,
    ):
        if kind not in ("numeric", "binary"):
            raise ValueError(f"unknown column kind {kind!r}")
        if kind == "binary":
            if offsets is None:
                raise ValueError("binary column requires offsets")
            if data.dtype != torch.uint8:
                raise ValueError("binary column data must be uint8")
        self.kind = kind
        self.data = data
        self.offsets = offsets
        self.validity = validity

    # ------------------------------------------------------------------ build
    @staticmethod
    def from_numeric(values: Union[torch.Tensor, np.ndarray, Sequence]) -> "Column":
        if isinstance(values, torch.Tensor):
            t = values
        elif isinstance(values, np.ndarray):
            arr = np.ascontiguousarray(values)
            if not arr.flags.writeable:  # e.g. zero-copy pyarrow buffers
                arr = arr.copy()
            t = torch.from_numpy(arr)
        else:
            vals = list(values)
            validity = None
            if any(v is None for v in vals):
                validity = torch.tensor([v is not None for v in vals])
                fill = next((v for v in vals if v is not None), 0)
                vals = [fill if v is None else v for v in vals]
            if vals and isinstance(vals[0], bool):
                t = torch.tensor(vals, dtype=torch.bool)
            elif all(isinstance(v, int) for v in vals):
                t = torch.tensor(vals, dtype=torch.int64)
            else:
                t = torch.tensor([float(v) for v in vals], dtype=torch.float64)
            if t.dim() != 1:
                t = t.reshape(-1)
            return Column("numeric", t, validity=validity)
        if t.dim() != 1:
            t = t.reshape(-1)
        return Column("numeric", t)

    @staticmethod
    def from_bytes(values: Iterable[bytes]) -> "Column":
        bufs: List[bytes] = []
        offsets = [0]
        total = 0
        for v in values:
            if isinstance(v, str):
                v = v.encode("utf-8")
            elif v is None:
                v = b""
            bufs.append(v)
            total += len(v)
            offsets.append(total)
        joined = b"".join(bufs)
        data = torch.from_numpy(np.frombuffer(joined, dtype=np.uint8).copy()) \
            if joined else torch.empty(0, dtype=torch.uint8)
        off = torch.tensor(offsets, dtype=torch.int64)
        return Column("binary", data, off)

    @staticmethod
    def from_strings(values: Iterable[Optional[str]]) -> "Column":
        return Column.from_bytes(
            (v.encode("utf-8") if isinstance(v, str) else (v or b"")) for v in values
        )

    # ------------------------------------------------------------------ props
    def __len__(self) -> int:
        if self.kind == "binary":
            return int(self.offsets.shape[0]) - 1
        return int(self.data.shape[0])

    @property
    def device(self) -> torch.device:
        return self.data.device

    @property
    def dtype(self):
        return self.data.dtype if self.kind == "numeric" else bytes

    def to(self, device, non_blocking: bool = False) -> "Column":
        return Column(
            self.kind,
            self.data.to(device, non_blocking=non_blocking),
            self.offsets.to(device, non_blocking=non_blocking)
            if self.offsets is not None else None,
            self.validity.to(device, non_blocking=non_blocking)
            if self.validity is not None else None,
        )

    # --------------------------------------------------------------- row ops
    def slice(self, start: int, length: int) -> "Column":
        if self.kind == "numeric":
            return Column(
                "numeric",
                self.data[start:start + length],
                validity=self.validity[start:start + length]
                if self.validity is not None else None,
            )
        off = self.offsets[start:start + length + 1]
        base = off[0].item()
        data = self.data[base:off[-1].item()]
        return Column(
            "binary",
            data,
            off - base,
            self.validity[start:start + length] if self.validity is not None else None,
        )

    def take(self, indices: torch.Tensor) -> "Column":
        """Gather rows by index tensor (device-side gather on GPU)."""
        if self.kind == "numeric":
            return Column(
                "numeric",
                self.data[indices],
                validity=self.validity[indices] if self.validity is not None else None,
            )
        idx = indices.to(torch.int64)
        if self.data.is_cuda:
            from . import ops
            out, new_off = ops.require_native().take_binary(
                self.data, self.offsets, idx.to(self.data.device))
            return Column("binary", out.contiguous(), new_off,
                          self.validity[idx]
                          if self.validity is not None else None)
        lengths_all = self.offsets[1:] - self.offsets[:-1]
        lengths = lengths_all[idx]
        new_off = torch.zeros(idx.shape[0] + 1, dtype=torch.int64, device=idx.device)
        torch.cumsum(lengths, 0, out=new_off[1:])
        total = int(new_off[-1].item())
        out = torch.empty(total, dtype=torch.uint8, device=self.data.device)
        # gather variable-length rows; vectorized via repeat_interleave of src starts
        if total:
            starts = self.offsets[:-1][idx]
            src_pos = (
                torch.repeat_interleave(starts - new_off[:-1], lengths)
                + torch.arange(total, dtype=torch.int64, device=idx.device)
            )
            out = self.data[src_pos]
        return Column(
            "binary",
            out,
            new_off,
            self.validity[idx] if self.validity is not None else None,
        )

    @staticmethod
    def concat(cols: Sequence["Column"]) -> "Column":
        first = cols[0]
        if first.kind == "numeric":
            data = torch.cat([c.data for c in cols])
            if any(c.validity is not None for c in cols):
                validity = torch.cat([
                    c.validity if c.validity is not None
                    else torch.ones(len(c), dtype=torch.bool, device=c.device)
                    for c in cols
                ])
            else:
                validity = None
            return Column("numeric", data, validity=validity)
        data = torch.cat([c.data for c in cols])
        sizes = [c.offsets[-1].item() for c in cols]
        parts = [cols[0].offsets]
        base = sizes[0]
        for c, s in zip(cols[1:], sizes[1:]):
            parts.append(c.offsets[1:] + base)
            base += s
        off = torch.cat(parts)
        if any(c.validity is not None for c in cols):
            validity = torch.cat([
                c.validity if c.validity is not None
                else torch.ones(len(c), dtype=torch.bool, device=c.device)
                for c in cols
            ])
        else:
            validity = None
        return Column("binary", data, off, validity)

    # ---------------------------------------------------------------- export
    def to_pylist(self) -> list:
        if self.kind == "numeric":
            vals = self.data.detach().to("cpu")
            if vals.dtype in (torch.bfloat16, torch.float16):
                vals = vals.to(torch.float32)
            out = vals.tolist()
        else:
            data = self.data.detach().to("cpu").numpy().tobytes()
            off = self.offsets.detach().to("cpu").tolist()
            out = [data[off[i]:off[i + 1]] for i in range(len(off) - 1)]
        if self.validity is not None:
            mask = self.validity.detach().to("cpu").tolist()
            out = [v if m else None for v, m in zip(out, mask)]
        return out

    def to_strlist(self) -> list:
        return [
            v.decode("utf-8", "replace") if isinstance(v, (bytes, bytearray)) else v
            for v in self.to_pylist()
        ]


class MessageBatch:
    """Named columns + optional input provenance — the unit of flow.

    Mirrors reference ``MessageBatch`` (lib.rs:243-246): zero-copy fan-out is
    achieved by sharing this object (columns are immutable by convention).
    """

    __slots__ = ("columns", "input_name")

    def __init__(self, columns: Dict[str, Column], input_name: Optional[str] = None):
        n = None
        for name, col in columns.items():
            ln = len(col)
            if n is None:
                n = ln
            elif ln != n:
                raise ValueError(
                    f"column {name!r} has {ln} rows, expected {n}"
                )
        self.columns = dict(columns)
        self.input_name = input_name

    # ------------------------------------------------------------------ build
    @staticmethod
    def from_binary(payloads: Sequence[bytes], input_name: Optional[str] = None
                    ) -> "MessageBatch":
        return MessageBatch(
            {DEFAULT_BINARY_VALUE_FIELD: Column.from_bytes(payloads)}, input_name
        )

    @staticmethod
    def from_dict(data: Dict[str, object], input_name: Optional[str] = None
                  ) -> "MessageBatch":
        cols: Dict[str, Column] = {}
        for name, values in data.items():
            if isinstance(values, Column):
                cols[name] = values
            elif isinstance(values, torch.Tensor):
                cols[name] = Column.from_numeric(values)
            elif isinstance(values, np.ndarray):
                cols[name] = Column.from_numeric(values)
            else:
                vals = list(values)
                if vals and isinstance(vals[0], (bytes, bytearray)):
                    cols[name] = Column.from_bytes(vals)
                elif vals and isinstance(vals[0], str):
                    cols[name] = Column.from_strings(vals)
                else:
                    cols[name] = Column.from_numeric(vals)
        return MessageBatch(cols, input_name)

    # ------------------------------------------------------------------ props
    @property
    def num_rows(self) -> int:
        for c in self.columns.values():
            return len(c)
        return 0

    def __len__(self) -> int:
        return self.num_rows

    @property
    def column_names(self) -> List[str]:
        return list(self.columns.keys())

    @property
    def device(self) -> torch.device:
        for c in self.columns.values():
            return c.device
        return torch.device("cpu")

    def column(self, name: str) -> Column:
        return self.columns[name]

    def binary_values(self) -> List[bytes]:
        """Payload bytes from the ``__value__`` column."""
        col = self.columns.get(DEFAULT_BINARY_VALUE_FIELD)
        if col is None:
            raise KeyError(DEFAULT_BINARY_VALUE_FIELD)
        return col.to_pylist()

    # ----------------------------------------------------------------- moves
    def to(self, device, non_blocking: bool = False) -> "MessageBatch":
        return MessageBatch(
            {k: c.to(device, non_blocking=non_blocking)
             for k, c in self.columns.items()},
            self.input_name,
        )

    # --------------------------------------------------------------- row ops
    def slice(self, start: int, length: int) -> "MessageBatch":
        return MessageBatch(
            {k: c.slice(start, length) for k, c in self.columns.items()},
            self.input_name,
        )

    def take(self, indices: torch.Tensor) -> "MessageBatch":
        # GPU fast path: all plain numeric columns gather in ONE kernel
        # launch (csrc gather_multi) instead of one launch per column.
        if indices.is_cuda:
            simple = [
                (k, c) for k, c in self.columns.items()
                if c.kind == "numeric" and c.validity is None
            ]
            if len(simple) > 1:
                from .ops import native_available, require_native
                if native_available():
                    nat = require_native()
                    outs = nat.gather_columns([c.data for _, c in simple],
                                              indices)
                    cols = {k: Column("numeric", t)
                            for (k, _), t in zip(simple, outs)}
                    for k, c in self.columns.items():
                        if k not in cols:
                            cols[k] = c.take(indices)
                    return MessageBatch(
                        {k: cols[k] for k in self.columns}, self.input_name)
        return MessageBatch(
            {k: c.take(indices) for k, c in self.columns.items()},
            self.input_name,
        )

    def with_columns(self, new_cols: Dict[str, Column]) -> "MessageBatch":
        cols = dict(self.columns)
        cols.update(new_cols)
        return MessageBatch(cols, self.input_name)

    def select(self, names: Sequence[str]) -> "MessageBatch":
        return MessageBatch({n: self.columns[n] for n in names}, self.input_name)

    def drop_meta(self) -> "MessageBatch":
        return MessageBatch(
            {k: v for k, v in self.columns.items()
             if not k.startswith("__meta_")},
            self.input_name,
        )

    # ---------------------------------------------------------------- export
    def to_pydict(self) -> Dict[str, list]:
        return {k: c.to_pylist() for k, c in self.columns.items()}

    def to_rows(self) -> List[dict]:
        d = self.to_pydict()
        names = list(d.keys())
        return [
            {n: d[n][i] for n in names} for i in range(self.num_rows)
        ]

    def to_json_lines(self) -> List[bytes]:
        rows = self.to_rows()
        out = []
        for r in rows:
            clean = {}
            for k, v in r.items():
                if isinstance(v, (bytes, bytearray)):
                    try:
                        clean[k] = v.decode("utf-8")
                    except UnicodeDecodeError:
                        clean[k] = v.hex()
                else:
                    clean[k] = v
            out.append(json.dumps(clean, separators=(",", ":")).encode())
        return out

    def __repr__(self) -> str:
        cols = ", ".join(
            f"{k}:{'bin' if c.kind == 'binary' else str(c.dtype).replace('torch.', '')}"
            for k, c in self.columns.items()
        )
        return f"MessageBatch[{self.num_rows} rows; {cols}]"


def split_batch(batch: MessageBatch, max_rows: int = DEFAULT_RECORD_BATCH
                ) -> List[MessageBatch]:
    """Re-chunk an oversized batch (reference lib.rs:441-467)."""
    n = batch.num_rows
    if n <= max_rows:
        return [batch]
    return [
        batch.slice(s, min(max_rows, n - s)) for s in range(0, n, max_rows)
    ]


def concat_batches(batches: Sequence[MessageBatch]) -> MessageBatch:
    if not batches:
        raise ValueError("concat of zero batches")
    if len(batches) == 1:
        return batches[0]
    names = batches[0].column_names
    for b in batches[1:]:
        if b.column_names != names:
            # union-by-name like the reference's schema coercion is out of
            # scope: require identical schemas, same as concat_batches in
            # arrow-rs (reference buffer/memory.rs:132).
            raise ValueError(
                f"schema mismatch in concat: {names} vs {b.column_names}"
            )
    return MessageBatch(
        {n: Column.concat([b.columns[n] for b in batches]) for n in names},
        batches[0].input_name,
    )

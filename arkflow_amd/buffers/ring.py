"""Device ring buffer: window state as views over preallocated HBM.

The reference concat-copies window contents on every emit
(`concat_batches` in buffer/memory.rs:132, buffer/window.rs:132). On GPU that
is a copy of the whole window per emit; the MI355X-native design (SURVEY §2.9
"device ring-buffer append — windows are views over device ring buffers")
appends each arriving batch ONCE into a preallocated per-column ring and
emits contiguous SLICES (zero-copy views; a wrapped window is at most two
slices concatenated).

Numeric columns ring; binary columns fall back to the batch-list path
(variable-length rows don't slice). Capacity doubles on overflow.
"""
from __future__ import annotations

from typing import Dict, List, Optional, Tuple

import torch

from ..batch import Column, MessageBatch


class DeviceRingBuffer:
    """Rows are addressed by ABSOLUTE index (monotonic); physical position =
    (abs - base) % capacity. [head, tail) is the live region."""

    def __init__(self, capacity: int = 1 << 16):
        self.capacity = capacity
        self.cols: Dict[str, torch.Tensor] = {}
        self.schema: Optional[List[Tuple[str, torch.dtype]]] = None
        self.base = 0
        self.head = 0
        self.tail = 0
        self.device = None

    @staticmethod
    def suitable(batch: MessageBatch) -> bool:
        return batch.num_rows > 0 and all(
            c.kind == "numeric" and c.validity is None
            for c in batch.columns.values())

    def _phys(self, abs_idx: int) -> int:
        return (abs_idx - self.base) % self.capacity

    def _ensure(self, batch: MessageBatch, extra: int) -> bool:
        schema = [(n, c.data.dtype) for n, c in batch.columns.items()]
        if self.schema is None:
            self.schema = schema
            self.device = batch.device
            while self.capacity < extra * 2:
                self.capacity *= 2
            for n, dt in schema:
                self.cols[n] = torch.empty(self.capacity, dtype=dt,
                                           device=self.device)
        elif schema != self.schema:
            return False
        live = self.tail - self.head
        if live + extra > self.capacity:
            new_cap = self.capacity
            while live + extra > new_cap:
                new_cap *= 2
            for n, dt in self.schema:
                nt = torch.empty(new_cap, dtype=dt, device=self.device)
                if live:
                    s = self._phys(self.head)
                    if s + live <= self.capacity:
                        nt[:live] = self.cols[n][s:s + live]
                    else:
                        first = self.capacity - s
                        nt[:first] = self.cols[n][s:]
                        nt[first:live] = self.cols[n][: live - first]
                self.cols[n] = nt
            self.base = self.head  # oldest live row now at physical 0
            self.capacity = new_cap
        return True

    def append(self, batch: MessageBatch) -> Optional[Tuple[int, int]]:
        """Append rows; returns the absolute [start, end) range, or None if
        the batch doesn't fit this ring (schema change / binary columns)."""
        if not self.suitable(batch):
            return None
        n = batch.num_rows
        if not self._ensure(batch, n):
            return None
        start = self.tail
        s = self._phys(start)
        for name, col in batch.columns.items():
            dst = self.cols[name]
            if s + n <= self.capacity:
                dst[s:s + n] = col.data
            else:
                first = self.capacity - s
                dst[s:s + first] = col.data[:first]
                dst[: n - first] = col.data[first:]
        self.tail = start + n
        return start, self.tail

    def release_before(self, abs_idx: int) -> None:
        """Rows < abs_idx become reusable."""
        self.head = max(self.head, min(abs_idx, self.tail))

    def slice(self, start: int, end: int) -> MessageBatch:
        """Batch over absolute [start, end): zero-copy view when physically
        contiguous, 2-piece concat when wrapped."""
        n = end - start
        cols = {}
        s = self._phys(start)
        for name, _ in self.schema:
            t = self.cols[name]
            if s + n <= self.capacity:
                cols[name] = Column("numeric", t[s:s + n])
            else:
                first = self.capacity - s
                cols[name] = Column("numeric",
                                    torch.cat([t[s:], t[: n - first]]))
        return MessageBatch(cols)

    def slice_many(self, ranges: List[Tuple[int, int]]) -> MessageBatch:
        """One batch over several absolute ranges (coalesces adjacency)."""
        if not ranges:
            raise ValueError("no ranges")
        merged: List[Tuple[int, int]] = []
        for s, e in sorted(ranges):
            if merged and s == merged[-1][1]:
                merged[-1] = (merged[-1][0], e)
            else:
                merged.append((s, e))
        if len(merged) == 1:
            return self.slice(*merged[0])
        from ..batch import concat_batches
        return concat_batches([self.slice(s, e) for s, e in merged])

"""Per-stream runtime metrics — the reference's 9 lock-free counters
(crates/arkflow-core/src/runtime.rs:204-231) plus GPU-side timings."""
from __future__ import annotations

import time
from dataclasses import dataclass, field
from typing import Dict


class RuntimeMetrics:
    __slots__ = (
        "input_batches", "input_messages", "processing_errors",
        "output_batches", "output_messages", "input_errors",
        "input_reconnects", "output_errors", "restarts",
        "stage_ns", "started_at", "wal_lag",
    )

    def __init__(self):
        self.input_batches = 0
        self.input_messages = 0
        self.processing_errors = 0
        self.output_batches = 0
        self.output_messages = 0
        self.input_errors = 0
        self.input_reconnects = 0
        self.output_errors = 0
        self.restarts = 0
        # per-stage wall-time accumulators (ns): input/process/output
        self.stage_ns: Dict[str, int] = {"input": 0, "process": 0, "output": 0}
        self.wal_lag = 0  # entries appended but not yet cursor-acked
        self.started_at = time.time()

    def snapshot(self) -> dict:
        """StreamMetricsSnapshot analog (reference control.rs:289)."""
        return {
            "input_batches": self.input_batches,
            "input_messages": self.input_messages,
            "processing_errors": self.processing_errors,
            "output_batches": self.output_batches,
            "output_messages": self.output_messages,
            "input_errors": self.input_errors,
            "input_reconnects": self.input_reconnects,
            "output_errors": self.output_errors,
            "restarts": self.restarts,
            "uptime_secs": time.time() - self.started_at,
            "stage_ms": {k: v / 1e6 for k, v in self.stage_ns.items()},
            "wal_lag": self.wal_lag,
        }


@dataclass
class ControlEvent:
    """reference control.rs:192 ControlEvent."""
    seq: int
    stream_id: str
    kind: str
    message: str = ""
    ts: float = field(default_factory=time.time)

    def to_dict(self) -> dict:
        return {"seq": self.seq, "stream_id": self.stream_id,
                "kind": self.kind, "message": self.message, "ts": self.ts}

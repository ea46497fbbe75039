"""Durable-pipeline soak: generate → sql(filter) → sliding window (device
ring) → sql(agg) with a segmented WAL, on GPU. Reports sustained rows/s and
error counters. Usage: python tools/soak_durable.py [seconds]
"""
import asyncio
import os
import sys
import tempfile

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import arkflow_amd as af
from arkflow_amd.config import EngineConfig

SECONDS = float(sys.argv[1]) if len(sys.argv) > 1 else 30.0


async def main():
    wal_dir = tempfile.mkdtemp(prefix="soak_wal_")
    cfg = EngineConfig.from_dict({
        "streams": [{
            "id": "soak",
            "input": {
                "type": "generate", "batch_size": 8192, "interval": "0ms",
                "fields": {
                    "k": {"dtype": "int64", "low": 0, "high": 4096},
                    "v": {"dtype": "float32", "low": 0.0, "high": 1.0},
                },
            },
            "durability": {
                "enabled": True, "path": wal_dir, "backend": "segment",
                "sync_policy": "group_commit",
                "extra": {"segment_strategy": "balanced"},
            },
            "buffer": {"type": "sliding_window", "window_size": 8,
                       "slide_size": 4},
            "pipeline": {"processors": [
                {"type": "sql",
                 "query": "SELECT k, count(*) AS n, sum(v) AS s FROM flow "
                          "WHERE v >= 0.1 GROUP BY k"},
            ]},
            "output": {"type": "drop"},
        }],
    })
    errs = cfg.validate()
    assert not errs, errs
    eng = af.Engine(cfg)
    cancel = asyncio.Event()
    task = asyncio.ensure_future(eng.run_with_cancellation(cancel))
    await asyncio.sleep(SECONDS)
    cancel.set()
    await asyncio.wait_for(task, 120)
    e = eng.runtime.get("soak")
    m = e.metrics
    print(f"state={e.state.value} rows_in={m.input_messages:,} "
          f"batches_out={m.output_batches:,} "
          f"rows/s={m.input_messages / SECONDS / 1e6:.1f}M "
          f"proc_errors={m.processing_errors} input_errors={m.input_errors} "
          f"output_errors={m.output_errors}")
    assert m.processing_errors == 0 and m.output_errors == 0


asyncio.run(main())

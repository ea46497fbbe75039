"""Mixed-workload soak: five different stream shapes concurrently —
fused flagship (direct mode), fused GROUP BY graph, durable WAL +
sliding window + agg,
JSON-decode + filter, BERT inference — through the full engine.
Usage: soak_mixed.py [seconds]"""
import asyncio
import json
import os
import sys
import tempfile

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import arkflow_amd as af
from arkflow_amd.config import EngineConfig

SECONDS = float(sys.argv[1]) if len(sys.argv) > 1 else 60.0


async def main():
    wal_dir = tempfile.mkdtemp(prefix="soak_wal_")
    doc = json.dumps({"v": 0.5, "k": 3, "tag": "abc"})
    cfg = EngineConfig.from_dict({"streams": [
        {   # fused flagship, direct mode
            "id": "fused",
            "input": {"type": "generate", "batch_size": 8192,
                      "interval": "0ms",
                      "fields": {**{f"f{i}": {"dtype": "float32"}
                                    for i in range(16)},
                                 "key": {"dtype": "int64", "low": 0,
                                         "high": 1024}}},
            "pipeline": {"thread_num": 1, "processors": [
                {"type": "sql", "query":
                 "SELECT * FROM flow WHERE f0 >= 0.2"},
                {"type": "inference", "model": "mlp_anomaly",
                 "columns": [f"f{i}" for i in range(16)]}]},
            "output": {"type": "drop"},
        },
        {   # fused GROUP BY step graph (whole-step hipGraph agg)
            "id": "fusedagg",
            "input": {"type": "generate", "batch_size": 8192,
                      "interval": "0ms",
                      "fields": {"f0": {"dtype": "float32", "low": 0.0,
                                        "high": 1.0},
                                 "key": {"dtype": "int64", "low": 0,
                                         "high": 1024}}},
            "pipeline": {"thread_num": 1, "processors": [
                {"type": "sql", "query":
                 "SELECT key, count(*) AS c, sum(f0) AS s FROM flow "
                 "WHERE f0 >= 0.2 GROUP BY key"}]},
            "output": {"type": "drop"},
        },
        {   # durable + windowed aggregation
            "id": "durable",
            "input": {"type": "generate", "batch_size": 8192,
                      "interval": "2ms",
                      "fields": {"k": {"dtype": "int64", "low": 0,
                                       "high": 4096},
                                 "v": {"dtype": "float32"}}},
            "durability": {"enabled": True, "path": wal_dir,
                           "backend": "segment",
                           "sync_policy": "group_commit"},
            "buffer": {"type": "sliding_window", "window_size": 4,
                       "slide_size": 2},
            "pipeline": {"thread_num": 2, "processors": [
                {"type": "sql", "query":
                 "SELECT k, count(*) AS c, sum(v) AS s FROM flow "
                 "GROUP BY k"}]},
            "output": {"type": "drop"},
        },
        {   # JSON decode + vrl + filter
            "id": "jsonl",
            "input": {"type": "generate", "batch_size": 2048,
                      "interval": "2ms", "context": doc},
            "pipeline": {"thread_num": 2, "processors": [
                {"type": "json_to_arrow"},
                {"type": "vrl",
                 "source": '.big = .v * 2.0\nif .k > 1 { .hot = true } '
                           'else { .hot = false }'},
                {"type": "sql", "query":
                 "SELECT * FROM flow WHERE big >= 0.5"}]},
            "output": {"type": "drop"},
        },
        {   # BERT inference
            "id": "bert",
            "dedicated_thread": True,
            "input": {"type": "generate", "batch_size": 2048,
                      "interval": "5ms",
                      "fields": {"token": {"dtype": "int64", "low": 0,
                                           "high": 30000}}},
            "pipeline": {"thread_num": 1, "processors": [
                {"type": "inference", "model": "bert_base",
                 "layers": 4}]},
            "output": {"type": "drop"},
        },
    ]})
    eng = af.Engine(cfg)
    for sc in cfg.streams:
        eng.runtime.register(sc)
    await eng.runtime.start_all()
    import faulthandler
    import time
    faulthandler.dump_traceback_later(SECONDS + 90, exit=True)
    t0 = time.perf_counter()
    left = SECONDS
    while left > 0:
        await asyncio.sleep(min(20, left))
        left -= 20
        print("  " + " ".join(
            f"{e.stream_id}={e.stream.metrics.input_messages/1e6:.1f}M"
            f"/e{e.stream.metrics.processing_errors + e.stream.metrics.output_errors + e.stream.metrics.input_errors}"
            for e in eng.runtime.entries.values()), flush=True)
    elapsed = time.perf_counter() - t0
    for e in eng.runtime.entries.values():
        m = e.stream.metrics
        errs = m.processing_errors + m.output_errors + m.input_errors
        print(f"{e.stream_id}: state={e.state.value} "
              f"in={m.input_messages/1e6:.1f}M "
              f"({m.input_messages/elapsed/1e6:.1f}M/s) out_batches="
              f"{m.output_batches} errors={errs}", flush=True)
    print("stopping...", flush=True)
    await asyncio.wait_for(eng.runtime.stop_all(), 90)
    print("stopped clean", flush=True)


asyncio.run(main())

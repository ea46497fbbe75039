"""Engine-path soak: N identical generate→sql(filter)→mlp streams through
the FULL engine (RuntimeManager, backpressure, ordered output) — reports
sustained rows/s. The fusable chain auto-fuses into whole-step hipGraphs
on GPU (stream.py fusable_chain). Usage: soak_engine.py [seconds] [streams]
"""
import asyncio
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import arkflow_amd as af
from arkflow_amd.config import EngineConfig

SECONDS = float(sys.argv[1]) if len(sys.argv) > 1 else 20.0
NSTREAMS = int(sys.argv[2]) if len(sys.argv) > 2 else 1
THREADED = len(sys.argv) > 3 and sys.argv[3] == "threads"


async def main():
    streams = []
    for s in range(NSTREAMS):
        streams.append({
            "id": f"soak{s}",
            "dedicated_thread": THREADED,
            "input": {"type": "generate", "batch_size": 8192,
                      "interval": "0ms",
                      "fields": {
                          **{f"f{i}": {"dtype": "float32"}
                             for i in range(16)},
                          "key": {"dtype": "int64", "low": 0, "high": 1024},
                      }},
            "pipeline": {"thread_num": 1, "processors": [
                {"type": "sql",
                 "query": "SELECT * FROM flow WHERE f0 >= 0.2"},
                {"type": "inference", "model": "mlp_anomaly",
                 "columns": [f"f{i}" for i in range(16)]},
            ]},
            "output": {"type": "drop"},
        })
    cfg = EngineConfig.from_dict({"streams": streams})
    eng = af.Engine(cfg)
    for sc in cfg.streams:
        eng.runtime.register(sc)
    await eng.runtime.start_all()
    import faulthandler
    import time
    faulthandler.dump_traceback_later(SECONDS + 30, exit=True)
    t0 = time.perf_counter()
    left = SECONDS
    while left > 0:
        await asyncio.sleep(min(5, left))
        left -= 5
        tot = sum(e.stream.metrics.input_messages
                  for e in eng.runtime.entries.values())
        print(f"  t={time.perf_counter()-t0:.0f}s rows={tot/1e6:.1f}M",
              flush=True)
    elapsed = time.perf_counter() - t0
    total_in = sum(e.stream.metrics.input_messages
                   for e in eng.runtime.entries.values())
    total_out = sum(e.stream.metrics.output_messages
                    for e in eng.runtime.entries.values())
    errs = sum(e.stream.metrics.processing_errors +
               e.stream.metrics.output_errors
               for e in eng.runtime.entries.values())
    fused = [type(e.stream.input).__name__
             for e in eng.runtime.entries.values()]
    print("stopping...", flush=True)
    await asyncio.wait_for(eng.runtime.stop_all(), 60)
    print(f"streams={NSTREAMS} threaded={THREADED} fused={fused[0]} "
          f"rows_in/s={total_in/elapsed/1e6:.1f}M "
          f"rows_out/s={total_out/elapsed/1e6:.1f}M errors={errs}")


asyncio.run(main())

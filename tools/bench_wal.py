"""WAL subsystem benchmark — the only metric with published reference numbers
(local append ~1µs, flush ~2ms, 500 MB/s; segment encode 500+
MB/s). Measures append latency, group-commit throughput, segment encode rate
and recovery."""
import os, shutil, sys, tempfile, time
sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
import asyncio
import torch
from arkflow_amd.batch import MessageBatch
from arkflow_amd.config import DurabilityConfig
from arkflow_amd.wal.store import encode_frame, serialize_batch
from arkflow_amd.wal.wal import Wal

root = tempfile.mkdtemp(prefix="walbench")
batch = MessageBatch.from_dict({
    "v": torch.rand(8192), "k": torch.randint(0, 1000, (8192,)),
    "s": [f"row-{i}" for i in range(8192)],
})
payload = serialize_batch(batch)
print(f"batch payload: {len(payload)/1024:.1f} KiB (8192 rows)")

# 1) serialization + frame encode rate (reference 'segment encoding 500+ MB/s')
t0 = time.perf_counter()
n = 200
for i in range(n):
    buf = encode_frame(i, serialize_batch(batch))
dt = time.perf_counter() - t0
print(f"serialize+frame: {n*len(payload)/dt/1e6:.0f} MB/s "
      f"({dt/n*1e6:.0f} µs/batch)")

async def staged_append(policy, sync_every):
    cfg = DurabilityConfig(enabled=True, path=os.path.join(root, policy),
                           sync_policy=policy, group_window_ms=2)
    wal = Wal.open(cfg, "bench")
    lat = []
    t0 = time.perf_counter()
    iters = 50 if policy == "per_entry" else 400
    for i in range(iters):
        t1 = time.perf_counter()
        await wal.append(batch)
        lat.append(time.perf_counter() - t1)
    await wal.flush_pending()
    dt = time.perf_counter() - t0
    await wal.close()
    lat.sort()
    print(f"{policy}: append p50 {lat[len(lat)//2]*1e6:.0f} µs, "
          f"throughput {iters*len(payload)/dt/1e6:.0f} MB/s")

asyncio.new_event_loop().run_until_complete(staged_append("group_commit", 0))
asyncio.new_event_loop().run_until_complete(staged_append("per_entry", 1))

# 3) segment store PUT throughput (reference S3: 450 MB/s @ 8 workers)
from arkflow_amd.wal.segment_store import SegmentWalStore
st = SegmentWalStore(os.path.join(root, "seg"), "bench", max_entries=64,
                     put_workers=4)
entries = [(i, payload) for i in range(400)]
t0 = time.perf_counter()
for i in range(0, 400, 64):
    st.append_batch(entries[i:i+64], sync=False)
st.close()
dt = time.perf_counter() - t0
print(f"segment store (4 PUT workers): "
      f"{400*len(payload)/dt/1e6:.0f} MB/s")

# 4) recovery rate
st2 = SegmentWalStore(os.path.join(root, "seg"), "bench")
t0 = time.perf_counter()
cnt = sum(1 for _ in st2.read_after(0))
dt = time.perf_counter() - t0
print(f"recovery: {cnt} entries, {cnt*len(payload)/dt/1e6:.0f} MB/s")
st2.close()
shutil.rmtree(root, ignore_errors=True)

# mmap store: per-entry durability via msync (the redb-analog fast path)
from arkflow_amd.wal.store import MmapWalStore
mdir = os.path.join(root, "mmap")
mst = MmapWalStore(mdir, stream_id="m")
frame = serialize_batch(batch)
t0 = time.perf_counter()
lat = []
for i in range(100):
    t1 = time.perf_counter()
    mst.append_batch([(i + 1, frame)], sync=True)
    lat.append(time.perf_counter() - t1)
dt = time.perf_counter() - t0
lat.sort()
print(f"mmap per_entry: append p50 {lat[50]*1e6:.0f} µs, "
      f"throughput {100*len(frame)/dt/1e6:.0f} MB/s")
mst.close()

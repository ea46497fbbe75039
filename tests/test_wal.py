"""WAL durability tests — the reference's key scenarios (stream/mod.rs
:814 group-commit flush on close, :1161 replay ordering [replayed, new],
:1257 replay > channel capacity without deadlock; wal/mod.rs corruption
tests)."""
import asyncio
import os

import pytest

from arkflow_amd.batch import MessageBatch
from arkflow_amd.config import DurabilityConfig, PipelineConfig, StreamConfig
from arkflow_amd.wal.store import (
    LocalWalStore,
    decode_frames,
    deserialize_batch,
    encode_frame,
    serialize_batch,
)
from arkflow_amd.wal.segment_store import SegmentWalStore
from arkflow_amd.wal.wal import Wal, WalAck


def _mk(vals, name="in"):
    return MessageBatch.from_dict(
        {"v": vals, "s": [f"r{v}" for v in vals]}, input_name=name)


def test_serialize_roundtrip():
    import torch
    b = MessageBatch.from_dict({
        "i": [1, 2, 3],
        "f": [1.5, 2.5, 3.5],
        "bf": torch.tensor([1.0, 2.0, 3.0], dtype=torch.bfloat16),
        "s": ["a", "bb", ""],
    }, input_name="src")
    r = deserialize_batch(serialize_batch(b))
    assert r.input_name == "src"
    assert r.column("i").to_pylist() == [1, 2, 3]
    assert r.column("f").to_pylist() == [1.5, 2.5, 3.5]
    assert r.column("bf").to_pylist() == [1.0, 2.0, 3.0]
    assert r.column("s").to_pylist() == [b"a", b"bb", b""]


def test_frame_torn_tail():
    good = encode_frame(1, b"hello") + encode_frame(2, b"world")
    torn = good + encode_frame(3, b"xx")[:-2]  # truncated tail
    out = list(decode_frames(torn))
    assert [(s, p) for s, p in out] == [(1, b"hello"), (2, b"world")]
    # corrupt CRC in the middle stops replay there
    bad = bytearray(good)
    bad[20] ^= 0xFF
    assert len(list(decode_frames(bytes(bad)))) <= 1


@pytest.mark.parametrize("store_cls", [LocalWalStore, SegmentWalStore])
def test_store_append_cursor_recovery(tmp_path, store_cls):
    st = store_cls(str(tmp_path), stream_id="s1")
    st.append_batch([(1, b"a"), (2, b"b"), (3, b"c")], sync=True)
    st.write_cursor(1)
    st.close()
    st2 = store_cls(str(tmp_path), stream_id="s1")
    got = list(st2.read_after(st2.cursor))
    assert [(s, p) for s, p in got] == [(2, b"b"), (3, b"c")]
    assert st2.max_seq == 3
    st2.close()


def test_wal_per_entry_and_replay(tmp_path, run):
    async def main():
        cfg = DurabilityConfig(enabled=True, path=str(tmp_path),
                               sync_policy="per_entry")
        wal = Wal.open(cfg, "s1")
        s1 = await wal.append(_mk([1]))
        s2 = await wal.append(_mk([2]))
        assert (s1, s2) == (1, 2)
        # ack the first → cursor advances past it
        await WalAck(wal, s1, _NopAck()).ack()
        await wal.close()
        wal2 = Wal.open(cfg, "s1")
        replayed = [b async for _, b in wal2.read_after_cursor()]
        assert len(replayed) == 1
        assert replayed[0].column("v").to_pylist() == [2]
        await wal2.close()

    run(main())


class _NopAck:
    async def ack(self):
        pass


def test_wal_group_commit_flush_on_close(tmp_path, run):
    """reference stream/mod.rs:814 — staged entries survive via close flush."""
    async def main():
        cfg = DurabilityConfig(enabled=True, path=str(tmp_path),
                               sync_policy="group_commit",
                               group_window_ms=10_000)  # never fires
        wal = Wal.open(cfg, "s1")
        await wal.append(_mk([1]))
        await wal.append(_mk([2]))
        await wal.close()  # must flush pending
        wal2 = Wal.open(cfg, "s1")
        replayed = [b async for _, b in wal2.read_after_cursor()]
        assert len(replayed) == 2
        await wal2.close()

    run(main())


def test_stream_recovery_replay_order(tmp_path, run):
    """Replay comes BEFORE new input: output = [replayed, new]
    (reference stream/mod.rs:1161)."""
    from arkflow_amd.pipeline import Pipeline
    from arkflow_amd.stream import Stream
    from arkflow_amd.wal.wal import Wal
    from tests.test_stream_engine import CountingOutput, StubInput

    async def main():
        cfg = DurabilityConfig(enabled=True, path=str(tmp_path),
                               sync_policy="per_entry")
        # pre-populate an un-acked WAL entry (simulates crash before ack)
        wal0 = Wal.open(cfg, "t")
        await wal0.append(_mk([100]))
        await wal0.close()

        sc = StreamConfig(id="t", input={"type": "memory"},
                          output={"type": "drop"},
                          pipeline=PipelineConfig(thread_num=2))
        inp = StubInput([_mk([1]), _mk([2])])
        out = CountingOutput()
        wal = Wal.open(cfg, "t")
        s = Stream(sc, inp, Pipeline([]), out, wal=wal)
        await asyncio.wait_for(s.run(asyncio.Event()), 10)
        vals = [b.column("v").to_pylist()[0] for b in out.batches]
        assert vals == [100, 1, 2]  # replayed first, then new input
        # everything acked → fresh WAL replays nothing
        wal2 = Wal.open(cfg, "t")
        assert [x async for x in wal2.read_after_cursor()] == []
        await wal2.close()

    run(main())


def test_replay_larger_than_channel_capacity(tmp_path, run):
    """Replay of many entries with bounded queues must not deadlock
    (reference stream/mod.rs:1257 — consumers started before replay)."""
    from arkflow_amd.pipeline import Pipeline
    from arkflow_amd.stream import Stream
    from tests.test_stream_engine import CountingOutput, StubInput

    async def main():
        cfg = DurabilityConfig(enabled=True, path=str(tmp_path),
                               sync_policy="per_entry")
        wal0 = Wal.open(cfg, "t")
        for i in range(100):  # > thread_num*4 = 8 queue capacity
            await wal0.append(_mk([i]))
        await wal0.close()
        sc = StreamConfig(id="t", input={"type": "memory"},
                          output={"type": "drop"},
                          pipeline=PipelineConfig(thread_num=2))
        out = CountingOutput()
        wal = Wal.open(cfg, "t")
        s = Stream(sc, StubInput([]), Pipeline([]), out, wal=wal)
        await asyncio.wait_for(s.run(asyncio.Event()), 15)
        assert out.rows == 100

    run(main())


def test_segment_store_reclaim(tmp_path):
    st = SegmentWalStore(str(tmp_path), stream_id="s1", max_entries=2)
    st.append_batch([(1, b"a"), (2, b"b")], sync=True)
    st.append_batch([(3, b"c"), (4, b"d")], sync=True)
    segs = [f for f in os.listdir(st.dir) if f.startswith("seg-")]
    assert len(segs) >= 2
    st.write_cursor(2)  # first segment fully acked → reclaimed
    segs2 = [f for f in os.listdir(st.dir) if f.startswith("seg-")]
    assert len(segs2) < len(segs)
    assert [s for s, _ in st.read_after(st.cursor)] == [3, 4]
    st.close()


def test_native_frame_codec_interop():
    """Native (csrc/wal_codec.cpp) and Python framing must be bit-identical
    and cross-readable, including torn-tail truncation."""
    nwal = pytest.importorskip("arkflow_amd._wal_native")
    import zlib
    from arkflow_amd.wal.store import decode_frames, encode_frame
    entries = [(1, b"alpha"), (2, b""), (3, bytes(range(256)) * 100)]
    blob = nwal.encode_frames(entries)
    assert blob == b"".join(encode_frame(s, p) for s, p in entries)
    assert list(decode_frames(blob)) == entries
    # torn tail: drop 3 bytes → last frame discarded by both decoders
    assert [s for s, _ in decode_frames(blob[:-3])] == [1, 2]
    # corrupt a body byte of frame 2 → truncates from there
    bad = bytearray(blob)
    bad[len(blob) - 50] ^= 0xFF
    assert [s for s, _ in decode_frames(bytes(bad))] == [1, 2]
    # crc32 matches zlib for arbitrary inits
    assert nwal.crc32(b"data", 1234) == zlib.crc32(b"data", 1234)


def test_chaos_flaky_output_at_least_once(tmp_path, run):
    """Intermittent output failures withhold acks; a restarted stream
    replays the failed batches from the WAL — at-least-once, nothing lost
    (reference stream/mod.rs:517-537 + §3.4 recovery)."""
    from arkflow_amd.pipeline import Pipeline
    from arkflow_amd.spi import Output
    from arkflow_amd.stream import Stream
    from tests.test_stream_engine import StubInput

    class FlakyOutput(Output):
        def __init__(self):
            self.rows = []
            self.calls = 0

        async def connect(self):
            pass

        async def write(self, batch):
            raise NotImplementedError

        async def write_batch(self, batches):
            self.calls += 1
            if self.calls % 3 == 0:  # every 3rd write fails AFTER recording
                raise RuntimeError("injected output failure")
            for b in batches:
                self.rows.extend(b.column("v").to_pylist())

        async def close(self):
            pass

    async def main():
        cfg = DurabilityConfig(enabled=True, path=str(tmp_path),
                               sync_policy="per_entry")
        out = FlakyOutput()
        wal = Wal.open(cfg, "c")
        sc = StreamConfig(id="c", input={"type": "memory"},
                          output={"type": "drop"},
                          pipeline=PipelineConfig(thread_num=2))
        s = Stream(sc, StubInput([_mk([i]) for i in range(30)]),
                   Pipeline([]), out, wal=wal)
        await asyncio.wait_for(s.run(asyncio.Event()), 15)
        first_run = set(out.rows)
        assert s.metrics.output_errors > 0
        assert first_run != set(range(30))  # some withheld

        # restart: replay-before-ingest redelivers only un-acked batches
        out2 = FlakyOutput()
        out2.calls = 1  # desync the failure phase so retries succeed
        wal2 = Wal.open(cfg, "c")
        s2 = Stream(sc, StubInput([]), Pipeline([]), out2, wal=wal2)
        await asyncio.wait_for(s2.run(asyncio.Event()), 15)
        delivered = first_run | set(out2.rows)
        missing = set(range(30)) - delivered
        assert not missing, f"lost rows {missing}"

    run(main())


def test_concurrent_cursor_writes_race_free(tmp_path):
    """write_cursor from many threads (executor acks) must not collide on
    the tmp-file replace (GPU-soak-caught race) for either store."""
    from concurrent.futures import ThreadPoolExecutor
    from arkflow_amd.registry import build_component

    for spec in ({"type": "local", "path": str(tmp_path / "l"),
                  "fsync": False},
                 {"type": "segment", "path": str(tmp_path / "s")}):
        st = build_component("wal_store", {**spec, "stream_id": "r"})
        st.append_batch([(i, b"x" * 100) for i in range(1, 65)], True)
        with ThreadPoolExecutor(16) as pool:
            list(pool.map(st.write_cursor, list(range(1, 80)) * 4))
        assert st.cursor == 79
        st.close()


def test_local_store_online_compaction(tmp_path):
    """The acked prefix is dropped online once the log exceeds the
    compaction threshold, without a restart."""
    import os
    from arkflow_amd.wal.store import LocalWalStore
    st = LocalWalStore(str(tmp_path), "oc", fsync=False)
    st.compact_bytes = 50_000
    payload = b"z" * 1000
    st.append_batch([(i, payload) for i in range(1, 101)], True)
    before = os.path.getsize(st.log_path)
    assert before > 50_000
    st.write_cursor(90)  # 90% acked → compaction triggers
    after = os.path.getsize(st.log_path)
    assert after < before / 5
    # the live tail survives and replays
    live = [s for s, _ in st.read_after(st.cursor)]
    assert live == list(range(91, 101))
    # appends continue cleanly on the reopened handle
    st.append_batch([(101, payload)], True)
    assert [s for s, _ in st.read_after(100)] == [101]
    st.close()


def test_segment_recovery_manifest_union_list(tmp_path):
    """Recovery = manifest ∪ directory LIST: a segment PUT that landed after
    the last manifest write (crash window) is still recovered; a corrupt
    tail inside one segment truncates only that segment (s3.rs semantics)."""
    import os
    from arkflow_amd.registry import build_component
    st = build_component("wal_store", {
        "type": "segment", "path": str(tmp_path), "stream_id": "m",
        "max_entries": 2})
    st.append_batch([(1, b"a"), (2, b"b")], True)   # seals segment 1
    st.append_batch([(3, b"c"), (4, b"d")], True)   # seals segment 2
    st.close()
    seg_dir = tmp_path / "m"
    # simulate crash-after-PUT-before-manifest: drop manifest references by
    # rewriting it to mention only the FIRST segment
    import json
    man = json.loads((seg_dir / "manifest.json").read_text())
    assert len(man["segments"]) >= 2
    dropped = man["segments"][1:]
    man["segments"] = man["segments"][:1]
    (seg_dir / "manifest.json").write_text(json.dumps(man))
    st2 = build_component("wal_store", {
        "type": "segment", "path": str(tmp_path), "stream_id": "m"})
    got = [s for s, _ in st2.read_after(0)]
    assert got == [1, 2, 3, 4], got  # LISTed segments recovered
    st2.close()
    # corrupt the tail of the last segment: its torn entry is dropped,
    # earlier entries survive
    last = sorted(p for p in os.listdir(seg_dir) if p.startswith("seg-"))[-1]
    raw = (seg_dir / last).read_bytes()
    (seg_dir / last).write_bytes(raw[:-3])
    st3 = build_component("wal_store", {
        "type": "segment", "path": str(tmp_path), "stream_id": "m"})
    got = [s for s, _ in st3.read_after(0)]
    assert got == [1, 2, 3], got
    st3.close()


def test_mmap_store_roundtrip_and_durability(tmp_path, run):
    """mmap WAL store: frames, torn-tail recovery, cursor compaction, and
    engine-level replay parity with the local store."""
    from arkflow_amd.registry import build_component
    st = build_component("wal_store", {"type": "mmap", "path": str(tmp_path),
                                       "stream_id": "m",
                                       "chunk_bytes": 1 << 20})
    st.append_batch([(i, bytes([i]) * (i * 100)) for i in range(1, 20)], True)
    assert [s for s, _ in st.read_after(10)] == list(range(11, 20))
    st.write_cursor(15)
    st.close()
    # reopen: compaction dropped the acked prefix, tail intact
    st2 = build_component("wal_store", {"type": "mmap", "path": str(tmp_path),
                                        "stream_id": "m",
                                        "chunk_bytes": 1 << 20})
    assert [s for s, _ in st2.read_after(st2.cursor)] == [16, 17, 18, 19]
    assert st2.max_seq == 19
    st2.append_batch([(20, b"new")], True)
    assert [s for s, _ in st2.read_after(18)] == [19, 20]
    st2.close()

    # full Wal on the mmap backend: staged appends + replay
    async def main():
        from arkflow_amd.config import DurabilityConfig
        from arkflow_amd.wal.wal import Wal
        cfg = DurabilityConfig(enabled=True, path=str(tmp_path),
                               backend="mmap", sync_policy="group_commit")
        wal = Wal.open(cfg, "w2")
        for i in range(30):
            await wal.append(_mk([i]))
        await wal.close()
        wal2 = Wal.open(cfg, "w2")
        out = [b async for _, b in wal2.read_after_cursor()]
        assert [b.column("v").to_pylist()[0] for b in out] == list(range(30))
        await wal2.close()

    run(main())


def test_open_window_drains_on_graceful_stop(tmp_path, run):
    """A graceful stop drains the open window (delivered + acked, nothing to
    replay); only a hard crash (no close — covered by
    test_stream_recovery_replay_order) leaves entries for the WAL.
    Reference §5: window contents are volatile, the WAL holds un-acked
    source batches."""
    from arkflow_amd.buffers.windows import SlidingWindowBuffer
    from arkflow_amd.pipeline import Pipeline
    from arkflow_amd.stream import Stream
    from tests.test_stream_engine import CountingOutput, StubInput

    async def main():
        cfg = DurabilityConfig(enabled=True, path=str(tmp_path),
                               sync_policy="per_entry")
        wal = Wal.open(cfg, "w")
        buf = SlidingWindowBuffer({"window_size": 100, "slide_size": 100})
        sc = StreamConfig(id="w", input={"type": "memory"},
                          output={"type": "drop"},
                          pipeline=PipelineConfig(thread_num=1))

        class SlowEOFInput(StubInput):
            async def read(self):
                if not self.batches:
                    await asyncio.sleep(10)  # stay open; cancel fires first
                return await super().read()

        inp = SlowEOFInput([_mk([i]) for i in range(5)])
        out = CountingOutput()
        s = Stream(sc, inp, Pipeline([]), out, wal=wal, buffer=buf)
        cancel = asyncio.Event()
        task = asyncio.ensure_future(s.run(cancel))
        await asyncio.sleep(0.5)  # batches ingested into the open window
        cancel.set()
        await asyncio.wait_for(task, 15)
        # drained on stop: one combined window batch, all rows, all acked
        assert out.rows == 5
        # cursor advanced → nothing replays on restart
        wal2 = Wal.open(cfg, "w")
        out2 = CountingOutput()
        s2 = Stream(sc, StubInput([]), Pipeline([]), out2, wal=wal2)
        await asyncio.wait_for(s2.run(asyncio.Event()), 15)
        assert out2.rows == 0

    run(main())


def test_frame_format_golden_bytes():
    """Pin the on-disk frame layout: [seq u64 BE|len u32 BE|tag|body|crc32 BE].
    Breaking this silently would orphan existing WALs. (CRC coverage was
    deliberately widened to the WHOLE frame incl. header after the
    random-corruption fuzz showed a flipped seq byte replayed as a wrong
    sequence number — golden updated with that format revision.)"""
    from arkflow_amd.wal.store import decode_frames, encode_frame
    golden = bytes.fromhex("00000000000000070000000752676f6c64656e9222f6e3")
    assert encode_frame(7, b"golden") == golden
    assert list(decode_frames(golden)) == [(7, b"golden")]
    nwal = pytest.importorskip("arkflow_amd._wal_native")
    assert nwal.encode_frames([(7, b"golden")]) == golden
    assert nwal.encode_frame_parts(7, [b"gol", b"den"]) == golden


def test_batch_payload_golden_bytes():
    """Pin the batch serialization layout (magic, input name, per-column
    dtype/kind headers, buffers)."""
    import torch
    from arkflow_amd.batch import Column
    from arkflow_amd.wal.store import deserialize_batch, serialize_batch
    b = MessageBatch(
        {"v": Column.from_numeric(torch.tensor([1.0], dtype=torch.float32)),
         "s": Column.from_strings(["x"])}, input_name="in")
    golden = bytes.fromhex(
        "4157414c0002696e00000002000176000000070000000000000001000000"
        "04666c6f617433320000803f000173010000000000000000010000000100"
        "0000107800000000000000000100000000000000")
    assert serialize_batch(b) == golden
    out = deserialize_batch(golden)
    assert out.input_name == "in"
    assert out.column("v").to_pylist() == [1.0]
    assert out.column("s").to_strlist() == ["x"]


def test_validity_mask_survives_wal_roundtrip():
    """Regression: deserialize_batch used to slice validity with [:-1],
    losing the last element of every replayed NULL mask (advisor r1)."""
    import torch
    from arkflow_amd.batch import Column
    b = MessageBatch.from_dict({"v": [1, 2, 3], "s": ["a", "b", "c"]})
    b.columns["v"].validity = torch.tensor([True, False, True])
    b.columns["s"].validity = torch.tensor([False, True, True])
    r = deserialize_batch(serialize_batch(b))
    assert r.column("v").validity.tolist() == [True, False, True]
    assert r.column("s").validity.tolist() == [False, True, True]


def test_mmap_store_append_after_reopen(tmp_path):
    """Regression: reopening with recovered data left the mapping size
    non-page-aligned, so the first append's rounded-up msync range fell
    outside the map and raised ValueError (advisor r1)."""
    from arkflow_amd.wal.store import MmapWalStore
    st = MmapWalStore(str(tmp_path), stream_id="m1", chunk_bytes=65536)
    st.append_batch([(1, b"x" * 100), (2, b"y" * 100)], sync=True)
    st.close()
    st2 = MmapWalStore(str(tmp_path), stream_id="m1", chunk_bytes=65536)
    # the close() truncated to _pos (non-aligned); this append must not raise
    st2.append_batch([(3, b"z" * 100)], sync=True)
    assert [s for s, _ in st2.read_after(0)] == [1, 2, 3]
    st2.close()


def test_mmap_store_online_compaction(tmp_path):
    """Regression: the compaction trigger gated on file.tell() which stays 0
    for mmap appends, so the acked prefix grew unbounded (advisor r1)."""
    from arkflow_amd.wal.store import MmapWalStore
    st = MmapWalStore(str(tmp_path), stream_id="m2", chunk_bytes=65536)
    st.compact_bytes = 4096
    payload = b"p" * 1024
    for seq in range(1, 9):
        st.append_batch([(seq, payload)], sync=True)
    assert st._log_size() > st.compact_bytes
    st.write_cursor(6)  # should trigger online compaction
    assert st._log_size() < 4096  # only seqs 7,8 remain
    assert [s for s, _ in st.read_after(0)] == [7, 8]
    st.close()


def test_manifest_cas_merges_on_lost_race(tmp_path):
    """Two stores sharing one object store: a lost conditional PUT reloads,
    MERGES (segment union, max cursor) and retries — the reference's
    PutMode precondition path (manifest.rs; s3.rs:939)."""
    from arkflow_amd.wal.object_store import DirObjectStore
    store = DirObjectStore(str(tmp_path / "shared"))
    a = SegmentWalStore(str(tmp_path), stream_id="a", max_entries=1,
                        store=store)
    b = SegmentWalStore(str(tmp_path), stream_id="b", max_entries=1,
                        store=store)
    # b's manifest etag is now stale the moment a writes
    a.append_batch([(1, b"pa")], sync=True)
    b.append_batch([(2, b"pb")], sync=True)  # CAS retry fires here
    a.close()
    b.close()
    # a fresh store over the same objects recovers BOTH segments
    c = SegmentWalStore(str(tmp_path), stream_id="c", store=store)
    got = sorted(s for s, _ in c.read_after(0))
    assert got == [1, 2]
    c.close()


def test_object_store_conditional_put(tmp_path):
    from arkflow_amd.wal.object_store import (
        DirObjectStore, PreconditionFailed)
    st = DirObjectStore(str(tmp_path))
    e1 = st.put("m", b"v1", if_none_match=True)
    with pytest.raises(PreconditionFailed):
        st.put("m", b"v2", if_none_match=True)  # already exists
    with pytest.raises(PreconditionFailed):
        st.put("m", b"v2", if_match="wrong-etag")
    e2 = st.put("m", b"v2", if_match=e1)
    assert st.get("m") == (b"v2", e2)
    assert st.list() == ["m"]
    st.delete("m")
    assert st.get("m") is None


@pytest.mark.skipif(not os.environ.get("MINIO_ENDPOINT"),
                    reason="MINIO_ENDPOINT not set")
@pytest.mark.timeout(300)
def test_s3_wal_matrix_against_minio(tmp_path):
    """Segment strategy × parallel PUT × compression against a real
    S3-compatible store (reference wal_optimization_e2e.rs, gated the same
    way)."""
    import uuid

    from arkflow_amd.wal.object_store import S3ObjectStore
    for strategy in ("low_latency", "balanced", "aggressive"):
        for workers in (1, 4):
            for compress in (False, True):
                prefix = f"t-{uuid.uuid4().hex[:8]}"
                store = S3ObjectStore.from_env("arkflow-test", prefix)
                store.ensure_bucket()
                preset = dict(
                    __import__("arkflow_amd.wal.segment_store",
                               fromlist=["SEGMENT_STRATEGIES"]
                               ).SEGMENT_STRATEGIES[strategy])
                st = SegmentWalStore(
                    str(tmp_path), stream_id=prefix,
                    max_entries=preset["max_entries"],
                    max_bytes=preset["max_bytes"],
                    flush_interval_secs=preset["flush_interval_secs"],
                    put_workers=workers, compress=compress, store=store)
                st.append_batch([(i, f"p{i}".encode() * 10)
                                 for i in range(1, 101)], sync=True)
                st.write_cursor(50)
                st.close()
                st2 = SegmentWalStore(str(tmp_path), stream_id=prefix,
                                      store=S3ObjectStore.from_env(
                                          "arkflow-test", prefix))
                got = sorted(s for s, _ in st2.read_after(st2.cursor))
                assert got == list(range(51, 101)), (strategy, workers,
                                                     compress)
                st2.close()


def test_s3_client_against_fake_server(tmp_path):
    """S3ObjectStore offline: a local HTTP server validates SigV4-shaped
    auth headers, conditional PUTs, and ListObjectsV2 XML."""
    import threading
    from http.server import BaseHTTPRequestHandler, HTTPServer

    from arkflow_amd.wal.object_store import PreconditionFailed, S3ObjectStore

    objects = {}

    class H(BaseHTTPRequestHandler):
        def log_message(self, *a):
            pass

        def _key(self):
            return self.path.split("?")[0].lstrip("/")

        def _check_auth(self):
            auth = self.headers.get("authorization", "")
            assert auth.startswith("AWS4-HMAC-SHA256 Credential="), auth
            assert "SignedHeaders=" in auth and "Signature=" in auth
            assert self.headers.get("x-amz-date")
            assert self.headers.get("x-amz-content-sha256")

        def do_PUT(self):
            self._check_auth()
            key = self._key()
            body = self.rfile.read(
                int(self.headers.get("content-length", 0)))
            cur = objects.get(key)
            if self.headers.get("If-None-Match") == "*" and cur is not None:
                self.send_response(412)
                self.end_headers()
                return
            want = self.headers.get("If-Match")
            if want is not None and (cur is None or cur[1] != want):
                self.send_response(412)
                self.end_headers()
                return
            import hashlib
            etag = hashlib.md5(body).hexdigest()
            objects[key] = (body, etag)
            self.send_response(200)
            self.send_header("ETag", f'"{etag}"')
            self.end_headers()

        def do_GET(self):
            self._check_auth()
            key = self._key()
            if "list-type=2" in (self.path.split("?") + [""])[1]:
                import urllib.parse
                q = urllib.parse.parse_qs(self.path.split("?")[1])
                prefix = q.get("prefix", [""])[0]
                keys = sorted(k for k in objects
                              if k.startswith("b/" + prefix)
                              or k.startswith(prefix))
                xml = "<ListBucketResult>" + "".join(
                    f"<Contents><Key>{k.split('/', 1)[1]}</Key></Contents>"
                    for k in keys) + \
                    "<IsTruncated>false</IsTruncated></ListBucketResult>"
                self.send_response(200)
                self.end_headers()
                self.wfile.write(xml.encode())
                return
            cur = objects.get(key)
            if cur is None:
                self.send_response(404)
                self.end_headers()
                return
            self.send_response(200)
            self.send_header("ETag", f'"{cur[1]}"')
            self.end_headers()
            self.wfile.write(cur[0])

        def do_DELETE(self):
            self._check_auth()
            objects.pop(self._key(), None)
            self.send_response(204)
            self.end_headers()

    srv = HTTPServer(("127.0.0.1", 0), H)
    threading.Thread(target=srv.serve_forever, daemon=True).start()
    try:
        st = S3ObjectStore(f"http://127.0.0.1:{srv.server_port}", "b",
                           prefix="wal", access_key="k", secret_key="s")
        e1 = st.put("manifest.json", b"v1", if_none_match=True)
        with pytest.raises(PreconditionFailed):
            st.put("manifest.json", b"v2", if_none_match=True)
        with pytest.raises(PreconditionFailed):
            st.put("manifest.json", b"v2", if_match="bogus")
        e2 = st.put("manifest.json", b"v2", if_match=e1)
        data, etag = st.get("manifest.json")
        assert data == b"v2" and etag == e2
        st.put("seg-0001.wal", b"abc")
        assert st.list("seg-") == ["seg-0001.wal"]
        st.delete("seg-0001.wal")
        assert st.get("seg-0001.wal") is None
        # the segment store end-to-end over the fake S3 server
        seg = SegmentWalStore(str(tmp_path), stream_id="x", max_entries=2,
                              store=S3ObjectStore(
                                  f"http://127.0.0.1:{srv.server_port}",
                                  "b", prefix="wal2", access_key="k",
                                  secret_key="s"))
        seg.append_batch([(1, b"p1"), (2, b"p2"), (3, b"p3")], sync=True)
        seg.close()
        seg2 = SegmentWalStore(str(tmp_path), stream_id="x",
                               store=S3ObjectStore(
                                   f"http://127.0.0.1:{srv.server_port}",
                                   "b", prefix="wal2", access_key="k",
                                   secret_key="s"))
        assert [s for s, _ in seg2.read_after(0)] == [1, 2, 3]
        seg2.close()
    finally:
        srv.shutdown()


@pytest.mark.parametrize("store_cls", [LocalWalStore, SegmentWalStore])
def test_store_random_corruption_fuzz(tmp_path, store_cls):
    """Property: flipping ONE random byte anywhere in a store's files never
    crashes recovery, and replay yields a clean prefix (or drops only the
    corrupted entry — CRC catches every flip; seq stays ordered)."""
    import random

    rng = random.Random(99)
    for case in range(25):
        root = tmp_path / f"c{case}"
        st = store_cls(str(root), stream_id="s1")
        entries = [(i + 1, bytes([i]) * (1 + i % 37)) for i in range(20)]
        st.append_batch(entries, sync=True)
        st.close()
        # flip one random byte in one random data file
        files = [p for p in root.rglob("*") if p.is_file()]
        victim = rng.choice(files)
        data = bytearray(victim.read_bytes())
        if not data:
            continue
        pos = rng.randrange(len(data))
        data[pos] ^= 0xA5
        victim.write_bytes(bytes(data))
        st2 = store_cls(str(root), stream_id="s1")
        got = list(st2.read_after(0))
        st2.close()
        seqs = [s for s, _ in got]
        assert seqs == sorted(seqs), (case, seqs)
        assert len(seqs) == len(set(seqs)), (case, seqs)
        # payload integrity for every survivor
        orig = dict(entries)
        for s, p in got:
            assert orig[s] == p or s not in orig, (case, s)

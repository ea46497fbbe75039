"""Multi-process distributed tests over gloo (world_size 2) — the CPU stand-in
for the RCCL path, as the reference gates broker tests behind containers
(SURVEY §4.6)."""
import os

import pytest
import torch
import torch.multiprocessing as mp


def _run_repartition(rank, world, port, results):
    os.environ.update({
        "RANK": str(rank), "WORLD_SIZE": str(world),
        "MASTER_ADDR": "127.0.0.1", "MASTER_PORT": str(port),
    })
    import torch.distributed as dist
    dist.init_process_group("gloo", rank=rank, world_size=world)
    from arkflow_amd.batch import MessageBatch
    from arkflow_amd.parallel.dist import repartition_by_key

    n = 100
    batch = MessageBatch.from_dict({
        "k": torch.arange(rank * n, (rank + 1) * n, dtype=torch.int64),
        "v": torch.arange(n, dtype=torch.float32) + rank * 1000,
        "s": [f"r{rank}-{i}" for i in range(n)],
    })
    out = repartition_by_key(batch, "k")
    # every key must land on exactly the rank selected by the hash
    z = out.column("k").data * 0x9E3779B97F4A7C15
    z = torch.bitwise_xor(z, z >> 30) * -0x40A7B892E31B1A47
    z = torch.bitwise_xor(z, z >> 27)
    dest = torch.remainder(z, world).abs()
    assert bool((dest == rank).all()), "row landed on wrong rank"
    # string column survives with matching k suffixes
    ks = out.column("k").to_pylist()
    ss = out.column("s").to_strlist()
    for k, s in zip(ks, ss):
        assert s.endswith(f"-{k % n}")
    results[rank] = out.num_rows
    dist.destroy_process_group()


@pytest.mark.timeout(120)
def test_repartition_two_ranks():
    port = 29531
    with mp.Manager() as mgr:
        results = mgr.dict()
        mp.spawn(_run_repartition, args=(2, port, results), nprocs=2,
                 join=True)
        total = results[0] + results[1]
        assert total == 200  # no rows lost or duplicated


def _run_bench_style(rank, world, port, results):
    os.environ.update({
        "RANK": str(rank), "WORLD_SIZE": str(world), "LOCAL_RANK": str(rank),
        "MASTER_ADDR": "127.0.0.1", "MASTER_PORT": str(port),
    })
    import torch.distributed as dist
    dist.init_process_group("gloo", rank=rank, world_size=world)
    from arkflow_amd.parallel.dist import all_reduce_scalar
    total = all_reduce_scalar(float(rank + 1), "sum")
    assert total == 3.0
    mx = all_reduce_scalar(float(rank), "max")
    assert mx == 1.0
    results[rank] = total
    dist.destroy_process_group()


@pytest.mark.timeout(120)
def test_all_reduce_scalar_two_ranks():
    with mp.Manager() as mgr:
        results = mgr.dict()
        mp.spawn(_run_bench_style, args=(2, 29532, results), nprocs=2,
                 join=True)
        assert results[0] == results[1] == 3.0


def _run_engine_sharded(rank, world, port, results):
    """Full engine with a repartition processor in the pipeline — the
    BASELINE config-4 shape (sharded streams + keyed repartition) on gloo."""
    os.environ.update({
        "RANK": str(rank), "WORLD_SIZE": str(world), "LOCAL_RANK": str(rank),
        "MASTER_ADDR": "127.0.0.1", "MASTER_PORT": str(port),
    })
    import asyncio
    import torch.distributed as dist
    dist.init_process_group("gloo", rank=rank, world_size=world)
    import arkflow_amd as af
    from arkflow_amd.config import EngineConfig

    n_batches = 6
    cfg = EngineConfig.from_dict({
        "streams": [{
            "id": "shard",
            "input": {"type": "generate", "batch_size": 512,
                      "count": 512 * n_batches, "interval": "0ms",
                      "seed": 100 + rank,
                      "fields": {
                          "session_id": {"dtype": "int64", "low": 0,
                                         "high": 64},
                          "value": {"dtype": "float32"},
                      }},
            # single worker: collectives must run in lockstep across ranks
            "pipeline": {"thread_num": 1, "processors": [
                {"type": "repartition", "key": "session_id"},
                {"type": "sql",
                 "query": "SELECT session_id, count(*) AS c, sum(value) AS s "
                          "FROM flow GROUP BY session_id"},
            ]},
            "output": {"type": "memory"},
        }]
    })
    eng = af.Engine(cfg)
    asyncio.new_event_loop().run_until_complete(
        asyncio.wait_for(eng.run_with_cancellation(), 60))
    # collect the sessions this rank saw: all must hash to this rank
    from arkflow_amd.stream import build_stream  # noqa: F401
    out_rows = eng.runtime.entries["shard"].metrics.output_messages
    results[rank] = out_rows
    dist.destroy_process_group()


@pytest.mark.timeout(180)
def test_engine_sharded_session_agg():
    with mp.Manager() as mgr:
        results = mgr.dict()
        mp.spawn(_run_engine_sharded, args=(2, 29533, results), nprocs=2,
                 join=True)
        # both ranks produced grouped outputs; union of sessions ≤ 64 per batch
        assert results[0] > 0 and results[1] > 0


def _run_repartition_w4(rank, world, port, results):
    os.environ.update({
        "RANK": str(rank), "WORLD_SIZE": str(world),
        "MASTER_ADDR": "127.0.0.1", "MASTER_PORT": str(port),
    })
    import torch.distributed as dist
    dist.init_process_group("gloo", rank=rank, world_size=world)
    from arkflow_amd.batch import MessageBatch
    from arkflow_amd.parallel.dist import repartition_by_key

    # uneven shards, one EMPTY rank — zero-length all_to_all splits must work
    n = 0 if rank == 3 else 37 * (rank + 1)
    batch = MessageBatch.from_dict({
        "k": torch.arange(rank * 1000, rank * 1000 + n, dtype=torch.int64),
        "v": torch.arange(n, dtype=torch.float32),
        "s": [f"r{rank}-{i}" for i in range(n)],
    })
    out = repartition_by_key(batch, "k")
    assert out.column("k").data.numel() == out.num_rows
    assert len(out.column("s")) == out.num_rows
    results[rank] = out.num_rows
    dist.destroy_process_group()


@pytest.mark.timeout(180)
def test_repartition_four_ranks_uneven():
    with mp.Manager() as mgr:
        results = mgr.dict()
        mp.spawn(_run_repartition_w4, args=(4, 29534, results), nprocs=4,
                 join=True)
        assert sum(results.values()) == 37 * (1 + 2 + 3)  # conservation


@pytest.mark.timeout(180)
def test_cli_path_shards_under_torchrun_env(tmp_path):
    """Regression: the *CLI path* — `python -m
    arkflow_amd --config …` under torchrun-style env vars — must join the
    process group and actually shard. r1 started N independent
    non-sharding ranks because init_from_env was never called."""
    import json
    import subprocess
    import sys

    cfg = tmp_path / "sharded.yaml"
    cfg.write_text("""
streams:
  - id: cli_shard
    input:
      type: generate
      batch_size: 64
      interval: 0ms
      count: 1280
      fields:
        k: {dtype: int64, low: 0, high: 1000}
        v: {dtype: float32, low: 0, high: 1}
    pipeline:
      thread_num: 1
      processors:
        - type: repartition
          key: k
    output:
      type: file
      path: %s/out_r${RANK}.jsonl
""" % tmp_path)

    world = 2
    procs = []
    for rank in range(world):
        env = dict(os.environ)
        env.update({
            "RANK": str(rank), "LOCAL_RANK": str(rank),
            "WORLD_SIZE": str(world), "MASTER_ADDR": "127.0.0.1",
            "MASTER_PORT": "29541",
        })
        procs.append(subprocess.Popen(
            [sys.executable, "-m", "arkflow_amd", "--config", str(cfg)],
            env=env, stdout=subprocess.PIPE, stderr=subprocess.PIPE))
    outs = [p.communicate(timeout=150) for p in procs]
    for p, (so, se) in zip(procs, outs):
        assert p.returncode == 0, se.decode()[-2000:]

    total = 0
    for rank in range(world):
        rows = [json.loads(l) for l in
                (tmp_path / f"out_r{rank}.jsonl").read_text().splitlines()]
        assert rows, f"rank {rank} produced no output"
        total += len(rows)
        ks = torch.tensor([r["k"] for r in rows], dtype=torch.int64)
        z = ks * 0x9E3779B97F4A7C15
        z = torch.bitwise_xor(z, z >> 30) * -0x40A7B892E31B1A47
        z = torch.bitwise_xor(z, z >> 27)
        dest = torch.remainder(z, world).abs()
        # every row this rank emitted belongs here by hash — proof the
        # repartition collective actually moved rows between processes
        assert bool((dest == rank).all())
    assert total == world * 1280  # nothing lost or duplicated


def _run_repartition_validity(rank, world, port, results):
    os.environ.update({
        "RANK": str(rank), "WORLD_SIZE": str(world),
        "MASTER_ADDR": "127.0.0.1", "MASTER_PORT": str(port),
    })
    import torch.distributed as dist
    dist.init_process_group("gloo", rank=rank, world_size=world)
    from arkflow_amd.batch import Column, MessageBatch
    from arkflow_amd.parallel.dist import repartition_by_key

    n = 60
    # NULL iff (k % 3 == 0) — a key-derived invariant every rank can check
    # after the shuffle regardless of where rows land
    k = torch.arange(rank * n, (rank + 1) * n, dtype=torch.int64)
    valid = (k % 3) != 0
    sc = Column.from_strings([f"x{int(x)}" for x in k])
    batch = MessageBatch({
        "k": Column("numeric", k),
        "v": Column("numeric", k.to(torch.float32), validity=valid),
        "s": Column("binary", sc.data, sc.offsets, valid),
    })
    out = repartition_by_key(batch, "k")
    ko = out.column("k").data
    for name in ("v", "s"):
        c = out.column(name)
        assert c.validity is not None, name
        assert torch.equal(c.validity, (ko % 3) != 0), name
    results[rank] = out.num_rows
    dist.destroy_process_group()


@pytest.mark.timeout(120)
def test_repartition_preserves_validity():
    port = 29537
    with mp.Manager() as mgr:
        results = mgr.dict()
        mp.spawn(_run_repartition_validity, args=(2, port, results),
                 nprocs=2, join=True)
        assert results[0] + results[1] == 120

"""Multi-GPU sharding: RCCL collectives over xGMI.

The reference has NO data-plane distribution — its scale-out
story is brokers + a control plane. The MI355X-native engine shards streams
across the 8 GPUs of a node, one process per GPU over torch.distributed
(backend "nccl" IS RCCL on ROCm), and repartitions keyed state (session
windows, join buffers) with all-to-all — the right collective for xGMI's
7 point-to-point links (one hop between any GPU pair; ring collectives are
per-link bound, all-to-all uses all links concurrently — SURVEY §5).

CPU CI runs the same code over gloo; gloo lacks all_to_all, so an
all-gather-based equivalent keeps multi-process tests green (same results,
different transport).
"""
from __future__ import annotations

import os
from typing import Dict, List, Optional

import torch
import torch.distributed as dist

from ..batch import Column, MessageBatch


def init_from_env() -> bool:
    """Initialize the process group from torchrun env vars if present."""
    if dist.is_initialized():
        return True
    if "RANK" not in os.environ or int(os.environ.get("WORLD_SIZE", 1)) <= 1:
        return False
    backend = "nccl" if torch.cuda.is_available() else "gloo"
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29500")
    dist.init_process_group(backend=backend)
    if backend == "nccl":
        torch.cuda.set_device(int(os.environ.get("LOCAL_RANK", 0)))
    return True


def rank() -> int:
    return dist.get_rank() if dist.is_initialized() else 0


def world_size() -> int:
    return dist.get_world_size() if dist.is_initialized() else 1


def _supports_all_to_all() -> bool:
    return dist.get_backend() == "nccl"


def _all_to_all_1d(t: torch.Tensor, out_splits: List[int],
                   in_splits: List[int]) -> torch.Tensor:
    """all_to_all_single with a gloo fallback (gather + select)."""
    if _supports_all_to_all():
        out = torch.empty(sum(out_splits), dtype=t.dtype, device=t.device)
        dist.all_to_all_single(out, t.contiguous(),
                               output_split_sizes=out_splits,
                               input_split_sizes=in_splits)
        return out
    # gloo: all_gather splits + payloads, pick my slice from each peer
    w = world_size()
    r = rank()
    sizes = torch.zeros(w, w, dtype=torch.int64)
    my_sizes = torch.tensor([in_splits], dtype=torch.int64)
    gathered = [torch.zeros(1, w, dtype=torch.int64) for _ in range(w)]
    dist.all_gather(gathered, my_sizes)
    sizes = torch.cat(gathered, 0)  # sizes[src][dst]
    max_n = int(max(1, t.numel()))
    n_all = [int(sizes[s].sum().item()) for s in range(w)]
    pad = max(max(n_all), 1)
    buf = torch.zeros(pad, dtype=t.dtype, device=t.device)
    buf[: t.numel()] = t
    bufs = [torch.zeros(pad, dtype=t.dtype, device=t.device)
            for _ in range(w)]
    dist.all_gather(bufs, buf)
    parts = []
    for src in range(w):
        start = int(sizes[src][:r].sum().item())
        ln = int(sizes[src][r].item())
        parts.append(bufs[src][start:start + ln])
    return torch.cat(parts) if parts else t[:0]


def repartition_by_key(batch: MessageBatch, key_column: str
                       ) -> MessageBatch:
    """Hash-repartition rows across ranks so equal keys land on one rank.

    dest = mix64(key) % world; rows are grouped by destination locally
    (order-preserving within a destination), then exchanged with one
    all-to-all per column buffer.
    """
    w = world_size()
    if w <= 1:
        return batch
    key_col = batch.column(key_column)
    if key_col.kind != "numeric":
        raise ValueError("repartition key must be numeric")
    keys = key_col.data.to(torch.int64)
    # splittable mix then mod world (python-side constant mirror of mix64)
    z = keys * 0x9E3779B97F4A7C15
    z = torch.bitwise_xor(z, z >> 30) * -0x40A7B892E31B1A47
    z = torch.bitwise_xor(z, z >> 27)
    dest = torch.remainder(z, w).to(torch.int64).abs()
    order = torch.argsort(dest, stable=True)
    sorted_dest = dest[order]
    in_splits = torch.bincount(sorted_dest, minlength=w).tolist()
    reordered = batch.take(order)

    # exchange split sizes (device tensors on nccl/RCCL, host on gloo)
    split_dev = torch.device("cuda") if dist.get_backend() == "nccl" \
        else torch.device("cpu")
    in_t = torch.tensor(in_splits, dtype=torch.int64, device=split_dev)
    size_mat = [torch.zeros(w, dtype=torch.int64, device=split_dev)
                for _ in range(w)]
    dist.all_gather(size_mat, in_t)
    out_splits = [int(size_mat[src][rank()].item()) for src in range(w)]

    # column-type consensus: a rank with an EMPTY shard cannot know a
    # column's kind/dtype (an empty list types as numeric float64), and gloo/
    # RCCL collectives hang or kill peers if ranks disagree on the exchange
    # protocol. One small all_gather settles (kind, dtype) per column.
    DT = [torch.float64, torch.float32, torch.bfloat16, torch.float16,
          torch.int64, torch.int32, torch.int16, torch.int8, torch.uint8,
          torch.bool]
    names = list(reordered.columns)
    local = []
    for name in names:
        col = reordered.columns[name]
        kind = 1 if col.kind == "binary" else 0
        dtc = -1 if kind else DT.index(col.data.dtype)
        local += [kind, dtc, 1 if len(col) else 0,
                  1 if col.validity is not None else 0]
    meta_dev = torch.device("cuda") if dist.get_backend() == "nccl" \
        else torch.device("cpu")  # nccl/RCCL collectives need device tensors
    lt = torch.tensor(local, dtype=torch.int64, device=meta_dev)
    gathered_meta = [torch.zeros_like(lt) for _ in range(w)]
    dist.all_gather(gathered_meta, lt)
    gathered_meta = [g.cpu() for g in gathered_meta]

    cols: Dict[str, Column] = {}
    for ci, name in enumerate(names):
        col = reordered.columns[name]
        nonempty = [g for g in gathered_meta if int(g[4 * ci + 2])]
        is_binary = any(int(g[4 * ci]) == 1 for g in nonempty) or (
            not nonempty and col.kind == "binary")
        # validity consensus: if ANY rank carries a validity mask, every
        # rank exchanges one (all-True where absent) — NULL bits must
        # survive the shuffle, and asymmetric exchanges deadlock
        any_validity = any(int(g[4 * ci + 3]) for g in gathered_meta)
        out_validity = None
        if any_validity:
            v = col.validity if col.validity is not None else \
                torch.ones(len(col), dtype=torch.bool)
            out_validity = _exchange_numeric(v, in_splits, out_splits)
        if is_binary:
            if col.kind != "binary":
                if len(col) != 0:
                    raise ValueError(f"repartition: column {name!r} kind "
                                     "differs across ranks")
                col = Column("binary", torch.empty(0, dtype=torch.uint8),
                             torch.zeros(1, dtype=torch.int64))
            out = _exchange_binary(col, in_splits, out_splits, w)
            cols[name] = Column(out.kind, out.data, out.offsets,
                                out_validity)
            continue
        # symmetric fold over NON-EMPTY ranks' dtypes (identical on every
        # rank) so all ranks pick the same wire dtype
        dts = [DT[int(g[4 * ci + 1])] for g in nonempty
               if int(g[4 * ci]) == 0]
        if dts:
            target = dts[0]
            for dt in dts[1:]:
                target = torch.promote_types(target, dt)
        else:
            target = col.data.dtype if col.kind == "numeric" \
                else torch.float64
        if col.kind != "numeric":  # binary-empty vs numeric consensus
            data = torch.empty(0, dtype=target)
        else:
            data = col.data.to(target)
        data = _exchange_numeric(data, in_splits, out_splits)
        cols[name] = Column("numeric", data, validity=out_validity)
    return MessageBatch(cols, batch.input_name)


def _exchange_numeric(t: torch.Tensor, in_splits, out_splits) -> torch.Tensor:
    # bf16/bool over gloo: view as int16/uint8
    orig = t.dtype
    if orig == torch.bfloat16:
        t = t.view(torch.int16)
    elif orig == torch.bool:
        t = t.to(torch.uint8)
    out = _all_to_all_1d(t, out_splits, in_splits)
    if orig == torch.bfloat16:
        out = out.view(torch.bfloat16)
    elif orig == torch.bool:
        out = out.to(torch.bool)
    return out


def _exchange_binary(col: Column, in_splits, out_splits, w) -> Column:
    lengths = (col.offsets[1:] - col.offsets[:-1]).to(torch.int64)
    out_lengths = _all_to_all_1d(lengths, out_splits, in_splits)
    # byte splits per destination
    byte_in = []
    pos = 0
    for s in in_splits:
        byte_in.append(int(lengths[pos:pos + s].sum().item()))
        pos += s
    byte_out = []
    pos = 0
    for s in out_splits:
        byte_out.append(int(out_lengths[pos:pos + s].sum().item()))
        pos += s
    data = _all_to_all_1d(col.data, byte_out, byte_in)
    offsets = torch.zeros(out_lengths.numel() + 1, dtype=torch.int64,
                          device=data.device)
    torch.cumsum(out_lengths, 0, out=offsets[1:])
    return Column("binary", data, offsets)


def all_reduce_scalar(v: float, op: str = "sum",
                      device: Optional[torch.device] = None) -> float:
    if not dist.is_initialized():
        return v
    t = torch.tensor([v], dtype=torch.float64,
                     device=device or torch.device("cpu"))
    dist.all_reduce(t, op={"sum": dist.ReduceOp.SUM,
                           "max": dist.ReduceOp.MAX,
                           "min": dist.ReduceOp.MIN}[op])
    return float(t.item())

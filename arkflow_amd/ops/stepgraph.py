"""Whole-step hipGraph capture for the fused hot chains.

The r1 flagship bench was host-dispatch-bound at the default batch (8192
rows: ~0.15 ms GPU inside a ~0.22 ms step — VERDICT weak #3/#5). The fix:
capture the ENTIRE step as one hipGraph and replay it. Round-2 final form
(profiles r2-21/r2-22):

  genfiltpack (ONE persistent kernel: generate + filter + compact +
  feature-pack, in-kernel grid barrier, survivor values regenerated from
  the counter RNG)  →  MLP GEMM chain            [FusedGenerateFilterInfer]
  genfiltpack  →  capture-safe hash-agg chain     [FusedGenerateAgg]
  proto decode →  featpack → MLP                  [FusedProtoMlp]

The only host interaction per step is a stream sync + ONE host-mapped
int32 read (surviving-row / group count) used to slice zero-copy views
out of the static output buffers; in direct mode those views materialize
lazily on first access (_LazyStepBatch — at ~28 µs of GPU per step the
view construction itself was the larger cost).

The returned batch aliases the graph's static buffers and is valid until
the next ``read()``/``step()`` — the same contract as the window ring's
zero-copy slices (buffers/ring.py). Downstream stages that retain batches
must copy (FusedStepSource clone mode).

Reference analog: the hot loop the reference runs as compiled Rust end to
end (stream/mod.rs:370-444 + DataFusion physical operators); here the whole
step becomes one device-side graph so Python dispatch cost is O(1) per step.
"""
from __future__ import annotations

from typing import Dict, Tuple

import torch

from ..batch import Column, MessageBatch
from . import require_native

_OPS = {"<": 0, "<=": 1, ">": 2, ">=": 3, "==": 4, "!=": 5}


class _LazyStepBatch(MessageBatch):
    """MessageBatch whose column views materialize on first access.

    The fused step's outputs are zero-copy slices of the graph's static
    buffers — building 18 Column objects per step costs more host time than
    the whole GPU step. Consumers that read columns get the identical views;
    consumers that only route/count/drop (the common sink path) skip the
    construction entirely. Engine mode (clone=True) always materializes —
    retained batches must detach from the static buffers before the next
    replay, exactly as before."""

    __slots__ = ("_build", "_cols", "_n", "_dev")

    def __init__(self, build, n, device, input_name=None):
        self._build = build
        self._cols = None
        self._n = n
        self._dev = device
        self.input_name = input_name

    @property
    def columns(self):
        if self._cols is None:
            self._cols = self._build()
        return self._cols

    @property
    def num_rows(self):
        return self._n

    @property
    def device(self):  # MessageBatch.device walks columns — don't
        return self._dev


class FusedGenerateFilterInfer:
    """generate(fields) → WHERE col OP scalar → mlp(score) as one hipGraph.

    fields: ordered {name: {dtype, low, high}} — float32 fields are
    generated with one fused rand+scale launch; int64 fields via randint.
    """

    def __init__(self, fields: Dict[str, dict], batch_size: int,
                 filter_col: str, op: str, scalar: float, mlp,
                 device: torch.device, seed: int = 0x5EED):
        self.device = torch.device(device)
        if self.device.type != "cuda":
            raise RuntimeError("FusedGenerateFilterInfer requires a GPU")
        self.n = int(batch_size)
        self.fields = dict(fields)
        self.filter_col = filter_col
        self.op = _OPS[op]
        self.scalar = float(scalar)
        self.mlp = mlp
        self.nat = require_native()
        torch.cuda.manual_seed(seed)

        names = list(fields)
        self.float_names = [f for f in names
                            if str(fields[f].get("dtype", "float32"))
                            not in ("int32", "int64")]
        self.int_names = [f for f in names if f not in self.float_names]
        if len(self.int_names) > 1:
            raise ValueError("fused generate supports at most one int64 "
                             "key column")

        # static buffers: generated block, key, compacted outputs, packed
        # MFMA operand, device row count, RNG replay counter
        nf = len(self.float_names)
        self.block = torch.zeros((nf, self.n), device=self.device,
                                 dtype=torch.float32)
        self.key = torch.zeros(self.n, device=self.device,
                               dtype=torch.int64) if self.int_names else None
        self.ctr = torch.zeros(2, device=self.device, dtype=torch.int64)
        self.ctr[0] = int(seed) & 0x7FFFFFFF
        self.outs: Dict[str, torch.Tensor] = {}
        for f in self.float_names:
            self.outs[f] = torch.zeros(self.n, device=self.device,
                                       dtype=torch.float32)
        for f in self.int_names:
            self.outs[f] = torch.zeros(self.n, device=self.device,
                                       dtype=torch.int64)
        self.feats = torch.zeros((self.n, mlp.dims[0]), device=self.device,
                                 dtype=torch.bfloat16)
        # count lives in host-mapped pinned memory: the filter kernel writes
        # it over PCIe, so reading it back is a stream sync + CPU load — no
        # hipMemcpyDtoH launch per step (was ~10 us of a ~100 us step)
        import os
        try:
            if os.environ.get("ARKFLOW_MAPPED_COUNT", "1") == "0":
                raise RuntimeError("disabled")
            self.count_host, self.count = self.nat.mapped_int32(1)
        except RuntimeError:
            self.count_host = None
            self.count = torch.zeros(1, device=self.device, dtype=torch.int32)
        self._lo = [float(fields[f].get("low", 0.0))
                    for f in self.float_names]
        self._width = [float(fields[f].get("high", 100.0)) -
                       float(fields[f].get("low", 0.0))
                       for f in self.float_names]
        if self.int_names:
            spec = fields[self.int_names[0]]
            self._key_lo = int(float(spec.get("low", 0.0)))
            hi = int(float(spec.get("high", 100.0)))
            self._key_range = max(hi - self._key_lo, 1)
        else:
            self._key_lo = self._key_range = 0
        # genfiltpack workspaces: per-block counts + the MONOTONIC grid
        # barrier counter (never reset — graph replays rely on it)
        self._gfp_counts = torch.zeros(256, device=self.device,
                                       dtype=torch.int32)
        self._gfp_bar = torch.zeros(1, device=self.device, dtype=torch.int64)
        self._graph = None
        self._scores = None

    def _front(self, feats) -> None:
        """generate → filter → compact (→ featpack): ONE persistent kernel
        when the batch fits the in-kernel grid barrier (≤256K rows), else
        the multi-kernel chain. Survivor values are REGENERATED from the
        counter RNG in the fused kernel — no staged block traffic."""
        import os
        if self.n <= 256 * 1024 and len(self.float_names) <= 32 \
                and self.filter_col in self.float_names \
                and os.environ.get("ARKFLOW_NO_GFP") != "1":
            self.nat.genfiltpack(
                self._lo, self._width, self._key_lo, self._key_range,
                self.n, self.float_names.index(self.filter_col), self.op,
                self.scalar, [self.outs[f] for f in self.float_names],
                self.outs[self.int_names[0]] if self.int_names else None,
                feats, self.count, self._gfp_counts, self._gfp_bar,
                self.ctr)
            return
        cols: Dict[str, torch.Tensor] = {
            f: self.block[i] for i, f in enumerate(self.float_names)}
        if self.key is not None:
            cols[self.int_names[0]] = self.key
        self.nat.gen_fields(self.block, self.key, self._lo, self._width,
                            self._key_lo, self._key_range, self.ctr)
        ordered = [cols[f] for f in self.fields]
        fidx = list(self.fields).index(self.filter_col)
        self.nat.filter_gather_capture(
            ordered, fidx, self.op, self.scalar,
            [self.outs[f] for f in self.fields], self.count)
        if feats is not None:
            self.nat.featpack([self.outs[f] for f in self.float_names],
                              feats)

    # ------------------------------------------------------------------ body
    def _body(self) -> torch.Tensor:
        """One step as a pure kernel chain (4 launches at default batch):
        genfiltpack (generate+filter+compact+featpack fused) →
        GEMM(relu) → GEMM(relu) → GEMV+f32. No host syncs, no torch
        elementwise tail — every stage is a stepfused.hip/gemm kernel.
        Garbage rows past `count` are scored too, then sliced off."""
        self._front(self.feats)
        return self.mlp._net(self.feats)

    def capture(self) -> None:
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            for _ in range(2):  # warm up allocator + kernels
                self._body()
        torch.cuda.current_stream().wait_stream(s)
        self._graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(self._graph, capture_error_mode="thread_local"):
            self._scores = self._body()

    def _kept(self) -> int:
        """Surviving-row count for the last replay on the CURRENT stream."""
        if self.count_host is not None:
            torch.cuda.current_stream().synchronize()
            return int(self.count_host[0])
        return int(self.count.item())

    def _make_batch(self, clone: bool = False, rows: int = -1
                    ) -> MessageBatch:
        kept = self._kept() if rows < 0 else rows
        cols = {f: Column("numeric", self.outs[f][:kept])
                for f in self.fields}
        cols["score"] = Column("numeric", self._scores[:kept])
        if clone:
            cols = {k: Column("numeric", c.data.clone())
                    for k, c in cols.items()}
        return MessageBatch(cols, input_name="generate")

    # ------------------------------------------------------------------ step
    def step(self) -> Tuple[MessageBatch, int]:
        """One replay. Returns (batch_of_views, kept_rows); the batch is
        valid until the next step()."""
        if self._graph is None:
            self.capture()
        self._graph.replay()
        batch = self._make_batch()
        return batch, batch.num_rows


class FusedGenerateAgg:
    """generate(fields) → WHERE col OP scalar → GROUP BY key AGG(...) as ONE
    hipGraph (BASELINE config 2 whole-step form). Shares the generator /
    capture-safe filter stages with FusedGenerateFilterInfer; the group-by
    stage is the capture-safe hash-agg chain (csrc/hash_agg.hip device-count
    kernels). Per step: one replay + one CPU read of the host-mapped group
    count. aggs: list of (fn, col, alias) with fn in count/sum/min/max/avg.
    """

    def __init__(self, fields: Dict[str, dict], batch_size: int,
                 filter_col: str, op: str, scalar: float, key_col: str,
                 aggs, device: torch.device, seed: int = 0x5EED,
                 g_cap: int = 2048, table_size: int = 8192):
        self.device = torch.device(device)
        if self.device.type != "cuda":
            raise RuntimeError("FusedGenerateAgg requires a GPU")
        self.n = int(batch_size)
        self.fields = dict(fields)
        self.filter_col = filter_col
        self.op = _OPS[op]
        self.scalar = float(scalar)
        self.key_col = key_col
        self.aggs = list(aggs)
        self.g_cap = int(g_cap)
        self.table_size = int(table_size)
        self.nat = require_native()
        torch.cuda.manual_seed(seed)

        names = list(fields)
        self.float_names = [f for f in names
                            if str(fields[f].get("dtype", "float32"))
                            not in ("int32", "int64")]
        self.int_names = [f for f in names if f not in self.float_names]
        if self.int_names != [key_col]:
            raise ValueError("fused agg needs exactly one int64 key column")
        self.block = torch.zeros((len(self.float_names), self.n),
                                 device=self.device, dtype=torch.float32)
        self.key = torch.zeros(self.n, device=self.device, dtype=torch.int64)
        self.ctr = torch.zeros(2, device=self.device, dtype=torch.int64)
        self.ctr[0] = int(seed) & 0x7FFFFFFF
        self.outs: Dict[str, torch.Tensor] = {}
        for f in self.float_names:
            self.outs[f] = torch.zeros(self.n, device=self.device,
                                       dtype=torch.float32)
        self.outs[key_col] = torch.zeros(self.n, device=self.device,
                                         dtype=torch.int64)
        import os
        # row count stays in DEVICE memory: the hash-agg chain re-reads it
        # per thread (host-mapped would mean per-thread PCIe reads); only
        # the GROUP count is host-mapped — that's the one the host reads.
        self.count = torch.zeros(1, device=self.device, dtype=torch.int32)
        self.count_host = None
        try:
            if os.environ.get("ARKFLOW_MAPPED_COUNT", "1") == "0":
                raise RuntimeError("disabled")
            self.gcount_host, self.gcount = self.nat.mapped_int32(1)
        except RuntimeError:
            self.gcount_host = None
            self.gcount = torch.zeros(1, device=self.device,
                                      dtype=torch.int32)
        self._lo = [float(fields[f].get("low", 0.0))
                    for f in self.float_names]
        self._width = [float(fields[f].get("high", 100.0)) -
                       float(fields[f].get("low", 0.0))
                       for f in self.float_names]
        spec = fields[key_col]
        self._key_lo = int(float(spec.get("low", 0.0)))
        hi = int(float(spec.get("high", 100.0)))
        self._key_range = max(hi - self._key_lo, 1)
        # one reduction per distinct (col, op!=count) pair
        ops_code = {"sum": 0, "avg": 0, "min": 1, "max": 2}
        pairs = []
        for fn, col, _ in self.aggs:
            if fn in ("sum", "min", "max", "avg"):
                pairs.append((col, "sum" if fn == "avg" else fn))
        self._val_list = list(dict.fromkeys(pairs))
        self._val_ops = [ops_code[o] for _, o in self._val_list]
        self._gfp_counts = torch.zeros(256, device=self.device,
                                       dtype=torch.int32)
        self._gfp_bar = torch.zeros(1, device=self.device, dtype=torch.int64)
        self._graph = None

    # shares the one-kernel generate+filter front with the infer graph
    _front = FusedGenerateFilterInfer._front

    def _body(self):
        self._front(None)
        uniq, counts, red = self.nat.hash_agg_capture(
            self.outs[self.key_col], self.count,
            [self.outs[c] for c, _ in self._val_list], self._val_ops,
            self.table_size, self.g_cap, self.gcount)
        return uniq, counts, red

    def capture(self) -> None:
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            for _ in range(2):
                self._body()
        torch.cuda.current_stream().wait_stream(s)
        self._graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(self._graph, capture_error_mode="thread_local"):
            self._uniq, self._counts, self._red = self._body()

    def _kept(self) -> int:
        """Group count for the last replay on the CURRENT stream."""
        if self.gcount_host is not None:
            torch.cuda.current_stream().synchronize()
            g = int(self.gcount_host[0])
        else:
            g = int(self.gcount.item())
        if g > self.g_cap:
            raise RuntimeError(
                f"group-by overflow: {g} groups > g_cap {self.g_cap}")
        return g

    def _make_batch(self, clone: bool = False, rows: int = -1
                    ) -> MessageBatch:
        g = self._kept() if rows < 0 else rows
        rmap = {pair: t for pair, t in zip(self._val_list, self._red)}
        cols: Dict[str, Column] = {}
        for fn, col, alias in self.aggs:
            if fn == "key":
                cols[alias] = Column("numeric", self._uniq[:g])
            elif fn == "count":
                cols[alias] = Column("numeric", self._counts[:g])
            elif fn == "avg":
                cols[alias] = Column(
                    "numeric",
                    rmap[(col, "sum")][:g].double() /
                    self._counts[:g].double())
            else:
                cols[alias] = Column("numeric", rmap[(col, fn)][:g])
        if clone:
            cols = {k: Column("numeric", c.data.clone())
                    for k, c in cols.items()}
        return MessageBatch(cols, input_name="generate")

    def step(self) -> MessageBatch:
        if self._graph is None:
            self.capture()
        self._graph.replay()
        return self._make_batch()


class FusedProtoMlp:
    """kafka-shaped protobuf payloads → GPU varint decode → MLP scoring as
    ONE hipGraph (BASELINE config 3). The payload batch is the static
    device-resident input (the bench's kafka stand-in re-reads the same
    wire bytes each step, as the eager path does); every replay re-runs
    decode + featpack + the MFMA MLP. Scalar-only schemas (no string
    fields: their output allocation needs a host readback). Decode errors
    are validated once at capture; row count is fixed, so a step has ZERO
    host syncs — the caller's event wait paces the stream.
    """

    def __init__(self, data: torch.Tensor, offsets: torch.Tensor,
                 fno, kind, isf, slot, n_int: int, n_float: int,
                 float_names, int_names, mlp, device: torch.device):
        self.nat = require_native()
        self.data, self.offsets = data, offsets
        self.args = (list(fno), list(kind), list(isf), list(slot),
                     n_int, n_float)
        self.float_names = list(float_names)
        self.int_names = list(int_names)
        self.mlp = mlp
        self.device = torch.device(device)
        self.n = int(offsets.numel() - 1)
        self.feats = torch.zeros((self.n, mlp.dims[0]), device=self.device,
                                 dtype=torch.bfloat16)
        self._graph = None

    def _body(self):
        fno, kind, isf, slot, n_int, n_float = self.args
        out_i, out_f, err, _s, _sum = self.nat.proto_decode(
            self.data, self.offsets, fno, kind, isf, slot, n_int, n_float,
            0, capture=True)
        self.nat.featpack([out_f[i] for i in range(len(self.float_names))],
                          self.feats)
        return out_i, out_f, err, self.mlp._net(self.feats)

    def capture(self) -> None:
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            for _ in range(2):
                out = self._body()
        torch.cuda.current_stream().wait_stream(s)
        if int(out[2].item()) != 0:
            raise RuntimeError("proto decode error in fused warmup")
        self._graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(self._graph, capture_error_mode="thread_local"):
            self._out_i, self._out_f, self._err, self._scores = self._body()

    def _cols(self):
        cols = {}
        for i, name in enumerate(self.int_names):
            cols[name] = Column("numeric", self._out_i[i])
        for i, name in enumerate(self.float_names):
            cols[name] = Column("numeric", self._out_f[i])
        cols["score"] = Column("numeric", self._scores)
        return cols

    def step(self) -> MessageBatch:
        if self._graph is None:
            self.capture()
        self._graph.replay()
        return _LazyStepBatch(self._cols, self.n, self.device,
                              input_name="kafka")


class FusedStepSource:
    """Input-SPI facade over FusedGenerateFilterInfer: `read()` yields the
    fully processed batch, so a Stream/bench drives the fused step through
    the normal (input, pipeline) shape with an empty pipeline.

    With two instances the source software-pipelines in ONE thread: each
    ``read()`` first launches the NEXT step's replay on the other
    instance's stream, then drains the PREVIOUS step's count readback —
    so step k+1's kernels run while step k's result is consumed. An
    instance's static buffers are re-replayed only one full read later,
    after the sequential engine loop has processed its batch; callers that
    retain batches across steps must copy (the ring-buffer contract)."""

    nonblocking = True  # read() completes in one GPU step — never waits
    # on external IO, so the engine's direct loop may skip its per-read
    # cancellation race

    def __init__(self, fused: FusedGenerateFilterInfer,
                 ninstances: int = 1, make_instance=None,
                 clone: bool = False):
        self.insts = [fused]
        self.streams = [torch.cuda.Stream()]
        self.clone = clone  # engine mode: queues/buffers retain batches
        for _ in range(max(ninstances, 1) - 1):
            self.insts.append(make_instance() if make_instance else fused)
            self.streams.append(torch.cuda.Stream())
        from collections import deque
        self._q = deque()  # replayed-but-not-consumed instance ring
        self._next = 0
        for inst, stream in zip(self.insts, self.streams):
            if inst._graph is None:
                with torch.cuda.stream(stream):
                    inst.capture()
                stream.synchronize()

    def _replay(self, i: int) -> None:
        with torch.cuda.stream(self.streams[i]):
            self.insts[i]._graph.replay()

    def _consume(self, i: int):
        with torch.cuda.stream(self.streams[i]):
            inst = self.insts[i]
            if self.clone:
                # engine mode: queues / window buffers retain batches past
                # the next replay — detach from the static buffers NOW
                return inst._make_batch(clone=True)
            # bench/direct mode: sync for the count, defer the (zero-copy)
            # column views until something actually reads them
            kept = inst._kept()
            return _LazyStepBatch(
                lambda k=kept: inst._make_batch(rows=k).columns, kept,
                inst.device, input_name="generate")

    def _launch_next(self) -> None:
        self._replay(self._next)
        self._q.append(self._next)
        self._next = (self._next + 1) % len(self.insts)

    async def read(self):
        from ..spi import NoopAck
        if len(self.insts) == 1:
            self._replay(0)
            return self._consume(0), NoopAck()
        # keep N-1 replays in flight while consuming the oldest: the newest
        # launch goes in BEFORE the oldest's sync so the GPU never drains.
        # An instance is re-replayed exactly one read() after its batch was
        # handed out — the documented buffer-validity contract.
        while len(self._q) < len(self.insts) - 1:
            self._launch_next()
        i = self._q.popleft()
        self._launch_next()
        batch = self._consume(i)
        return batch, NoopAck()

    async def connect(self):  # pragma: no cover - trivial
        pass

    async def close(self):  # pragma: no cover - trivial
        pass

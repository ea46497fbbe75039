"""MLP anomaly detector — the native ML-inference model the reference only
promises via its python-processor escape hatch (reference README.md:16-18,
processor/python.rs:47-98). Random-init weights (no network in this env);
forward = chained fused GEMM+bias+ReLU kernels, bf16 MFMA on GPU."""
from __future__ import annotations

from typing import List, Sequence

import torch

from ..ops import nn as opsnn


class MlpAnomalyDetector:
    def __init__(self, in_features: int, hidden: Sequence[int] = (256, 256),
                 device: torch.device = torch.device("cpu"), seed: int = 1234):
        self.device = torch.device(device)
        self.in_features = in_features
        gen = torch.Generator().manual_seed(seed)
        # hidden dims rounded up to 32 so every GEMM K is a multiple of 32
        # (MFMA kernel contract); the raw input is zero-padded to match.
        dims = [_pad32(in_features), *[_pad32(h) for h in hidden], 1]
        self.weights: List[torch.Tensor] = []
        self.biases: List[torch.Tensor] = []
        for i in range(len(dims) - 1):
            k = dims[i]
            n = dims[i + 1]
            w = torch.randn(n, k, generator=gen) * (k ** -0.5)
            b = torch.randn(n, generator=gen) * 0.01
            self.weights.append(w.to(self.device, torch.bfloat16))
            self.biases.append(b.to(self.device, torch.float32))
        self.dims = dims
        self.use_graph = True
        self._graphs = {}  # padded-cap → (graph, static_in, static_out)
        self._head_bias = None  # scalar bias of the N=1 scoring head

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        """x: [n, in_features] float → anomaly score [n] float32.

        On GPU the whole forward is hipGraph-captured over a padded static
        batch (row count varies after upstream filters; scores are per-row so
        stale pad rows are harmless and sliced off)."""
        if self.use_graph and self.device.type == "cuda":
            return self._forward_graphed(x)
        return self._forward_eager(x)

    def _forward_eager(self, x: torch.Tensor) -> torch.Tensor:
        h = x.to(self.device, torch.bfloat16)
        pad = _pad32(self.in_features) - self.in_features
        if pad:
            h = torch.nn.functional.pad(h, (0, pad))
        return self._net(h)

    def _net(self, h: torch.Tensor) -> torch.Tensor:
        for i, (w, b) in enumerate(zip(self.weights, self.biases)):
            last = i == len(self.weights) - 1
            if last and h.is_cuda and w.shape[0] == 1:
                # scoring head: one fused GEMV+downcast launch instead of a
                # hipBLASLt N=1 GEMM + bf16→f32 copy
                from ..ops import require_native
                if self._head_bias is None:
                    self._head_bias = float(b[0].item())
                return require_native().gemv_bf16_f32(
                    h.contiguous(), w.reshape(-1).contiguous(),
                    self._head_bias)
            act = "relu" if not last else "none"
            h = opsnn.linear_bf16(h, w, b, act=act)
        return h.reshape(-1).to(torch.float32)

    def _forward_graphed(self, x: torch.Tensor) -> torch.Tensor:
        n = x.shape[0]
        cap = self._cap_for(n)
        entry = self._graphs.get(cap)
        if entry is None:
            static_in = torch.zeros(cap, _pad32(self.in_features),
                                    device=self.device, dtype=torch.bfloat16)
            s = torch.cuda.Stream()
            s.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(s):
                for _ in range(2):
                    self._net(static_in)
            torch.cuda.current_stream().wait_stream(s)
            graph = torch.cuda.CUDAGraph()
            with torch.cuda.graph(graph, capture_error_mode="thread_local"):
                static_out = self._net(static_in)
            entry = (graph, static_in, static_out)
            self._graphs[cap] = entry
        graph, static_in, static_out = entry
        static_in[:n, : self.in_features].copy_(
            x.to(self.device, torch.bfloat16))
        graph.replay()
        return static_out[:n].clone()

    @staticmethod
    def _cap_for(n: int) -> int:
        cap = 4096
        while cap < n:
            cap <<= 1
        return cap


def _pad32(k: int) -> int:
    return (k + 31) // 32 * 32

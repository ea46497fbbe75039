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

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        """x: [n, in_features] float → anomaly score [n] float32."""
        h = x.to(self.device, torch.bfloat16)
        pad = _pad32(self.in_features) - self.in_features
        if pad:
            h = torch.nn.functional.pad(h, (0, pad))
        for i, (w, b) in enumerate(zip(self.weights, self.biases)):
            act = "relu" if i < len(self.weights) - 1 else "none"
            h = opsnn.linear_bf16(h, w, b, act=act)
        return h.reshape(-1).to(torch.float32)


def _pad32(k: int) -> int:
    return (k + 31) // 32 * 32

"""LDS row-pad sweep for the fused attention kernel (NOTES: 590K bank
conflicts/dispatch at pad=4)."""
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
import torch

from arkflow_amd.ops import require_native

nat = require_native()
B, H, S, D = 64, 12, 128, 64
qkv = torch.randn(B, S, 3, H, D, device="cuda", dtype=torch.bfloat16)
scale = D ** -0.5
ref = None
for pad in (4, 8, 16, 20):
    out = nat.attention_qkv_bf16(qkv, scale, pad)
    if ref is None:
        q = qkv[:, :, 0].permute(0, 2, 1, 3).float()
        k = qkv[:, :, 1].permute(0, 2, 1, 3).float()
        v = qkv[:, :, 2].permute(0, 2, 1, 3).float()
        p = torch.softmax(q @ k.transpose(-1, -2) * scale, -1)
        ref = (p @ v).permute(0, 2, 1, 3).reshape(B, S, H * D)
    err = (out.float() - ref).abs().max().item()
    for _ in range(20):
        nat.attention_qkv_bf16(qkv, scale, pad)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(300):
        nat.attention_qkv_bf16(qkv, scale, pad)
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / 300
    print(f"pad={pad}: {dt*1e6:.1f} us  maxerr {err:.4f}", flush=True)

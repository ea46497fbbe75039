"""Standalone GEMM microbench: our gemm_bf16 vs torch.matmul (hipBLASLt)."""
import sys
import time

import torch

sys.path.insert(0, __import__("os").path.join(__import__("os").path.dirname(__file__), ".."))
from arkflow_amd import ops

nat = ops.require_native()
dev = torch.device("cuda:0")


def bench(fn, iters=30):
    for _ in range(5):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


shapes = [(8192, 768, 768), (8192, 2304, 768), (8192, 3072, 768),
          (8192, 768, 3072), (4096, 4096, 4096), (8192, 8192, 8192)]
for M, N, K in shapes:
    A = torch.randn(M, K, device=dev, dtype=torch.bfloat16)
    Bt = torch.randn(N, K, device=dev, dtype=torch.bfloat16)
    t_ours = bench(lambda: nat.gemm_bf16(A, Bt, None, 0))
    B = Bt.T.contiguous()
    t_torch = bench(lambda: A @ B)
    fl = 2.0 * M * N * K
    print(f"M{M} N{N} K{K}: ours {t_ours*1e6:8.1f}us {fl/t_ours/1e12:7.1f}TF"
          f" | torch {t_torch*1e6:8.1f}us {fl/t_torch/1e12:7.1f}TF",
          flush=True)
